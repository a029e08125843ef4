"""Architecture registry: HF `architectures[0]` string -> sharded model class."""

from __future__ import annotations

from typing import Dict, Type

MODEL_REGISTRY: Dict[str, type] = {}


def register_model(*arch_names: str):
    def deco(cls):
        for name in arch_names:
            MODEL_REGISTRY[name] = cls
        return cls

    return deco


def get_model_class(arch_name: str):
    if arch_name not in MODEL_REGISTRY:
        raise KeyError(
            f"architecture {arch_name!r} not supported; known: {sorted(MODEL_REGISTRY)}"
        )
    return MODEL_REGISTRY[arch_name]
