"""Selective safetensors shard loading for a layer range.

Parity with the reference's src/parallax/server/shard_loader.py:342 — map
global->local layer keys, open only the safetensors files that contain tensors
for this shard's layer range, and route each tensor through the model's
load_hf_weight (which handles fused-QKV stacking and TP sharding). No network:
model_path is a local HF-format directory.
"""

from __future__ import annotations

import json
import os
from typing import Dict, List, Optional, Set

from ..utils.logging_config import get_logger

logger = get_logger("server.shard_loader")


def list_safetensors_files(model_path: str) -> List[str]:
    idx = os.path.join(model_path, "model.safetensors.index.json")
    if os.path.exists(idx):
        with open(idx) as f:
            weight_map: Dict[str, str] = json.load(f)["weight_map"]
        return sorted(set(weight_map.values())), weight_map
    single = os.path.join(model_path, "model.safetensors")
    if os.path.exists(single):
        return ["model.safetensors"], None
    files = sorted(f for f in os.listdir(model_path) if f.endswith(".safetensors"))
    if not files:
        raise FileNotFoundError(f"no safetensors files under {model_path}")
    return files, None


def load_shard_weights(model, model_path: str) -> int:
    """Load the weights this shard needs. Returns the number of tensors loaded."""
    from safetensors import safe_open

    files, weight_map = list_safetensors_files(model_path)

    def wants(name: str) -> bool:
        return model.map_global_layer(name) is not None

    needed_files: Set[str] = set()
    if weight_map is not None:
        for name, fname in weight_map.items():
            if wants(name):
                needed_files.add(fname)
    else:
        needed_files = set(files)

    loaded = 0
    for fname in sorted(needed_files):
        path = os.path.join(model_path, fname)
        with safe_open(path, framework="pt", device="cpu") as f:
            for name in f.keys():
                if not wants(name):
                    continue
                if model.load_hf_weight(name, f.get_tensor(name)):
                    loaded += 1
    logger.info(
        "loaded %d tensors from %d file(s) for layers [%d,%d)",
        loaded, len(needed_files), model.start_layer, model.end_layer,
    )
    return loaded


def selective_file_list(model_path: str, start_layer: int, end_layer: int) -> List[str]:
    """Which safetensors files cover a layer range (the download planner for the
    decentralized layer; parity with utils/model_download.py:79 selective pick)."""
    files, weight_map = list_safetensors_files(model_path)
    if weight_map is None:
        return files
    keep: Set[str] = set()
    for name, fname in weight_map.items():
        if name.startswith("model.layers."):
            g = int(name.split(".")[2])
            if start_layer <= g < end_layer:
                keep.add(fname)
        else:
            keep.add(fname)  # embeddings / norms / lm_head travel with endpoints
    return sorted(keep)
