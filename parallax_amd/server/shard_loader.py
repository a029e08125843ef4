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


def load_shard_weights(model, model_path: str, lora_path: str = None,
                       lora_scale: float = 1.0) -> int:
    """Load the weights this shard needs (optionally fusing a LoRA adapter).
    Returns the number of tensors loaded."""
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
    if lora_path:
        loaded += fuse_lora(model, lora_path, lora_scale)
    return loaded


def fuse_lora(model, lora_path: str, scale: float = 1.0) -> int:
    """Fuse a PEFT-format LoRA adapter into the base weights at load time
    (reference shard_loader.py:114-228 behavior: W += scale * B @ A). The
    adapter never exists as separate serving state — fused weights keep the
    decode hot path unchanged."""
    import json as _json

    import torch
    from safetensors import safe_open

    cfg_path = os.path.join(lora_path, "adapter_config.json")
    if os.path.exists(cfg_path):
        with open(cfg_path) as f:
            acfg = _json.load(f)
        r = acfg.get("r", 8)
        alpha = acfg.get("lora_alpha", r)
        scale = scale * alpha / max(1, r)

    adapter_file = None
    for cand in ("adapter_model.safetensors", "adapter.safetensors"):
        if os.path.exists(os.path.join(lora_path, cand)):
            adapter_file = os.path.join(lora_path, cand)
            break
    if adapter_file is None:
        raise FileNotFoundError(f"no adapter safetensors under {lora_path}")

    pairs: Dict[str, Dict[str, "torch.Tensor"]] = {}
    with safe_open(adapter_file, framework="pt", device="cpu") as f:
        for name in f.keys():
            # base_model.model.<hf name>.lora_A.weight / lora_B.weight
            if ".lora_A." in name:
                base, kind = name.split(".lora_A."), "A"
            elif ".lora_B." in name:
                base, kind = name.split(".lora_B."), "B"
            else:
                continue
            key = base[0].replace("base_model.model.", "")
            pairs.setdefault(key, {})[kind] = f.get_tensor(name)

    fused = 0
    for key, ab in pairs.items():
        if "A" not in ab or "B" not in ab:
            continue
        delta = (ab["B"].float() @ ab["A"].float()) * scale
        # route the DELTA through the model's normal weight router by fetching
        # the current weight, adding, and re-loading
        hf_name = key + ".weight"
        cur = _read_current_weight(model, hf_name, delta.shape)
        if cur is None:
            continue
        if model.load_hf_weight(hf_name, (cur + delta).to(torch.float32)):
            fused += 1
    logger.info("fused %d LoRA deltas from %s (scale %.3f)", fused, lora_path, scale)
    return fused


def _read_current_weight(model, hf_name: str, shape):
    """Reconstruct the full (unsharded) current weight for a fusable target.
    Works for the llama-family fused projections at tp_size 1 (LoRA fusion with
    TP sharding re-loads from the base checkpoint first)."""
    import torch

    local = model.map_global_layer(hf_name)
    if local is None:
        return None
    parts = local.split(".")
    stem = parts[-2]
    stacked = getattr(model, "_STACKED", {})
    try:
        if stem in stacked:
            target, idx = stacked[stem]
            module = model
            for p in parts[:-2] + [target]:
                module = module[int(p)] if p.isdigit() else getattr(module, p)
            off = sum(module.shard_sizes[:idx])
            return module.weight.data[off : off + shape[0]].float()
        module = model
        for p in parts[:-1]:
            module = module[int(p)] if p.isdigit() else getattr(module, p)
        return module.weight.data.float()
    except (AttributeError, IndexError, KeyError):
        return None


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
