"""Qwen3-MoE: Qwen3 attention (qk-norm) with a routed-expert MLP on every layer
(reference analogue: src/parallax/models/qwen3_moe.py). HF weight names for the
experts (mlp.experts.<e>.gate_proj / up_proj / down_proj and mlp.gate.weight)
are remapped into the stacked FusedMoE tensors on load."""

from __future__ import annotations

from typing import Optional

import torch

from .config import ModelConfig
from .llama import LlamaDecoderLayer, LlamaShardModel
from .moe import MoEBlock
from .registry import register_model


class MoEDecoderLayer(LlamaDecoderLayer):
    def __init__(self, cfg: ModelConfig, layer_idx: int, local_layer_idx: int):
        super().__init__(cfg, layer_idx, local_layer_idx)
        if cfg.is_moe_layer(layer_idx):
            self.mlp = MoEBlock(cfg)


@register_model("Qwen3MoeForCausalLM", "Qwen2MoeForCausalLM")
class Qwen3MoEShardModel(LlamaShardModel):
    decoder_layer_cls = MoEDecoderLayer

    def load_hf_weight(self, name: str, tensor: torch.Tensor) -> bool:
        routed = self._route_moe_weight(name, tensor)
        if routed is not None:
            return routed
        return super().load_hf_weight(name, tensor)

    def _route_moe_weight(self, name: str, tensor: torch.Tensor) -> Optional[bool]:
        """Handle 'model.layers.<g>.mlp.experts.<e>.{gate,up,down}_proj.weight'
        and 'model.layers.<g>.mlp.gate.weight' (router)."""
        if ".mlp.experts." not in name and ".mlp.gate." not in name and \
                ".mlp.shared_expert" not in name:
            return None
        local = self.map_global_layer(name)
        if local is None:
            return False
        parts = local.split(".")
        layer = self.layers[int(parts[1])]
        if not isinstance(layer.mlp, MoEBlock):
            return False
        t = tensor.to(torch.bfloat16)
        inter = layer.mlp.experts.intermediate_size
        if parts[3] == "gate":  # router
            layer.mlp.experts.router.weight.data.copy_(t)
            return True
        if parts[3] == "experts":
            if parts[4] == "gate_up_proj":  # fused [E, 2I, H] (transformers >= 5)
                layer.mlp.experts.load_fused_gate_up(t)
                return True
            if parts[4] == "down_proj":
                layer.mlp.experts.load_fused_down(t)
                return True
            e = int(parts[4])
            proj = parts[5]
            if proj == "gate_proj":
                layer.mlp.experts.load_expert_gate(e, t)
            elif proj == "up_proj":
                layer.mlp.experts.load_expert_up(e, t)
            elif proj == "down_proj":
                layer.mlp.experts.load_expert_down(e, t)
            return True
        if parts[3].startswith("shared_expert"):
            if parts[3] == "shared_expert_gate":
                return True  # qwen2-moe shared gate: not modeled (sums directly)
            proj, leaf = parts[4], parts[5]
            mlp = layer.mlp.shared
            if mlp is None:
                return False
            if proj == "gate_proj":
                mlp.gate_up_proj.load_full_weight_part(0, t)
            elif proj == "up_proj":
                mlp.gate_up_proj.load_full_weight_part(1, t)
            elif proj == "down_proj":
                mlp.down_proj.load_full_weight(t)
            return True
        return False
