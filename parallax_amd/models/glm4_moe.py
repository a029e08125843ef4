"""GLM4-MoE: llama-style GQA attention with partial rotary (rot = head_dim/2),
optional per-head qk-norm, and a DeepSeek-style sigmoid-scored MoE with
e_score_correction_bias + one shared expert (reference analogue:
src/parallax/models/glm4_moe.py)."""

from __future__ import annotations

import torch

from .config import ModelConfig
from .llama import LlamaDecoderLayer, LlamaShardModel
from .moe import MoEBlock
from .registry import register_model


class Glm4MoeDecoderLayer(LlamaDecoderLayer):
    def __init__(self, cfg: ModelConfig, layer_idx: int, local_layer_idx: int):
        super().__init__(cfg, layer_idx, local_layer_idx)
        if cfg.is_moe_layer(layer_idx):
            self.mlp = MoEBlock(cfg)


@register_model("Glm4MoeForCausalLM")
class Glm4MoeShardModel(LlamaShardModel):
    decoder_layer_cls = Glm4MoeDecoderLayer

    def load_hf_weight(self, name: str, tensor: torch.Tensor) -> bool:
        local = self.map_global_layer(name)
        if local is None:
            return False
        parts = local.split(".")
        if len(parts) >= 4 and parts[0] == "layers" and parts[2] == "mlp":
            layer = self.layers[int(parts[1])]
            if isinstance(layer.mlp, MoEBlock):
                return self._load_moe(layer.mlp, parts[3:], tensor.to(torch.bfloat16))
        return super().load_hf_weight(name, tensor)

    def _load_moe(self, moe: MoEBlock, parts, t: torch.Tensor) -> bool:
        inter = moe.experts.intermediate_size
        if parts[0] == "gate":
            if parts[1] == "weight":
                moe.experts.router.weight.data.copy_(t)
            else:  # e_score_correction_bias
                moe.experts.router.e_score_correction_bias.data.copy_(t.float())
            return True
        if parts[0] == "experts":
            if parts[1] == "gate_up_proj":
                moe.experts.w_gate_up.data.copy_(t)
                return True
            if parts[1] == "down_proj":
                moe.experts.w_down.data.copy_(t)
                return True
            e, proj = int(parts[1]), parts[2]
            if proj == "gate_proj":
                moe.experts.w_gate_up.data[e, :inter].copy_(t)
            elif proj == "up_proj":
                moe.experts.w_gate_up.data[e, inter:].copy_(t)
            elif proj == "down_proj":
                moe.experts.w_down.data[e].copy_(t)
            return True
        if parts[0] == "shared_experts":
            proj = parts[1]
            if moe.shared is None:
                return False
            if proj == "gate_proj":
                moe.shared.gate_up_proj.load_full_weight_part(0, t)
            elif proj == "up_proj":
                moe.shared.gate_up_proj.load_full_weight_part(1, t)
            elif proj == "down_proj":
                moe.shared.down_proj.load_full_weight(t)
            return True
        return False
