"""MiniMax-M2: GQA attention with FULL-WIDTH q/k RMSNorm (one norm over the
concatenated head dim, unlike qwen3's per-head norm) + sigmoid-routed MoE with
e_score_correction_bias on every layer (reference analogue:
src/parallax/models/minimax.py; HF modeling_minimax_m2).

The attention/MoE machinery is the shared Llama/FusedMoE stack — this module
contributes the weight-name routing (the bias lives at mlp.e_score_correction_bias,
not under the gate) and the registry entry; qk_norm_full is set by
ModelConfig.from_hf_config for model_type minimax_m2."""

from __future__ import annotations

import torch

from .moe import MoEBlock
from .qwen3_moe import Qwen3MoEShardModel
from .registry import register_model


@register_model("MiniMaxM2ForCausalLM")
class MiniMaxM2ShardModel(Qwen3MoEShardModel):
    def load_hf_weight(self, name: str, tensor: torch.Tensor) -> bool:
        if ".mlp.e_score_correction_bias" in name:
            local = self.map_global_layer(name)
            if local is None:
                return False
            layer = self.layers[int(local.split(".")[1])]
            if not isinstance(layer.mlp, MoEBlock):
                return False
            layer.mlp.experts.router.e_score_correction_bias.data.copy_(
                tensor.float()
            )
            return True
        return super().load_hf_weight(name, tensor)
