"""Step-3.5: Llama-family block with per-head q/k RMSNorm, interleaved
sliding/full attention layers, an optional head-wise attention output gate
(out_h *= sigmoid(g_proj(x)_h)) and a MoE stack with a shared expert.

Reference analogue: src/parallax/models/step3p5.py:26-208 (MLX wrapper over
mlx_lm's step3p5: q_norm/k_norm, per-layer is_sliding, use_head_wise_attn_gate
-> g_proj, Step3p5MLP dense / Step3p5MoE with share_expert). No offline HF
oracle for this family exists in this environment (transformers 5.15 has no
step3p5), so coverage is determinism + chunked-prefill/decode-consistency
tests (tests/test_step3p5.py); weight names follow the HF convention the
reference's loader consumes (model.layers.N.self_attn.g_proj.weight etc.).
"""

from __future__ import annotations

import torch

from .config import ModelConfig
from .llama import LlamaAttention, LlamaDecoderLayer, LlamaMLP, LlamaShardModel
from .moe import MoEBlock
from .registry import register_model
from ..parallel.layers import ColumnParallelLinear


class Step3p5Attention(LlamaAttention):
    def __init__(self, cfg: ModelConfig, layer_idx: int, local_layer_idx: int):
        super().__init__(cfg, layer_idx, local_layer_idx)
        self.g_proj = None
        if cfg.use_attn_gate:
            # one sigmoid gate logit per (sharded) query head
            self.g_proj = ColumnParallelLinear(
                cfg.hidden_size, cfg.num_heads, bias=False
            )

    def _project_out(self, attn: torch.Tensor, x: torch.Tensor, T: int):
        if self.g_proj is not None:
            gate = torch.sigmoid(self.g_proj(x))          # [T, H]
            attn = attn.view(T, self.num_heads, self.head_dim) * gate.unsqueeze(-1)
        return self.o_proj(attn.reshape(T, self.num_heads * self.head_dim))


class Step3p5DecoderLayer(LlamaDecoderLayer):
    def __init__(self, cfg: ModelConfig, layer_idx: int, local_layer_idx: int):
        super().__init__(cfg, layer_idx, local_layer_idx)
        self.self_attn = Step3p5Attention(cfg, layer_idx, local_layer_idx)
        if cfg.is_moe_layer(layer_idx):
            self.mlp = MoEBlock(cfg)
        else:
            self.mlp = LlamaMLP(cfg)


@register_model("Step3p5ForCausalLM")
class Step3p5ShardModel(LlamaShardModel):
    decoder_layer_cls = Step3p5DecoderLayer
