"""Qwen3.5: the qwen3_next hybrid stack (3:1 gated-DeltaNet : gated full
attention, partial rotary 0.25, Gemma-style (1+w) norms) with SPLIT DeltaNet
input projections — in_proj_qkv ([q|k|v] plain concat, no per-kv-head
interleave) plus separate in_proj_z / in_proj_b / in_proj_a — and a dense MLP
in the base model (the MoE variant registers the same shard with routed
experts). Reference analogue: src/parallax/models/qwen3_5.py."""

from __future__ import annotations

import torch
import torch.nn as nn

from .config import ModelConfig
from .qwen3_next import (
    GatedDeltaNet,
    Qwen3NextDecoderLayer,
    Qwen3NextShardModel,
)
from .registry import register_model


class Qwen35GatedDeltaNet(GatedDeltaNet):
    def __init__(self, cfg: ModelConfig, layer_idx: int, linear_layer_idx: int):
        super().__init__(cfg, layer_idx, linear_layer_idx)
        h = cfg.hidden_size
        del self.in_proj_qkvz
        del self.in_proj_ba
        self.in_proj_qkv = nn.Linear(h, 2 * self.key_dim + self.value_dim, bias=False)
        self.in_proj_z = nn.Linear(h, self.value_dim, bias=False)
        self.in_proj_b = nn.Linear(h, self.hv, bias=False)
        self.in_proj_a = nn.Linear(h, self.hv, bias=False)

    def _project(self, x: torch.Tensor):
        T = x.shape[0]
        qkv = self.in_proj_qkv(x)
        q, k, v = torch.split(
            qkv, [self.key_dim, self.key_dim, self.value_dim], dim=-1
        )
        q = q.view(T, self.hk, self.dk)
        k = k.view(T, self.hk, self.dk)
        v = v.view(T, self.hv, self.dv)
        z = self.in_proj_z(x).view(T, self.hv, self.dv)
        b = self.in_proj_b(x)
        a = self.in_proj_a(x)
        return q, k, v, z, b, a


class Qwen35DecoderLayer(Qwen3NextDecoderLayer):
    deltanet_cls = Qwen35GatedDeltaNet


@register_model("Qwen3_5ForCausalLM", "Qwen3_5MoeForCausalLM")
class Qwen35ShardModel(Qwen3NextShardModel):
    hybrid_layer_cls = Qwen35DecoderLayer
