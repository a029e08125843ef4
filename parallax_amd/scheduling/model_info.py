"""Model sizing + roofline math for the cluster scheduler.

Behavior parity with the reference's scheduling/model_info.py:95-193 (per-layer
FLOPs/IO with the MoE expected-activated-experts correction, embedding/lm_head
endpoint costs); fresh implementation keyed off our ModelConfig."""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Optional

from ..models.config import ModelConfig


@dataclass
class ModelInfo:
    name: str
    cfg: ModelConfig
    param_bytes_per_element: int = 2  # bf16

    # -- parameter sizes -----------------------------------------------------

    def embedding_io_bytes(self) -> int:
        return self.cfg.embedding_bytes()

    def lm_head_io_bytes(self) -> int:
        return self.cfg.lm_head_bytes() or self.cfg.embedding_bytes()

    def decoder_layer_io_bytes(self, layer_idx: int = -1) -> int:
        """Bytes of parameters read per decode step for one layer. For MoE
        layers only the activated experts are read."""
        cfg = self.cfg
        full = cfg.per_layer_param_bytes(layer_idx)
        if cfg.is_moe and (layer_idx < 0 or cfg.is_moe_layer(layer_idx)):
            inter = cfg.moe_intermediate_size or cfg.intermediate_size
            expert_bytes = 3 * cfg.hidden_size * inter * self.param_bytes_per_element
            routed_total = expert_bytes * cfg.num_experts
            activated = expert_bytes * max(1, cfg.num_experts_per_tok)
            return full - routed_total + activated
        return full

    def decoder_layer_param_bytes(self, layer_idx: int = -1) -> int:
        """Bytes of parameters STORED for one layer (all experts resident)."""
        return self.cfg.per_layer_param_bytes(layer_idx)

    def decoder_layer_flops(self, context_len: int = 1024, batch: int = 1) -> float:
        """FLOPs for one decode token per layer (attention GEMMs + KV dot +
        activated-expert FFN)."""
        cfg = self.cfg
        h = cfg.hidden_size
        attn_proj = 2 * h * (cfg.num_heads + 2 * cfg.num_kv_heads) * cfg.head_dim \
            + 2 * cfg.num_heads * cfg.head_dim * h
        attn_kv = 4 * cfg.num_heads * cfg.head_dim * context_len
        if cfg.is_moe:
            inter = cfg.moe_intermediate_size or cfg.intermediate_size
            ffn = 6 * h * inter * max(1, cfg.num_experts_per_tok)
            if cfg.num_shared_experts:
                ffn += 6 * h * inter * cfg.num_shared_experts
        else:
            ffn = 6 * h * cfg.intermediate_size
        return float(batch * (attn_proj + attn_kv + ffn))

    def kv_bytes_per_token_per_layer(self) -> int:
        cfg = self.cfg
        if cfg.is_mla:
            return (cfg.kv_lora_rank + cfg.qk_rope_head_dim) * self.param_bytes_per_element
        return 2 * cfg.num_kv_heads * cfg.head_dim * self.param_bytes_per_element

    @property
    def num_layers(self) -> int:
        return self.cfg.num_layers

    def total_param_bytes(self) -> int:
        total = self.cfg.embedding_bytes() + self.cfg.lm_head_bytes()
        for i in range(self.cfg.num_layers):
            total += self.cfg.per_layer_param_bytes(i)
        return total

    @classmethod
    def from_config(cls, name: str, cfg: ModelConfig) -> "ModelInfo":
        return cls(name=name, cfg=cfg)
