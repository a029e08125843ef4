"""Llama-family decoder (covers LlamaForCausalLM and Qwen2ForCausalLM — the
flagship DeepSeek-R1-Distill-Llama-8B is this architecture).

Fresh MI355X-first design (reference analogue: src/parallax/models/llama.py /
qwen2.py): RMSNorm / fused-add-RMSNorm, RoPE, KV scatter, paged attention and
SwiGLU all route through parallax_amd.ops (HIP kernels on GPU); QKV and gate/up
are fused single GEMMs (hipBLASLt); TP via column/row-parallel layers with RCCL
all-reduce over xGMI.
"""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn as nn

from .. import ops
from ..parallel.comm import get_comm
from ..parallel.layers import (
    ColumnParallelLinear,
    MergedColumnParallelLinear,
    RowParallelLinear,
    VocabEmbedding,
)
from .config import ModelConfig
from .forward_meta import ForwardMeta
from .registry import register_model
from .rope import build_rope_cache_for


class RMSNorm(nn.Module):
    def __init__(self, hidden_size: int, eps: float):
        super().__init__()
        self.weight = nn.Parameter(torch.empty(hidden_size), requires_grad=False)
        self.eps = eps

    def forward(self, x: torch.Tensor, residual: Optional[torch.Tensor] = None):
        if residual is None:
            return ops.rmsnorm(x, self.weight, self.eps)
        return ops.fused_add_rmsnorm(x, residual, self.weight, self.eps)


class LlamaAttention(nn.Module):
    def __init__(self, cfg: ModelConfig, layer_idx: int, local_layer_idx: int):
        super().__init__()
        comm = get_comm()
        self.layer_idx = layer_idx
        self.local_layer_idx = local_layer_idx
        self.head_dim = cfg.head_dim
        self.num_heads = cfg.num_heads // comm.tp_size
        self.num_kv_heads = max(1, cfg.num_kv_heads // comm.tp_size)
        self.scale = self.head_dim ** -0.5
        self.sliding_window = (
            cfg.sliding_window if cfg.layer_type(layer_idx) == "sliding_attention" else -1
        )
        self.qk_norm = cfg.qk_norm
        h = cfg.hidden_size
        self.qkv_proj = MergedColumnParallelLinear(
            h,
            [cfg.num_heads * cfg.head_dim, cfg.num_kv_heads * cfg.head_dim,
             cfg.num_kv_heads * cfg.head_dim],
            bias=cfg.attention_bias,
        )
        self.o_proj = RowParallelLinear(
            cfg.num_heads * cfg.head_dim, h, bias=cfg.o_proj_bias
        )
        self.qk_norm_full = cfg.qk_norm_full
        if self.qk_norm:
            if self.qk_norm_full:
                # minimax-m2: one RMSNorm over the whole concatenated q (and k)
                # width — couples heads, so TP sharding would change the math
                assert comm.tp_size == 1, "full-width qk_norm incompatible with TP"
                self.q_norm = RMSNorm(cfg.num_heads * cfg.head_dim, cfg.rms_norm_eps)
                self.k_norm = RMSNorm(cfg.num_kv_heads * cfg.head_dim, cfg.rms_norm_eps)
            else:
                self.q_norm = RMSNorm(cfg.head_dim, cfg.rms_norm_eps)
                self.k_norm = RMSNorm(cfg.head_dim, cfg.rms_norm_eps)
        self.sinks: Optional[nn.Parameter] = None
        if cfg.attention_sinks:
            self.sinks = nn.Parameter(
                torch.empty(self.num_heads, dtype=torch.float32), requires_grad=False
            )

    def forward(self, x: torch.Tensor, meta: ForwardMeta, rope_cache: torch.Tensor):
        T = x.shape[0]
        qkv = self.qkv_proj(x)
        q, k, v = self.qkv_proj.split_output(qkv)
        # strided views into the fused QKV buffer — no contiguous copies; the
        # HIP kernels take a row stride
        q = q.view(T, self.num_heads, self.head_dim)
        k = k.view(T, self.num_kv_heads, self.head_dim)
        v = v.view(T, self.num_kv_heads, self.head_dim)
        if self.qk_norm:
            if self.qk_norm_full:
                q = self.q_norm(q.reshape(T, -1)).view(T, self.num_heads, self.head_dim)
                k = self.k_norm(k.reshape(T, -1)).view(T, self.num_kv_heads, self.head_dim)
            else:
                q = self.q_norm(q.contiguous())
                k = self.k_norm(k.contiguous())
            v = v.contiguous()
        k_cache, v_cache = meta.kv_cache.layer(self.local_layer_idx)
        ops.rope_and_cache(
            q, k, v, k_cache, v_cache, meta.positions, rope_cache,
            meta.slot_mapping,
        )

        sinks = self.sinks.float() if self.sinks is not None else None
        if meta.is_prefill:
            attn = ops.prefill_attention(
                q, k_cache, v_cache, meta.block_tables, meta.seq_lens,
                meta.query_lens, self.scale, self.sliding_window, sinks=sinks,
            )
        else:
            attn = ops.paged_attention_decode(
                q, k_cache, v_cache, meta.block_tables, meta.seq_lens,
                self.scale, self.sliding_window, sinks=sinks,
                max_seq_len=meta.max_seq_len or None,
            )
        return self._project_out(attn, x, T)

    def _project_out(self, attn: torch.Tensor, x: torch.Tensor, T: int):
        """Output projection; subclasses may gate `attn` first (step3p5)."""
        return self.o_proj(attn.reshape(T, self.num_heads * self.head_dim))


class LlamaMLP(nn.Module):
    def __init__(self, cfg: ModelConfig, intermediate_size: Optional[int] = None):
        super().__init__()
        inter = intermediate_size or cfg.intermediate_size
        self.gate_up_proj = MergedColumnParallelLinear(
            cfg.hidden_size, [inter, inter], bias=cfg.mlp_bias
        )
        self.down_proj = RowParallelLinear(inter, cfg.hidden_size, bias=cfg.mlp_bias)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.down_proj(ops.silu_and_mul(self.gate_up_proj(x)))


class LlamaDecoderLayer(nn.Module):
    def __init__(self, cfg: ModelConfig, layer_idx: int, local_layer_idx: int):
        super().__init__()
        self.self_attn = LlamaAttention(cfg, layer_idx, local_layer_idx)
        self.mlp = LlamaMLP(cfg)
        self.input_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps)
        self.post_attention_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps)

    def forward(self, x, residual, meta: ForwardMeta, rope_cache):
        if residual is None:
            residual = x
            x = self.input_layernorm(x)
        else:
            x, residual = self.input_layernorm(x, residual)
        x = self.self_attn(x, meta, rope_cache)
        x, residual = self.post_attention_layernorm(x, residual)
        x = self.mlp(x)
        return x, residual


@register_model("LlamaForCausalLM", "Qwen2ForCausalLM", "MistralForCausalLM")
class LlamaShardModel(nn.Module):
    """A contiguous layer range [start_layer, end_layer) of a Llama-family model.
    First shard owns the embedding; last shard owns final norm + lm_head
    (reference base_executor.py:124-125 stage-role convention)."""

    decoder_layer_cls = LlamaDecoderLayer

    def __init__(self, cfg: ModelConfig, start_layer: int = 0, end_layer: Optional[int] = None):
        super().__init__()
        self.cfg = cfg
        self.start_layer = start_layer
        self.end_layer = end_layer if end_layer is not None else cfg.num_layers
        self.is_first = start_layer == 0
        self.is_last = self.end_layer == cfg.num_layers

        if self.is_first:
            self.embed_tokens = VocabEmbedding(cfg.vocab_size, cfg.hidden_size)
        self.layers = nn.ModuleList(
            self._build_layers(cfg, start_layer, self.end_layer)
        )
        if self.is_last:
            self.norm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps)
            self.lm_head = ColumnParallelLinear(cfg.hidden_size, cfg.vocab_size,
                                                gather_output=True)
        self.register_buffer(
            "rope_cache",
            build_rope_cache_for(cfg, rot_dim=cfg.rot_dim),
            persistent=False,
        )

    def _build_layers(self, cfg: ModelConfig, start: int, end: int):
        return [
            self.decoder_layer_cls(cfg, g, i)
            for i, g in enumerate(range(start, end))
        ]

    @property
    def num_local_layers(self) -> int:
        return len(self.layers)

    def embed(self, input_ids: torch.Tensor) -> torch.Tensor:
        assert self.is_first, "only the first shard embeds tokens"
        return self.embed_tokens(input_ids)

    def forward(self, hidden: torch.Tensor, meta: ForwardMeta) -> torch.Tensor:
        """hidden: [T, hidden] — embeddings on the first shard, the previous
        stage's output elsewhere. Returns [T, hidden] post final-norm on the
        last shard, pre-norm activations otherwise."""
        residual = None
        for layer in self.layers:
            hidden, residual = layer(hidden, residual, meta, self.rope_cache)
        if self.is_last:
            if residual is None:  # post-norm families (minimax) fold residuals
                hidden = self.norm(hidden)
            else:
                hidden, _ = self.norm(hidden, residual)
        elif residual is not None:
            hidden = hidden + residual
        return hidden

    def compute_logits(self, hidden: torch.Tensor) -> torch.Tensor:
        assert self.is_last, "only the last shard computes logits"
        # lm_head is a ColumnParallelLinear (gathers under TP) and already
        # routes through ops.linear -> the tuned hipBLASLt path
        return self.lm_head(
            hidden.to(self.lm_head.weight.dtype).contiguous()
        ).float()

    # -- weight loading ---------------------------------------------------------

    _STACKED = {
        "q_proj": ("qkv_proj", 0),
        "k_proj": ("qkv_proj", 1),
        "v_proj": ("qkv_proj", 2),
        "gate_proj": ("gate_up_proj", 0),
        "up_proj": ("gate_up_proj", 1),
    }

    def map_global_layer(self, name: str) -> Optional[str]:
        """'model.layers.<g>.rest' -> 'layers.<local>.rest' (None if out of range);
        parity with the reference shard loader's key remap (shard_loader.py:229)."""
        if name.startswith("model.layers."):
            parts = name.split(".")
            g = int(parts[2])
            if not (self.start_layer <= g < self.end_layer):
                return None
            return ".".join(["layers", str(g - self.start_layer)] + parts[3:])
        if name.startswith("model.embed_tokens."):
            if self.is_first:
                return name.replace("model.", "", 1)
            if self.is_last and self.cfg.tie_word_embeddings:
                return name.replace("model.embed_tokens", "lm_head")
            return None
        if name.startswith("model.norm."):
            return name.replace("model.", "", 1) if self.is_last else None
        if name.startswith("lm_head."):
            return name if self.is_last and not self.cfg.tie_word_embeddings else None
        return None

    def load_hf_weight(self, name: str, tensor: torch.Tensor) -> bool:
        """Route one HF-named tensor into this shard. Returns True if consumed."""
        local = self.map_global_layer(name)
        if local is None:
            return False
        parts = local.split(".")
        leaf = parts[-1]  # 'weight' or 'bias'
        stem = parts[-2]
        tensor = tensor.to(torch.bfloat16)
        if stem in self._STACKED:
            target_name, idx = self._STACKED[stem]
            module = self._resolve(parts[:-2] + [target_name])
            module.load_full_weight_part(idx, tensor if leaf == "weight" else None,
                                         tensor if leaf == "bias" else None)
            return True
        module = self._resolve(parts[:-1])
        if isinstance(module, (ColumnParallelLinear, RowParallelLinear)):
            module.load_full_weight(tensor if leaf == "weight" else None,
                                    tensor if leaf == "bias" else None)
        else:
            getattr(module, leaf).data.copy_(tensor)
        return True

    def _resolve(self, path_parts):
        mod = self
        for p in path_parts:
            mod = mod[int(p)] if p.isdigit() else getattr(mod, p)
        return mod

    @torch.no_grad()
    def init_random(self, seed: int = 1234) -> None:
        """Random-init all parameters (synthetic-weight benchmarking; no network
        for checkpoints). Norm weights -> 1, linears -> N(0, 0.02/sqrt(2L))."""
        g = torch.Generator().manual_seed(seed)
        std = 0.02 / math.sqrt(2 * max(1, self.cfg.num_layers))
        for name, p in self.named_parameters():
            if "norm" in name.lower():
                p.data.fill_(1.0)
            elif name.endswith("bias"):
                p.data.zero_()
            elif "embed" in name or "lm_head" in name:
                p.data.copy_(
                    torch.randn(p.shape, generator=g, dtype=torch.float32).mul_(0.02)
                    .to(p.dtype)
                )
            else:
                p.data.copy_(
                    torch.randn(p.shape, generator=g, dtype=torch.float32).mul_(std)
                    .to(p.dtype)
                )


