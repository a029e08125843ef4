"""MiniMax-Text-01 / M1: hybrid lightning (linear) attention + softmax MoE.

Reference analogue: src/parallax/models/minimax.py. Alternating layer stack
(layer_types): 'linear_attention' layers run lightning attention — a
decay-weighted linear attention with per-head slope rates, silu'd fused QKV, a
full-width RMSNorm and a sigmoid output gate — over a per-slot recurrent
state S[H, d, d] (k^T v accumulator) held in the LinearStateCache;
'full_attention' layers are plain GQA over the paged KV cache. Every layer has
a Mixtral-style softmax top-k MoE, and the residual stream is the MiniMax
norm-first variant: residual = NORMED hidden, combined with configurable
alpha/beta factors.

Prefill replicates the HF 256-token block recursion exactly (intra-block
decay matrix + inter-block decayed state), so chunked prefill at any boundary
is bit-consistent with a single-shot prefill; decode is one batched recurrence
step (graph-capturable einsums over the slot-indexed state)."""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from .config import ModelConfig
from .forward_meta import ForwardMeta
from .llama import LlamaAttention, LlamaShardModel, RMSNorm
from .moe import FusedMoE
from .registry import register_model


class LightningAttention(nn.Module):
    def __init__(self, cfg: ModelConfig, layer_idx: int, linear_layer_idx: int):
        super().__init__()
        self.linear_layer_idx = linear_layer_idx
        self.num_heads = cfg.num_heads
        self.head_dim = cfg.head_dim
        self.block_size = int(cfg.raw.get("block_size", 256))
        h = cfg.hidden_size
        H, d = self.num_heads, self.head_dim
        self.qkv_proj = nn.Linear(h, 3 * H * d, bias=False)
        self.out_proj = nn.Linear(H * d, h, bias=False)
        self.output_gate = nn.Linear(h, H * d, bias=False)
        self.norm = RMSNorm(H * d, cfg.rms_norm_eps)
        # per-head decay slope (deterministic function of layer/head index,
        # HF get_slope_rate) — recomputed here, the HF buffers are ignored
        base = 1.0 / (2.0 ** (8.0 / H))
        factor = 1.0 - layer_idx / (cfg.num_layers - 1 + 1e-5) + 1e-5
        slope = torch.tensor(
            [base ** (i + 1) * factor for i in range(H)], dtype=torch.float32
        )
        self.register_buffer("slope_rate", slope, persistent=False)

    def forward(self, x: torch.Tensor, meta: ForwardMeta, rope_cache) -> torch.Tensor:
        T = x.shape[0]
        H, d = self.num_heads, self.head_dim
        qkv = F.silu(self.qkv_proj(x)).view(T, H, 3 * d)
        q, k, v = torch.split(qkv, [d, d, d], dim=-1)
        rec = meta.linear_cache.recurrent_states[self.linear_layer_idx]
        slots = meta.linear_slots
        if meta.is_prefill:
            core = self._prefill(q, k, v, meta, rec, slots)
        else:
            core = self._decode(q, k, v, rec, slots)
        out = self.norm(core.reshape(T, H * d).to(x.dtype))
        out = torch.sigmoid(self.output_gate(x).float()).to(out.dtype) * out
        return self.out_proj(out)

    def _decode(self, q, k, v, rec, slots):
        """One recurrence step for all sequences: S <- ratio*S + k^T v;
        out = q S. Batched einsums, hipGraph-capturable."""
        S = rec[slots]                                   # [B, H, d, d] fp32
        qf, kf, vf = q.float(), k.float(), v.float()
        ratio = torch.exp(-self.slope_rate).view(1, -1, 1, 1)
        S = S * ratio + torch.einsum("bhk,bhv->bhkv", kf, vf)
        out = torch.einsum("bhk,bhkv->bhv", qf, S)
        rec[slots] = S
        return out.to(q.dtype)

    def _prefill(self, q, k, v, meta: ForwardMeta, rec, slots):
        """HF block recursion (block_size tokens at a time) continued from the
        cached state, so chunk boundaries do not change the math."""
        B = self.block_size
        sr = self.slope_rate.view(-1, 1, 1)              # [H,1,1]
        outs = []
        t0 = 0
        for i in range(meta.batch_size):
            L = int(meta.query_lens[i])
            slot = int(slots[i])
            S = rec[slot].clone()                        # [H, d, d] fp32
            qf = q[t0:t0 + L].float().transpose(0, 1)    # [H, L, d]
            kf = k[t0:t0 + L].float().transpose(0, 1)
            vf = v[t0:t0 + L].float().transpose(0, 1)
            out = torch.empty_like(qf)
            for s0 in range(0, L, B):
                e = min(s0 + B, L)
                n = e - s0
                r = torch.arange(n, dtype=torch.float32, device=q.device)
                qb, kb, vb = qf[:, s0:e], kf[:, s0:e], vf[:, s0:e]
                # intra-block: (Q K^T ∘ exp(-slope (i-j)), i>=j) V
                diag = (r.view(1, -1, 1) - r.view(1, 1, -1)) * sr
                diag = torch.where(diag >= 0, torch.exp(-diag),
                                   torch.zeros_like(diag))
                intra = torch.matmul(torch.matmul(qb, kb.transpose(-1, -2)) * diag, vb)
                # inter-block: decayed query against the carried state
                qdec = torch.exp(-sr * (r.view(1, -1, 1) + 1.0))
                inter = torch.matmul(qb * qdec, S)
                out[:, s0:e] = inter + intra
                # fold this block into the state
                kdec = torch.exp(-sr * (n - 1 - r).view(1, -1, 1))
                S = S * torch.exp(-sr * n) + torch.matmul(
                    (kb * kdec).transpose(-1, -2), vb
                )
            rec[slot] = S
            outs.append(out.transpose(0, 1).to(q.dtype))
            t0 += L
        return torch.cat(outs, dim=0)


class MiniMaxMoE(nn.Module):
    """Mixtral-style MoE: softmax over all experts, top-k, renormalize."""

    def __init__(self, cfg: ModelConfig):
        super().__init__()
        self.experts = FusedMoE(cfg)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.experts(x)


class MiniMaxDecoderLayer(nn.Module):
    """MiniMax residual flow: the residual is the NORMED hidden state, and
    residual/output are combined with alpha/beta factors (HF
    MiniMaxDecoderLayer.forward)."""

    def __init__(self, cfg: ModelConfig, layer_idx: int, local_layer_idx: int,
                 kv_layer_idx: int, linear_layer_idx: int):
        super().__init__()
        raw = cfg.raw
        self.is_linear = cfg.layer_type(layer_idx) == "linear_attention"
        if self.is_linear:
            self.linear_attn = LightningAttention(cfg, layer_idx, linear_layer_idx)
            self.attn_alpha = float(raw.get("linear_attn_alpha_factor", 1.0))
            self.attn_beta = float(raw.get("linear_attn_beta_factor", 1.0))
        else:
            self.self_attn = LlamaAttention(cfg, layer_idx, kv_layer_idx)
            self.attn_alpha = float(raw.get("full_attn_alpha_factor", 1.0))
            self.attn_beta = float(raw.get("full_attn_beta_factor", 1.0))
        self.mlp_alpha = float(raw.get("mlp_alpha_factor", 1.0))
        self.mlp_beta = float(raw.get("mlp_beta_factor", 1.0))
        self.mlp = MiniMaxMoE(cfg)
        self.input_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps)
        self.post_attention_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps)

    def forward(self, x, residual, meta, rope_cache):
        if residual is not None:  # entered from a dual-stream (llama) layer
            x = x + residual
        h = self.input_layernorm(x)
        if self.is_linear:
            a = self.linear_attn(h, meta, rope_cache)
        else:
            a = self.self_attn(h, meta, rope_cache)
        x = h * self.attn_alpha + a * self.attn_beta
        h = self.post_attention_layernorm(x)
        x = h * self.mlp_alpha + self.mlp(h) * self.mlp_beta
        return x, None


@register_model("MiniMaxForCausalLM", "MiniMaxM1ForCausalLM")
class MiniMaxShardModel(LlamaShardModel):
    decoder_layer_cls = None  # constructed in _build_layers

    def _build_layers(self, cfg: ModelConfig, start: int, end: int):
        layers = []
        kv_idx = 0
        lin_idx = 0
        for i, g in enumerate(range(start, end)):
            if cfg.layer_type(g) == "linear_attention":
                layers.append(MiniMaxDecoderLayer(cfg, g, i, 0, lin_idx))
                lin_idx += 1
            else:
                layers.append(MiniMaxDecoderLayer(cfg, g, i, kv_idx, 0))
                kv_idx += 1
        return layers

    # the decay buffers in the HF checkpoint are deterministic functions of
    # (layer, head) recomputed at init — accept and ignore them
    _DERIVED = ("slope_rate", "query_decay", "key_decay", "diagonal_decay")

    def load_hf_weight(self, name: str, tensor: torch.Tensor) -> bool:
        local = self.map_global_layer(name)
        if local is None:
            return False
        parts = local.split(".")
        if len(parts) >= 4 and parts[0] == "layers" and parts[2] == "self_attn":
            layer = self.layers[int(parts[1])]
            if layer.is_linear:
                la: LightningAttention = layer.linear_attn
                sub = parts[3]
                if sub in self._DERIVED:
                    return True
                if sub in ("qkv_proj", "out_proj", "output_gate", "norm"):
                    getattr(la, sub).weight.data.copy_(
                        tensor.to(getattr(la, sub).weight.dtype)
                    )
                    return True
                return False
        if len(parts) >= 4 and parts[0] == "layers" and parts[2] == "mlp":
            layer = self.layers[int(parts[1])]
            t = tensor.to(torch.bfloat16)
            if parts[3] == "gate":
                layer.mlp.experts.router.weight.data.copy_(t)
                return True
            if parts[3] == "experts":
                if parts[4] == "gate_up_proj":   # fused [E, 2I, H]
                    layer.mlp.experts.load_fused_gate_up(t)
                    return True
                if parts[4] == "down_proj":      # fused [E, H, I]
                    layer.mlp.experts.load_fused_down(t)
                    return True
                e, proj = int(parts[4]), parts[5]
                if proj in ("gate_proj", "w1"):
                    layer.mlp.experts.load_expert_gate(e, t)
                elif proj in ("up_proj", "w3"):
                    layer.mlp.experts.load_expert_up(e, t)
                elif proj in ("down_proj", "w2"):
                    layer.mlp.experts.load_expert_down(e, t)
                else:
                    return False
                return True
            return False
        return super().load_hf_weight(name, tensor)
