"""Qwen3-Next: hybrid stack — 3/4 gated-DeltaNet linear-attention layers +
1/4 gated full-attention layers (partial rotary, qk-norm), MoE with gated
shared expert.

Reference analogue: src/parallax/models/qwen3_next.py (hybrid linear+full
attention with conv/recurrent state slots, cache/linear_cache.py). Fresh
design: state lives in LinearStateCache slots (conv window + fp32 recurrent
state per request); decode is a fully batched single-step recurrence
(graph-capturable), prefill is a per-request scan. A fused HIP step kernel is a
later optimization — the per-token recurrence math is tiny next to the MoE.
"""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import ops
from ..parallel.layers import MergedColumnParallelLinear
from .config import ModelConfig
from .forward_meta import ForwardMeta
from .llama import LlamaAttention, LlamaMLP, LlamaShardModel, RMSNorm
from .moe import FusedMoE
from .registry import register_model


def _l2norm(x: torch.Tensor, eps: float = 1e-6) -> torch.Tensor:
    return x * torch.rsqrt((x.float() ** 2).sum(-1, keepdim=True) + eps).to(x.dtype)


class GatedRMSNorm(nn.Module):
    """RMSNorm then multiply by silu(gate) (HF Qwen3NextRMSNormGated)."""

    def __init__(self, dim: int, eps: float):
        super().__init__()
        self.weight = nn.Parameter(torch.empty(dim), requires_grad=False)
        self.eps = eps

    def forward(self, x: torch.Tensor, gate: torch.Tensor) -> torch.Tensor:
        xf = x.float()
        xf = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + self.eps)
        xf = xf * self.weight.float()
        return (xf * F.silu(gate.float())).to(x.dtype)


class GatedDeltaNet(nn.Module):
    def __init__(self, cfg: ModelConfig, layer_idx: int, linear_layer_idx: int):
        super().__init__()
        self.linear_layer_idx = linear_layer_idx
        self.hk = cfg.linear_num_key_heads
        self.hv = cfg.linear_num_value_heads
        self.dk = cfg.linear_key_head_dim
        self.dv = cfg.linear_value_head_dim
        self.kernel = cfg.linear_conv_kernel_dim
        self.key_dim = self.hk * self.dk
        self.value_dim = self.hv * self.dv
        self.conv_dim = 2 * self.key_dim + self.value_dim
        h = cfg.hidden_size
        self.in_proj_qkvz = nn.Linear(h, 2 * self.key_dim + 2 * self.value_dim,
                                      bias=False)
        self.in_proj_ba = nn.Linear(h, 2 * self.hv, bias=False)
        self.conv_weight = nn.Parameter(
            torch.empty(self.conv_dim, self.kernel), requires_grad=False
        )
        self.dt_bias = nn.Parameter(torch.empty(self.hv), requires_grad=False)
        self.A_log = nn.Parameter(torch.empty(self.hv), requires_grad=False)
        self.norm = GatedRMSNorm(self.dv, cfg.rms_norm_eps)
        self.out_proj = nn.Linear(self.value_dim, h, bias=False)

    def _split_qkvz_ba(self, qkvz: torch.Tensor, ba: torch.Tensor):
        """HF fix_query_key_value_ordering on packed [T, ...] tensors."""
        T = qkvz.shape[0]
        g = self.hv // self.hk
        qkvz = qkvz.view(T, self.hk, 2 * self.dk + 2 * g * self.dv)
        ba = ba.view(T, self.hk, 2 * g)
        q, k, v, z = torch.split(
            qkvz, [self.dk, self.dk, g * self.dv, g * self.dv], dim=2
        )
        b, a = torch.split(ba, [g, g], dim=2)
        v = v.reshape(T, self.hv, self.dv)
        z = z.reshape(T, self.hv, self.dv)
        b = b.reshape(T, self.hv)
        a = a.reshape(T, self.hv)
        return q, k, v, z, b, a

    def _project(self, x: torch.Tensor):
        """-> (q [T,hk,dk], k [T,hk,dk], v [T,hv,dv], z [T,hv,dv], b, a [T,hv]).
        Subclasses (qwen3_5: split in_proj_qkv/z/b/a) override this."""
        return self._split_qkvz_ba(self.in_proj_qkvz(x), self.in_proj_ba(x))

    def forward(self, x: torch.Tensor, meta: ForwardMeta, rope_cache) -> torch.Tensor:
        T = x.shape[0]
        q, k, v, z, b, a = self._project(x)
        mixed = torch.cat(
            [q.reshape(T, -1), k.reshape(T, -1), v.reshape(T, -1)], dim=-1
        )  # [T, conv_dim]

        conv_states = meta.linear_cache.conv_states[self.linear_layer_idx]
        rec_states = meta.linear_cache.recurrent_states[self.linear_layer_idx]
        slots = meta.linear_slots

        if meta.is_prefill:
            mixed = self._conv_prefill(mixed, meta, conv_states, slots)
        else:
            mixed = self._conv_decode(mixed, conv_states, slots)

        q, k, v = torch.split(
            mixed, [self.key_dim, self.key_dim, self.value_dim], dim=-1
        )
        q = _l2norm(q.view(T, self.hk, self.dk))
        k = _l2norm(k.view(T, self.hk, self.dk))
        v = v.view(T, self.hv, self.dv)
        g_ratio = self.hv // self.hk
        if g_ratio > 1:
            q = q.repeat_interleave(g_ratio, dim=1)
            k = k.repeat_interleave(g_ratio, dim=1)

        beta = b.float().sigmoid()                                   # [T, Hv]
        gdecay = -self.A_log.float().exp() * F.softplus(a.float() + self.dt_bias.float())
        scale = self.dk ** -0.5

        if meta.is_prefill:
            core = self._delta_prefill(q, k, v, gdecay, beta, scale, meta,
                                       rec_states, slots)
        else:
            core = self._delta_decode(q, k, v, gdecay, beta, scale,
                                      rec_states, slots)

        out = self.norm(core.reshape(T * self.hv, self.dv),
                        z.reshape(T * self.hv, self.dv))
        return self.out_proj(out.view(T, self.value_dim))

    # -- causal conv front-end ---------------------------------------------------

    def _conv_decode(self, mixed, conv_states, slots):
        """Batched single-token conv window update (graph-capturable)."""
        state = conv_states[slots]                       # [B, conv_dim, K-1]
        window = torch.cat([state.to(mixed.dtype), mixed.unsqueeze(-1)], dim=-1)
        out = (window.float() * self.conv_weight.float()).sum(-1)
        conv_states[slots] = window[..., 1:].to(conv_states.dtype)
        return F.silu(out).to(mixed.dtype)

    def _conv_prefill(self, mixed, meta: ForwardMeta, conv_states, slots):
        outs = []
        t0 = 0
        for i in range(meta.batch_size):
            L = int(meta.query_lens[i])
            slot = int(slots[i])
            seq = mixed[t0 : t0 + L].T                  # [conv_dim, L]
            state = conv_states[slot].to(seq.dtype)     # [conv_dim, K-1]
            padded = torch.cat([state, seq], dim=-1)    # [conv_dim, K-1+L]
            out = F.conv1d(
                padded.float().unsqueeze(0),
                self.conv_weight.float().unsqueeze(1),
                groups=self.conv_dim,
            )[0]                                         # [conv_dim, L]
            conv_states[slot] = padded[:, -(self.kernel - 1):].to(conv_states.dtype)
            outs.append(F.silu(out).T.to(mixed.dtype))
            t0 += L
        return torch.cat(outs, dim=0)

    # -- gated delta rule ------------------------------------------------------------

    def _delta_decode(self, q, k, v, g, beta, scale, rec_states, slots):
        """One recurrence step for every sequence at once.
        S <- S * exp(g);  out_mem = k^T S;  delta = beta (v - out_mem);
        S <- S + k^T delta;  out = (q*scale)^T S   (per head)."""
        S = rec_states[slots].float()                    # [B, Hv, dk, dv]
        qf, kf, vf = q.float(), k.float(), v.float()
        S = S * g.exp().unsqueeze(-1).unsqueeze(-1)
        kv_mem = torch.einsum("bhk,bhkv->bhv", kf, S)
        delta = (vf - kv_mem) * beta.unsqueeze(-1)
        S = S + torch.einsum("bhk,bhv->bhkv", kf, delta)
        out = torch.einsum("bhk,bhkv->bhv", qf * scale, S)
        rec_states[slots] = S
        return out.to(q.dtype)

    def _delta_prefill(self, q, k, v, g, beta, scale, meta: ForwardMeta,
                       rec_states, slots):
        outs = []
        t0 = 0
        for i in range(meta.batch_size):
            L = int(meta.query_lens[i])
            slot = int(slots[i])
            S = rec_states[slot].float()                 # [Hv, dk, dv]
            qf = q[t0 : t0 + L].float() * scale
            kf = k[t0 : t0 + L].float()
            vf = v[t0 : t0 + L].float()
            gf = g[t0 : t0 + L].exp()
            bf = beta[t0 : t0 + L]
            out = torch.empty(L, self.hv, self.dv, dtype=torch.float32,
                              device=q.device)
            for t in range(L):
                S = S * gf[t].unsqueeze(-1).unsqueeze(-1)
                kv_mem = torch.einsum("hk,hkv->hv", kf[t], S)
                delta = (vf[t] - kv_mem) * bf[t].unsqueeze(-1)
                S = S + torch.einsum("hk,hv->hkv", kf[t], delta)
                out[t] = torch.einsum("hk,hkv->hv", qf[t], S)
            rec_states[slot] = S
            outs.append(out.to(q.dtype))
            t0 += L
        return torch.cat(outs, dim=0)


class Qwen3NextAttention(LlamaAttention):
    """Full-attention layers: fused [q | output-gate] projection, qk-norm,
    partial rotary; output multiplied by sigmoid(gate)."""

    def __init__(self, cfg: ModelConfig, layer_idx: int, local_layer_idx: int):
        super().__init__(cfg, layer_idx, local_layer_idx)
        h = cfg.hidden_size
        # q_proj emits query + gate
        self.qkv_proj = MergedColumnParallelLinear(
            h,
            [2 * cfg.num_heads * cfg.head_dim, cfg.num_kv_heads * cfg.head_dim,
             cfg.num_kv_heads * cfg.head_dim],
            bias=cfg.attention_bias,
        )

    def forward(self, x: torch.Tensor, meta: ForwardMeta, rope_cache: torch.Tensor):
        T = x.shape[0]
        qg, k, v = self.qkv_proj.split_output(self.qkv_proj(x))
        qg = qg.view(T, self.num_heads, 2 * self.head_dim)
        q, gate = qg[..., : self.head_dim], qg[..., self.head_dim :]
        k = k.view(T, self.num_kv_heads, self.head_dim)
        v = v.view(T, self.num_kv_heads, self.head_dim)
        q = self.q_norm(q.contiguous())
        k = self.k_norm(k.contiguous())
        v = v.contiguous()
        k_cache, v_cache = meta.kv_cache.layer(self.local_layer_idx)
        ops.rope_and_cache(q, k, v, k_cache, v_cache, meta.positions, rope_cache,
                           meta.slot_mapping)
        if meta.is_prefill:
            attn = ops.prefill_attention(
                q, k_cache, v_cache, meta.block_tables, meta.seq_lens,
                meta.query_lens, self.scale,
            )
        else:
            attn = ops.paged_attention_decode(
                q, k_cache, v_cache, meta.block_tables, meta.seq_lens, self.scale,
                max_seq_len=meta.max_seq_len or None,
            )
        attn = attn * torch.sigmoid(gate.float()).to(attn.dtype)
        return self.o_proj(attn.reshape(T, self.num_heads * self.head_dim))


class Qwen3NextMoE(nn.Module):
    """Softmax top-k MoE + shared expert scaled by a sigmoid gate."""

    def __init__(self, cfg: ModelConfig):
        super().__init__()
        self.experts = FusedMoE(cfg)
        self.shared = (
            LlamaMLP(cfg, intermediate_size=cfg.shared_expert_intermediate_size)
            if cfg.shared_expert_intermediate_size else None
        )
        self.shared_gate = (
            nn.Linear(cfg.hidden_size, 1, bias=False) if self.shared else None
        )

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        y = self.experts(x)
        if self.shared is not None:
            y = y + torch.sigmoid(self.shared_gate(x).float()).to(x.dtype) * self.shared(x)
        return y


class Qwen3NextDecoderLayer(nn.Module):
    deltanet_cls = GatedDeltaNet

    def __init__(self, cfg: ModelConfig, layer_idx: int, local_layer_idx: int,
                 kv_layer_idx: int, linear_layer_idx: int):
        super().__init__()
        self.is_linear = cfg.layer_type(layer_idx) == "linear_attention"
        if self.is_linear:
            self.linear_attn = self.deltanet_cls(cfg, layer_idx, linear_layer_idx)
        else:
            self.self_attn = Qwen3NextAttention(cfg, layer_idx, kv_layer_idx)
        self.mlp = Qwen3NextMoE(cfg) if cfg.is_moe_layer(layer_idx) else LlamaMLP(cfg)
        self.input_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps)
        self.post_attention_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps)

    def forward(self, x, residual, meta, rope_cache):
        if residual is None:
            residual = x
            x = self.input_layernorm(x)
        else:
            x, residual = self.input_layernorm(x, residual)
        if self.is_linear:
            x = self.linear_attn(x, meta, rope_cache)
        else:
            x = self.self_attn(x, meta, rope_cache)
        x, residual = self.post_attention_layernorm(x, residual)
        x = self.mlp(x)
        return x, residual


@register_model("Qwen3NextForCausalLM")
class Qwen3NextShardModel(LlamaShardModel):
    def __init__(self, cfg: ModelConfig, start_layer: int = 0, end_layer=None):
        # build layers manually: hybrid stacks index the paged-KV cache by
        # full-attention layer count and the linear cache by deltanet count
        self._pending_cfg = cfg
        super().__init__(cfg, start_layer, end_layer)

    decoder_layer_cls = None  # constructed in _build_layers
    hybrid_layer_cls = Qwen3NextDecoderLayer

    def _build_layers(self, cfg: ModelConfig, start: int, end: int):
        layers = []
        kv_idx = 0
        lin_idx = 0
        for i, g in enumerate(range(start, end)):
            if cfg.layer_type(g) == "linear_attention":
                layers.append(self.hybrid_layer_cls(cfg, g, i, 0, lin_idx))
                lin_idx += 1
            else:
                layers.append(self.hybrid_layer_cls(cfg, g, i, kv_idx, 0))
                kv_idx += 1
        return layers

    # -- weight loading ---------------------------------------------------------

    # HF Qwen3NextRMSNorm is Gemma-style `x * (1 + w)` with zero-init weights;
    # our RMSNorm multiplies by w directly -> add 1 on load. The gated deltanet
    # norm (linear_attn.norm) is plain ones-init and excluded.
    _GEMMA_NORMS = (
        "input_layernorm.weight", "post_attention_layernorm.weight",
        "q_norm.weight", "k_norm.weight",
    )

    def load_hf_weight(self, name: str, tensor: torch.Tensor) -> bool:
        if (name.endswith(self._GEMMA_NORMS) and ".linear_attn." not in name) or \
                name == "model.norm.weight":
            tensor = tensor.float() + 1.0
        local = self.map_global_layer(name)
        if local is None:
            return False
        parts = local.split(".")
        t = tensor.to(torch.bfloat16)
        if len(parts) >= 3 and parts[0] == "layers":
            layer = self.layers[int(parts[1])]
            if parts[2] == "linear_attn":
                la: GatedDeltaNet = layer.linear_attn
                sub = parts[3]
                if sub == "conv1d":
                    la.conv_weight.data.copy_(t.squeeze(1))
                elif sub in ("dt_bias", "A_log"):
                    getattr(la, sub).data.copy_(tensor.float().to(torch.bfloat16))
                elif sub == "norm":
                    la.norm.weight.data.copy_(t)
                elif sub in ("in_proj_qkvz", "in_proj_ba", "out_proj",
                             "in_proj_qkv", "in_proj_z", "in_proj_b",
                             "in_proj_a"):
                    getattr(la, sub).weight.data.copy_(t)
                else:
                    return False
                return True
            if parts[2] == "self_attn" and parts[3] == "q_proj":
                layer.self_attn.qkv_proj.load_full_weight_part(
                    0, t if parts[4] == "weight" else None,
                    t if parts[4] == "bias" else None)
                return True
            if parts[2] == "mlp" and isinstance(layer.mlp, Qwen3NextMoE):
                return self._load_moe_weight(layer, parts[3:], t)
        return super().load_hf_weight(name, tensor)

    def _load_moe_weight(self, layer, parts, t) -> bool:
        mlp = layer.mlp
        if not isinstance(mlp, Qwen3NextMoE):
            return False
        if parts[0] == "gate":
            mlp.experts.router.weight.data.copy_(t)
            return True
        if parts[0] == "experts":
            if parts[1] == "gate_up_proj":
                mlp.experts.load_fused_gate_up(t)
            elif parts[1] == "down_proj":
                mlp.experts.load_fused_down(t)
            else:
                return False
            return True
        if parts[0] == "shared_expert":
            proj = parts[1]
            if proj == "gate_proj":
                mlp.shared.gate_up_proj.load_full_weight_part(0, t)
            elif proj == "up_proj":
                mlp.shared.gate_up_proj.load_full_weight_part(1, t)
            elif proj == "down_proj":
                mlp.shared.down_proj.load_full_weight(t)
            return True
        if parts[0] == "shared_expert_gate":
            mlp.shared_gate.weight.data.copy_(t)
            return True
        return False
