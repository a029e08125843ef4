"""DeepSeek-V3 family: MLA attention (compressed latent KV cache) + sigmoid
group-limited MoE routing with shared experts.

Reference analogue: src/parallax/models/deepseek_v3.py (MLA over a compressed
cache) — fresh MI355X design: decode runs ABSORBED (q projected into latent
space; attention over the [kv_lora_rank + rope] cache via the MLA HIP kernel);
prefill runs non-absorbed with latents gathered from the paged cache and
expanded through kv_b (standard MHA math). Covers DeepSeek-V2/V3/R1 and
Kimi-K2 (same architecture class).
"""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import ops
from ..parallel.comm import get_comm
from ..parallel.layers import ColumnParallelLinear, RowParallelLinear
from .config import ModelConfig
from .forward_meta import ForwardMeta
from .llama import LlamaMLP, LlamaShardModel, RMSNorm
from .moe import MoEBlock
from .registry import register_model


class MLAAttention(nn.Module):
    def __init__(self, cfg: ModelConfig, layer_idx: int, local_layer_idx: int):
        super().__init__()
        comm = get_comm()
        self.local_layer_idx = local_layer_idx
        self.num_heads = cfg.num_heads // comm.tp_size
        self.dn = cfg.qk_nope_head_dim
        self.dr = cfg.qk_rope_head_dim
        self.dv = cfg.v_head_dim
        self.r = cfg.kv_lora_rank
        self.q_lora_rank = cfg.q_lora_rank
        self.scale = (self.dn + self.dr) ** -0.5
        # deepseek yarn: softmax scale gets mscale^2 (HF yarn_apply_mscale)
        rs = cfg.rope_scaling or {}
        if rs.get("rope_type", rs.get("type", "default")) != "default":
            factor = rs.get("factor", 1.0)
            mscale_all_dim = rs.get("mscale_all_dim", 0)
            if mscale_all_dim and factor > 1:
                m = 0.1 * mscale_all_dim * math.log(factor) + 1.0
                self.scale *= m * m
        self.rope_neox = not cfg.rope_interleave
        h = cfg.hidden_size
        H = cfg.num_heads  # full head count for weight shapes; TP shards q_b/kv_b/o

        if self.q_lora_rank:
            self.q_a_proj = nn.Linear(h, self.q_lora_rank, bias=False)
            self.q_a_layernorm = RMSNorm(self.q_lora_rank, cfg.rms_norm_eps)
            self.q_b_proj = ColumnParallelLinear(
                self.q_lora_rank, H * (self.dn + self.dr)
            )
        else:
            self.q_proj = ColumnParallelLinear(h, H * (self.dn + self.dr))
        self.kv_a_proj_with_mqa = nn.Linear(h, self.r + self.dr, bias=False)
        self.kv_a_layernorm = RMSNorm(self.r, cfg.rms_norm_eps)
        self.kv_b_proj = ColumnParallelLinear(self.r, H * (self.dn + self.dv))
        self.o_proj = RowParallelLinear(H * self.dv, h)

        # absorbed-decode weights, derived from kv_b_proj after load:
        #   W_UK [H, dn, r] (k_nope = latent @ W_UK^T), W_UV [H, dv, r]
        self.register_buffer(
            "w_uk", torch.empty(self.num_heads, self.dn, self.r), persistent=False
        )
        self.register_buffer(
            "w_uv", torch.empty(self.num_heads, self.dv, self.r), persistent=False
        )

    def finalize_weights(self) -> None:
        w = self.kv_b_proj.weight.data.view(self.num_heads, self.dn + self.dv, self.r)
        self.w_uk.copy_(w[:, : self.dn])
        self.w_uv.copy_(w[:, self.dn :])

    def forward(self, x: torch.Tensor, meta: ForwardMeta, rope_cache: torch.Tensor):
        T = x.shape[0]
        H = self.num_heads
        if self.q_lora_rank:
            q = self.q_b_proj(self.q_a_layernorm(
                ops.linear(x, self.q_a_proj.weight)))
        else:
            q = self.q_proj(x)
        q = q.view(T, H, self.dn + self.dr)
        q_nope, q_pe = q[..., : self.dn], q[..., self.dn :]

        kv_a = ops.linear(x, self.kv_a_proj_with_mqa.weight)
        latent = self.kv_a_layernorm(kv_a[:, : self.r])
        k_pe = kv_a[:, self.r :].unsqueeze(1)  # [T, 1, dr]

        q_pe = q_pe.contiguous()
        k_pe = k_pe.contiguous()
        ops.rope_inplace(q_pe, k_pe, meta.positions, rope_cache,
                         is_neox=self.rope_neox)

        cache = meta.mla_cache.layer(self.local_layer_idx)
        ops.mla_reshape_and_cache(latent, k_pe.squeeze(1), cache, meta.slot_mapping)

        if meta.is_prefill:
            attn = self._prefill_attention(q_nope, q_pe, meta, cache)
        else:
            # absorbed: q_latent[b,h] = q_nope[b,h] @ W_UK[h]^T... (dn x r)
            q_latent = torch.einsum(
                "bhd,hdr->bhr", q_nope.float(), self.w_uk.float()
            ).to(q_nope.dtype)
            out_latent = ops.mla_paged_attention_decode(
                q_latent.contiguous(), q_pe.contiguous(), cache,
                meta.block_tables, meta.seq_lens, self.scale,
                max_seq_len=meta.max_seq_len or None,
            )
            attn = torch.einsum(
                "bhr,hvr->bhv", out_latent.float(), self.w_uv.float()
            ).to(x.dtype)
        return self.o_proj(attn.reshape(T, H * self.dv))

    def _prefill_sparse_mask(self, i, t0, QL, L, meta):
        """Hook for sparse-attention subclasses (DSA): additive [QL, L] mask
        applied on top of the causal mask during prefill. None = dense."""
        return None

    def _prefill_attention(self, q_nope, q_pe, meta: ForwardMeta, cache):
        """Absorbed, L-chunked online-softmax prefill over the COMPRESSED
        cache: q_nope folds through w_uk so QK^T and PV both contract against
        the latent directly (same math as the absorbed decode kernel) —
        no [L, H, dn+dv] expansion and no [QL, L] full-logits buffer, which
        is what makes 256k-context prefill (BASELINE Kimi-K2 config) fit.
        Sparse subclasses (DSA) still take the dense masked path below."""
        H, dn, dv, r = self.num_heads, self.dn, self.dv, self.r
        bs = cache.shape[1]
        out = torch.empty(
            q_nope.shape[0], H, dv, dtype=q_nope.dtype, device=q_nope.device
        )
        t0 = 0
        for i in range(meta.batch_size):
            QL = int(meta.query_lens[i])
            L = int(meta.seq_lens[i])
            nb = (L + bs - 1) // bs
            entries = cache[meta.block_tables[i, :nb].long()].reshape(nb * bs, -1)[:L]
            lat, kpe_ctx = entries[:, :r], entries[:, r:]        # [L, r], [L, dr]
            qi_n = q_nope[t0 : t0 + QL].float()                  # [QL, H, dn]
            qi_p = q_pe[t0 : t0 + QL].float()                    # [QL, H, dr]
            extra = self._prefill_sparse_mask(i, t0, QL, L, meta)
            if extra is not None:  # DSA (deepseek_v32) top-k restriction
                kv = torch.einsum("lr,hdr->lhd", lat.float(),
                                  torch.cat([self.w_uk, self.w_uv], dim=1).float())
                k_nope_ctx, v_ctx = kv[:, :, :dn], kv[:, :, dn:]
                logits = (
                    torch.einsum("qhd,lhd->hql", qi_n, k_nope_ctx)
                    + torch.einsum("qhd,ld->hql", qi_p, kpe_ctx.float())
                ) * self.scale
                qpos = torch.arange(L - QL, L, device=logits.device).unsqueeze(-1)
                kpos = torch.arange(L, device=logits.device).unsqueeze(0)
                logits.masked_fill_((kpos > qpos).unsqueeze(0), float("-inf"))
                logits = logits + extra.unsqueeze(0)
                p = logits.softmax(dim=-1)
                out[t0 : t0 + QL] = (
                    torch.einsum("hql,lhv->qhv", p, v_ctx).to(out.dtype)
                )
                t0 += QL
                continue

            q_lat = torch.einsum("qhd,hdr->qhr", qi_n, self.w_uk.float())
            dev = q_lat.device
            qpos = torch.arange(L - QL, L, device=dev).view(1, -1, 1)
            # chunk so the [H, QL, Lc] score buffer stays ~<=2 GB
            lc_size = max(1024, (1 << 29) // max(1, H * QL))
            m = torch.full((H, QL, 1), float("-inf"), device=dev)
            lsum = torch.zeros(H, QL, 1, device=dev)
            acc = torch.zeros(QL, H, r, device=dev)
            for c0 in range(0, L, lc_size):
                c1 = min(c0 + lc_size, L)
                latc = lat[c0:c1].float()
                s = (
                    torch.einsum("qhr,lr->hql", q_lat, latc)
                    + torch.einsum("qhd,ld->hql", qi_p, kpe_ctx[c0:c1].float())
                ) * self.scale
                kpos = torch.arange(c0, c1, device=dev).view(1, 1, -1)
                s.masked_fill_(kpos > qpos, float("-inf"))
                m_new = torch.maximum(m, s.amax(dim=-1, keepdim=True))
                p = torch.exp(s - m_new)
                resc = torch.exp(m - m_new)
                lsum = lsum * resc + p.sum(dim=-1, keepdim=True)
                acc = acc * resc.squeeze(-1).permute(1, 0).unsqueeze(-1) \
                    + torch.einsum("hql,lr->qhr", p, latc)
                m = m_new
            out_lat = acc / lsum.squeeze(-1).permute(1, 0).unsqueeze(-1)
            out[t0 : t0 + QL] = torch.einsum(
                "qhr,hvr->qhv", out_lat, self.w_uv.float()
            ).to(out.dtype)
            t0 += QL
        return out


class DeepseekV3DecoderLayer(nn.Module):
    def __init__(self, cfg: ModelConfig, layer_idx: int, local_layer_idx: int):
        super().__init__()
        self.self_attn = MLAAttention(cfg, layer_idx, local_layer_idx)
        self.mlp = MoEBlock(cfg) if cfg.is_moe_layer(layer_idx) else LlamaMLP(cfg)
        self.input_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps)
        self.post_attention_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps)

    def forward(self, x, residual, meta, rope_cache):
        if residual is None:
            residual = x
            x = self.input_layernorm(x)
        else:
            x, residual = self.input_layernorm(x, residual)
        x = self.self_attn(x, meta, rope_cache)
        x, residual = self.post_attention_layernorm(x, residual)
        x = self.mlp(x)
        return x, residual


@register_model("DeepseekV3ForCausalLM", "DeepseekV2ForCausalLM", "KimiK2ForCausalLM")
class DeepseekV3ShardModel(LlamaShardModel):
    decoder_layer_cls = DeepseekV3DecoderLayer

    def __init__(self, cfg: ModelConfig, start_layer: int = 0, end_layer=None):
        super().__init__(cfg, start_layer, end_layer)
        # MLA ropes only the decoupled qk_rope dims
        from .rope import build_rope_cache_for

        self.rope_cache = build_rope_cache_for(cfg, rot_dim=cfg.qk_rope_head_dim)

    def finalize_weights(self) -> None:
        for layer in self.layers:
            layer.self_attn.finalize_weights()

    @torch.no_grad()
    def init_random(self, seed: int = 1234) -> None:
        super().init_random(seed)
        self.finalize_weights()

    # -- HF weight routing -------------------------------------------------------

    def load_hf_weight(self, name: str, tensor: torch.Tensor) -> bool:
        local = self.map_global_layer(name)
        if local is None:
            return False
        parts = local.split(".")
        t = tensor.to(torch.bfloat16)
        if len(parts) >= 4 and parts[0] == "layers" and parts[2] == "mlp":
            layer = self.layers[int(parts[1])]
            if isinstance(layer.mlp, MoEBlock):
                return self._load_moe(layer.mlp, parts[3:], t)
            # dense layer mlp -> llama path
            return super().load_hf_weight(name, tensor)
        if len(parts) >= 4 and parts[2] == "self_attn":
            layer = self.layers[int(parts[1])]
            attn: MLAAttention = layer.self_attn
            sub, leaf = parts[3], parts[4]
            if sub in ("q_a_proj", "kv_a_proj_with_mqa"):
                getattr(attn, sub).weight.data.copy_(t)
            elif sub in ("q_a_layernorm", "kv_a_layernorm"):
                getattr(attn, sub).weight.data.copy_(t)
            elif sub in ("q_b_proj", "q_proj", "kv_b_proj"):
                getattr(attn, sub).load_full_weight(
                    t if leaf == "weight" else None,
                    t if leaf == "bias" else None,
                )
            elif sub == "o_proj":
                attn.o_proj.load_full_weight(t)
            else:
                return False
            return True
        return super().load_hf_weight(name, tensor)

    def _load_moe(self, moe: MoEBlock, parts, t: torch.Tensor) -> bool:
        inter = moe.experts.intermediate_size
        if parts[0] == "gate":
            if parts[1] == "weight":
                moe.experts.router.weight.data.copy_(t)
            elif parts[1] == "e_score_correction_bias":
                moe.experts.router.e_score_correction_bias.data.copy_(t.float())
            return True
        if parts[0] == "experts":
            if parts[1] == "gate_up_proj":  # fused [E, 2I, H] (transformers >= 5)
                moe.experts.load_fused_gate_up(t)
                return True
            if parts[1] == "down_proj":     # fused [E, H, I]
                moe.experts.load_fused_down(t)
                return True
            e, proj = int(parts[1]), parts[2]
            if proj == "gate_proj":
                moe.experts.load_expert_gate(e, t)
            elif proj == "up_proj":
                moe.experts.load_expert_up(e, t)
            elif proj == "down_proj":
                moe.experts.load_expert_down(e, t)
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
