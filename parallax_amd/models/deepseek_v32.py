"""DeepSeek-V3.2: DeepSeek-V3 MLA + MoE with DSA (DeepSeek Sparse Attention).

A lightweight per-layer indexer (its own wq_b/wk/weights_proj projections,
reference analogue: src/parallax/models/deepseek_v32.py + kernels/dsa/) scores
every cached token and restricts the main MLA attention to each query's top-k
token positions. The indexer key cache is one shared vector per token per
layer (paged alongside the compressed MLA cache, see MLAKVCache.index_caches);
decode runs the absorbed MLA kernel in its sparse (topk_indices) variant.

Parity notes vs HF transformers modeling_deepseek_v32:
- the indexer applies NON-interleaved (half-split) rope to the first
  qk_rope_head_dim dims of its q/k while the main MLA attention stays
  interleaved;
- scores = sum_h w_h * relu(q_h . k) * head_dim^-0.5 with
  w = weights_proj(x) * n_heads^-0.5, computed in fp32;
- rows whose whole context fits in index_topk fall back to dense attention
  (identical math, and the sparse kernel's -1-leading-row convention).
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import ops
from .config import ModelConfig
from .deepseek_v3 import DeepseekV3DecoderLayer, DeepseekV3ShardModel, MLAAttention
from .forward_meta import ForwardMeta
from .registry import register_model

_FMIN = torch.finfo(torch.float32).min


class DSAIndexer(nn.Module):
    def __init__(self, cfg: ModelConfig):
        super().__init__()
        self.n_heads = cfg.index_n_heads
        self.head_dim = cfg.index_head_dim
        self.topk = cfg.index_topk
        self.rope_dim = cfg.qk_rope_head_dim
        self.wq_b = nn.Linear(cfg.q_lora_rank, self.n_heads * self.head_dim, bias=False)
        self.wk = nn.Linear(cfg.hidden_size, self.head_dim, bias=False)
        self.k_norm = nn.LayerNorm(self.head_dim, eps=1e-6)
        self.weights_proj = nn.Linear(cfg.hidden_size, self.n_heads, bias=False)
        # fold head_dim^-0.5 (score scale) and n_heads^-0.5 (weight scale)
        # into the per-head weights: relu(c*s) = c*relu(s) for c > 0
        self.weight_scale = self.n_heads**-0.5 * self.head_dim**-0.5

    def _rope_neox(self, t: torch.Tensor, cs: torch.Tensor) -> torch.Tensor:
        """Half-split rope on the first rope_dim dims of t [T, H, D];
        cs = rope_cache[positions] laid out [cos(half) | sin(half)]."""
        half = self.rope_dim // 2
        cos = cs[:, :half].unsqueeze(1).float()
        sin = cs[:, half:].unsqueeze(1).float()
        r, rest = t[..., : self.rope_dim].float(), t[..., self.rope_dim :]
        x1, x2 = r[..., :half], r[..., half:]
        rot = torch.cat([x1 * cos - x2 * sin, x2 * cos + x1 * sin], dim=-1)
        return torch.cat([rot.to(t.dtype), rest], dim=-1)

    def project(self, x: torch.Tensor, q_resid: torch.Tensor,
                positions: torch.Tensor, rope_cache: torch.Tensor):
        """Returns (q_index [T,H,D], k_index [T,D], head_weights [T,H])."""
        T = x.shape[0]
        cs = rope_cache[positions.long()]
        q = self.wq_b(q_resid).view(T, self.n_heads, self.head_dim)
        q = self._rope_neox(q, cs)
        # fp32 LayerNorm regardless of model dtype (HF computes it on the
        # bf16 tensor; fp32 here is a superset in precision and keeps
        # torch.layer_norm happy with mixed dtypes)
        k = F.layer_norm(
            self.wk(x).float(), (self.head_dim,),
            self.k_norm.weight.float(), self.k_norm.bias.float(),
            self.k_norm.eps,
        ).to(x.dtype).unsqueeze(1)
        k = self._rope_neox(k, cs).squeeze(1)
        w = self.weights_proj(x).float() * self.weight_scale
        return q, k, w


class DSAMLAAttention(MLAAttention):
    def __init__(self, cfg: ModelConfig, layer_idx: int, local_layer_idx: int):
        super().__init__(cfg, layer_idx, local_layer_idx)
        assert self.q_lora_rank, "DeepSeek-V3.2 requires q_lora_rank"
        self.indexer = DSAIndexer(cfg)
        self._idx_state = None  # (q_idx, w_idx, idx_cache) during prefill

    def forward(self, x: torch.Tensor, meta: ForwardMeta, rope_cache: torch.Tensor):
        T = x.shape[0]
        H = self.num_heads
        q_resid = self.q_a_layernorm(self.q_a_proj(x))
        q = self.q_b_proj(q_resid).view(T, H, self.dn + self.dr)
        q_nope, q_pe = q[..., : self.dn], q[..., self.dn :]

        kv_a = self.kv_a_proj_with_mqa(x)
        latent = self.kv_a_layernorm(kv_a[:, : self.r])
        k_pe = kv_a[:, self.r :].unsqueeze(1)

        q_pe = q_pe.contiguous()
        k_pe = k_pe.contiguous()
        ops.rope_inplace(q_pe, k_pe, meta.positions, rope_cache,
                         is_neox=self.rope_neox)

        cache = meta.mla_cache.layer(self.local_layer_idx)
        ops.mla_reshape_and_cache(latent, k_pe.squeeze(1), cache, meta.slot_mapping)

        # indexer: rope q/k (non-interleaved), write k to the paged index cache
        idx_cache = meta.mla_cache.index_layer(self.local_layer_idx)
        q_idx, k_idx, w_idx = self.indexer.project(
            x, q_resid, meta.positions, rope_cache
        )
        ops.store_indexer_cache(k_idx, idx_cache, meta.slot_mapping)

        if meta.is_prefill:
            self._idx_state = (q_idx, w_idx, idx_cache)
            attn = self._prefill_attention(q_nope, q_pe, meta, cache)
            self._idx_state = None
        else:
            scores = self._decode_scores(q_idx, w_idx, idx_cache, meta)
            topk_indices = self._decode_topk(scores, meta.seq_lens)
            q_latent = torch.einsum(
                "bhd,hdr->bhr", q_nope.float(), self.w_uk.float()
            ).to(q_nope.dtype)
            out_latent = ops.dsa_paged_attention_decode(
                q_latent.contiguous(), q_pe.contiguous(), cache,
                meta.block_tables, meta.seq_lens, topk_indices, self.scale,
                max_seq_len=meta.max_seq_len or None,
            )
            attn = torch.einsum(
                "bhr,hvr->bhv", out_latent.float(), self.w_uv.float()
            ).to(x.dtype)
        return self.o_proj(attn.reshape(T, H * self.dv))

    def _decode_scores(self, q_idx, w_idx, idx_cache, meta) -> torch.Tensor:
        """[B, msl] fp32 weighted relu scores over the paged indexer cache.
        Fully batched (no per-row host loops) so it is hipGraph-capturable;
        shapes are fixed by meta.max_seq_len."""
        B = q_idx.shape[0]
        bs = idx_cache.shape[1]
        msl = meta.max_seq_len or int(meta.seq_lens.max())
        npages = min((msl + bs - 1) // bs, meta.block_tables.shape[1])
        msl = min(msl, npages * bs)
        if q_idx.is_cuda and q_idx.dtype == torch.bfloat16 \
                and idx_cache.dim() == 3:
            # HIP MFMA score kernel: streams the paged index cache directly
            # (no [B, msl, Di] gather materialization — at 256k ctx that
            # gather alone is 64 MB/req/layer)
            return ops.dsa_indexer_scores(
                q_idx, idx_cache, w_idx.float(), meta.block_tables,
                meta.seq_lens, max_ctx=msl,
            )
        keys = idx_cache[meta.block_tables[:, :npages].long()].reshape(
            B, npages * bs, -1
        )[:, :msl]                                       # [B, msl, Di]
        s = torch.relu(torch.einsum("bhd,btd->bht", q_idx.float(), keys.float()))
        scores = torch.einsum("bh,bht->bt", w_idx, s)    # [B, msl]
        t = torch.arange(msl, device=scores.device)
        valid = t.view(1, -1) < meta.seq_lens.view(B, 1)
        return scores.masked_fill(~valid, float("-inf"))

    def _decode_topk(self, scores: torch.Tensor, seq_lens: torch.Tensor) -> torch.Tensor:
        """[B, index_topk] int32; rows with ctx <= topk use the dense fallback
        (-1 in column 0) — identical math, no gather cost."""
        B, max_ctx = scores.shape
        k = self.indexer.topk
        if max_ctx <= k:
            return torch.full((B, 1), -1, dtype=torch.int32, device=scores.device)
        idx = scores.topk(k, dim=-1).indices.to(torch.int32)
        dense = (seq_lens.to(scores.device) <= k).unsqueeze(1)
        return torch.where(dense, torch.full_like(idx, -1), idx)

    def _prefill_sparse_mask(self, i: int, t0: int, QL: int, L: int, meta):
        """Additive [QL, L] mask from the indexer's top-k per query row.
        Matches HF: non-selected causally-valid keys get fp32 min (not -inf),
        so rows degrade identically in edge cases."""
        q_idx, w_idx, idx_cache = self._idx_state
        bs = idx_cache.shape[1]
        nb = (L + bs - 1) // bs
        keys = idx_cache[meta.block_tables[i, :nb].long()].reshape(
            nb * bs, -1
        )[:L].float()                                   # [L, Di]
        qi = q_idx[t0 : t0 + QL].float()                # [QL, Hi, Di]
        s = torch.relu(torch.einsum("qhd,ld->qhl", qi, keys))
        s = torch.einsum("qh,qhl->ql", w_idx[t0 : t0 + QL], s)  # [QL, L] fp32
        qpos = torch.arange(L - QL, L, device=s.device).unsqueeze(-1)
        kpos = torch.arange(L, device=s.device).unsqueeze(0)
        causal = kpos > qpos
        s.masked_fill_(causal, float("-inf"))
        k = min(self.indexer.topk, L)
        picks = s.topk(k, dim=-1).indices
        allowed = torch.zeros(QL, L, dtype=torch.bool, device=s.device)
        allowed.scatter_(1, picks, True)
        allowed &= ~causal
        return torch.where(allowed, 0.0, _FMIN)


class DeepseekV32DecoderLayer(DeepseekV3DecoderLayer):
    def __init__(self, cfg: ModelConfig, layer_idx: int, local_layer_idx: int):
        super().__init__(cfg, layer_idx, local_layer_idx)
        self.self_attn = DSAMLAAttention(cfg, layer_idx, local_layer_idx)


@register_model("DeepseekV32ForCausalLM")
class DeepseekV32ShardModel(DeepseekV3ShardModel):
    decoder_layer_cls = DeepseekV32DecoderLayer

    def load_hf_weight(self, name: str, tensor: torch.Tensor) -> bool:
        local = self.map_global_layer(name)
        if local is not None and ".self_attn.indexer." in local:
            parts = local.split(".")
            attn: DSAMLAAttention = self.layers[int(parts[1])].self_attn
            sub, leaf = parts[4], parts[5]
            mod = getattr(attn.indexer, sub, None)
            if mod is None:
                return False
            getattr(mod, leaf).data.copy_(tensor.to(getattr(mod, leaf).dtype))
            return True
        return super().load_hf_weight(name, tensor)
