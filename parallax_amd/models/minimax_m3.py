"""MiniMax-M3: GQA with MSA block-sparse attention (per-KV-head lightning
indexer) + sigmoid-routed MoE with clamped-swiglu experts and shared experts.

Reference analogue: src/parallax/models/minimax_m3.py + kernels/msa/. Every
attention layer tagged 'minimax_m3_sparse' in layer_types carries a small
indexer (per-KV-head query branch, one shared key per token cached in the
paged index cache): scores are max-pooled into index_block_size key blocks,
the index_local_blocks blocks around the query are always kept, and the
top-index_topk_blocks blocks per KV head feed the block-sparse attention.
Decode expands the selected blocks to explicit token positions and runs the
MSA HIP kernel with per-KV-head position lists ([B, Hk, P]); prefill builds
the HF-identical additive block mask over a dense torch attention.

All RMSNorms in this family are Gemma-style x*(1+w) with zero-init weights
(+1 applied on load). MoE: sigmoid scores + e_score_correction_bias, top-k
renormalized, routed output scaled by routed_scaling_factor, plus a clamped
dense shared expert; 'dense' mlp_layer_types entries use a clamped dense MLP
of dense_intermediate_size."""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import ops
from .config import ModelConfig
from .forward_meta import ForwardMeta
from .llama import LlamaAttention, LlamaDecoderLayer, LlamaShardModel, RMSNorm
from .moe import FusedMoE
from .registry import register_model

_FMIN = torch.finfo(torch.float32).min


def _clamped_swiglu(gate_up: torch.Tensor, limit: float, alpha: float) -> torch.Tensor:
    gate, up = gate_up.chunk(2, dim=-1)
    gate = gate.clamp(max=limit)
    up = up.clamp(min=-limit, max=limit)
    return (up + 1.0) * (gate * torch.sigmoid(gate * alpha))


class M3DenseMLP(nn.Module):
    def __init__(self, cfg: ModelConfig, intermediate_size: int):
        super().__init__()
        self.limit = float(cfg.raw.get("swiglu_limit", 7.0))
        self.alpha = float(cfg.raw.get("swiglu_alpha", 1.702))
        self.gate_up_proj = nn.Linear(cfg.hidden_size, 2 * intermediate_size, bias=False)
        self.down_proj = nn.Linear(intermediate_size, cfg.hidden_size, bias=False)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.down_proj(_clamped_swiglu(self.gate_up_proj(x), self.limit, self.alpha))


class M3MoE(nn.Module):
    """Sigmoid router + bias, clamped routed experts (routed_scaling handled
    inside MoERouter's weight scaling) + clamped shared expert."""

    def __init__(self, cfg: ModelConfig):
        super().__init__()
        self.experts = FusedMoE(cfg, act_limit=float(cfg.raw.get("swiglu_limit", 7.0)))
        self.shared = M3DenseMLP(
            cfg, int(cfg.raw.get("shared_intermediate_size", cfg.intermediate_size))
        )

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.experts(x) + self.shared(x)


class M3Indexer(nn.Module):
    """Per-KV-head block selection branch (HF MiniMaxM3VLIndexer)."""

    def __init__(self, cfg: ModelConfig):
        super().__init__()
        self.n_heads = cfg.index_n_heads
        self.head_dim = cfg.index_head_dim
        self.block = cfg.index_block_size
        self.topk_blocks = cfg.index_topk_blocks
        self.local_blocks = cfg.index_local_blocks
        self.rot = min(cfg.rot_dim, self.head_dim)
        self.q_proj = nn.Linear(cfg.hidden_size, self.n_heads * self.head_dim, bias=False)
        self.k_proj = nn.Linear(cfg.hidden_size, self.head_dim, bias=False)
        self.q_norm = RMSNorm(self.head_dim, cfg.rms_norm_eps)
        self.k_norm = RMSNorm(self.head_dim, cfg.rms_norm_eps)

    def _rope(self, t: torch.Tensor, cs: torch.Tensor) -> torch.Tensor:
        """HF slices the MAIN partial-rotary cos/sin to the first
        min(rot_dim, index_head_dim) dims and rotate_halves that span —
        replicate exactly (cs = rope_cache[positions], [cos(half)|sin(half)])."""
        R = self.rot
        half_w = cs.shape[-1] // 2
        cos = torch.cat([cs[:, :half_w], cs[:, :half_w]], dim=-1)[:, :R]
        sin = torch.cat([cs[:, half_w:], cs[:, half_w:]], dim=-1)[:, :R]
        cos = cos.unsqueeze(1).float()
        sin = sin.unsqueeze(1).float()
        r, rest = t[..., :R].float(), t[..., R:]
        x1, x2 = r[..., : R // 2], r[..., R // 2 :]
        rot = torch.cat([-x2, x1], dim=-1)
        return torch.cat([(r * cos + rot * sin).to(t.dtype), rest], dim=-1)

    def project(self, x: torch.Tensor, positions: torch.Tensor, rope_cache):
        """-> (q_idx [T, Hi, Di], k_idx [T, Di]) post-norm, post-rope."""
        T = x.shape[0]
        cs = rope_cache[positions.long()]
        q = self.q_norm(self.q_proj(x).view(T, self.n_heads, self.head_dim))
        q = self._rope(q, cs)
        k = self.k_norm(self.k_proj(x)).unsqueeze(1)
        k = self._rope(k, cs).squeeze(1)
        return q, k


class M3SparseAttention(LlamaAttention):
    def __init__(self, cfg: ModelConfig, layer_idx: int, local_layer_idx: int):
        super().__init__(cfg, layer_idx, local_layer_idx)
        assert cfg.index_n_heads == cfg.num_kv_heads, \
            "M3 indexer emits one block selection per KV head"
        self.indexer = M3Indexer(cfg)

    def forward(self, x: torch.Tensor, meta: ForwardMeta, rope_cache: torch.Tensor):
        T = x.shape[0]
        qkv = self.qkv_proj(x)
        q, k, v = self.qkv_proj.split_output(qkv)
        q = self.q_norm(q.view(T, self.num_heads, self.head_dim).contiguous())
        k = self.k_norm(k.view(T, self.num_kv_heads, self.head_dim).contiguous())
        v = v.view(T, self.num_kv_heads, self.head_dim).contiguous()
        k_cache, v_cache = meta.kv_cache.layer(self.local_layer_idx)
        ops.rope_and_cache(q, k, v, k_cache, v_cache, meta.positions, rope_cache,
                           meta.slot_mapping)

        idx_cache = meta.kv_cache.index_layer(self.local_layer_idx)
        q_idx, k_idx = self.indexer.project(x, meta.positions, rope_cache)
        ops.store_indexer_cache(k_idx, idx_cache, meta.slot_mapping)

        if meta.is_prefill:
            attn = self._sparse_prefill(q, q_idx, meta, k_cache, v_cache, idx_cache)
        else:
            positions = self._decode_positions(q_idx, idx_cache, meta)
            attn = ops.msa_paged_attention_decode(
                q, k_cache, v_cache, meta.block_tables, meta.seq_lens,
                positions, self.scale,
            )
        return self.o_proj(attn.reshape(T, self.num_heads * self.head_dim))

    def _block_select(self, scores: torch.Tensor, q_block: torch.Tensor):
        """scores [..., NB] fp32 (-inf at masked blocks); q_block [...]: the
        query's own block index. Returns top-k block ids, -1 at -inf picks."""
        idx = self.indexer
        if idx.local_blocks > 0:
            local = torch.arange(idx.local_blocks, device=scores.device)
            local_idx = (q_block.unsqueeze(-1) - local).clamp(min=0)
            scores = scores.scatter(-1, local_idx, float("inf"))
        k = min(idx.topk_blocks, scores.shape[-1])
        vals, picks = scores.topk(k, dim=-1)
        return picks.masked_fill(vals == float("-inf"), -1)

    def _decode_positions(self, q_idx: torch.Tensor, idx_cache, meta) -> torch.Tensor:
        """[B, Hk, topk*block] int32 token positions, -1 padded. Vectorized
        over the batch (shape fixed by meta.max_seq_len: graph-capturable)."""
        idx = self.indexer
        B = q_idx.shape[0]
        bs_page = idx_cache.shape[1]
        msl = meta.max_seq_len or int(meta.seq_lens.max())
        npages = min((msl + bs_page - 1) // bs_page, meta.block_tables.shape[1])
        msl = min(msl, npages * bs_page)
        keys = idx_cache[meta.block_tables[:, :npages].long()].reshape(
            B, npages * bs_page, -1
        )[:, :msl]                                        # [B, msl, Di]
        scores = torch.einsum("bhd,btd->bht", q_idx.float(), keys.float())
        t = torch.arange(msl, device=scores.device)
        valid = t.view(1, 1, -1) < meta.seq_lens.view(B, 1, 1)
        scores = scores.masked_fill(~valid, float("-inf"))
        NB = (msl + idx.block - 1) // idx.block
        pad = NB * idx.block - msl
        if pad:
            scores = F.pad(scores, (0, pad), value=float("-inf"))
        block_scores = scores.view(B, idx.n_heads, NB, idx.block).amax(-1)
        q_block = ((meta.seq_lens.long() - 1) // idx.block).view(B, 1).expand(
            B, idx.n_heads
        )
        picks = self._block_select(block_scores, q_block)  # [B, Hk, K]
        offs = torch.arange(idx.block, device=picks.device)
        pos = picks.unsqueeze(-1) * idx.block + offs       # [B, Hk, K, block]
        bad = (picks.unsqueeze(-1) < 0) | (
            pos >= meta.seq_lens.view(B, 1, 1, 1)
        )
        pos = pos.masked_fill(bad, -1)
        return pos.reshape(B, idx.n_heads, -1).to(torch.int32).contiguous()

    def _sparse_prefill(self, q, q_idx, meta: ForwardMeta, k_cache, v_cache,
                        idx_cache):
        """Dense torch attention under the HF-identical additive block mask
        (selected blocks get 0, everything else fp32 min on top of causal)."""
        idx = self.indexer
        Hq, Hk, D = self.num_heads, self.num_kv_heads, self.head_dim
        G = Hq // Hk
        bs_page = k_cache.shape[2]
        out = torch.empty(q.shape[0], Hq, D, dtype=q.dtype, device=q.device)
        t0 = 0
        for i in range(meta.batch_size):
            QL = int(meta.query_lens[i])
            L = int(meta.seq_lens[i])
            npages = (L + bs_page - 1) // bs_page
            tabs = meta.block_tables[i, :npages].long()
            kf = (
                k_cache[tabs].permute(0, 2, 1, 3).reshape(npages * bs_page, Hk, D)[:L]
            ).float()                                     # [L, Hk, D]
            vf = (  # V cache is transposed [.., Hk, D, bs]
                v_cache[tabs].permute(0, 3, 1, 2).reshape(npages * bs_page, Hk, D)[:L]
            ).float()
            keys_i = idx_cache[tabs].reshape(npages * bs_page, -1)[:L].float()
            qi = q_idx[t0 : t0 + QL].float()              # [QL, Hi, Di]
            iscores = torch.einsum("qhd,td->hqt", qi, keys_i)  # [Hk, QL, L]
            qpos = torch.arange(L - QL, L, device=q.device).view(1, -1, 1)
            kpos = torch.arange(L, device=q.device).view(1, 1, -1)
            causal = kpos > qpos
            iscores = iscores.masked_fill(causal, float("-inf"))
            NB = (L + idx.block - 1) // idx.block
            pad = NB * idx.block - L
            if pad:
                iscores = F.pad(iscores, (0, pad), value=float("-inf"))
            block_scores = iscores.view(Hk, QL, NB, idx.block).amax(-1)
            q_block = (torch.arange(L - QL, L, device=q.device) // idx.block).view(
                1, -1
            ).expand(Hk, QL)
            picks = self._block_select(block_scores, q_block)  # [Hk, QL, K]
            keep_blocks = torch.zeros(Hk, QL, NB + 1, dtype=torch.bool,
                                      device=q.device)
            keep_blocks.scatter_(-1, picks.masked_fill(picks < 0, NB), True)
            keep = keep_blocks[..., :NB].repeat_interleave(idx.block, dim=-1)[..., :L]
            keep = keep & ~causal                          # [Hk, QL, L]
            mask = torch.where(keep, 0.0, _FMIN).repeat_interleave(G, dim=0)
            qf = q[t0 : t0 + QL].float()                   # [QL, Hq, D]
            kfe = kf.repeat_interleave(G, dim=1)           # [L, Hq, D]
            vfe = vf.repeat_interleave(G, dim=1)
            logits = torch.einsum("qhd,lhd->hql", qf, kfe) * self.scale + mask
            p = logits.softmax(dim=-1)
            out[t0 : t0 + QL] = torch.einsum("hql,lhv->qhv", p, vfe).to(out.dtype)
            t0 += QL
        return out


class M3DecoderLayer(LlamaDecoderLayer):
    def __init__(self, cfg: ModelConfig, layer_idx: int, local_layer_idx: int):
        super().__init__(cfg, layer_idx, local_layer_idx)
        if cfg.layer_types and cfg.layer_types[layer_idx] == "minimax_m3_sparse":
            self.self_attn = M3SparseAttention(cfg, layer_idx, local_layer_idx)
        mlp_types = cfg.raw.get("mlp_layer_types")
        if (mlp_types is None or mlp_types[layer_idx] == "sparse") and cfg.is_moe:
            self.mlp = M3MoE(cfg)
        else:
            self.mlp = M3DenseMLP(
                cfg, int(cfg.raw.get("dense_intermediate_size",
                                     cfg.intermediate_size)))


@register_model("MiniMaxM3ForCausalLM", "MiniMaxM3VLForCausalLM")
class MiniMaxM3ShardModel(LlamaShardModel):
    decoder_layer_cls = M3DecoderLayer

    # every RMSNorm in this family is Gemma-style x*(1+w) with zero-init
    _GEMMA_NORMS = (
        "input_layernorm.weight", "post_attention_layernorm.weight",
        "q_norm.weight", "k_norm.weight",
    )

    def load_hf_weight(self, name: str, tensor: torch.Tensor) -> bool:
        if name.endswith(self._GEMMA_NORMS) or name == "model.norm.weight":
            tensor = tensor.float() + 1.0
        local = self.map_global_layer(name)
        if local is None:
            return False
        parts = local.split(".")
        if len(parts) >= 3 and parts[0] == "layers":
            layer = self.layers[int(parts[1])]
            t = tensor.to(torch.bfloat16)
            if parts[2] == "self_attn" and parts[3] == "indexer":
                mod = getattr(layer.self_attn.indexer, parts[4], None)
                if mod is None:
                    return False
                mod.weight.data.copy_(tensor.to(mod.weight.dtype))
                return True
            if parts[2] == "mlp":
                mlp = layer.mlp
                if isinstance(mlp, M3DenseMLP):
                    if parts[3] in ("gate_up_proj", "down_proj"):
                        getattr(mlp, parts[3]).weight.data.copy_(t)
                        return True
                    return False
                if parts[3] == "gate":
                    if parts[4] == "weight":
                        mlp.experts.router.weight.data.copy_(t)
                    elif parts[4] == "e_score_correction_bias":
                        mlp.experts.router.e_score_correction_bias.data.copy_(
                            tensor.float())
                    return True
                if parts[3] == "experts":
                    if parts[4] == "gate_up_proj":
                        mlp.experts.load_fused_gate_up(t)
                    elif parts[4] == "down_proj":
                        mlp.experts.load_fused_down(t)
                    else:
                        return False
                    return True
                if parts[3] == "shared_experts":
                    if parts[4] in ("gate_up_proj", "down_proj"):
                        getattr(mlp.shared, parts[4]).weight.data.copy_(t)
                        return True
                    return False
                return False
        return super().load_hf_weight(name, tensor)
