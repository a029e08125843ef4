"""Eager PyTorch reference implementations of every custom op.

These are the numerics oracle for the HIP/CDNA4 kernels (tests compare the HIP
kernel against these in fp32) and the CPU execution path for GPU-less plumbing
tests. They are NOT the GPU serving path — on a GPU box the HIP extension is
required and ops fail loudly if it is missing (see ops/__init__.py).

Op inventory mirrors the reference's src/parallax_extensions/ops.py kernel set
(paged_attention v1/v2, reshape_and_cache, mla_paged_attention, ...) re-designed
for the [num_blocks, num_kv_heads, block_size, head_dim] MI355X layout.
"""

from __future__ import annotations

from typing import List, Optional, Tuple

import torch


# -- normalization -----------------------------------------------------------------


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-6) -> torch.Tensor:
    xf = x.float()
    var = xf.pow(2).mean(dim=-1, keepdim=True)
    out = xf * torch.rsqrt(var + eps) * weight.float()
    return out.to(x.dtype)


def fused_add_rmsnorm(
    x: torch.Tensor, residual: torch.Tensor, weight: torch.Tensor, eps: float = 1e-6
) -> Tuple[torch.Tensor, torch.Tensor]:
    """residual' = x + residual; out = rmsnorm(residual')."""
    new_residual = (x.float() + residual.float()).to(x.dtype)
    return rmsnorm(new_residual, weight, eps), new_residual


# -- rotary embedding ----------------------------------------------------------------


def build_rope_cache(
    max_positions: int,
    rot_dim: int,
    base: float = 10000.0,
    scaling_factor: float = 1.0,
    device: torch.device = torch.device("cpu"),
    inv_freq: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """[max_positions, rot_dim] fp32 cache: cos in [:, :rot_dim//2], sin in [:, rot_dim//2:].
    Precomputed on host per CDNA guide (on-device trig turns RoPE VALU-bound)."""
    if inv_freq is None:
        inv_freq = 1.0 / (
            base ** (torch.arange(0, rot_dim, 2, dtype=torch.float32) / rot_dim)
        )
    t = torch.arange(max_positions, dtype=torch.float32) / scaling_factor
    freqs = torch.outer(t, inv_freq)  # [P, rot_dim/2]
    return torch.cat([freqs.cos(), freqs.sin()], dim=-1).to(device)


def rope_inplace(
    q: torch.Tensor,           # [T, Hq, D]
    k: torch.Tensor,           # [T, Hk, D]
    positions: torch.Tensor,   # [T] int
    cos_sin: torch.Tensor,     # [P, rot_dim] fp32
    is_neox: bool = True,
) -> None:
    rot_dim = cos_sin.shape[-1]
    half = rot_dim // 2
    cs = cos_sin[positions.long()]          # [T, rot_dim]
    cos = cs[:, :half].unsqueeze(1)         # [T, 1, half]
    sin = cs[:, half:].unsqueeze(1)

    def _apply(t: torch.Tensor) -> None:
        # clone: .float() on an fp32 tensor is a VIEW — without the clone the
        # first store below would clobber x1/x2 before o2 is evaluated
        rot = t[..., :rot_dim].clone().float()
        if is_neox:
            x1, x2 = rot[..., :half], rot[..., half:]
            o1 = x1 * cos - x2 * sin
            o2 = x2 * cos + x1 * sin
            t[..., :half] = o1.to(t.dtype)
            t[..., half:rot_dim] = o2.to(t.dtype)
        else:  # interleaved (GPT-J style)
            x1, x2 = rot[..., 0::2], rot[..., 1::2]
            o1 = x1 * cos - x2 * sin
            o2 = x2 * cos + x1 * sin
            t[..., 0:rot_dim:2] = o1.to(t.dtype)
            t[..., 1:rot_dim:2] = o2.to(t.dtype)

    _apply(q)
    if k is not None:
        _apply(k)


# -- KV cache scatter ------------------------------------------------------------------


def reshape_and_cache(
    k: torch.Tensor,             # [T, Hk, D]
    v: torch.Tensor,             # [T, Hk, D]
    k_cache: torch.Tensor,       # [num_blocks, Hk, block_size, D]
    v_cache: torch.Tensor,       # [num_blocks, Hk, D, block_size] (transposed)
    slot_mapping: torch.Tensor,  # [T] int (block*bs + off); -1 = skip (padding)
) -> None:
    block_size = k_cache.shape[2]
    valid = slot_mapping >= 0
    slots = slot_mapping[valid].long()
    blk = slots // block_size
    off = slots % block_size
    k_cache[blk, :, off] = k[valid].to(k_cache.dtype)
    v_cache[blk, :, :, off] = v[valid].to(v_cache.dtype)


def mla_reshape_and_cache(
    kv_latent: torch.Tensor,     # [T, lora_rank]
    k_rope: torch.Tensor,        # [T, rope_dim]
    cache: torch.Tensor,         # [num_blocks, block_size, lora_rank + rope_dim]
    slot_mapping: torch.Tensor,
) -> None:
    block_size = cache.shape[1]
    lora_rank = kv_latent.shape[-1]
    valid = slot_mapping >= 0
    slots = slot_mapping[valid].long()
    blk, off = slots // block_size, slots % block_size
    cache[blk, off, :lora_rank] = kv_latent[valid].to(cache.dtype)
    cache[blk, off, lora_rank:] = k_rope[valid].to(cache.dtype)


# -- attention ---------------------------------------------------------------------------


def _gather_kv(
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    block_table: torch.Tensor,  # [max_blocks] int
    seq_len: int,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Gather one request's KV as [seq_len, Hk, D] (V cache is transposed)."""
    block_size = k_cache.shape[2]
    nb = (seq_len + block_size - 1) // block_size
    blocks = block_table[:nb].long()
    k = k_cache[blocks].transpose(1, 2).reshape(nb * block_size, *k_cache.shape[1:2], k_cache.shape[3])
    v = v_cache[blocks].permute(0, 3, 1, 2).reshape(nb * block_size, *v_cache.shape[1:2], v_cache.shape[2])
    return k[:seq_len], v[:seq_len]


def paged_attention_decode(
    q: torch.Tensor,             # [B, Hq, D]
    k_cache: torch.Tensor,       # [num_blocks, Hk, bs, D]
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,  # [B, max_blocks] int
    seq_lens: torch.Tensor,      # [B] int  (context INCLUDING the current token)
    scale: float,
    sliding_window: int = -1,
    softcap: float = 0.0,
    sinks: Optional[torch.Tensor] = None,  # [Hq] attention-sink logits
) -> torch.Tensor:
    B, Hq, D = q.shape
    Hk = k_cache.shape[1]
    group = Hq // Hk
    out = torch.empty_like(q, dtype=q.dtype)
    for i in range(B):
        L = int(seq_lens[i])
        k, v = _gather_kv(k_cache, v_cache, block_tables[i], L)  # [L, Hk, D]
        qi = q[i].float()                                        # [Hq, D]
        kf = k.float().transpose(0, 1)                           # [Hk, L, D]
        vf = v.float().transpose(0, 1)
        kf = kf.repeat_interleave(group, dim=0)                  # [Hq, L, D]
        vf = vf.repeat_interleave(group, dim=0)
        logits = torch.einsum("hd,hld->hl", qi, kf) * scale      # [Hq, L]
        if softcap > 0:
            logits = softcap * torch.tanh(logits / softcap)
        if sliding_window > 0:
            first_valid = max(0, L - sliding_window)
            logits[:, :first_valid] = float("-inf")
        if sinks is not None:
            logits = torch.cat([sinks.float().unsqueeze(-1), logits], dim=-1)
        p = torch.softmax(logits, dim=-1)
        if sinks is not None:
            p = p[:, 1:]
        out[i] = torch.einsum("hl,hld->hd", p, vf).to(q.dtype)
    return out


def prefill_attention(
    q: torch.Tensor,             # [T, Hq, D] packed varlen new tokens
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,  # [B, max_blocks]
    seq_lens: torch.Tensor,      # [B] total context length (prefix + new)
    query_lens: torch.Tensor,    # [B] new-token count per request
    scale: float,
    sliding_window: int = -1,
    softcap: float = 0.0,
    sinks: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """Causal varlen prefill over paged KV (new tokens attend to full prefix +
    causally to each other). KV for the new tokens must already be scattered
    into the cache (reshape_and_cache runs first)."""
    T, Hq, D = q.shape
    Hk = k_cache.shape[1]
    group = Hq // Hk
    out = torch.empty_like(q)
    t0 = 0
    for i in range(len(query_lens)):
        QL = int(query_lens[i])
        L = int(seq_lens[i])
        prefix = L - QL
        k, v = _gather_kv(k_cache, v_cache, block_tables[i], L)
        qi = q[t0 : t0 + QL].float().transpose(0, 1)       # [Hq, QL, D]
        kf = k.float().transpose(0, 1).repeat_interleave(group, dim=0)  # [Hq, L, D]
        vf = v.float().transpose(0, 1).repeat_interleave(group, dim=0)
        logits = torch.einsum("hqd,hld->hql", qi, kf) * scale
        if softcap > 0:
            logits = softcap * torch.tanh(logits / softcap)
        # causal mask: query t0+j (absolute pos prefix+j) sees keys <= prefix+j
        qpos = torch.arange(prefix, L, device=q.device).unsqueeze(-1)   # [QL,1]
        kpos = torch.arange(0, L, device=q.device).unsqueeze(0)          # [1,L]
        mask = kpos > qpos
        if sliding_window > 0:
            mask |= kpos <= (qpos - sliding_window)
        logits.masked_fill_(mask.unsqueeze(0), float("-inf"))
        if sinks is not None:
            logits = torch.cat(
                [sinks.float().view(Hq, 1, 1).expand(Hq, QL, 1), logits], dim=-1
            )
        p = torch.softmax(logits, dim=-1)
        if sinks is not None:
            p = p[..., 1:]
        out[t0 : t0 + QL] = torch.einsum("hql,hld->hqd", p, vf).transpose(0, 1).to(q.dtype)
        t0 += QL
    return out


def mla_paged_attention_decode(
    q_latent: torch.Tensor,      # [B, Hq, lora_rank]  (q absorbed into latent space)
    q_rope: torch.Tensor,        # [B, Hq, rope_dim]
    cache: torch.Tensor,         # [num_blocks, bs, lora_rank + rope_dim]
    block_tables: torch.Tensor,
    seq_lens: torch.Tensor,
    scale: float,
) -> torch.Tensor:
    """DeepSeek MLA decode over the compressed cache:
    softmax(scale * (q_latent . latent^T + q_rope . rope^T)) . latent
    (reference ops.py:73-121). Output is [B, Hq, lora_rank] — the caller applies
    the out-projection from latent space."""
    B, Hq, R = q_latent.shape
    bs = cache.shape[1]
    out = torch.empty_like(q_latent)
    for i in range(B):
        L = int(seq_lens[i])
        nb = (L + bs - 1) // bs
        entries = cache[block_tables[i, :nb].long()].reshape(nb * bs, -1)[:L].float()
        latent, rope = entries[:, :R], entries[:, R:]
        logits = (
            q_latent[i].float() @ latent.T + q_rope[i].float() @ rope.T
        ) * scale                                            # [Hq, L]
        p = torch.softmax(logits, dim=-1)
        out[i] = (p @ latent).to(q_latent.dtype)
    return out


# -- activations ------------------------------------------------------------------------


def silu_and_mul(x: torch.Tensor) -> torch.Tensor:
    """x = [gate | up] on the last dim; returns silu(gate) * up."""
    half = x.shape[-1] // 2
    gate, up = x[..., :half].float(), x[..., half:].float()
    return (torch.nn.functional.silu(gate) * up).to(x.dtype)


def gelu_and_mul(x: torch.Tensor) -> torch.Tensor:
    half = x.shape[-1] // 2
    gate, up = x[..., :half].float(), x[..., half:].float()
    return (torch.nn.functional.gelu(gate, approximate="tanh") * up).to(x.dtype)


# -- sampling -----------------------------------------------------------------------------


def sample_tokens(
    logits: torch.Tensor,        # [B, vocab] (any device)
    temperatures,                # [B] floats (list or CPU tensor)
    top_ps,
    top_ks,                      # ints, -1 = off
    min_ps,
    generator: Optional[torch.Generator] = None,
) -> torch.Tensor:
    """Batched temperature / top-k / top-p / min-p filtering + sampling.

    Fully vectorized — sampling params are HOST values (never .item()'d off the
    device), so the hot path issues no device syncs. Greedy rows
    (temperature == 0) take argmax."""
    B, V = logits.shape
    temps = [float(t) for t in temperatures]
    tps = [float(p) for p in top_ps]
    tks = [int(k) for k in top_ks]
    mps = [float(m) for m in min_ps]
    dev = logits.device
    logits = logits.float()

    greedy = [t <= 0.0 for t in temps]
    need_filter = any(
        (not g) and (0 < k < V or p < 1.0 or m > 0.0)
        for g, k, p, m in zip(greedy, tks, tps, mps)
    )
    if all(greedy):
        return logits.argmax(dim=-1)

    t_gpu = torch.tensor(
        [max(t, 1e-6) for t in temps], dtype=torch.float32, device=dev
    ).unsqueeze(-1)
    if not need_filter:
        # gumbel-max: argmax(logits/T + G), G ~ Gumbel(0,1). Identical
        # distribution to softmax+multinomial but pure elementwise+argmax —
        # torch.multinomial over a [B, 128k] prob matrix is the single most
        # expensive sampler kernel on ROCm
        u = torch.rand(B, V, device=dev, generator=generator)
        gumbel = -torch.log(
            (-torch.log(u.clamp_min(1e-20))).clamp_min(1e-20)
        )
        sampled = (logits / t_gpu + gumbel).argmax(dim=-1)
        if not any(greedy):
            return sampled
        greedy_mask = torch.tensor(greedy, dtype=torch.bool, device=dev)
        return torch.where(greedy_mask, logits.argmax(dim=-1), sampled)

    out = logits.argmax(dim=-1)  # greedy default for every row
    probs = torch.softmax(logits / t_gpu, dim=-1)
    if need_filter:
        sorted_p, idx = torch.sort(probs, dim=-1, descending=True)
        cum = torch.cumsum(sorted_p, dim=-1)
        tp_g = torch.tensor(tps, dtype=torch.float32, device=dev).unsqueeze(-1)
        keep = (cum - sorted_p) < tp_g                       # top-p (first always kept)
        tk = torch.tensor(
            [k if 0 < k < V else V for k in tks], dtype=torch.long, device=dev
        ).unsqueeze(-1)
        keep &= torch.arange(V, device=dev).unsqueeze(0) < tk  # top-k
        mp_g = torch.tensor(mps, dtype=torch.float32, device=dev).unsqueeze(-1)
        keep &= sorted_p >= mp_g * sorted_p[:, :1]             # min-p
        keep[:, 0] = True
        sorted_p = torch.where(keep, sorted_p, torch.zeros_like(sorted_p))
        sorted_p = sorted_p / sorted_p.sum(dim=-1, keepdim=True)
        picked = torch.multinomial(sorted_p, 1, generator=generator)
        sampled = idx.gather(-1, picked).squeeze(-1)
    else:
        sampled = torch.multinomial(probs, 1, generator=generator).squeeze(-1)

    greedy_mask = torch.tensor(greedy, dtype=torch.bool, device=dev)
    return torch.where(greedy_mask, out, sampled)


def apply_penalties(
    logits: torch.Tensor,            # [B, vocab]
    output_token_ids: List[List[int]],
    prompt_token_ids: List[List[int]],
    repetition_penalties: torch.Tensor,
    presence_penalties: torch.Tensor,
    frequency_penalties: torch.Tensor,
) -> torch.Tensor:
    logits = logits.clone()
    for i in range(logits.shape[0]):
        rp = float(repetition_penalties[i])
        if rp != 1.0:
            seen = torch.tensor(
                sorted(set(prompt_token_ids[i]) | set(output_token_ids[i])),
                dtype=torch.long, device=logits.device,
            )
            if seen.numel():
                vals = logits[i, seen]
                logits[i, seen] = torch.where(vals > 0, vals / rp, vals * rp)
        pp, fp = float(presence_penalties[i]), float(frequency_penalties[i])
        if (pp != 0.0 or fp != 0.0) and output_token_ids[i]:
            ids = torch.tensor(output_token_ids[i], dtype=torch.long, device=logits.device)
            counts = torch.bincount(ids, minlength=logits.shape[1]).to(logits.dtype)
            logits[i] -= pp * (counts > 0).to(logits.dtype) + fp * counts
    return logits


# -- sparse attention (DSA: DeepSeek V3.2 top-k tokens; MSA: MiniMax block top-k) --


def dsa_paged_attention_decode(
    q_latent: torch.Tensor,      # [B, Hq, lora_rank]
    q_rope: torch.Tensor,        # [B, Hq, rope_dim]
    cache: torch.Tensor,         # [num_blocks, bs, lora_rank + rope_dim]
    block_tables: torch.Tensor,
    seq_lens: torch.Tensor,
    topk_indices: torch.Tensor,  # [B, index_topk] int; -1 padded; row starting
                                 # with -1 => dense fallback (reference ops.py:182)
    scale: float,
) -> torch.Tensor:
    B, Hq, R = q_latent.shape
    bs = cache.shape[1]
    out = torch.empty_like(q_latent)
    for i in range(B):
        L = int(seq_lens[i])
        nb = (L + bs - 1) // bs
        entries = cache[block_tables[i, :nb].long()].reshape(nb * bs, -1)[:L].float()
        idx = topk_indices[i]
        if int(idx[0]) >= 0:
            keep = idx[(idx >= 0) & (idx < L)].long()
            entries = entries[keep]
        latent, rope = entries[:, :R], entries[:, R:]
        logits = (
            q_latent[i].float() @ latent.T + q_rope[i].float() @ rope.T
        ) * scale
        p = torch.softmax(logits, dim=-1)
        out[i] = (p @ latent).to(q_latent.dtype)
    return out


def dsa_indexer_scores(
    q_index: torch.Tensor,       # [B, index_heads, index_dim]
    index_cache: torch.Tensor,   # [nb, bs, index_heads, index_dim] or
                                 # [nb, bs, index_dim] (keys shared across
                                 # heads, as in DeepSeek-V3.2 where only the
                                 # indexer query is multi-headed)
    head_weights: torch.Tensor,  # [B, index_heads] per-head score weights
    block_tables: torch.Tensor,
    seq_lens: torch.Tensor,
) -> torch.Tensor:
    """Weighted q.k scores over the full context for top-k selection
    (reference dsa_indexer, ops.py:325-367). Returns [B, max_ctx] with -inf
    past each row's length; top-k itself is torch.topk on the result."""
    B, Hi, Di = q_index.shape
    bs = index_cache.shape[1]
    shared_k = index_cache.dim() == 3
    max_ctx = int(seq_lens.max())
    scores = torch.full((B, max_ctx), float("-inf"), dtype=torch.float32,
                        device=q_index.device)
    for i in range(B):
        L = int(seq_lens[i])
        nb = (L + bs - 1) // bs
        keys = index_cache[block_tables[i, :nb].long()].reshape(nb * bs, -1, Di)[:L]
        if shared_k:
            s = torch.einsum("hd,ld->hl", q_index[i].float(),
                             keys.squeeze(1).float())
        else:
            s = torch.einsum("hd,lhd->hl", q_index[i].float(), keys.float())
        s = torch.relu(s)  # per DeepSeek-V3.2 indexer: ReLU before head-weighting
        scores[i, :L] = torch.einsum("h,hl->l", head_weights[i].float(), s)
    return scores


def store_indexer_cache(
    index_keys: torch.Tensor,    # [T, index_heads, index_dim] or [T, index_dim]
    index_cache: torch.Tensor,   # [num_blocks(+1 trash), bs, ...]
    slot_mapping: torch.Tensor,
) -> None:
    """Scatter indexer keys by slot. hipGraph-capture safe: instead of
    boolean compaction (data-dependent shapes) pad tokens (slot -1) are
    redirected into the cache's LAST block, which the engine over-allocates
    as a trash row (MLAKVCache/PagedKVCache index_caches)."""
    bs = index_cache.shape[1]
    slots = slot_mapping.long()
    trash = (index_cache.shape[0] - 1) * bs
    slots = torch.where(slots >= 0, slots, torch.full_like(slots, trash))
    index_cache[slots // bs, slots % bs] = index_keys.to(index_cache.dtype)


def msa_block_scores(
    q: torch.Tensor,             # [B, Hq, D]
    k_cache: torch.Tensor,       # [num_blocks, Hk, bs, D]
    block_tables: torch.Tensor,
    seq_lens: torch.Tensor,
    sparse_block: int,
) -> torch.Tensor:
    """Mean-pooled K per sparse block scored against the mean query
    (reference msa_indexer phase 1). Returns [B, max_sparse_blocks] (-inf pad)."""
    B, Hq, D = q.shape
    Hk = k_cache.shape[1]
    bs = k_cache.shape[2]
    max_blocks = (int(seq_lens.max()) + sparse_block - 1) // sparse_block
    out = torch.full((B, max_blocks), float("-inf"), dtype=torch.float32)
    qm = q.float().mean(dim=1)  # [B, D]
    for i in range(B):
        L = int(seq_lens[i])
        nb = (L + bs - 1) // bs
        k = (
            k_cache[block_tables[i, :nb].long()]
            .transpose(1, 2).reshape(nb * bs, Hk, D)[:L]
            .float().mean(dim=1)
        )  # [L, D]
        nsb = (L + sparse_block - 1) // sparse_block
        for sb in range(nsb):
            seg = k[sb * sparse_block : min((sb + 1) * sparse_block, L)]
            out[i, sb] = seg.mean(dim=0) @ qm[i]
    return out


def msa_topk_tokens(
    block_scores: torch.Tensor,  # [B, max_sparse_blocks]
    seq_lens: torch.Tensor,
    sparse_block: int,
    topk_blocks: int,
    init_blocks: int = 1,
    local_blocks: int = 2,
) -> torch.Tensor:
    """Top-k sparse blocks per sequence expanded to token positions, always
    keeping the first `init_blocks` and last `local_blocks` (reference
    msa_block_topk_tokens). Returns [B, max_positions] int64, -1 padded."""
    B = block_scores.shape[0]
    rows = []
    for i in range(B):
        L = int(seq_lens[i])
        nsb = (L + sparse_block - 1) // sparse_block
        keep = set(range(min(init_blocks, nsb)))
        keep |= set(range(max(0, nsb - local_blocks), nsb))
        remaining = [b for b in range(nsb) if b not in keep]
        if remaining and topk_blocks > 0:
            sc = block_scores[i, remaining]
            take = min(topk_blocks, len(remaining))
            top = torch.topk(sc, take).indices.tolist()
            keep |= {remaining[t] for t in top}
        pos = []
        for b in sorted(keep):
            pos.extend(range(b * sparse_block, min((b + 1) * sparse_block, L)))
        rows.append(torch.tensor(sorted(pos), dtype=torch.int64))
    max_p = max(r.numel() for r in rows)
    out = torch.full((B, max_p), -1, dtype=torch.int64)
    for i, r in enumerate(rows):
        out[i, : r.numel()] = r
    return out


def msa_paged_attention_decode(
    q: torch.Tensor,             # [B, Hq, D]
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,
    seq_lens: torch.Tensor,
    token_positions: torch.Tensor,  # [B, max_positions] int, -1 padded; or
                                    # [B, Hk, max_positions] for one selection
                                    # per kv head (minimax-m3)
    scale: float,
) -> torch.Tensor:
    """Exact attention over explicit token positions mapped through block
    tables (reference msa_paged_attention, ops.py:594-662)."""
    B, Hq, D = q.shape
    Hk = k_cache.shape[1]
    group = Hq // Hk
    per_head = token_positions.dim() == 3
    out = torch.empty_like(q)
    for i in range(B):
        L = int(seq_lens[i])
        k, v = _gather_kv(k_cache, v_cache, block_tables[i], L)
        for h in range(Hk):
            pos = token_positions[i, h] if per_head else token_positions[i]
            keep = pos[(pos >= 0) & (pos < L)].long()
            kf = k.float()[keep, h]                      # [P, D]
            vf = v.float()[keep, h]
            qh = q[i, h * group:(h + 1) * group].float() # [G, D]
            logits = (qh @ kf.T) * scale
            p = torch.softmax(logits, dim=-1)
            out[i, h * group:(h + 1) * group] = (p @ vf).to(q.dtype)
    return out
