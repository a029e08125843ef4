"""DSA / MSA sparse attention: CPU reference semantics + GPU kernels vs
reference (the reference's dsa/msa kernel test coverage)."""

import math

import pytest
import torch

from parallax_amd import ops
from parallax_amd.ops import reference as ref


def _mla_setup(B, H, ctxs, R=512, DR=64, BS=32, seed=0, device="cpu"):
    g = torch.Generator(device=device).manual_seed(seed)
    max_blocks = (max(ctxs) + BS - 1) // BS
    NB = B * max_blocks + 1
    dt = torch.bfloat16 if device != "cpu" else torch.float32
    cache = torch.randn(NB, BS, R + DR, generator=g, dtype=dt, device=device)
    bt = (
        torch.arange(B * max_blocks, dtype=torch.int32, device=device)
        .reshape(B, max_blocks) + 1
    )
    ql = torch.randn(B, H, R, generator=g, dtype=dt, device=device) * 0.1
    qp = torch.randn(B, H, DR, generator=g, dtype=dt, device=device) * 0.1
    sl = torch.tensor(ctxs, dtype=torch.int32, device=device)
    return ql, qp, cache, bt, sl


def _topk_for(ctxs, k, device="cpu", seed=1):
    g = torch.Generator().manual_seed(seed)
    rows = []
    for L in ctxs:
        take = min(k, L)
        idx = torch.randperm(L, generator=g)[:take].sort().values
        rows.append(torch.cat([idx, torch.full((k - take,), -1, dtype=torch.long)]))
    return torch.stack(rows).to(device)


def test_dsa_cpu_matches_dense_when_full():
    """top-k covering the whole context == dense MLA."""
    ql, qp, cache, bt, sl = _mla_setup(2, 8, [40, 17], R=32, DR=16)
    idx = torch.stack([
        torch.arange(48, dtype=torch.long),
        torch.cat([torch.arange(17), torch.full((31,), -1, dtype=torch.long)]),
    ])
    scale = 0.1
    sparse = ref.dsa_paged_attention_decode(ql, qp, cache, bt, sl, idx, scale)
    dense = ref.mla_paged_attention_decode(ql, qp, cache, bt, sl, scale)
    # row 0: indices 0..47 but L=40 -> extra clipped; equals dense
    torch.testing.assert_close(sparse, dense, atol=1e-4, rtol=1e-4)


def test_dsa_cpu_dense_fallback_row():
    ql, qp, cache, bt, sl = _mla_setup(1, 4, [30], R=32, DR=16)
    idx = torch.full((1, 8), -1, dtype=torch.long)  # row starts -1 => dense
    out = ref.dsa_paged_attention_decode(ql, qp, cache, bt, sl, idx, 0.1)
    dense = ref.mla_paged_attention_decode(ql, qp, cache, bt, sl, 0.1)
    torch.testing.assert_close(out, dense, atol=1e-4, rtol=1e-4)


def test_msa_pipeline_cpu():
    """block scores -> topk tokens -> sparse attention selects a superset of
    init+local blocks and matches dense when everything is kept."""
    torch.manual_seed(3)
    B, Hq, Hk, D, BS = 2, 4, 2, 64, 16
    ctxs = [120, 40]
    max_blocks = (max(ctxs) + BS - 1) // BS
    NB = B * max_blocks + 1
    kc = torch.randn(NB, Hk, BS, D)
    vc = torch.randn(NB, Hk, D, BS)
    bt = (torch.arange(B * max_blocks, dtype=torch.int32).reshape(B, max_blocks) + 1)
    q = torch.randn(B, Hq, D) * 0.2
    sl = torch.tensor(ctxs, dtype=torch.int32)
    scores = ref.msa_block_scores(q, kc, vc_dummy := bt, sl, sparse_block=32) \
        if False else ref.msa_block_scores(q, kc, bt, sl, sparse_block=32)
    assert scores.shape[1] == 4  # ceil(120/32)
    pos = ref.msa_topk_tokens(scores, sl, sparse_block=32, topk_blocks=1)
    # init block 0 and last 2 blocks always kept
    assert (pos[0][:32] == torch.arange(32)).all()
    # keeping everything == dense
    pos_all = ref.msa_topk_tokens(scores, sl, 32, topk_blocks=99)
    sparse = ref.msa_paged_attention_decode(q, kc, vc, bt, sl, pos_all, 0.125)
    dense = ref.paged_attention_decode(q, kc, vc, bt, sl, 0.125)
    torch.testing.assert_close(sparse, dense, atol=1e-4, rtol=1e-4)


def test_indexer_cache_roundtrip():
    torch.manual_seed(4)
    T, Hi, Di, BS, NB = 10, 2, 32, 8, 4
    keys = torch.randn(T, Hi, Di)
    cache = torch.zeros(NB, BS, Hi, Di)
    slots = torch.randperm(NB * BS)[:T]
    ref.store_indexer_cache(keys, cache, slots)
    got = cache[slots // BS, slots % BS]
    torch.testing.assert_close(got, keys)
    # scores: q == key of position p makes p the argmax for that row
    bt = torch.arange(NB, dtype=torch.int32).unsqueeze(0)
    sl = torch.tensor([T], dtype=torch.int32)
    cache2 = torch.zeros(1 * NB + 1, BS, Hi, Di)
    slots2 = torch.arange(T) + BS  # block 1 onward
    bt2 = (torch.arange(NB, dtype=torch.int32) + 1).unsqueeze(0)
    ref.store_indexer_cache(keys, cache2, slots2)
    scores = ref.dsa_indexer_scores(
        keys[3].unsqueeze(0) * 3, cache2, torch.ones(1, Hi), bt2, sl
    )
    assert int(scores[0, :T].argmax()) == 3


@pytest.mark.gpu
def test_dsa_kernel_gpu():
    ctxs = [300, 77]
    ql, qp, cache, bt, sl = _mla_setup(2, 64, ctxs, device="cuda")
    idx = _topk_for(ctxs, 64, device="cuda")
    scale = 1.0 / math.sqrt(192)
    out = ops.dsa_paged_attention_decode(ql, qp, cache, bt, sl,
                                         idx.to(torch.int32), scale)
    expect = ref.dsa_paged_attention_decode(
        ql.float().cpu(), qp.float().cpu(), cache.float().cpu(), bt.cpu(),
        sl.cpu(), idx.cpu(), scale,
    )
    torch.testing.assert_close(out.float().cpu(), expect, atol=6e-2, rtol=6e-2)


@pytest.mark.gpu
def test_dsa_kernel_gpu_dense_fallback():
    ctxs = [200]
    ql, qp, cache, bt, sl = _mla_setup(1, 32, ctxs, device="cuda")
    idx = torch.full((1, 16), -1, dtype=torch.int32, device="cuda")
    scale = 1.0 / math.sqrt(192)
    out = ops.dsa_paged_attention_decode(ql, qp, cache, bt, sl, idx, scale)
    dense = ops.mla_paged_attention_decode(ql, qp, cache, bt, sl, scale)
    torch.testing.assert_close(out.float(), dense.float(), atol=2e-2, rtol=2e-2)


@pytest.mark.gpu
def test_msa_kernel_gpu():
    torch.manual_seed(5)
    Hk, G, D, BS = 2, 4, 128, 32
    ctxs = [500, 90]
    B, Hq = len(ctxs), G * 2
    max_blocks = (max(ctxs) + BS - 1) // BS
    NB = B * max_blocks + 1
    kc = torch.randn(NB, Hk, BS, D, dtype=torch.bfloat16, device="cuda")
    vc = torch.randn(NB, Hk, D, BS, dtype=torch.bfloat16, device="cuda")
    bt = (torch.arange(B * max_blocks, dtype=torch.int32, device="cuda")
          .reshape(B, max_blocks) + 1)
    q = torch.randn(B, Hq, D, dtype=torch.bfloat16, device="cuda")
    sl = torch.tensor(ctxs, dtype=torch.int32, device="cuda")
    pos = _topk_for(ctxs, 128, device="cuda")
    out = ops.msa_paged_attention_decode(q, kc, vc, bt, sl,
                                         pos.to(torch.int32), 0.088)
    expect = ref.msa_paged_attention_decode(
        q.float().cpu(), kc.float().cpu(), vc.float().cpu(), bt.cpu(),
        sl.cpu(), pos.cpu(), 0.088,
    )
    torch.testing.assert_close(out.float().cpu(), expect, atol=4e-2, rtol=4e-2)


@pytest.mark.gpu
def test_msa_kernel_gpu_per_head_positions():
    """[B, Hk, P] position lists: each kv head attends its own block set
    (minimax-m3 per-KV-head indexer contract)."""
    torch.manual_seed(9)
    Hk, G, D, BS = 2, 2, 128, 32
    ctxs = [420, 130]
    B, Hq = len(ctxs), G * Hk
    max_blocks = (max(ctxs) + BS - 1) // BS
    NB = B * max_blocks + 1
    kc = torch.randn(NB, Hk, BS, D, dtype=torch.bfloat16, device="cuda")
    vc = torch.randn(NB, Hk, D, BS, dtype=torch.bfloat16, device="cuda")
    bt = (torch.arange(B * max_blocks, dtype=torch.int32, device="cuda")
          .reshape(B, max_blocks) + 1)
    q = torch.randn(B, Hq, D, dtype=torch.bfloat16, device="cuda")
    sl = torch.tensor(ctxs, dtype=torch.int32, device="cuda")
    P = 96
    pos = torch.full((B, Hk, P), -1, dtype=torch.int32, device="cuda")
    g = torch.Generator().manual_seed(3)
    for i, L in enumerate(ctxs):
        for h in range(Hk):
            n = min(P, L)
            sel = torch.randperm(L, generator=g)[:n].sort().values
            pos[i, h, :n] = sel.to(torch.int32).cuda()
    out = ops.msa_paged_attention_decode(q, kc, vc, bt, sl, pos, 0.088)
    expect = ref.msa_paged_attention_decode(
        q.float().cpu(), kc.float().cpu(), vc.float().cpu(), bt.cpu(),
        sl.cpu(), pos.cpu(), 0.088,
    )
    torch.testing.assert_close(out.float().cpu(), expect, atol=4e-2, rtol=4e-2)


@pytest.mark.gpu
def test_dsa_indexer_scores_kernel_gpu():
    """HIP MFMA indexer score pass vs the fp32 torch reference (VERDICT item
    5; reference Metal analogue dsa_indexer.metal)."""
    torch.manual_seed(0)
    B, Hi, Di, bs = 3, 64, 128, 32
    seq_lens = torch.tensor([70, 128, 15], dtype=torch.int32, device="cuda")
    max_blocks = 4
    nb = B * max_blocks + 1
    cache = (torch.randn(nb, bs, Di, device="cuda") * 0.3).bfloat16()
    bt = torch.arange(B * max_blocks, dtype=torch.int32, device="cuda") \
        .reshape(B, max_blocks).contiguous()
    q = (torch.randn(B, Hi, Di, device="cuda") * 0.3).bfloat16()
    w = torch.rand(B, Hi, device="cuda")

    got = ops.dsa_indexer_scores(q, cache, w, bt, seq_lens)
    exp = ref.dsa_indexer_scores(q.float().cpu(), cache.float().cpu(),
                                 w.cpu(), bt.cpu(), seq_lens.cpu())
    for i in range(B):
        L = int(seq_lens[i])
        torch.testing.assert_close(got[i, :L].cpu(), exp[i, :L],
                                   atol=5e-2, rtol=5e-2)
        assert (got[i, L:].cpu() < -1e29).all()


@pytest.mark.gpu
def test_store_indexer_cache_kernel_gpu():
    torch.manual_seed(1)
    T, Di, bs, nb = 10, 128, 16, 4  # last block = trash
    cache = torch.zeros(nb, bs, Di, dtype=torch.bfloat16, device="cuda")
    keys = (torch.randn(T, Di, device="cuda") * 0.5).bfloat16()
    slots = torch.tensor([0, 5, 17, 31, -1, 40, 2, -1, 33, 47],
                         dtype=torch.int64, device="cuda")
    ops.store_indexer_cache(keys, cache, slots)
    flat = cache.reshape(-1, Di)
    for t, s in enumerate(slots.tolist()):
        if s >= 0:
            torch.testing.assert_close(flat[s], keys[t])
    # pad rows landed in the trash block, not in live slots
    assert (flat[: (nb - 1) * bs][~torch.isin(
        torch.arange((nb - 1) * bs, device="cuda"),
        slots[slots >= 0])].float().abs().sum(dim=-1) == 0).all()


@pytest.mark.gpu
def test_msa_indexer_kernels_gpu():
    """HIP block-score + top-k expansion vs the torch reference (per-row
    position SETS compared; widths may differ by padding)."""
    torch.manual_seed(2)
    B, Hq, Hk, D, bs = 2, 8, 2, 128, 16
    seq_lens = torch.tensor([150, 83], dtype=torch.int32, device="cuda")
    max_blocks = 10
    kc = (torch.randn(B * max_blocks + 1, Hk, bs, D, device="cuda") * 0.3).bfloat16()
    bt = torch.arange(B * max_blocks, dtype=torch.int32, device="cuda") \
        .reshape(B, max_blocks).contiguous()
    q = (torch.randn(B, Hq, D, device="cuda") * 0.3).bfloat16()

    sparse_block = 32
    got = ops.msa_block_scores(q, kc, bt, seq_lens, sparse_block)
    exp = ref.msa_block_scores(q.float().cpu(), kc.float().cpu(), bt.cpu(),
                               seq_lens.cpu(), sparse_block)
    for i in range(B):
        nsb = (int(seq_lens[i]) + sparse_block - 1) // sparse_block
        torch.testing.assert_close(got[i, :nsb].cpu(), exp[i, :nsb],
                                   atol=5e-2, rtol=5e-2)

    pos_g = ops.msa_topk_tokens(got, seq_lens, sparse_block, topk_blocks=2)
    pos_e = ref.msa_topk_tokens(exp, seq_lens.cpu(), sparse_block,
                                topk_blocks=2)
    for i in range(B):
        sg = set(pos_g[i][pos_g[i] >= 0].tolist())
        se = set(pos_e[i][pos_e[i] >= 0].tolist())
        assert sg == se, f"row {i}: {sorted(sg)[:8]} vs {sorted(se)[:8]}"
