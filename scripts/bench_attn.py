"""Paged decode attention microbenchmark: effective KV bandwidth across
(batch, ctx, partition) configs. Run on a GPU box:
    python scripts/bench_attn.py
"""

import math
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from parallax_amd import ops


def run_case(B, Hk, G, D, ctx, max_seq_len, BS=32, iters=50):
    torch.manual_seed(0)
    Hq = Hk * G
    max_blocks = (max(max_seq_len, ctx) + BS - 1) // BS
    nb = B * ((ctx + BS - 1) // BS) + 1
    kc = torch.randn(nb, Hk, BS, D, dtype=torch.bfloat16, device="cuda")
    vc = torch.randn(nb, Hk, D, BS, dtype=torch.bfloat16, device="cuda")
    bt = torch.zeros(B, max_blocks, dtype=torch.int32, device="cuda")
    nblk = (ctx + BS - 1) // BS
    bt[:, :nblk] = (
        torch.arange(B * nblk, dtype=torch.int32, device="cuda").reshape(B, nblk) + 1
    )
    q = torch.randn(B, Hq, D, dtype=torch.bfloat16, device="cuda")
    seq_lens = torch.full((B,), ctx, dtype=torch.int32, device="cuda")
    scale = 1.0 / math.sqrt(D)

    for _ in range(5):
        ops.paged_attention_decode(q, kc, vc, bt, seq_lens, scale, max_seq_len=max_seq_len)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        ops.paged_attention_decode(q, kc, vc, bt, seq_lens, scale, max_seq_len=max_seq_len)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    kv_bytes = B * ctx * Hk * D * 2 * 2
    print(
        f"B={B:4d} Hk={Hk} G={G:2d} D={D} ctx={ctx:6d} msl={max_seq_len:6d} "
        f"-> {dt*1e6:9.1f} us  {kv_bytes/dt/1e12:6.2f} TB/s"
    )


if __name__ == "__main__":
    # flagship decode shape (8B: Hk=8, G=4) at various batch/ctx
    for B, ctx in [(64, 520), (128, 520), (256, 520), (64, 2048), (64, 8192), (8, 8192), (1, 131072)]:
        run_case(B, 8, 4, 128, ctx, max_seq_len=8192 if ctx <= 8192 else ctx)
    # VERDICT target shape: batch 512, ctx 4096 (>=5.5 TB/s effective)
    run_case(512, 8, 4, 128, 4096, 4096, iters=20)
    run_case(512, 8, 4, 128, 640, 1024)
    # effect of the graph-mode fixed partition count (msl >> ctx)
    for msl in [512, 1024, 2048, 8192, 32768]:
        run_case(64, 8, 4, 128, 520, max_seq_len=msl)
    # 70B-class shape (Hk=8, G=8 at TP1) and GQA16
    run_case(64, 8, 8, 128, 2048, 8192)
    run_case(64, 2, 16, 128, 2048, 8192)
