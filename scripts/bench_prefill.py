"""Prefill attention + MoE kernel microbenchmarks (GPU box)."""

import math
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from parallax_amd import ops


def bench_prefill(B, S, Hq, Hk, D=128, BS=32, iters=20):
    torch.manual_seed(0)
    max_blocks = (S + BS - 1) // BS
    NB = B * max_blocks + 1
    kc = torch.randn(NB, Hk, BS, D, dtype=torch.bfloat16, device="cuda")
    vc = torch.randn(NB, Hk, BS, D, dtype=torch.bfloat16, device="cuda")
    bt = (
        torch.arange(B * max_blocks, dtype=torch.int32, device="cuda")
        .reshape(B, max_blocks) + 1
    )
    q = torch.randn(B * S, Hq, D, dtype=torch.bfloat16, device="cuda")
    sl = torch.full((B,), S, dtype=torch.int32, device="cuda")
    ql = torch.full((B,), S, dtype=torch.int32, device="cuda")
    scale = 1.0 / math.sqrt(D)
    for _ in range(3):
        ops.prefill_attention(q, kc, vc, bt, sl, ql, scale)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        ops.prefill_attention(q, kc, vc, bt, sl, ql, scale)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    # causal flops: 2 GEMMs, ~S^2/2 effective
    flops = B * Hq * (2 * 2 * S * S * D) / 2
    print(f"prefill B={B} S={S} Hq={Hq} Hk={Hk} -> {dt*1e3:8.3f} ms  {flops/dt/1e12:7.1f} TF")


def bench_moe(T, E, k, H, I, iters=20):
    torch.manual_seed(0)
    x = torch.randn(T, H, dtype=torch.bfloat16, device="cuda") * 0.3
    w_gu = torch.randn(E, 2 * I, H, dtype=torch.bfloat16, device="cuda") * 0.02
    w_dn = torch.randn(E, H, I, dtype=torch.bfloat16, device="cuda") * 0.02
    ids = torch.randint(0, E, (T, k), device="cuda").long()
    w = torch.rand(T, k, dtype=torch.float32, device="cuda")
    for _ in range(3):
        ops.fused_moe_forward(x, w_gu, w_dn, ids, w)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        ops.fused_moe_forward(x, w_gu, w_dn, ids, w)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    # traffic: activated expert weights (upper bound: all E)
    n_active = len(torch.unique(ids))
    wbytes = n_active * (2 * I * H + H * I) * 2
    flops = T * k * (2 * H * 2 * I + 2 * I * H)
    print(
        f"moe T={T} E={E} k={k} H={H} I={I} active={n_active} -> {dt*1e3:8.3f} ms  "
        f"{wbytes/dt/1e12:5.2f} TB/s(w)  {flops/dt/1e12:6.1f} TF"
    )


if __name__ == "__main__":
    # 8B-class prefill shapes
    bench_prefill(1, 2048, 32, 8)
    bench_prefill(4, 2048, 32, 8)
    bench_prefill(1, 8192, 32, 8)
    bench_prefill(16, 512, 32, 8)
    # DeepSeek-V3-class MoE decode/prefill shapes (per layer)
    bench_moe(64, 256, 8, 7168, 2048)
    bench_moe(512, 256, 8, 7168, 2048)
    bench_moe(4096, 256, 8, 7168, 2048)
    # Qwen3-MoE-class
    bench_moe(256, 128, 8, 4096, 1536)
