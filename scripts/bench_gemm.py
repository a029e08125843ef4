"""Skinny GEMM vs hipBLASLt at decode shapes."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from parallax_amd import ops

def bench(M, N, K, iters=50):
    x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda") * 0.2
    w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda") * 0.02
    for fn, name in [(lambda: ops.linear(x, w), "ours"),
                     (lambda: torch.nn.functional.linear(x, w), "blaslt")]:
        for _ in range(5): fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iters): fn()
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / iters
        gb = N * K * 2 / dt / 1e12
        print(f"M={M:4d} N={N:6d} K={K:6d} {name:7s} {dt*1e6:9.1f} us  {gb:5.2f} TB/s(w)")

for M in (64, 128, 256):
    bench(M, 6144, 4096)    # qkv
    bench(M, 4096, 4096)    # o
    bench(M, 28672, 4096)   # gate_up
    bench(M, 4096, 14336)   # down
    bench(M, 128256, 4096)  # lm_head
