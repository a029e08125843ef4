"""Exhaustive hipBLASLt search for the decode GEMM shapes: what is the
library's true ceiling at M=512? Run with a big budget, e.g.
  PARALLAX_LT_TUNE_CAP=100000 PARALLAX_LT_TUNE_MS=20000 \
      python scripts/search_gemm.py
"""

import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from parallax_amd import ops

SHAPES = [  # (M, N, K, label) — DeepSeek-R1-Distill-Llama-8B decode shapes
    (512, 6144, 4096, "qkv@512"),
    (512, 4096, 4096, "o_proj@512"),
    (512, 28672, 4096, "gate_up@512"),
    (512, 4096, 14336, "down@512"),
    (1024, 6144, 4096, "qkv@1024"),
    (1024, 4096, 4096, "o_proj@1024"),
    (1024, 28672, 4096, "gate_up@1024"),
    (1024, 4096, 14336, "down@1024"),
]


def main():
    assert torch.cuda.is_available()
    out = []
    for M, N, K, label in SHAPES:
        x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
        w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda")
        y = ops.linear(x, w)  # triggers the (budgeted) tuning
        torch.cuda.synchronize()
        iters = 50
        t0 = time.perf_counter()
        for _ in range(iters):
            y = ops.linear(x, w)
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / iters
        tf = 2.0 * M * N * K / dt / 1e12
        gbs = (N * K * 2 + M * K * 2 + M * N * 2) / dt / 1e9
        out.append({"label": label, "M": M, "N": N, "K": K,
                    "us": round(dt * 1e6, 1), "tflops": round(tf, 1),
                    "gbps": round(gbs, 1)})
        print(json.dumps(out[-1]))
    print(json.dumps({"total_us": round(sum(o["us"] for o in out), 1)}))


if __name__ == "__main__":
    main()
