"""Microbench: DSA indexer score pass at long context (VERDICT item 5's
128k-ctx datapoint). Compares the HIP MFMA kernel against the batched torch
composition it replaced and reports effective index-cache stream bandwidth.

Run on a GPU box:  python scripts/bench_indexer.py [--ctx 131072] [--batch 4]
"""

import argparse
import json
import time

import torch

import os
import sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from parallax_amd import ops


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--ctx", type=int, default=131072)
    ap.add_argument("--batch", type=int, default=4)
    ap.add_argument("--heads", type=int, default=64)
    ap.add_argument("--dim", type=int, default=128)
    ap.add_argument("--block-size", type=int, default=32)
    ap.add_argument("--iters", type=int, default=20)
    args = ap.parse_args()

    assert torch.cuda.is_available()
    dev = "cuda"
    B, Hi, Di, bs = args.batch, args.heads, args.dim, args.block_size
    max_blocks = (args.ctx + bs - 1) // bs
    nb = B * max_blocks + 1
    torch.manual_seed(0)
    cache = (torch.randn(nb, bs, Di, device=dev) * 0.3).bfloat16()
    bt = torch.arange(B * max_blocks, dtype=torch.int32, device=dev).reshape(
        B, max_blocks
    ).contiguous()
    q = (torch.randn(B, Hi, Di, device=dev) * 0.3).bfloat16()
    w = torch.rand(B, Hi, device=dev)
    seq_lens = torch.full((B,), args.ctx, dtype=torch.int32, device=dev)

    def timed(fn):
        for _ in range(3):
            fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.iters):
            fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / args.iters

    t_hip = timed(lambda: ops.dsa_indexer_scores(q, cache, w, bt, seq_lens,
                                                 max_ctx=args.ctx))

    def torch_path():
        keys = cache[bt.long()].reshape(B, max_blocks * bs, -1)[:, : args.ctx]
        s = torch.relu(torch.einsum("bhd,btd->bht", q.float(), keys.float()))
        return torch.einsum("bh,bht->bt", w, s)

    t_ref = timed(torch_path)

    stream_gb = B * args.ctx * Di * 2 / 1e9
    flops = 2.0 * B * args.ctx * Hi * Di
    print(json.dumps({
        "bench": "dsa_indexer_scores",
        "ctx": args.ctx, "batch": B, "heads": Hi, "dim": Di,
        "hip_ms": round(t_hip * 1e3, 3),
        "torch_ms": round(t_ref * 1e3, 3),
        "speedup": round(t_ref / t_hip, 2),
        "hip_stream_tbps": round(stream_gb / t_hip / 1e3, 2),
        "hip_tflops": round(flops / t_hip / 1e12, 1),
    }))


if __name__ == "__main__":
    main()
