"""Decode-step time breakdown: engine wall vs graph replay vs sampler vs
bookkeeping. Run on a GPU box:
    python scripts/profile_decode.py --batch 64
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from parallax_amd.models.config import ModelConfig
from parallax_amd.server.engine import Engine, EngineArgs
from parallax_amd.server.sampling_params import SamplingParams

from bench import deepseek_r1_distill_llama_8b  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=64)
    ap.add_argument("--prompt-len", type=int, default=512)
    ap.add_argument("--steps", type=int, default=32)
    args = ap.parse_args()

    cfg = ModelConfig.from_hf_config(deepseek_r1_distill_llama_8b())
    eng = Engine(cfg, EngineArgs(max_batch_size=max(128, args.batch)), random_weights=True)
    g = torch.Generator().manual_seed(7)
    sp = SamplingParams(max_new_tokens=100000, ignore_eos=True)
    for _ in range(args.batch):
        eng.submit(torch.randint(0, cfg.vocab_size, (args.prompt_len,), generator=g).tolist(), sp)
    while True:
        eng.step()
        if all(r.prefill_done and r.num_output_tokens >= 1 for r in eng.scheduler.running.values()):
            break
    for _ in range(4):
        eng.step()
    torch.cuda.synchronize()

    # (a) full engine step
    t0 = time.perf_counter()
    for _ in range(args.steps):
        eng.step()
    torch.cuda.synchronize()
    t_engine = (time.perf_counter() - t0) / args.steps

    # (b) pure graph replay of the captured bucket
    runner = eng.graph_runner
    bucket = runner.bucket_for(args.batch)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        runner._graphs[bucket].replay()
    torch.cuda.synchronize()
    t_replay = (time.perf_counter() - t0) / args.steps

    # (c) sampler alone on live logits
    logits = runner._outputs[bucket][: args.batch].clone()
    reqs = list(eng.scheduler.running.values())[: args.batch]
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        eng.sampler.sample(logits, reqs)
    torch.cuda.synchronize()
    t_sampler = (time.perf_counter() - t0) / args.steps

    # (d) graph input refresh only (bookkeeping python + H2D)
    t0 = time.perf_counter()
    for _ in range(args.steps):
        eng._decode_one(reqs)
    torch.cuda.synchronize()
    t_decode_one = (time.perf_counter() - t0) / args.steps

    print(
        f"batch={args.batch} engine_step={t_engine*1e3:.3f}ms "
        f"graph_replay={t_replay*1e3:.3f}ms decode_one={t_decode_one*1e3:.3f}ms "
        f"sampler={t_sampler*1e3:.3f}ms "
        f"other={(t_engine-t_decode_one-t_sampler)*1e3:.3f}ms"
    )


if __name__ == "__main__":
    main()
