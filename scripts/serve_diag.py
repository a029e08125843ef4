"""Isolate serving-path overhead layer by layer.

Phase A: engine direct — submit N requests, loop step() in this thread.
Phase B: EngineServer thread — same load through submit()+stream queues.
Both use the same (96 in / 128 out) shape as scripts/serve_e2e.sh, so the
difference vs the HTTP benchmark numbers is the FastAPI/SSE layer.
"""
import os
import sys
import threading
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from bench import MODELS
from parallax_amd.models.config import ModelConfig
from parallax_amd.parallel.comm import init_distributed
from parallax_amd.server.engine import Engine, EngineArgs
from parallax_amd.server.engine_server import EngineServer
from parallax_amd.server.sampling_params import SamplingParams

N, IN_LEN, OUT_LEN = 512, 96, 128


def make_engine():
    comm = init_distributed(pp_size=1, tp_size=1)
    cfg = ModelConfig.from_hf_config(MODELS["deepseek-r1-distill-llama-8b"][0]())
    args = EngineArgs(max_batch_size=768, max_model_len=4096,
                      dtype=torch.bfloat16)
    eng = Engine(cfg, args, comm=comm, random_weights=True)
    eng.warmup_serving()
    return eng


def prompts(seed=0):
    g = torch.Generator().manual_seed(seed)
    return [torch.randint(10, 50000, (IN_LEN,), generator=g).tolist()
            for _ in range(N)]


def phase_a(eng):
    sp = SamplingParams(temperature=1.0, max_new_tokens=OUT_LEN,
                        ignore_eos=True)
    t0 = time.monotonic()
    for i, p in enumerate(prompts()):
        eng.submit(p, sp, rid=f"a{i}")
    done = 0
    ttft = {}
    while eng.has_work:
        outs = eng.step()
        now = time.monotonic()
        for o in outs:
            if o.rid not in ttft:
                ttft[o.rid] = now - t0
        done += len(outs)
    dt = time.monotonic() - t0
    tt = sorted(ttft.values())
    print(f"A engine-direct: {done / dt:.0f} tok/s  wall={dt:.1f}s "
          f"ttft p50={tt[len(tt) // 2] * 1e3:.0f}ms", flush=True)


def phase_b(eng):
    srv = EngineServer(eng)
    srv.start()
    sp = SamplingParams(temperature=1.0, max_new_tokens=OUT_LEN,
                        ignore_eos=True)
    t0 = time.monotonic()
    streams = [srv.submit(p, sp, rid=f"b{i}")
               for i, p in enumerate(prompts(1))]
    ttfts = []

    counts = []

    def drain(st):
        first, n = None, 0
        while True:
            o = st.out_queue.get()
            if o is None:
                break
            if first is None:
                first = time.monotonic() - t0
            n += 1
        ttfts.append(first)
        counts.append(n)

    ths = [threading.Thread(target=drain, args=(s,)) for s in streams]
    for t in ths:
        t.start()
    for t in ths:
        t.join()
    dt = time.monotonic() - t0
    srv.stop()
    done = sum(counts)
    tt = sorted(t for t in ttfts if t is not None)
    print(f"B engine-server: {done / dt:.0f} tok/s  wall={dt:.1f}s "
          f"ttft p50={tt[len(tt) // 2] * 1e3:.0f}ms", flush=True)


if __name__ == "__main__":
    eng = make_engine()
    phase_a(eng)
    phase_b(eng)
