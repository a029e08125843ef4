"""Serving benchmark harness — the reference's measurement methodology
(src/backend/benchmark/benchmark_serving.py): random dataset, Poisson/fixed
request rate, streaming requests against an OpenAI endpoint, metrics =
request throughput, output/total token throughput, mean/median/std/percentile
TTFT, TPOT, ITL and E2E latency. Fresh asyncio implementation.

  python -m parallax_amd.benchmark.benchmark_serving \
      --base-url http://127.0.0.1:3000 --num-prompts 100 --request-rate 16
"""

from __future__ import annotations

import argparse
import asyncio
import json
import random
import time
from dataclasses import dataclass, field
from typing import List, Optional

import httpx
import numpy as np


@dataclass
class RequestResult:
    success: bool = False
    ttft_s: float = 0.0
    itl_s: List[float] = field(default_factory=list)
    e2e_s: float = 0.0
    output_tokens: int = 0
    prompt_tokens: int = 0
    error: str = ""


def build_random_prompts(num: int, input_len: int, vocab: int = 30000,
                         seed: int = 0) -> List[str]:
    rng = random.Random(seed)
    return [
        " ".join(str(rng.randrange(vocab)) for _ in range(input_len))
        for _ in range(num)
    ]


def build_dataset(name: str, num: int, input_len: int, output_len: int,
                  dataset_path: Optional[str], seed: int = 0):
    """Returns (prompts, output_lens). Dataset modes (reference
    benchmark_serving.py:122-144 dataset breadth):
    - random: fixed-length random-token prompts
    - sharegpt: a LOCAL ShareGPT-format JSON (offline env: no downloader);
      human turns become prompts, the following reply's length caps output
    - synthetic-sharegpt: offline lognormal length mix approximating the
      ShareGPT distribution (no file needed)"""
    rng = random.Random(seed)
    if name == "random":
        return build_random_prompts(num, input_len, seed=seed), [output_len] * num
    if name == "sharegpt":
        if not dataset_path:
            raise SystemExit("--dataset-path required for --dataset-name sharegpt")
        with open(dataset_path) as f:
            data = json.load(f)
        pairs = []
        for conv in data:
            turns = conv.get("conversations") or conv.get("items") or []
            for i in range(len(turns) - 1):
                if turns[i].get("from") in ("human", "user"):
                    prompt = turns[i].get("value", "")
                    reply = turns[i + 1].get("value", "")
                    if prompt and reply:
                        pairs.append((prompt, max(1, len(reply.split()))))
        if not pairs:
            raise SystemExit(f"no usable turns in {dataset_path}")
        rng.shuffle(pairs)
        pairs = [pairs[i % len(pairs)] for i in range(num)]
        return [p for p, _ in pairs], [o for _, o in pairs]
    if name == "synthetic-sharegpt":
        prompts, outs = [], []
        for _ in range(num):
            ilen = max(4, min(4096, int(rng.lognormvariate(4.9, 1.0))))
            olen = max(4, min(1024, int(rng.lognormvariate(4.8, 0.9))))
            prompts.append(" ".join(str(rng.randrange(30000))
                                    for _ in range(ilen)))
            outs.append(olen)
        return prompts, outs
    raise SystemExit(f"unknown dataset {name}")


async def one_request(client: httpx.AsyncClient, base_url: str, prompt: str,
                      output_len: int, backend: str = "chat") -> RequestResult:
    """backend: "chat" (/v1/chat/completions) or "completions"
    (/v1/completions) — the reference ships per-backend request functions
    (backend_request_func.py); both endpoints stream SSE here."""
    res = RequestResult()
    t0 = time.perf_counter()
    last_t = t0
    if backend == "completions":
        url = f"{base_url}/v1/completions"
        body = {"model": "bench", "stream": True, "prompt": prompt,
                "max_tokens": output_len, "temperature": 1.0,
                "ignore_eos": True}
    else:
        url = f"{base_url}/v1/chat/completions"
        body = {"model": "bench", "stream": True,
                "messages": [{"role": "user", "content": prompt}],
                "max_tokens": output_len, "temperature": 1.0,
                "ignore_eos": True}
    try:
        async with client.stream("POST", url, json=body) as r:
            if r.status_code != 200:
                res.error = f"http {r.status_code}"
                return res
            async for line in r.aiter_lines():
                if not line.startswith("data: "):
                    continue
                payload = line[6:]
                if payload == "[DONE]":
                    break
                # only the final usage chunk needs parsing; token chunks
                # just need a timestamp (at 20k tok/s a json.loads per chunk
                # makes the CLIENT the bottleneck of the measurement)
                if '"usage"' in payload:
                    msg = json.loads(payload)
                    if msg.get("usage"):
                        res.output_tokens = msg["usage"]["completion_tokens"]
                        res.prompt_tokens = msg["usage"]["prompt_tokens"]
                        continue
                now = time.perf_counter()
                if res.ttft_s == 0.0:
                    res.ttft_s = now - t0
                else:
                    res.itl_s.append(now - last_t)
                last_t = now
        res.e2e_s = time.perf_counter() - t0
        res.success = res.output_tokens > 0 or res.ttft_s > 0
    except (httpx.HTTPError, json.JSONDecodeError) as e:
        res.error = str(e)
    return res


async def run_benchmark(base_url: str, num_prompts: int, request_rate: float,
                        input_len: int, output_len: int, seed: int = 0,
                        dataset: str = "random",
                        dataset_path: Optional[str] = None,
                        backend: str = "chat") -> dict:
    prompts, out_lens = build_dataset(
        dataset, num_prompts, input_len, output_len, dataset_path, seed
    )
    rng = random.Random(seed)
    results: List[RequestResult] = []
    t_start = time.perf_counter()
    # default pool caps at 100 connections: with streaming responses held
    # open for the whole generation, that silently throttles the benchmark to
    # 100 concurrent requests and inflates TTFT by the queueing delay
    limits = httpx.Limits(max_connections=None, max_keepalive_connections=None)
    async with httpx.AsyncClient(timeout=600.0, limits=limits) as client:
        tasks = []
        for prompt, olen in zip(prompts, out_lens):
            tasks.append(asyncio.create_task(
                one_request(client, base_url, prompt, olen, backend)))
            if request_rate != float("inf"):
                # Poisson arrivals at the requested rate (reference behavior)
                await asyncio.sleep(rng.expovariate(request_rate))
        results = list(await asyncio.gather(*tasks))
    wall = time.perf_counter() - t_start

    ok = [r for r in results if r.success]
    if not ok:
        return {"error": "all requests failed",
                "examples": [r.error for r in results[:3]]}

    def stats(xs, scale=1e3):
        xs = np.array(xs) * scale
        return {
            "mean": float(xs.mean()), "median": float(np.median(xs)),
            "std": float(xs.std()), "p90": float(np.percentile(xs, 90)),
            "p99": float(np.percentile(xs, 99)),
        }

    total_output = sum(r.output_tokens for r in ok)
    total_tokens = total_output + sum(r.prompt_tokens for r in ok)
    tpots = [r.e2e_s - r.ttft_s for r in ok if r.output_tokens > 1]
    tpot_per_tok = [
        (r.e2e_s - r.ttft_s) / (r.output_tokens - 1)
        for r in ok if r.output_tokens > 1
    ]
    itls = [x for r in ok for x in r.itl_s]
    return {
        "completed": len(ok),
        "failed": len(results) - len(ok),
        "duration_s": round(wall, 3),
        "request_throughput_rps": round(len(ok) / wall, 3),
        "output_token_throughput_tps": round(total_output / wall, 2),
        "total_token_throughput_tps": round(total_tokens / wall, 2),
        "ttft_ms": stats([r.ttft_s for r in ok]),
        "tpot_ms": stats(tpot_per_tok) if tpot_per_tok else None,
        "itl_ms": stats(itls) if itls else None,
        "e2el_ms": stats([r.e2e_s for r in ok]),
    }


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--base-url", default="http://127.0.0.1:3000")
    ap.add_argument("--num-prompts", type=int, default=100)
    ap.add_argument("--request-rate", type=float, default=16.0,
                    help="req/s Poisson; inf = all at once")
    ap.add_argument("--input-len", type=int, default=512)
    ap.add_argument("--output-len", type=int, default=128)
    ap.add_argument("--seed", type=int, default=0)
    ap.add_argument("--dataset-name", default="random",
                    choices=["random", "sharegpt", "synthetic-sharegpt"])
    ap.add_argument("--dataset-path", default=None,
                    help="local ShareGPT-format JSON (offline env)")
    ap.add_argument("--backend", default="chat",
                    choices=["chat", "completions"])
    args = ap.parse_args()
    result = asyncio.run(run_benchmark(
        args.base_url, args.num_prompts, args.request_rate,
        args.input_len, args.output_len, args.seed,
        dataset=args.dataset_name, dataset_path=args.dataset_path,
        backend=args.backend,
    ))
    print(json.dumps(result, indent=2))


if __name__ == "__main__":
    main()
