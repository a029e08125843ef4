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


async def one_request(client: httpx.AsyncClient, base_url: str, prompt: str,
                      output_len: int) -> RequestResult:
    res = RequestResult()
    t0 = time.perf_counter()
    last_t = t0
    try:
        async with client.stream(
            "POST", f"{base_url}/v1/chat/completions",
            json={
                "model": "bench", "stream": True,
                "messages": [{"role": "user", "content": prompt}],
                "max_tokens": output_len, "temperature": 1.0,
                "ignore_eos": True,
            },
        ) as r:
            if r.status_code != 200:
                res.error = f"http {r.status_code}"
                return res
            async for line in r.aiter_lines():
                if not line.startswith("data: "):
                    continue
                payload = line[6:]
                if payload == "[DONE]":
                    break
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
                        input_len: int, output_len: int, seed: int = 0) -> dict:
    prompts = build_random_prompts(num_prompts, input_len, seed=seed)
    rng = random.Random(seed)
    results: List[RequestResult] = []
    t_start = time.perf_counter()
    async with httpx.AsyncClient(timeout=600.0) as client:
        tasks = []
        for prompt in prompts:
            tasks.append(asyncio.create_task(
                one_request(client, base_url, prompt, output_len)))
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
    args = ap.parse_args()
    result = asyncio.run(run_benchmark(
        args.base_url, args.num_prompts, args.request_rate,
        args.input_len, args.output_len, args.seed,
    ))
    print(json.dumps(result, indent=2))


if __name__ == "__main__":
    main()
