"""Driver-contract guard: `python bench.py --cpu` must emit one valid JSON
line with the agreed schema for every named model config (the driver parses
this exact shape into BENCH_rNN.json / SCALE_rNN.json)."""

import json
import subprocess
import sys

import pytest

REQUIRED_KEYS = {
    "metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
    "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config",
}


@pytest.mark.parametrize("model", [
    "deepseek-r1-distill-llama-8b", "qwen2-0.5b", "qwen2-72b",
    "deepseek-v3", "kimi-k2",
])
def test_bench_cpu_contract(model):
    r = subprocess.run(
        [sys.executable, "bench.py", "--cpu", "--steps", "2", "--warmup", "1",
         "--batch-per-gpu", "4", "--model", model],
        capture_output=True, text=True, timeout=300, cwd=".",
    )
    assert r.returncode == 0, r.stderr[-2000:]
    lines = [l for l in r.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, r.stdout
    j = json.loads(lines[0])
    assert REQUIRED_KEYS <= set(j.keys())
    assert j["metric"] == "output_tokens_per_sec"
    assert j["unit"] == "tokens/s"
    assert j["higher_is_better"] is True
    assert j["scaling"] == "weak"
    assert j["data"] == "synthetic"
    assert j["value"] > 0 and j["ms_per_step"] > 0
    cfg = j["config"]
    assert {"model", "global_batch", "seq_len", "parallelism"} <= set(cfg)
    assert cfg["global_batch"] == 4


def test_bench_shared_prefix_contract():
    r = subprocess.run(
        [sys.executable, "bench.py", "--cpu", "--steps", "2", "--warmup", "1",
         "--batch-per-gpu", "4", "--shared-prefix", "64"],
        capture_output=True, text=True, timeout=300, cwd=".",
    )
    assert r.returncode == 0, r.stderr[-2000:]
    j = json.loads([l for l in r.stdout.splitlines() if l.startswith("{")][0])
    assert j["config"]["shared_prefix"] == 64


@pytest.mark.parametrize("mode,expect", [("auto", "dp2"), ("pp", "pp2")])
def test_bench_two_rank_contract(mode, expect):
    """The driver's SCALE runs launch bench.py under torch.distributed.run:
    both the DP-replica mode (auto for models that fit one GPU) and forced PP
    must produce the contract line with the right parallelism tag."""
    port = 29750 + (0 if mode == "auto" else 1)
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(port), "bench.py", "--cpu", "--steps", "2",
         "--warmup", "1", "--batch-per-gpu", "4", "--parallelism", mode],
        capture_output=True, text=True, timeout=600, cwd=".",
    )
    assert r.returncode == 0, r.stderr[-2000:]
    lines = [l for l in r.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, r.stdout
    j = json.loads(lines[0])
    assert j["config"]["parallelism"] == expect
    assert j["config"]["global_batch"] == 8
    assert j["n_gpus"] == 0  # cpu run
