"""End-to-end serving on the GPU engine: EngineServer + FastAPI app
(in-process TestClient — no sockets) over a small bf16 model with graphs,
warmup and the async-pipelined step loop, exercising non-stream, SSE
streaming, stop tokens and abort-on-validation paths."""

import json

import pytest
import torch

fastapi = pytest.importorskip("fastapi")
from fastapi.testclient import TestClient

from parallax_amd.models.config import ModelConfig
from parallax_amd.server.engine import Engine, EngineArgs
from parallax_amd.server.engine_server import EngineServer
from parallax_amd.server.http_frontend import create_app
from parallax_amd.server.tokenizer_util import TokenizerWrapper

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def gpu_client():
    if not torch.cuda.is_available():
        pytest.skip("needs a GPU")
    cfg = ModelConfig(
        architecture="LlamaForCausalLM", vocab_size=512, hidden_size=256,
        num_layers=2, num_heads=4, num_kv_heads=2, head_dim=64,
        intermediate_size=512, max_position_embeddings=2048,
        eos_token_ids=[2],
    )
    eng = Engine(cfg, EngineArgs(max_batch_size=16, max_model_len=1024,
                                 num_kv_blocks=128, dtype=torch.bfloat16),
                 random_weights=True)
    eng.warmup_serving()
    server = EngineServer(eng)
    server.start()
    tok = TokenizerWrapper(vocab_size=cfg.vocab_size)
    app = create_app(server, tok, model_name="gpu-serving-smoke")
    with TestClient(app) as c:
        yield c
    server.stop()


def test_completion_roundtrip(gpu_client):
    r = gpu_client.post("/v1/completions", json={
        "prompt": [5, 9, 13, 40], "max_tokens": 8, "temperature": 0.0,
        "ignore_eos": True,
    })
    assert r.status_code == 200
    body = r.json()
    assert body["usage"]["completion_tokens"] == 8
    assert body["choices"][0]["finish_reason"] == "length"


def test_streaming_usage_chunk(gpu_client):
    got_usage = None
    with gpu_client.stream("POST", "/v1/completions", json={
        "prompt": [7, 8, 9], "max_tokens": 6, "temperature": 1.0,
        "ignore_eos": True, "stream": True,
    }) as r:
        assert r.status_code == 200
        for line in r.iter_lines():
            if line.startswith("data: ") and '"usage"' in line:
                msg = json.loads(line[6:])
                if msg.get("usage"):
                    got_usage = msg["usage"]
    assert got_usage and got_usage["completion_tokens"] == 6
    assert got_usage["ttft_ms"] >= 0


def test_over_context_rejected(gpu_client):
    r = gpu_client.post("/v1/completions", json={
        "prompt": list(range(3, 1020)), "max_tokens": 64,
    })
    assert r.status_code == 400


def test_concurrent_batch(gpu_client):
    import concurrent.futures as cf

    def one(i):
        return gpu_client.post("/v1/completions", json={
            "prompt": [3 + i, 9, 13], "max_tokens": 5,
            "temperature": 0.0, "ignore_eos": True,
        }).json()["usage"]["completion_tokens"]

    with cf.ThreadPoolExecutor(8) as ex:
        counts = list(ex.map(one, range(8)))
    assert counts == [5] * 8


def test_stats_shape(gpu_client):
    s = gpu_client.get("/stats").json()
    assert s["engine_steps"] > 0 and s["total_output_tokens"] > 0
