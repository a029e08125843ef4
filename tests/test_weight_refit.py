"""Runtime weight refit: reload from disk changes outputs and resets the
prefix cache (reference §3.5 weight-refit path)."""

import json
import os

import pytest
import torch

from parallax_amd.models.config import ModelConfig
from parallax_amd.server.engine import Engine, EngineArgs
from parallax_amd.server.sampling_params import SamplingParams

transformers = pytest.importorskip("transformers")
safetensors = pytest.importorskip("safetensors")


def save_checkpoint(tmpdir, hf, hf_cfg):
    from safetensors.torch import save_file

    os.makedirs(tmpdir, exist_ok=True)
    with open(os.path.join(tmpdir, "config.json"), "w") as f:
        json.dump(hf_cfg.to_dict() | {"architectures": ["LlamaForCausalLM"]}, f)
    save_file({k: v.contiguous() for k, v in hf.state_dict().items()},
              os.path.join(tmpdir, "model.safetensors"))


def test_refit_changes_output_and_resets_prefix_cache(tmp_path):
    torch.manual_seed(0)
    hf_cfg = transformers.LlamaConfig(
        vocab_size=256, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=256, tie_word_embeddings=False,
    )
    hf_a = transformers.LlamaForCausalLM(hf_cfg).eval()
    torch.manual_seed(999)
    hf_b = transformers.LlamaForCausalLM(hf_cfg).eval()
    ckpt_a, ckpt_b = str(tmp_path / "a"), str(tmp_path / "b")
    save_checkpoint(ckpt_a, hf_a, hf_cfg)
    save_checkpoint(ckpt_b, hf_b, hf_cfg)

    cfg = ModelConfig.from_pretrained(ckpt_a)
    eng = Engine(cfg, EngineArgs(block_size=8, num_kv_blocks=128,
                                 dtype=torch.float32), model_path=ckpt_a)
    prompt = list(range(10, 40))
    sp = SamplingParams(temperature=0.0, max_new_tokens=4, ignore_eos=True)
    out_a = list(eng.generate([prompt], [sp]).values())[0]
    assert eng.cache_manager.radix.num_cached_blocks > 0  # prefix published

    n = eng.update_weights_from_disk(ckpt_b)
    assert n > 0
    assert eng.cache_manager.radix.num_cached_blocks == 0  # cache invalidated
    out_b = list(eng.generate([prompt], [sp]).values())[0]
    assert out_a != out_b  # new weights, new tokens

    # and back: refit restores the original behavior
    eng.update_weights_from_disk(ckpt_a)
    out_a2 = list(eng.generate([prompt], [sp]).values())[0]
    assert out_a2 == out_a


def test_shard_loader_selective_files(tmp_path):
    """selective_file_list picks only the shards covering a layer range."""
    from parallax_amd.server.shard_loader import selective_file_list

    idx = {
        "weight_map": {
            "model.embed_tokens.weight": "a.safetensors",
            "model.layers.0.mlp.up_proj.weight": "a.safetensors",
            "model.layers.1.mlp.up_proj.weight": "b.safetensors",
            "model.layers.2.mlp.up_proj.weight": "c.safetensors",
            "lm_head.weight": "c.safetensors",
        }
    }
    os.makedirs(tmp_path / "m", exist_ok=True)
    with open(tmp_path / "m" / "model.safetensors.index.json", "w") as f:
        json.dump(idx, f)
    files = selective_file_list(str(tmp_path / "m"), 1, 2)
    assert "b.safetensors" in files and "a.safetensors" in files  # endpoints travel


def test_http_update_weights_endpoint(tmp_path):
    """POST /update_weights reloads the checkpoint through the serving stack
    (reference /weight/refit -> node reload path, single-host variant)."""
    from fastapi.testclient import TestClient

    from parallax_amd.server.engine_server import EngineServer
    from parallax_amd.server.http_frontend import create_app
    from parallax_amd.server.tokenizer_util import TokenizerWrapper

    torch.manual_seed(5)
    hf_cfg = transformers.LlamaConfig(
        vocab_size=128, hidden_size=32, intermediate_size=64,
        num_hidden_layers=1, num_attention_heads=2, num_key_value_heads=2,
        head_dim=16, max_position_embeddings=128, rope_theta=10000.0,
        tie_word_embeddings=False,
    )
    hf = transformers.LlamaForCausalLM(hf_cfg).eval()
    ckpt = str(tmp_path / "ckpt")
    save_checkpoint(ckpt, hf, hf_cfg)
    cfg = ModelConfig.from_hf_config(
        hf_cfg.to_dict() | {"architectures": ["LlamaForCausalLM"]}
    )
    eng = Engine(cfg, EngineArgs(block_size=8, num_kv_blocks=64,
                                 dtype=torch.float32),
                 model_path=ckpt)
    server = EngineServer(eng)
    server.start()
    tok = TokenizerWrapper(vocab_size=cfg.vocab_size)
    app = create_app(server, tok, "tiny")
    try:
        with TestClient(app) as c:
            r = c.post("/update_weights", json={"model_path": ckpt})
            assert r.status_code == 200
            assert r.json()["updated_tensors"] > 0
    finally:
        server.stop()
