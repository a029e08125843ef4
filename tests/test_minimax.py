"""MiniMax-Text-01 (hybrid lightning attention + softmax MoE, norm-first
alpha/beta residuals) greedy parity vs HF transformers on CPU fp32."""

import pytest
import torch

transformers = pytest.importorskip("transformers")

from parallax_amd.models import get_model_class
from parallax_amd.models.config import ModelConfig
from parallax_amd.server.engine import Engine, EngineArgs
from parallax_amd.server.sampling_params import SamplingParams


@pytest.fixture(scope="module")
def tiny_minimax():
    torch.manual_seed(31)
    hf_cfg = transformers.MiniMaxConfig(
        vocab_size=256, hidden_size=64, intermediate_size=96,
        num_hidden_layers=4, num_attention_heads=4, num_key_value_heads=2,
        num_local_experts=4, num_experts_per_tok=2,
        layer_types=["linear_attention", "full_attention"] * 2,
        block_size=8,  # small lightning block so prefill spans several blocks
        max_position_embeddings=512,
        rope_parameters={"rope_type": "default", "rope_theta": 10000.0},
        tie_word_embeddings=False,
        full_attn_alpha_factor=0.9, full_attn_beta_factor=1.1,
        linear_attn_alpha_factor=1.05, linear_attn_beta_factor=0.95,
        mlp_alpha_factor=0.98, mlp_beta_factor=1.02,
    )
    hf = transformers.MiniMaxForCausalLM(hf_cfg).eval()
    cfg = ModelConfig.from_hf_config(
        hf_cfg.to_dict() | {"architectures": ["MiniMaxForCausalLM"]}
    )
    return hf, cfg


def _engine_with(hf, cfg, **kw):
    eng = Engine(cfg, EngineArgs(block_size=8, num_kv_blocks=128,
                                 dtype=torch.float32, max_batch_size=8, **kw))
    m = get_model_class(cfg.architecture)(cfg).eval()
    for name, t in hf.state_dict().items():
        m.load_hf_weight(name, t)
    m = m.float()
    eng.model = m
    return eng


def test_config(tiny_minimax):
    _, cfg = tiny_minimax
    assert cfg.has_linear_layers
    assert cfg.linear_num_value_heads == 4 and cfg.linear_key_head_dim == 16
    assert cfg.is_moe and cfg.scoring_func == "softmax"


def test_weight_load_complete(tiny_minimax):
    hf, cfg = tiny_minimax
    m = get_model_class(cfg.architecture)(cfg)
    missed = [n for n, t in hf.state_dict().items() if not m.load_hf_weight(n, t)]
    assert missed == [], f"unrouted: {missed[:10]}"


def test_greedy_parity(tiny_minimax):
    hf, cfg = tiny_minimax
    prompt = [7, 42, 99, 5, 81, 23, 150, 3, 66, 12, 9, 200, 41, 6, 88, 13, 54, 2]
    with torch.no_grad():
        ref = hf.generate(
            torch.tensor([prompt]), max_new_tokens=6, do_sample=False
        )[0][len(prompt):].tolist()
    eng = _engine_with(hf, cfg)
    out = eng.generate(
        [prompt], [SamplingParams(temperature=0.0, max_new_tokens=6, ignore_eos=True)]
    )
    assert list(out.values())[0] == ref


def test_chunked_prefill_parity(tiny_minimax):
    """Chunk boundary (12) deliberately not aligned to the lightning block
    size (8): the carried state must make both paths identical."""
    hf, cfg = tiny_minimax
    prompt = list(range(3, 33))
    with torch.no_grad():
        ref = hf.generate(
            torch.tensor([prompt]), max_new_tokens=4, do_sample=False
        )[0][len(prompt):].tolist()
    eng = _engine_with(hf, cfg, prefill_chunk_size=12)
    out = eng.generate(
        [prompt], [SamplingParams(temperature=0.0, max_new_tokens=4, ignore_eos=True)]
    )
    assert list(out.values())[0] == ref
