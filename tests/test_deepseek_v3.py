"""DeepSeek-V3 (MLA + sigmoid group-limited MoE) greedy parity vs HF
transformers on CPU fp32 — validates absorbed-decode math, the compressed MLA
cache path, and the MoE router."""

import pytest
import torch

transformers = pytest.importorskip("transformers")

from parallax_amd.models import get_model_class
from parallax_amd.models.config import ModelConfig
from parallax_amd.server.engine import Engine, EngineArgs
from parallax_amd.server.sampling_params import SamplingParams


@pytest.fixture(scope="module")
def tiny_dsv3():
    torch.manual_seed(7)
    hf_cfg = transformers.DeepseekV3Config(
        vocab_size=256,
        hidden_size=128,
        intermediate_size=256,
        moe_intermediate_size=64,
        num_hidden_layers=4,
        num_attention_heads=4,
        num_key_value_heads=4,
        n_routed_experts=8,
        num_experts_per_tok=2,
        n_shared_experts=1,
        n_group=2,
        topk_group=1,
        routed_scaling_factor=1.5,
        norm_topk_prob=True,
        first_k_dense_replace=1,
        q_lora_rank=48,
        kv_lora_rank=32,
        qk_nope_head_dim=32,
        qk_rope_head_dim=16,
        v_head_dim=32,
        max_position_embeddings=512,
        rope_theta=10000.0,
        tie_word_embeddings=False,
        rope_interleave=False,
        attention_bias=False,
    )
    hf = transformers.DeepseekV3ForCausalLM(hf_cfg).eval()
    cfg = ModelConfig.from_hf_config(
        hf_cfg.to_dict() | {"architectures": ["DeepseekV3ForCausalLM"]}
    )
    return hf, cfg


def test_config_parse(tiny_dsv3):
    _, cfg = tiny_dsv3
    assert cfg.is_mla and cfg.is_moe
    assert cfg.kv_lora_rank == 32 and cfg.qk_rope_head_dim == 16
    assert not cfg.is_moe_layer(0) and cfg.is_moe_layer(1)
    assert cfg.scoring_func == "sigmoid" or cfg.raw.get("scoring_func") is None


def test_weight_load_complete(tiny_dsv3):
    hf, cfg = tiny_dsv3
    m = get_model_class(cfg.architecture)(cfg)
    missed = [n for n, t in hf.state_dict().items() if not m.load_hf_weight(n, t)]
    assert missed == [], f"unrouted weights: {missed[:10]}"


def test_greedy_parity(tiny_dsv3):
    hf, cfg = tiny_dsv3
    prompt = [7, 42, 99, 5, 81, 23, 150, 3]
    with torch.no_grad():
        ref = hf.generate(
            torch.tensor([prompt]), max_new_tokens=6, do_sample=False
        )[0][len(prompt):].tolist()
    eng = Engine(cfg, EngineArgs(block_size=8, num_kv_blocks=128, dtype=torch.float32))
    m = get_model_class(cfg.architecture)(cfg).eval()
    for name, t in hf.state_dict().items():
        m.load_hf_weight(name, t)
    m = m.float()
    m.finalize_weights()
    eng.model = m
    out = eng.generate(
        [prompt], [SamplingParams(temperature=0.0, max_new_tokens=6, ignore_eos=True)]
    )
    assert list(out.values())[0] == ref


def test_greedy_parity_chunked(tiny_dsv3):
    hf, cfg = tiny_dsv3
    prompt = list(range(3, 43))
    with torch.no_grad():
        ref = hf.generate(
            torch.tensor([prompt]), max_new_tokens=4, do_sample=False
        )[0][len(prompt):].tolist()
    eng = Engine(cfg, EngineArgs(block_size=8, num_kv_blocks=256,
                                 dtype=torch.float32, prefill_chunk_size=16))
    m = get_model_class(cfg.architecture)(cfg).eval()
    for name, t in hf.state_dict().items():
        m.load_hf_weight(name, t)
    m = m.float()
    m.finalize_weights()
    eng.model = m
    out = eng.generate(
        [prompt], [SamplingParams(temperature=0.0, max_new_tokens=4, ignore_eos=True)]
    )
    assert list(out.values())[0] == ref


def test_mla_prefix_cache_reuse(tiny_dsv3):
    """Kimi-K2-style prefix reuse: shared-prefix requests hit the block-radix
    cache over the compressed MLA latent cache (BASELINE config 5 behavior)."""
    hf, cfg = tiny_dsv3
    eng = Engine(cfg, EngineArgs(block_size=8, num_kv_blocks=256,
                                 dtype=torch.float32))
    m = get_model_class(cfg.architecture)(cfg).eval()
    for name, t in hf.state_dict().items():
        m.load_hf_weight(name, t)
    m = m.float()
    m.finalize_weights()
    eng.model = m
    shared = list(range(10, 42))  # 32 shared tokens = 4 full blocks
    sp = SamplingParams(temperature=0.0, max_new_tokens=3, ignore_eos=True)
    out1 = list(eng.generate([shared + [1, 2]], [sp]).values())[0]
    hits0 = eng.cache_manager.radix.hit_tokens
    out2 = list(eng.generate([shared + [3, 4, 5]], [sp]).values())[0]
    assert eng.cache_manager.radix.hit_tokens > hits0  # prefix actually reused
    # and reuse does not change the result
    eng2 = Engine(cfg, EngineArgs(block_size=8, num_kv_blocks=256,
                                  dtype=torch.float32,
                                  enable_prefix_cache=False))
    eng2.model = m
    ref2 = list(eng2.generate([shared + [3, 4, 5]], [sp]).values())[0]
    assert out2 == ref2
