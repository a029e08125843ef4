"""Qwen3-Next (hybrid gated-DeltaNet + gated full attention + MoE) greedy
parity vs HF transformers on CPU fp32, including chunked prefill through the
conv/recurrent state slots."""

import pytest
import torch

transformers = pytest.importorskip("transformers")

from parallax_amd.models import get_model_class
from parallax_amd.models.config import ModelConfig
from parallax_amd.server.engine import Engine, EngineArgs
from parallax_amd.server.sampling_params import SamplingParams


@pytest.fixture(scope="module")
def tiny_next():
    torch.manual_seed(21)
    hf_cfg = transformers.Qwen3NextConfig(
        vocab_size=256, hidden_size=64, intermediate_size=128,
        moe_intermediate_size=32, shared_expert_intermediate_size=32,
        num_hidden_layers=4, num_attention_heads=4, num_key_value_heads=2,
        head_dim=32, num_experts=4, num_experts_per_tok=2, norm_topk_prob=True,
        decoder_sparse_step=1, mlp_only_layers=[],
        linear_num_key_heads=2, linear_num_value_heads=4,
        linear_key_head_dim=16, linear_value_head_dim=16,
        linear_conv_kernel_dim=4,
        layer_types=["linear_attention", "full_attention"] * 2,
        max_position_embeddings=512, tie_word_embeddings=False,
        rope_parameters={"rope_type": "default", "rope_theta": 10000.0,
                         "partial_rotary_factor": 0.25},
    )
    hf = transformers.Qwen3NextForCausalLM(hf_cfg).eval()
    cfg = ModelConfig.from_hf_config(
        hf_cfg.to_dict() | {"architectures": ["Qwen3NextForCausalLM"]}
    )
    return hf, cfg


def test_config(tiny_next):
    _, cfg = tiny_next
    assert cfg.has_linear_layers
    assert cfg.layer_type(0) == "linear_attention"
    assert cfg.layer_type(1) == "full_attention"
    assert cfg.linear_num_value_heads == 4 and cfg.linear_key_head_dim == 16


def test_weight_load_complete(tiny_next):
    hf, cfg = tiny_next
    m = get_model_class(cfg.architecture)(cfg)
    missed = [n for n, t in hf.state_dict().items() if not m.load_hf_weight(n, t)]
    assert missed == [], f"unrouted: {missed[:10]}"


def _engine_with_weights(hf, cfg, **kw):
    eng = Engine(cfg, EngineArgs(block_size=8, num_kv_blocks=128,
                                 dtype=torch.float32, max_batch_size=8, **kw))
    m = get_model_class(cfg.architecture)(cfg).eval()
    for name, t in hf.state_dict().items():
        m.load_hf_weight(name, t)
    eng.model = m.float()
    return eng


def test_greedy_parity(tiny_next):
    hf, cfg = tiny_next
    prompt = [7, 42, 99, 5, 81, 23, 150, 3]
    with torch.no_grad():
        ref = hf.generate(torch.tensor([prompt]), max_new_tokens=6,
                          do_sample=False)[0][len(prompt):].tolist()
    eng = _engine_with_weights(hf, cfg)
    out = eng.generate(
        [prompt], [SamplingParams(temperature=0.0, max_new_tokens=6, ignore_eos=True)]
    )
    assert list(out.values())[0] == ref


def test_greedy_parity_chunked_and_batched(tiny_next):
    hf, cfg = tiny_next
    prompts = [list(range(3, 43)), [9, 9, 9, 10, 11]]
    refs = []
    for p in prompts:
        with torch.no_grad():
            refs.append(hf.generate(torch.tensor([p]), max_new_tokens=4,
                                    do_sample=False)[0][len(p):].tolist())
    eng = _engine_with_weights(hf, cfg, prefill_chunk_size=16)
    out = eng.generate(
        prompts,
        [SamplingParams(temperature=0.0, max_new_tokens=4, ignore_eos=True)] * 2,
    )
    assert list(out.values()) == refs


def test_slot_recycling(tiny_next):
    """Linear-state slots are reset between requests (no state bleed)."""
    hf, cfg = tiny_next
    eng = _engine_with_weights(hf, cfg)
    p = [5, 6, 7, 8]
    sp = [SamplingParams(temperature=0.0, max_new_tokens=3, ignore_eos=True)]
    out1 = list(eng.generate([p], sp).values())[0]
    out2 = list(eng.generate([p], sp).values())[0]
    assert out1 == out2


# -- Qwen3.5 (split DeltaNet projections, dense MLP) ---------------------------


@pytest.fixture(scope="module")
def tiny_q35():
    torch.manual_seed(23)
    from transformers.models.qwen3_5.configuration_qwen3_5 import Qwen3_5TextConfig

    hf_cfg = Qwen3_5TextConfig(
        vocab_size=256, hidden_size=64, intermediate_size=96,
        num_hidden_layers=4, num_attention_heads=4, num_key_value_heads=2,
        head_dim=16, linear_num_key_heads=2, linear_num_value_heads=4,
        linear_key_head_dim=16, linear_value_head_dim=16,
        linear_conv_kernel_dim=3,
        layer_types=["linear_attention", "full_attention"] * 2,
        max_position_embeddings=512, tie_word_embeddings=False,
        rope_theta=10000.0,
    )
    hf = transformers.Qwen3_5ForCausalLM(hf_cfg).eval()
    cfg = ModelConfig.from_hf_config(
        hf_cfg.to_dict() | {"architectures": ["Qwen3_5ForCausalLM"]}
    )
    return hf, cfg


def test_q35_weight_load_complete(tiny_q35):
    hf, cfg = tiny_q35
    assert cfg.qk_norm and cfg.has_linear_layers
    m = get_model_class(cfg.architecture)(cfg)
    missed = [n for n, t in hf.state_dict().items() if not m.load_hf_weight(n, t)]
    assert missed == [], f"unrouted: {missed[:10]}"


def test_q35_greedy_parity(tiny_q35):
    hf, cfg = tiny_q35
    prompt = [7, 42, 99, 5, 81, 23, 150, 3, 66, 12]
    with torch.no_grad():
        ref = hf.generate(
            torch.tensor([prompt]), max_new_tokens=6, do_sample=False
        )[0][len(prompt):].tolist()
    eng = _engine_with_weights(hf, cfg)
    out = eng.generate(
        [prompt], [SamplingParams(temperature=0.0, max_new_tokens=6, ignore_eos=True)]
    )
    assert list(out.values())[0] == ref


def test_q35_chunked_prefill_parity(tiny_q35):
    hf, cfg = tiny_q35
    prompt = list(range(3, 30))
    with torch.no_grad():
        ref = hf.generate(
            torch.tensor([prompt]), max_new_tokens=4, do_sample=False
        )[0][len(prompt):].tolist()
    eng = _engine_with_weights(hf, cfg, prefill_chunk_size=8)
    out = eng.generate(
        [prompt], [SamplingParams(temperature=0.0, max_new_tokens=4, ignore_eos=True)]
    )
    assert list(out.values())[0] == ref


def test_q35_moe_greedy_parity():
    """Qwen3.5-MoE: split DeltaNet projections + qwen3_next-style MoE with a
    sigmoid-gated shared expert."""
    torch.manual_seed(29)
    from transformers.models.qwen3_5_moe.configuration_qwen3_5_moe import (
        Qwen3_5MoeTextConfig,
    )

    hf_cfg = Qwen3_5MoeTextConfig(
        vocab_size=256, hidden_size=64, intermediate_size=96,
        moe_intermediate_size=32, shared_expert_intermediate_size=32,
        num_experts=4, num_experts_per_tok=2, norm_topk_prob=True,
        num_hidden_layers=4, num_attention_heads=4, num_key_value_heads=2,
        head_dim=16, linear_num_key_heads=2, linear_num_value_heads=4,
        linear_key_head_dim=16, linear_value_head_dim=16,
        linear_conv_kernel_dim=3,
        layer_types=["linear_attention", "full_attention"] * 2,
        max_position_embeddings=512, tie_word_embeddings=False,
        rope_theta=10000.0,
    )
    hf = transformers.Qwen3_5MoeForCausalLM(hf_cfg).eval()
    cfg = ModelConfig.from_hf_config(
        hf_cfg.to_dict() | {"architectures": ["Qwen3_5MoeForCausalLM"]}
    )
    assert cfg.is_moe and cfg.has_linear_layers
    prompt = [7, 42, 99, 5, 81, 23, 150, 3, 66, 12]
    with torch.no_grad():
        ref = hf.generate(
            torch.tensor([prompt]), max_new_tokens=5, do_sample=False
        )[0][len(prompt):].tolist()
    m = get_model_class(cfg.architecture)(cfg)
    missed = [n for n, t in hf.state_dict().items() if not m.load_hf_weight(n, t)]
    assert missed == [], f"unrouted: {missed[:10]}"
    eng = _engine_with_weights(hf, cfg)
    out = eng.generate(
        [prompt], [SamplingParams(temperature=0.0, max_new_tokens=5, ignore_eos=True)]
    )
    assert list(out.values())[0] == ref


def test_async_late_finish_matches_sync_hybrid(tiny_next, monkeypatch):
    """Hybrid (linear-state) stacks under the async pipeline with
    value-dependent finishes: a zombie step writes a finished request's
    conv/recurrent state slot, which must stay allocated until the in-flight
    event drains — results must equal the sync engine exactly."""
    from parallax_amd.server import engine as engine_mod

    hf, cfg = tiny_next
    prompts = [[7, 42, 99, 5, 81], [3, 9, 13, 2]]

    def run(async_on, sps):
        monkeypatch.setattr(engine_mod, "ASYNC_DECODE", async_on)
        eng = _engine_with_weights(hf, cfg)
        return eng.generate(prompts, sps), eng

    probe = [SamplingParams(temperature=0.0, max_new_tokens=10,
                            ignore_eos=True)] * 2
    base, _ = run(False, probe)
    stop = list(base.values())[0][3]  # stop request 0 mid-stream

    sps = [SamplingParams(temperature=0.0, max_new_tokens=10, ignore_eos=True,
                          stop_token_ids=[stop])] * 2
    a, eng_a = run(True, sps)
    b, _ = run(False, sps)
    assert list(a.values()) == list(b.values())
    # all slots drained: linear-state slots and KV blocks back in the pools
    assert not eng_a.scheduler.running
    assert eng_a._inflight is None and not eng_a._deferred_free
