"""DeepSeek-V3.2 (MLA + DSA sparse attention) greedy parity vs HF transformers
on CPU fp32 — validates the indexer (rope, relu scoring, head weighting), the
paged indexer-key cache, prefill top-k masking, and the sparse decode path.

index_topk is set SMALLER than the prompt so decode genuinely runs sparse
(not the dense fallback)."""

import pytest
import torch

transformers = pytest.importorskip("transformers")

from parallax_amd.models import get_model_class
from parallax_amd.models.config import ModelConfig
from parallax_amd.server.engine import Engine, EngineArgs
from parallax_amd.server.sampling_params import SamplingParams


@pytest.fixture(scope="module")
def tiny_dsv32():
    torch.manual_seed(11)
    hf_cfg = transformers.DeepseekV32Config(
        vocab_size=256,
        hidden_size=128,
        intermediate_size=256,
        moe_intermediate_size=64,
        num_hidden_layers=3,
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
        index_n_heads=4,
        index_head_dim=32,
        index_topk=6,
        max_position_embeddings=512,
        rope_theta=10000.0,
        tie_word_embeddings=False,
        rope_interleave=True,  # deepseek default: GPT-J pairwise main-attn rope
        attention_bias=False,
    )
    hf = transformers.DeepseekV32ForCausalLM(hf_cfg).eval()
    cfg = ModelConfig.from_hf_config(
        hf_cfg.to_dict() | {"architectures": ["DeepseekV32ForCausalLM"]}
    )
    return hf, cfg


def _engine_with(hf, cfg, **eargs):
    eng = Engine(cfg, EngineArgs(block_size=8, num_kv_blocks=256,
                                 dtype=torch.float32, **eargs))
    m = get_model_class(cfg.architecture)(cfg).eval()
    for name, t in hf.state_dict().items():
        m.load_hf_weight(name, t)
    m = m.float()
    m.finalize_weights()
    eng.model = m
    return eng


def test_config_parse(tiny_dsv32):
    _, cfg = tiny_dsv32
    assert cfg.is_mla and cfg.is_moe and cfg.is_dsa
    assert cfg.index_n_heads == 4 and cfg.index_head_dim == 32
    assert cfg.index_topk == 6
    assert cfg.rope_interleave
    assert cfg.scoring_func == "sigmoid" or cfg.raw.get("scoring_func") is None


def test_weight_load_complete(tiny_dsv32):
    hf, cfg = tiny_dsv32
    m = get_model_class(cfg.architecture)(cfg)
    missed = [n for n, t in hf.state_dict().items() if not m.load_hf_weight(n, t)]
    assert missed == [], f"unrouted weights: {missed[:10]}"


def test_greedy_parity_sparse(tiny_dsv32):
    """Prompt (20) and decode contexts exceed index_topk=6, so both prefill
    masking and decode run genuinely sparse."""
    hf, cfg = tiny_dsv32
    prompt = [7, 42, 99, 5, 81, 23, 150, 3, 77, 12, 9, 200, 41, 6, 88, 13,
              54, 2, 190, 66]
    with torch.no_grad():
        ref = hf.generate(
            torch.tensor([prompt]), max_new_tokens=6, do_sample=False
        )[0][len(prompt):].tolist()
    eng = _engine_with(hf, cfg)
    out = eng.generate(
        [prompt], [SamplingParams(temperature=0.0, max_new_tokens=6, ignore_eos=True)]
    )
    assert list(out.values())[0] == ref


def test_greedy_parity_chunked_prefill(tiny_dsv32):
    """Chunked prefill must score indexer keys across chunk boundaries (the
    cached keys from earlier chunks feed later chunks' top-k)."""
    hf, cfg = tiny_dsv32
    prompt = list(range(3, 35))
    with torch.no_grad():
        ref = hf.generate(
            torch.tensor([prompt]), max_new_tokens=4, do_sample=False
        )[0][len(prompt):].tolist()
    eng = _engine_with(hf, cfg, prefill_chunk_size=16)
    out = eng.generate(
        [prompt], [SamplingParams(temperature=0.0, max_new_tokens=4, ignore_eos=True)]
    )
    assert list(out.values())[0] == ref


def test_short_prompt_dense_fallback(tiny_dsv32):
    """Context <= index_topk: decode takes the dense fallback (-1 row) and
    must still match HF (which selects every token)."""
    hf, cfg = tiny_dsv32
    prompt = [5, 9, 33]
    with torch.no_grad():
        ref = hf.generate(
            torch.tensor([prompt]), max_new_tokens=3, do_sample=False
        )[0][len(prompt):].tolist()
    eng = _engine_with(hf, cfg)
    out = eng.generate(
        [prompt], [SamplingParams(temperature=0.0, max_new_tokens=3, ignore_eos=True)]
    )
    assert list(out.values())[0] == ref


def test_dsa_engine_mixed_lifecycle(tiny_dsv32, monkeypatch):
    """DSA (MLA + sparse indexer) engine under mixed finishes and aborts:
    stop tokens, an abort mid-stream, and async-on/off equivalence — the
    sparse index cache and latent cache must drain cleanly."""
    from parallax_amd.server import engine as engine_mod

    hf, cfg = tiny_dsv32

    def run(async_on):
        monkeypatch.setattr(engine_mod, "ASYNC_DECODE", async_on)
        eng = _engine_with(hf, cfg)
        sp_a = SamplingParams(temperature=0.0, max_new_tokens=8,
                              ignore_eos=True)
        eng.submit([7, 42, 99, 5, 81, 23, 150, 3], sp_a, rid="a")
        eng.submit([3, 9, 13, 2, 7, 7, 7, 1], sp_a, rid="b")
        tokens, finish = {}, {}
        for i in range(200):
            if i == 4:
                eng.abort("b")
            for out in eng.step():
                if out.token_id >= 0:
                    tokens.setdefault(out.rid, []).append(out.token_id)
                if out.finished:
                    finish[out.rid] = out.finish_reason
            if not eng.has_work:
                break
        assert not eng.has_work
        assert set(finish) == {"a", "b"}
        assert finish["b"] == "abort" and finish["a"] == "length"
        assert not eng.scheduler.running and eng._inflight is None
        return tokens["a"]

    assert run(True) == run(False)
