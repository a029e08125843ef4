"""Sampler unit tests: filtering semantics (top-k / top-p / min-p), greedy
rows, penalty math, gumbel-max distribution — mirrors the reference's sampler
tests (sampling/sampler.py) against our vectorized torch implementation."""

import torch

from parallax_amd.ops import reference as ref
from parallax_amd.server.request import InitialRequest
from parallax_amd.server.sampler import Sampler
from parallax_amd.server.sampling_params import SamplingParams


def _logits(rows):
    return torch.tensor(rows, dtype=torch.float32)


def test_greedy_rows_take_argmax():
    lg = _logits([[0.1, 2.0, 0.3], [5.0, 1.0, 0.0]])
    out = ref.sample_tokens(lg, [0.0, 0.0], [1.0, 1.0], [-1, -1], [0.0, 0.0])
    assert out.tolist() == [1, 0]


def test_top_k_one_is_greedy():
    g = torch.Generator().manual_seed(0)
    lg = _logits([[0.0, 3.0, 1.0, 2.0]] * 50)
    out = ref.sample_tokens(lg, [1.0] * 50, [1.0] * 50, [1] * 50, [0.0] * 50,
                            generator=g)
    assert (out == 1).all()


def test_top_p_excludes_tail():
    """With top_p just above the best token's mass, only the top-1/2 survive."""
    g = torch.Generator().manual_seed(1)
    # softmax probs ~ [0.643, 0.236, 0.087, 0.032]
    lg = _logits([[2.0, 1.0, 0.0, -1.0]] * 2000)
    out = ref.sample_tokens(lg, [1.0] * 2000, [0.7] * 2000, [-1] * 2000,
                            [0.0] * 2000, generator=g)
    assert set(out.tolist()) <= {0, 1}


def test_min_p_filters_relative():
    g = torch.Generator().manual_seed(2)
    lg = _logits([[2.0, 1.0, -3.0, -3.0]] * 2000)
    out = ref.sample_tokens(lg, [1.0] * 2000, [1.0] * 2000, [-1] * 2000,
                            [0.5] * 2000, generator=g)
    # min_p=0.5: only tokens with p >= 0.5 * p_max survive
    assert set(out.tolist()) <= {0, 1}


def test_gumbel_matches_softmax_distribution():
    g = torch.Generator().manual_seed(3)
    lg = _logits([[2.0, 1.0, 0.0, -1.0]] * 20000)
    out = ref.sample_tokens(lg, [1.0] * 20000, [1.0] * 20000, [-1] * 20000,
                            [0.0] * 20000, generator=g)
    emp = torch.bincount(out, minlength=4).float() / 20000
    exp = torch.softmax(torch.tensor([2.0, 1.0, 0.0, -1.0]), 0)
    assert (emp - exp).abs().max() < 0.02


def test_penalties():
    lg = _logits([[1.0, 1.0, 1.0, 1.0]])
    out = ref.apply_penalties(
        lg.clone(), [[1, 1, 2]], [[0]],
        torch.tensor([2.0]), torch.tensor([0.5]), torch.tensor([0.25]),
    )
    # token 1 seen twice: rep 1.0/2=0.5, presence -0.5, frequency -0.25*2
    assert abs(out[0, 1].item() - (0.5 - 0.5 - 0.5)) < 1e-5
    # token 2 seen once
    assert abs(out[0, 2].item() - (0.5 - 0.5 - 0.25)) < 1e-5
    # token 0 only in the prompt: repetition applies, presence/frequency don't
    assert abs(out[0, 0].item() - 0.5) < 1e-5
    # unseen token 3 untouched
    assert abs(out[0, 3].item() - 1.0) < 1e-5


def test_sampler_logprobs_only_for_requesting_rows():
    s = Sampler(torch.device("cpu"), seed=0)
    reqs = []
    for i, lp in enumerate([True, False]):
        r = InitialRequest(rid=f"r{i}", prompt_token_ids=[1],
                           sampling_params=SamplingParams(
                               temperature=0.0, logprobs=lp))
        reqs.append(r)
    lg = _logits([[0.0, 3.0, 1.0], [0.0, 3.0, 1.0]])
    out = s.sample_with_logprobs(lg, reqs)
    assert out[0][0] == 1 and out[1][0] == 1
    assert out[0][1] is not None and out[0][1] <= 0.0
    assert out[1][1] is None


def test_logit_bias_forces_token():
    """A large positive bias makes the token win at temperature 0."""
    import torch

    from parallax_amd.server.request import InitialRequest
    from parallax_amd.server.sampler import Sampler
    from parallax_amd.server.sampling_params import SamplingParams

    s = Sampler(torch.device("cpu"), seed=0)
    logits = torch.randn(2, 32)
    reqs = [
        InitialRequest(rid="a", prompt_token_ids=[1],
                       sampling_params=SamplingParams(
                           temperature=0.0, logit_bias={7: 100.0})),
        InitialRequest(rid="b", prompt_token_ids=[1],
                       sampling_params=SamplingParams(temperature=0.0)),
    ]
    toks, _ = s.sample_device(logits, reqs)
    assert toks[0].item() == 7
    assert toks[1].item() == logits[1].argmax().item()


def test_logit_bias_openai_parse():
    from parallax_amd.server.sampling_params import SamplingParams

    sp = SamplingParams.from_openai({"logit_bias": {"42": -5, "7": 3.5}})
    assert sp.logit_bias == {42: -5.0, 7: 3.5}
    rt = SamplingParams.from_dict(sp.to_dict())
    assert rt.logit_bias == sp.logit_bias


def test_per_request_seed_reproducible():
    """Same seed -> same sampled stream regardless of batch composition."""
    import torch

    from parallax_amd.models.config import ModelConfig
    from parallax_amd.parallel.comm import CommContext
    from parallax_amd.server.engine import Engine, EngineArgs
    from parallax_amd.server.sampling_params import SamplingParams

    cfg = ModelConfig(
        architecture="LlamaForCausalLM", vocab_size=211, hidden_size=64,
        num_layers=2, num_heads=4, num_kv_heads=2, head_dim=16,
        intermediate_size=128, max_position_embeddings=256, eos_token_ids=[],
    )

    def make():
        comm = CommContext(world_size=1, rank=0, pp_size=1, tp_size=1,
                           pp_rank=0, tp_rank=0,
                           device=torch.device("cpu"))
        return Engine(cfg, EngineArgs(block_size=8, num_kv_blocks=128,
                                      dtype=torch.float32, seed=0),
                      comm=comm, random_weights=True)

    sp_seeded = SamplingParams(temperature=0.9, max_new_tokens=6,
                               ignore_eos=True, seed=1234)
    sp_other = SamplingParams(temperature=0.9, max_new_tokens=6,
                              ignore_eos=True)
    # run 1: alone; run 2: alongside another request (different batch shape)
    a = make().generate([[5, 9, 13]], [sp_seeded])
    eng = make()
    b = eng.generate([[5, 9, 13], [7, 8]], [sp_seeded, sp_other])
    assert list(a.values())[0] == list(b.values())[0]
    # a different seed diverges
    sp2 = SamplingParams(temperature=0.9, max_new_tokens=6,
                         ignore_eos=True, seed=99)
    c = make().generate([[5, 9, 13]], [sp2])
    assert list(c.values())[0] != list(a.values())[0]
