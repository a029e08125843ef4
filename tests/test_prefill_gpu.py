"""MFMA prefill attention kernel vs fp32 reference (asymmetric random data —
catches operand/output transposes per the CDNA guide's methodology)."""

import math

import pytest
import torch

pytestmark = pytest.mark.gpu

from parallax_amd import ops
from parallax_amd.ops import reference as ref


def _setup(query_lens, seq_lens, Hq, Hk, D, BS=32, seed=0):
    g = torch.Generator(device="cuda").manual_seed(seed)
    B = len(query_lens)
    max_blocks = (max(seq_lens) + BS - 1) // BS
    NB = B * max_blocks + 1
    kc = torch.randn(NB, Hk, BS, D, generator=g, dtype=torch.bfloat16, device="cuda")
    vc = torch.randn(NB, Hk, D, BS, generator=g, dtype=torch.bfloat16, device="cuda")
    bt = (
        torch.arange(B * max_blocks, dtype=torch.int32, device="cuda").reshape(B, max_blocks)
        + 1
    )
    T = sum(query_lens)
    q = torch.randn(T, Hq, D, generator=g, dtype=torch.bfloat16, device="cuda")
    return q, kc, vc, bt, torch.tensor(seq_lens, dtype=torch.int32, device="cuda"), \
        torch.tensor(query_lens, dtype=torch.int32, device="cuda")


def _run(q, kc, vc, bt, sl, ql, D, **kw):
    scale = 1.0 / math.sqrt(D)
    out = ops.prefill_attention(q, kc, vc, bt, sl, ql, scale, **kw)
    expect = ref.prefill_attention(
        q.float().cpu(), kc.float().cpu(), vc.float().cpu(), bt.cpu(),
        sl.cpu(), ql.cpu(), scale,
        **{k: (v.cpu() if torch.is_tensor(v) else v) for k, v in kw.items()},
    )
    torch.testing.assert_close(
        out.float().cpu(), expect.float().cpu(), atol=4e-2, rtol=4e-2
    )


@pytest.mark.parametrize("D", [128, 64])
def test_prefill_no_prefix(D):
    # pure prefill: seq_len == query_len, varlen batch incl. non-multiple-of-32
    q, kc, vc, bt, sl, ql = _setup([64, 33, 100], [64, 33, 100], 8, 2, D)
    _run(q, kc, vc, bt, sl, ql, D)


def test_prefill_with_prefix():
    # chunked prefill: earlier tokens already cached (prefix 96 and 32)
    q, kc, vc, bt, sl, ql = _setup([32, 64], [128, 96], 4, 4, 128)
    _run(q, kc, vc, bt, sl, ql, 128)


def test_prefill_single_token_rows():
    q, kc, vc, bt, sl, ql = _setup([1, 1, 5], [40, 1, 5], 8, 2, 128)
    _run(q, kc, vc, bt, sl, ql, 128)


def test_prefill_long():
    q, kc, vc, bt, sl, ql = _setup([1024], [1024], 4, 4, 128, seed=3)
    _run(q, kc, vc, bt, sl, ql, 128)


def test_prefill_sliding_window():
    q, kc, vc, bt, sl, ql = _setup([128], [256], 8, 2, 128, seed=4)
    _run(q, kc, vc, bt, sl, ql, 128, sliding_window=64)


def test_prefill_softcap_sinks():
    q, kc, vc, bt, sl, ql = _setup([64], [64], 8, 2, 128, seed=5)
    sinks = torch.randn(8, dtype=torch.float32, device="cuda")
    _run(q, kc, vc, bt, sl, ql, 128, softcap=20.0, sinks=sinks)


def test_prefill_engine_parity_gpu_vs_cpu():
    """Greedy tokens from the GPU engine (HIP prefill+decode, bf16) match the
    CPU fp32 engine on a tiny model (loose but end-to-end)."""
    from parallax_amd.models.config import ModelConfig
    from parallax_amd.server.engine import Engine, EngineArgs
    from parallax_amd.server.sampling_params import SamplingParams
    from parallax_amd.models import get_model_class

    cfg = ModelConfig(
        architecture="LlamaForCausalLM", vocab_size=256, hidden_size=512,
        num_layers=2, num_heads=4, num_kv_heads=2, head_dim=128,
        intermediate_size=1024, max_position_embeddings=2048, eos_token_ids=[],
    )
    prompts = [[3, 17, 42, 99, 5] * 9, [7] * 40]  # 45 and 40 tokens
    sp = [SamplingParams(temperature=0.0, max_new_tokens=4, ignore_eos=True)] * 2

    eng_gpu = Engine(cfg, EngineArgs(num_kv_blocks=256), random_weights=True)
    out_gpu = list(eng_gpu.generate(prompts, sp).values())

    import parallax_amd.parallel.comm as comm_mod
    ctx = comm_mod.get_comm()
    cpu_ctx = comm_mod.CommContext(
        world_size=1, rank=0, pp_size=1, tp_size=1, pp_rank=0, tp_rank=0,
        device=torch.device("cpu"),
    )
    eng_cpu = Engine(
        cfg, EngineArgs(num_kv_blocks=256, dtype=torch.float32), comm=cpu_ctx,
        random_weights=True,
    )
    comm_mod.set_comm(ctx)
    out_cpu = list(eng_cpu.generate(prompts, sp).values())
    # random init + bf16-vs-fp32: expect identical argmax on a tiny model
    assert out_gpu == out_cpu
