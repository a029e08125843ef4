"""MLA decode kernel numerics vs fp32 reference + DeepSeek GPU engine smoke."""

import math

import pytest
import torch

pytestmark = pytest.mark.gpu

from parallax_amd import ops
from parallax_amd.ops import reference as ref


def _setup(B, H, ctxs, R=512, DR=64, BS=32, seed=0):
    g = torch.Generator(device="cuda").manual_seed(seed)
    max_blocks = (max(ctxs) + BS - 1) // BS
    NB = B * max_blocks + 1
    cache = torch.randn(NB, BS, R + DR, generator=g, dtype=torch.bfloat16,
                        device="cuda")
    bt = (
        torch.arange(B * max_blocks, dtype=torch.int32, device="cuda")
        .reshape(B, max_blocks) + 1
    )
    ql = torch.randn(B, H, R, generator=g, dtype=torch.bfloat16, device="cuda") * 0.1
    qp = torch.randn(B, H, DR, generator=g, dtype=torch.bfloat16, device="cuda") * 0.1
    sl = torch.tensor(ctxs, dtype=torch.int32, device="cuda")
    return ql, qp, cache, bt, sl


@pytest.mark.parametrize("H,ctxs", [
    (128, [100, 37]),          # deepseek-v3 head count, short ctx
    (128, [1025, 300]),        # partitioned path
    (32, [64]),                # single head block
    (16, [1, 500]),            # H < HBLOCK (padded columns)
])
def test_mla_decode_kernel(H, ctxs):
    ql, qp, cache, bt, sl = _setup(len(ctxs), H, ctxs)
    scale = 1.0 / math.sqrt(128 + 64)
    out = ops.mla_paged_attention_decode(ql, qp, cache, bt, sl, scale)
    expect = ref.mla_paged_attention_decode(
        ql.float().cpu(), qp.float().cpu(), cache.float().cpu(), bt.cpu(),
        sl.cpu(), scale,
    )
    torch.testing.assert_close(
        out.float().cpu(), expect.float().cpu(), atol=6e-2, rtol=6e-2
    )


def test_deepseek_engine_gpu_smoke():
    """Tiny DeepSeek-shaped model (kv_lora 512/rope 64 geometry) decodes on the
    HIP MLA kernel end-to-end."""
    from parallax_amd.models.config import ModelConfig
    from parallax_amd.server.engine import Engine, EngineArgs
    from parallax_amd.server.sampling_params import SamplingParams

    cfg = ModelConfig(
        architecture="DeepseekV3ForCausalLM", model_type="deepseek_v3",
        vocab_size=512, hidden_size=256, num_layers=2, num_heads=16,
        num_kv_heads=16, head_dim=64, intermediate_size=512,
        moe_intermediate_size=128, num_experts=4, num_experts_per_tok=2,
        num_shared_experts=1, first_k_dense_layers=1, scoring_func="sigmoid",
        n_group=0, topk_group=0,
        q_lora_rank=128, kv_lora_rank=512, qk_nope_head_dim=64,
        qk_rope_head_dim=64, v_head_dim=64,
        max_position_embeddings=2048, eos_token_ids=[],
    )
    eng = Engine(cfg, EngineArgs(num_kv_blocks=128), random_weights=True)
    out = eng.generate(
        [[1, 2, 3, 4, 5, 6, 7, 8], [9] * 40],
        [SamplingParams(temperature=0.0, max_new_tokens=5, ignore_eos=True)] * 2,
    )
    assert all(len(v) == 5 for v in out.values())


@pytest.mark.gpu
def test_mla_decode_fp8_cache():
    """FP8-E4M3 compressed cache: quantize-on-store + dequantize-in-staging
    must track the bf16 kernel within fp8 rounding error."""
    torch.manual_seed(3)
    B, H, R, DR, BS = 4, 32, 512, 64, 32
    ctxs = [300, 77, 512, 40]
    max_blocks = (max(ctxs) + BS - 1) // BS
    NB = B * max_blocks + 1
    latent = torch.randn(sum(ctxs), R, dtype=torch.bfloat16, device="cuda") / 4
    rope = torch.randn(sum(ctxs), DR, dtype=torch.bfloat16, device="cuda") / 4
    cache8 = torch.zeros(NB, BS, R + DR, dtype=torch.float8_e4m3fn, device="cuda")
    cache16 = torch.zeros(NB, BS, R + DR, dtype=torch.bfloat16, device="cuda")
    bt = (torch.arange(B * max_blocks, dtype=torch.int32, device="cuda")
          .reshape(B, max_blocks) + 1)
    slots = []
    t0 = 0
    for i, L in enumerate(ctxs):
        for p in range(L):
            blk = int(bt[i, p // BS])
            slots.append(blk * BS + p % BS)
        t0 += L
    sm = torch.tensor(slots, dtype=torch.int64, device="cuda")
    ops.mla_reshape_and_cache(latent, rope, cache8, sm)
    ops.mla_reshape_and_cache(latent, rope, cache16, sm)
    sl = torch.tensor(ctxs, dtype=torch.int32, device="cuda")
    ql = torch.randn(B, H, R, dtype=torch.bfloat16, device="cuda") / 8
    qp = torch.randn(B, H, DR, dtype=torch.bfloat16, device="cuda") / 8
    scale = 1.0 / math.sqrt(R + DR)
    out8 = ops.mla_paged_attention_decode(ql, qp, cache8, bt, sl, scale)
    out16 = ops.mla_paged_attention_decode(ql, qp, cache16, bt, sl, scale)
    # fp8 rounding on both K and V sides: loose tolerance, but the softmax
    # keeps outputs in the same range
    torch.testing.assert_close(out8.float(), out16.float(), atol=0.12, rtol=0.12)
    # and the fp8 store actually quantized (bytes differ from bf16 view)
    assert cache8.dtype == torch.float8_e4m3fn
