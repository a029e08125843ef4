"""Grouped-GEMM MoE kernel vs the CPU reference expert loop."""

import pytest
import torch

pytestmark = pytest.mark.gpu

from parallax_amd import ops


def _cpu_moe(x, w_gu, w_down, topk_ids, topk_w):
    T, H = x.shape
    inter = w_gu.shape[1] // 2
    out = torch.zeros(T, H, dtype=torch.float32)
    for t in range(T):
        for j in range(topk_ids.shape[1]):
            e = int(topk_ids[t, j])
            w = float(topk_w[t, j])
            h = x[t].float() @ w_gu[e].float().T
            gate, up = h[:inter], h[inter:]
            act = torch.nn.functional.silu(gate) * up
            out[t] += w * (act @ w_down[e].float().T)
    return out


@pytest.mark.parametrize("T,E,k,H,I", [
    (16, 8, 2, 128, 64),
    (64, 32, 8, 256, 128),
    (3, 4, 2, 64, 64),     # tiny batch
    (128, 4, 1, 128, 192),  # heavy per-expert load (multi-tile segments)
])
def test_moe_forward(T, E, k, H, I):
    torch.manual_seed(0)
    x = (torch.randn(T, H, dtype=torch.bfloat16, device="cuda") * 0.5)
    w_gu = torch.randn(E, 2 * I, H, dtype=torch.bfloat16, device="cuda") * 0.05
    w_down = torch.randn(E, H, I, dtype=torch.bfloat16, device="cuda") * 0.05
    topk_ids = torch.stack(
        [torch.randperm(E, device="cuda")[:k] for _ in range(T)]
    ).long()
    topk_w = torch.rand(T, k, dtype=torch.float32, device="cuda")
    out = ops.fused_moe_forward(x, w_gu, w_down, topk_ids, topk_w)
    expect = _cpu_moe(x.float().cpu(), w_gu.float().cpu(), w_down.float().cpu(),
                      topk_ids.cpu(), topk_w.cpu())
    torch.testing.assert_close(out.float().cpu(), expect, atol=5e-2, rtol=5e-2)


def _quant_rows(x):
    """Emulate the kernel's per-row activation quantization in torch."""
    amax = x.float().abs().amax(dim=-1).clamp_min(1e-8)
    scale = amax / 448.0
    q = (x.float() / scale.unsqueeze(-1)).clamp(-448, 448).to(torch.float8_e4m3fn)
    return q, scale


def _cpu_moe_fp8(x, w_gu_q, w_gu_s, w_dn_q, w_dn_s, topk_ids, topk_w):
    """fp32 emulation of the W8A8 kernel: fp8-quantized operands, fp32
    accumulate, per-row h quantization between the two GEMMs."""
    T, H = x.shape
    inter = w_gu_q.shape[1] // 2
    out = torch.zeros(T, H, dtype=torch.float32)
    xq, xs = _quant_rows(x)
    for t in range(T):
        for j in range(topk_ids.shape[1]):
            e = int(topk_ids[t, j])
            w = float(topk_w[t, j])
            wg = w_gu_q[e].float() * w_gu_s[e].unsqueeze(-1)
            h = (xq[t].float() * xs[t]) @ wg.T
            gate, up = h[:inter], h[inter:]
            act = (torch.nn.functional.silu(gate) * up).to(torch.bfloat16)
            hq, hs = _quant_rows(act.unsqueeze(0))
            wd = w_dn_q[e].float() * w_dn_s[e].unsqueeze(-1)
            out[t] += w * ((hq[0].float() * hs[0]) @ wd.T)
    return out


@pytest.mark.parametrize("T,E,k,H,I", [
    (16, 8, 2, 128, 64),
    (64, 32, 8, 256, 128),
    (3, 4, 2, 64, 64),
])
def test_moe_forward_fp8(T, E, k, H, I):
    """W8A8 kernel vs a torch emulation with identical quantization (tight)
    and vs the unquantized fp32 reference (loose — quantization error)."""
    torch.manual_seed(0)
    x = (torch.randn(T, H, dtype=torch.bfloat16, device="cuda") * 0.5)
    w_gu = torch.randn(E, 2 * I, H, dtype=torch.bfloat16, device="cuda") * 0.05
    w_down = torch.randn(E, H, I, dtype=torch.bfloat16, device="cuda") * 0.05
    topk_ids = torch.stack(
        [torch.randperm(E, device="cuda")[:k] for _ in range(T)]
    ).long()
    topk_w = torch.rand(T, k, dtype=torch.float32, device="cuda")
    q_gu, s_gu = ops.quantize_fp8_weight(w_gu)
    q_dn, s_dn = ops.quantize_fp8_weight(w_down)
    out = ops.fused_moe_forward_fp8(x, q_gu, s_gu, q_dn, s_dn, topk_ids, topk_w)
    emu = _cpu_moe_fp8(x.float().cpu(), q_gu.cpu(), s_gu.cpu(), q_dn.cpu(),
                       s_dn.cpu(), topk_ids.cpu(), topk_w.cpu())
    torch.testing.assert_close(out.float().cpu(), emu, atol=3e-2, rtol=3e-2)
    fp32 = _cpu_moe(x.float().cpu(), w_gu.float().cpu(), w_down.float().cpu(),
                    topk_ids.cpu(), topk_w.cpu())
    rel = (out.float().cpu() - fp32).norm() / fp32.norm().clamp_min(1e-6)
    assert rel < 0.08, f"fp8 path rel error vs fp32 reference {rel:.3f}"


def test_moe_fp8_engine_decode():
    """DeepSeek-V3-class engine with fp8 expert weights decodes through the
    graph-captured path (BASELINE fp8 MFMA config, reduced layer count)."""
    from parallax_amd.models.config import ModelConfig
    from parallax_amd.server.engine import Engine, EngineArgs
    from parallax_amd.server.sampling_params import SamplingParams

    cfg = ModelConfig.from_hf_config({
        "architectures": ["DeepseekV3ForCausalLM"], "model_type": "deepseek_v3",
        "vocab_size": 512, "hidden_size": 256, "num_hidden_layers": 3,
        "num_attention_heads": 8, "num_key_value_heads": 8,
        "intermediate_size": 512, "moe_intermediate_size": 128,
        "n_routed_experts": 16, "num_experts_per_tok": 4, "n_shared_experts": 1,
        "n_group": 4, "topk_group": 2, "routed_scaling_factor": 2.5,
        "norm_topk_prob": True, "first_k_dense_replace": 1,
        "q_lora_rank": 128, "kv_lora_rank": 512, "qk_nope_head_dim": 64,
        "qk_rope_head_dim": 64, "v_head_dim": 64, "rms_norm_eps": 1e-6,
        "rope_theta": 10000.0, "max_position_embeddings": 2048,
    })
    eng = Engine(cfg, EngineArgs(num_kv_blocks=128, moe_weight_dtype="fp8"),
                 random_weights=True)
    out = eng.generate(
        [[1, 2, 3, 4, 5], [7] * 20],
        [SamplingParams(temperature=0.0, max_new_tokens=6, ignore_eos=True)] * 2,
    )
    assert all(len(v) == 6 for v in out.values())


def test_moe_engine_graph_capture():
    """MoE model decodes through the graph-captured path."""
    from parallax_amd.models.config import ModelConfig
    from parallax_amd.server.engine import Engine, EngineArgs
    from parallax_amd.server.sampling_params import SamplingParams

    cfg = ModelConfig(
        architecture="Qwen3MoeForCausalLM", model_type="qwen3_moe",
        vocab_size=512, hidden_size=256, num_layers=2, num_heads=4,
        num_kv_heads=2, head_dim=64, intermediate_size=512,
        moe_intermediate_size=128, num_experts=8, num_experts_per_tok=2,
        max_position_embeddings=2048, eos_token_ids=[],
    )
    eng = Engine(cfg, EngineArgs(num_kv_blocks=128), random_weights=True)
    out = eng.generate(
        [[1, 2, 3, 4, 5], [7] * 20],
        [SamplingParams(temperature=0.0, max_new_tokens=6, ignore_eos=True)] * 2,
    )
    assert eng.graph_runner is not None and len(eng.graph_runner._graphs) > 0
    assert all(len(v) == 6 for v in out.values())
