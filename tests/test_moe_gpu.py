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
