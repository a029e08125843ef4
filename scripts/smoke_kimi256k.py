"""Instrumented Kimi-K2 256k prefix-reuse smoke (stage timings) — the
standalone twin of tests/test_engine_oracle_gpu.py::test_kimi_k2_256k_...,
used to find where the 256k path spends time on hardware."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from parallax_amd.models.config import ModelConfig
from parallax_amd.server.engine import Engine, EngineArgs
from parallax_amd.server.sampling_params import SamplingParams


def main():
    CTX = int(os.environ.get("SMOKE_CTX", 256 * 1024))
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
    t0 = time.time()
    eng = Engine(
        cfg,
        EngineArgs(block_size=32, num_kv_blocks=(CTX // 32) * 2 + 64,
                   max_batch_size=4, max_num_tokens_per_batch=16384,
                   prefill_chunk_size=16384, max_model_len=CTX + 512,
                   enable_prefix_cache=True),
        random_weights=True,
    )
    print(f"init {time.time()-t0:.1f}s", flush=True)
    g = torch.Generator().manual_seed(7)
    prefix = torch.randint(0, cfg.vocab_size, (CTX,), generator=g).tolist()
    sp = SamplingParams(temperature=0.0, max_new_tokens=4, ignore_eos=True)

    t0 = time.time()
    r1 = eng.submit(prefix + [7, 8, 9], sp)
    steps = 0
    while eng.has_work:
        eng.step()
        steps += 1
        if steps % 4 == 0:
            torch.cuda.synchronize()
            print(f"  step {steps} t={time.time()-t0:.1f}s", flush=True)
    torch.cuda.synchronize()
    print(f"req1 {steps} steps {time.time()-t0:.1f}s", flush=True)

    t0 = time.time()
    r2 = eng.submit(prefix + [10, 11, 12], sp)
    eng.step()
    cached = eng.cache_manager.get(r2).num_cached_tokens
    while eng.has_work:
        eng.step()
    torch.cuda.synchronize()
    print(f"req2 cached={cached}/{CTX} in {time.time()-t0:.1f}s", flush=True)
    assert cached >= CTX - 32


if __name__ == "__main__":
    main()
