"""Flagship benchmark: continuous-batching decode throughput (output tokens/s)
of DeepSeek-R1-Distill-Llama-8B (Llama-3.1-8B architecture), bf16, synthetic
data, random-init weights — the BASELINE.json PP=1 headline config.

Single GPU by default. With --gpus N (launched via torch.distributed.run, one
rank per GPU) the model runs pipeline-parallel over N stages with the global
batch scaled by N (weak scaling) and N micro-batches in flight.

A "step" = one engine decode iteration: every running request advances one
token (forward through all local layers + paged attention + sampling + commit).
Prefill happens before the timed region (it is the TTFT story, not the decode
throughput story); warmup decode steps precede timing; the timed region is
bracketed by barrier + torch.cuda.synchronize on both sides; the reported time
is the MAX across ranks.
"""

import argparse
import json
import os
import time

import torch


def deepseek_r1_distill_llama_8b() -> dict:
    # architecture of deepseek-ai/DeepSeek-R1-Distill-Llama-8B (= Llama-3.1-8B)
    return {
        "architectures": ["LlamaForCausalLM"],
        "model_type": "llama",
        "vocab_size": 128256,
        "hidden_size": 4096,
        "num_hidden_layers": 32,
        "num_attention_heads": 32,
        "num_key_value_heads": 8,
        "head_dim": 128,
        "intermediate_size": 14336,
        "rms_norm_eps": 1e-5,
        "rope_theta": 500000.0,
        "max_position_embeddings": 131072,
        "tie_word_embeddings": False,
        "torch_dtype": "bfloat16",
        "eos_token_id": 128001,
    }


def qwen2_05b() -> dict:
    return {
        "architectures": ["Qwen2ForCausalLM"],
        "model_type": "qwen2",
        "vocab_size": 151936,
        "hidden_size": 896,
        "num_hidden_layers": 24,
        "num_attention_heads": 14,
        "num_key_value_heads": 2,
        "head_dim": 64,
        "intermediate_size": 4864,
        "rms_norm_eps": 1e-6,
        "rope_theta": 1000000.0,
        "max_position_embeddings": 32768,
        "qkv_bias": True,
        "tie_word_embeddings": True,
        "torch_dtype": "bfloat16",
    }


def qwen2_72b() -> dict:
    return {
        "architectures": ["Qwen2ForCausalLM"],
        "model_type": "qwen2",
        "vocab_size": 152064,
        "hidden_size": 8192,
        "num_hidden_layers": 80,
        "num_attention_heads": 64,
        "num_key_value_heads": 8,
        "head_dim": 128,
        "intermediate_size": 29568,
        "rms_norm_eps": 1e-6,
        "rope_theta": 1000000.0,
        "max_position_embeddings": 32768,
        "qkv_bias": True,
        "tie_word_embeddings": False,
        "torch_dtype": "bfloat16",
    }


def deepseek_v3() -> dict:
    # DeepSeek-V3/R1 671B (also the Kimi-K2 architecture class). bf16 weights
    # need ~168 GB/GPU at PP=8 — run with --gpus 8.
    return {
        "architectures": ["DeepseekV3ForCausalLM"],
        "model_type": "deepseek_v3",
        "vocab_size": 129280,
        "hidden_size": 7168,
        "num_hidden_layers": 61,
        "num_attention_heads": 128,
        "num_key_value_heads": 128,
        "intermediate_size": 18432,
        "moe_intermediate_size": 2048,
        "n_routed_experts": 256,
        "num_experts_per_tok": 8,
        "n_shared_experts": 1,
        "n_group": 8,
        "topk_group": 4,
        "routed_scaling_factor": 2.5,
        "norm_topk_prob": True,
        "first_k_dense_replace": 3,
        "q_lora_rank": 1536,
        "kv_lora_rank": 512,
        "qk_nope_head_dim": 128,
        "qk_rope_head_dim": 64,
        "v_head_dim": 128,
        "rms_norm_eps": 1e-6,
        "rope_theta": 10000.0,
        "max_position_embeddings": 163840,
        "tie_word_embeddings": False,
        "torch_dtype": "bfloat16",
    }


def kimi_k2() -> dict:
    """Kimi-K2 (1T MoE): DeepSeek-V3 architecture class with 384 routed
    experts — the BASELINE prefix-reuse config (Kimi-K2 PP=8 @ 256k ctx)."""
    cfg = deepseek_v3()
    cfg.update({
        "architectures": ["KimiK2ForCausalLM"],
        "model_type": "kimi_k2",
        "vocab_size": 163840,
        "n_routed_experts": 384,
        "num_experts_per_tok": 8,
        "n_group": 1,
        "topk_group": 1,
        "first_k_dense_replace": 1,
        "max_position_embeddings": 262144,
    })
    return cfg


MODELS = {
    "deepseek-r1-distill-llama-8b": (deepseek_r1_distill_llama_8b, "DeepSeek-R1-Distill-Llama-8B"),
    "qwen2-0.5b": (qwen2_05b, "Qwen2-0.5B"),
    "qwen2-72b": (qwen2_72b, "Qwen2-72B"),
    "deepseek-v3": (deepseek_v3, "DeepSeek-V3-671B"),
    "kimi-k2": (kimi_k2, "Kimi-K2"),
}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=128)
    ap.add_argument("--warmup", type=int, default=16)
    ap.add_argument("--model", default="deepseek-r1-distill-llama-8b", choices=MODELS)
    ap.add_argument("--batch-per-gpu", type=int, default=1024,
                    help="decode batch per GPU (global batch = N * this). "
                         "Measured sweep (profiles/README.md round 2): 128 -> "
                         "15.6k tok/s @ 574 ms TTFT, 512 -> 27.1k @ 1.7 s, "
                         "1024 -> 31.1k @ 3.2 s, 2048 -> 33.5k @ 6.3 s; 1024 "
                         "is the default throughput/latency point")
    ap.add_argument("--prompt-len", type=int, default=512)
    ap.add_argument("--block-size", type=int, default=32)
    ap.add_argument("--micro-batches", type=int, default=0,
                    help="0 = one per pipeline stage")
    ap.add_argument("--kv-dtype", default="auto", choices=["auto", "fp8"])
    ap.add_argument("--weight-dtype", default="auto",
                    choices=["auto", "bf16", "fp8"],
                    help="auto: fp8 W8A8 expert weights for the MoE configs "
                         "whose BASELINE entry names fp8 (deepseek-v3, "
                         "kimi-k2), bf16 otherwise")
    ap.add_argument("--parallelism", default="auto", choices=["auto", "dp", "pp"],
                    help="auto: DP replicas when the model fits one GPU "
                         "(288 GB HBM3E), PP layer split otherwise")
    ap.add_argument("--shared-prefix", type=int, default=0,
                    help="tokens of prompt shared across all requests "
                         "(exercises block-radix prefix reuse; BASELINE "
                         "Kimi-K2 config)")
    ap.add_argument("--layers", type=int, default=0,
                    help="override num_hidden_layers (reduced-layer single-GPU "
                         "runs of the PP=8 configs; headline runs use 0 = full)")
    ap.add_argument("--cpu", action="store_true", help="tiny CPU plumbing run")
    args = ap.parse_args()

    from parallax_amd.models.config import ModelConfig
    from parallax_amd.parallel.comm import CommContext, init_distributed
    from parallax_amd.server.engine import Engine, EngineArgs
    from parallax_amd.server.sampling_params import SamplingParams

    world = int(os.environ.get("WORLD_SIZE", "1"))
    n_gpus = args.gpus if world == 1 else world

    cfg_fn, model_name = MODELS[args.model]
    cfg = ModelConfig.from_hf_config(cfg_fn())
    if args.layers:
        cfg.num_layers = args.layers
        model_name += f"-{args.layers}L"

    # parallelism: the scheduler serves replicas (DP) whenever a whole model
    # fits one 288 GB GPU — that is the reference's multi-pipeline deployment
    # for the PP=1 configs — and splits layers (PP) only when it must
    mode = args.parallelism
    if mode == "auto":
        params_gb = (
            cfg.num_layers * cfg.per_layer_param_bytes()
            + cfg.embedding_bytes() + cfg.lm_head_bytes()
        ) / 1e9
        mode = "dp" if (world == 1 or params_gb < 190.0) else "pp"
    comm_world = init_distributed(pp_size=world, tp_size=1)
    if mode == "dp" and world > 1:
        # independent engine per rank; torch.distributed only for barriers
        comm = CommContext(world_size=1, rank=0, pp_size=1, tp_size=1,
                           pp_rank=0, tp_rank=0, device=comm_world.device)
    else:
        mode = "pp" if world > 1 else mode
        comm = comm_world
    if args.cpu:
        cfg.num_layers = 2
        cfg.vocab_size = 1024
        cfg.hidden_size = 256
        cfg.num_heads, cfg.num_kv_heads, cfg.head_dim = 4, 2, 64
        cfg.intermediate_size = 512

    use_gpu = torch.cuda.is_available() and not args.cpu
    global_batch = args.batch_per_gpu * world
    engine_batch = args.batch_per_gpu if (mode == "dp") else global_batch
    total_new_tokens = args.warmup + args.steps + 8
    eargs = EngineArgs(
        block_size=args.block_size,
        max_batch_size=max(128, engine_batch),
        max_num_tokens_per_batch=max(16384, engine_batch * 2),
        micro_batches=args.micro_batches or (1 if mode == "dp" else world),
        dtype=torch.bfloat16 if use_gpu else torch.float32,
        num_kv_blocks=None if use_gpu else 4096,
        enable_prefix_cache=args.shared_prefix > 0,
        kv_cache_dtype=args.kv_dtype,
        seed=0,
    )
    wdtype = args.weight_dtype
    if wdtype == "auto":
        wdtype = "fp8" if args.model in ("deepseek-v3", "kimi-k2") and use_gpu else "bf16"
    if wdtype == "fp8":
        eargs.moe_weight_dtype = "fp8"
        eargs.linear_weight_dtype = "fp8"   # dense GEMM route too (BASELINE)
    engine = Engine(cfg, eargs, comm=comm, random_weights=True)
    if use_gpu:
        # pre-tune hipBLASLt algo picks for the Ms this run will see:
        # prefill chunks, decode batch (graph bucket), per-chunk sampling rows
        chunk_m = min(eargs.max_num_tokens_per_batch,
                      engine_batch * args.prompt_len)
        sample_m = max(1, eargs.max_num_tokens_per_batch // max(1, args.prompt_len))
        engine.warmup_gemms([chunk_m, engine_batch, sample_m])

    # synthetic prompts, unique tokens so nothing prefix-shares (per-replica
    # seeds in DP mode so replicas do not share content)
    g = torch.Generator().manual_seed(1234 + 7 + comm_world.rank * 1000)
    sp = SamplingParams(
        temperature=1.0, top_p=1.0, top_k=-1,
        max_new_tokens=total_new_tokens, ignore_eos=True,
    )
    if comm.rank == 0:
        shared = torch.randint(
            0, cfg.vocab_size, (args.shared_prefix,), generator=g
        ).tolist() if args.shared_prefix else []
        for i in range(engine_batch):
            tail = max(1, args.prompt_len - len(shared))
            prompt = shared + torch.randint(
                0, cfg.vocab_size, (tail,), generator=g
            ).tolist()
            engine.submit(prompt, sp)

    def sync():
        if use_gpu:
            torch.cuda.synchronize()
        comm_world.barrier()

    # prefill everything (untimed; ends when all requests are decoding).
    # Per-request TTFT = submit -> first sampled token (all requests are
    # submitted at t0, so this is the saturated-arrival TTFT distribution).
    t_prefill0 = time.perf_counter()
    first_token_s = {}
    while True:
        for out in engine.step():
            if out.rid not in first_token_s and out.token_id >= 0:
                first_token_s[out.rid] = time.perf_counter() - t_prefill0
        running = engine.scheduler.running
        if running and all(r.prefill_done and r.num_output_tokens >= 1 for r in running.values()):
            break
        if not engine.has_work:
            raise RuntimeError("all requests finished during prefill phase?")
    sync()
    prefill_s = time.perf_counter() - t_prefill0
    ttfts = sorted(first_token_s.values())
    p50_ttft_ms = round(ttfts[len(ttfts) // 2] * 1000, 1) if ttfts else None

    for _ in range(args.warmup):
        engine.step()
    sync()

    t0 = time.perf_counter()
    for _ in range(args.steps):
        outs = engine.step()
        assert len(outs) == engine_batch, f"batch shrank: {len(outs)}"
    sync()
    elapsed = time.perf_counter() - t0

    # max over ranks
    if comm_world.world_size > 1:
        import torch.distributed as dist

        t = torch.tensor([elapsed], dtype=torch.float64)
        if comm_world.device.type == "cuda":
            t = t.to(comm_world.device)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    tokens = global_batch * args.steps
    value = tokens / elapsed
    if comm_world.rank == 0:
        result = {
            "metric": "output_tokens_per_sec",
            "value": round(value, 2),
            "unit": "tokens/s",
            "n_gpus": world if use_gpu else 0,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if use_gpu else "fp32",
            "data": "synthetic",
            "p50_ttft_ms": p50_ttft_ms,
            "config": {
                "model": model_name,
                "global_batch": global_batch,
                "seq_len": args.prompt_len,
                "parallelism": f"{mode if world > 1 else 'pp'}{world}"
                if world > 1 else "pp1",
                "kv_dtype": args.kv_dtype,
                "weight_dtype": wdtype,
                "shared_prefix": args.shared_prefix,
                "micro_batches": eargs.micro_batches,
                "prefill_s": round(prefill_s, 3),
                "p50_ttft_ms": p50_ttft_ms,
            },
        }
        print(json.dumps(result))


if __name__ == "__main__":
    main()
