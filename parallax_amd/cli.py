"""parallax_amd CLI: run / join / serve / chat (reference analogue: cli.py).

  run    — start the scheduler service (control plane + gateway)
  join   — start a worker node agent that joins a scheduler
  serve  — single-host serving: engine + OpenAI frontend in one process
           (PP over local GPUs via torch.distributed.run when --gpus > 1)
  chat   — minimal terminal chat client against any OpenAI endpoint
"""

from __future__ import annotations

import argparse
import json
import os

from .utils.logging_config import get_logger

logger = get_logger("cli")


def cmd_run(args) -> None:
    import uvicorn

    from .utils.banner import print_banner

    print_banner("scheduler")

    from .backend.service import SchedulerService, create_backend_app

    svc = SchedulerService()
    if args.model_path:
        from .models.config import ModelConfig

        with open(os.path.join(args.model_path, "config.json")) as f:
            hf_cfg = json.load(f)
        svc.init_model(args.model_name or args.model_path, hf_cfg,
                       min_nodes=args.min_nodes, allocator=args.allocator,
                       routing=args.routing)
    app = create_backend_app(svc)
    uvicorn.run(app, host=args.host, port=args.port)


def cmd_join(args) -> None:
    import threading

    import torch
    import uvicorn

    from .p2p.head_frontend import create_head_app
    from .p2p.node_agent import NodeAgent
    from .server.tokenizer_util import TokenizerWrapper

    device = torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")
    dtype = torch.bfloat16 if device.type == "cuda" else torch.float32
    agent = NodeAgent(
        args.scheduler_url, node_id=args.node_id, host=args.host,
        http_port=args.port, model_path=args.model_path,
        random_weights=args.model_path is None, device=device, dtype=dtype,
        num_kv_blocks=args.num_kv_blocks, block_size=args.block_size,
    )
    agent.join(timeout_s=args.join_timeout)
    agent.start()
    logger.info("node %s serving layers [%d,%d)", agent.node_id,
                agent.assignment["start_layer"], agent.assignment["end_layer"])
    if agent.assignment["start_layer"] == 0:
        tok = TokenizerWrapper(args.model_path,
                               vocab_size=agent.cfg.vocab_size)
        app = create_head_app(agent, tok, args.model_name or "model")
        uvicorn.run(app, host=args.host, port=args.port)
    else:
        threading.Event().wait()  # worker runs in background threads


def cmd_serve(args) -> None:
    import torch
    import uvicorn

    from .utils.banner import print_banner

    if int(os.environ.get("RANK", "0")) == 0:
        print_banner("serve")

    from .models.config import ModelConfig
    from .parallel.comm import init_distributed
    from .server.engine import Engine, EngineArgs
    from .server.engine_server import EngineServer
    from .server.http_frontend import create_app
    from .server.tokenizer_util import TokenizerWrapper

    world = int(os.environ.get("WORLD_SIZE", "1"))
    tp = max(1, getattr(args, "tp_size", 1))
    if world % tp != 0:
        raise SystemExit(f"--tp-size {tp} must divide the launched world "
                         f"size {world} (torchrun --nproc-per-node)")
    comm = init_distributed(pp_size=world // tp, tp_size=tp)
    if args.model_path:
        cfg = ModelConfig.from_pretrained(args.model_path)
    else:
        from bench import MODELS  # synthetic configs

        cfg = ModelConfig.from_hf_config(MODELS[args.model][0]())
    eargs = engine_args_from_cli(args, world=world)
    engine = Engine(cfg, eargs, comm=comm, model_path=args.model_path,
                    random_weights=args.model_path is None,
                    lora_path=getattr(args, "lora_path", None))
    if torch.cuda.is_available():
        engine.warmup_serving()
    tok = TokenizerWrapper(args.model_path, vocab_size=cfg.vocab_size)
    if world > 1:
        # constrained decoding masks logits on the SAMPLING rank (the last
        # pipeline stage), so every rank needs the id->text table up front
        engine.set_grammar_vocab(tok.vocab_strings())
    if comm.rank == 0:
        server = EngineServer(engine)
        server.start()
        app = create_app(server, tok, args.model_name or args.model or "model")
        uvicorn.run(app, host=args.host, port=args.port)
    else:
        # non-head ranks run the SPMD step loop; step() blocks in the ingress
        # broadcast until rank 0 steps, so this does not spin while idle
        while True:
            engine.step()


def cmd_chat(args) -> None:
    import json as _json

    import httpx

    url = args.url.rstrip("/") + "/v1/chat/completions"
    history = []
    print(f"chatting with {url} (ctrl-d to exit)")
    while True:
        try:
            user = input("you> ")
        except EOFError:
            break
        history.append({"role": "user", "content": user})
        print("assistant> ", end="", flush=True)
        text = ""
        with httpx.stream("POST", url, json={
            "model": "default", "messages": history,
            "max_tokens": args.max_tokens, "stream": True,
        }, timeout=600) as r:
            for line in r.iter_lines():
                if not line.startswith("data: ") or line == "data: [DONE]":
                    continue
                msg = _json.loads(line[6:])
                for c in msg.get("choices", []):
                    delta = (c.get("delta") or {}).get("content") or ""
                    if delta:
                        text += delta
                        print(delta, end="", flush=True)
        print()
        history.append({"role": "assistant", "content": text})


def add_engine_args(p: argparse.ArgumentParser) -> None:
    """Worker/engine flag surface (reference server_args.py breadth): every
    EngineArgs knob is reachable from the CLI."""
    p.add_argument("--block-size", type=int, default=32,
                   help="KV page size in tokens")
    p.add_argument("--max-batch-size", type=int, default=128)
    p.add_argument("--max-num-tokens-per-batch", type=int, default=16384)
    p.add_argument("--prefill-chunk-size", type=int, default=8192)
    p.add_argument("--max-model-len", type=int, default=8192,
                   help="context ceiling (sizes graph buffers)")
    p.add_argument("--cache-memory-fraction", type=float, default=0.80,
                   help="fraction of free HBM for the KV cache")
    p.add_argument("--num-kv-blocks", type=int, default=None,
                   help="explicit KV block count (overrides the fraction)")
    p.add_argument("--kv-cache-dtype", default="auto",
                   choices=["auto", "fp8"], help="fp8 = e4m3 KV storage")
    p.add_argument("--moe-weight-dtype", default="auto",
                   choices=["auto", "fp8"], help="fp8 = W8A8 expert weights")
    p.add_argument("--linear-weight-dtype", default="auto",
                   choices=["auto", "fp8"],
                   help="fp8 = W8A8 dense GEMMs (lm_head stays full precision)")
    p.add_argument("--dtype", default="auto",
                   choices=["auto", "bfloat16", "float16", "float32"])
    p.add_argument("--micro-batches", type=int, default=0,
                   help="decode micro-batches in flight (0 = one per stage)")
    p.add_argument("--disable-prefix-cache", action="store_true",
                   help="turn off the block-radix prefix cache")
    p.add_argument("--disable-graphs", action="store_true",
                   help="turn off hipGraph-captured decode")
    p.add_argument("--request-timeout", type=float, default=600.0,
                   help="per-request abort timeout (s)")
    p.add_argument("--seed", type=int, default=0, help="sampling seed")
    p.add_argument("--enable-routing-stats", action="store_true",
                   help="per-expert MoE routing counters (GET /stats)")
    p.add_argument("--decode-priority", action="store_true",
                   help="form decode batches before prefills (lower TPOT "
                        "under arrival bursts at the cost of TTFT)")
    p.add_argument("--tp-size", type=int, default=1,
                   help="tensor-parallel degree; ranks = tp x pp "
                        "(torchrun --nproc-per-node must equal tp*pp)")
    p.add_argument("--start-layer", type=int, default=None,
                   help="explicit layer range start (decentralized mode)")
    p.add_argument("--end-layer", type=int, default=None)
    p.add_argument("--lora-path", default=None,
                   help="LoRA adapter directory fused at load time")
    p.add_argument("--log-level", default=None,
                   help="logging level (DEBUG/INFO/WARNING)")


def engine_args_from_cli(args, world: int = 1):
    import torch

    from .server.engine import EngineArgs

    if args.log_level:
        import logging

        logging.getLogger("parallax_amd").setLevel(args.log_level.upper())
    use_gpu = torch.cuda.is_available()
    if args.dtype == "auto":
        dtype = torch.bfloat16 if use_gpu else torch.float32
    else:
        dtype = getattr(torch, args.dtype)
    return EngineArgs(
        block_size=args.block_size,
        max_batch_size=args.max_batch_size,
        max_num_tokens_per_batch=args.max_num_tokens_per_batch,
        prefill_chunk_size=args.prefill_chunk_size,
        cache_memory_fraction=args.cache_memory_fraction,
        num_kv_blocks=args.num_kv_blocks if args.num_kv_blocks
        else (None if use_gpu else 4096),
        micro_batches=args.micro_batches or world,
        enable_prefix_cache=not args.disable_prefix_cache,
        dtype=dtype,
        seed=args.seed,
        request_timeout_s=args.request_timeout,
        start_layer=args.start_layer,
        end_layer=args.end_layer,
        max_model_len=args.max_model_len,
        enable_graphs=not args.disable_graphs,
        kv_cache_dtype=args.kv_cache_dtype,
        enable_routing_stats=args.enable_routing_stats,
        prefill_priority=not args.decode_priority,
        moe_weight_dtype=args.moe_weight_dtype,
        linear_weight_dtype=args.linear_weight_dtype,
    )


def main(argv=None) -> None:
    p = argparse.ArgumentParser(prog="parallax_amd")
    sub = p.add_subparsers(dest="cmd", required=True)

    pr = sub.add_parser("run", help="start the scheduler service")
    pr.add_argument("--host", default="0.0.0.0")
    pr.add_argument("--port", type=int, default=3001)
    pr.add_argument("--model-path", default=None)
    pr.add_argument("--model-name", default=None)
    pr.add_argument("--min-nodes", type=int, default=1)
    pr.add_argument("--allocator", default="greedy",
                    choices=["greedy", "dp"],
                    help="layer allocator (greedy look-ahead or memoized DP)")
    pr.add_argument("--routing", default="round_robin",
                    choices=["round_robin", "random", "latency"],
                    help="request routing across pipelines")
    pr.set_defaults(fn=cmd_run)

    pj = sub.add_parser("join", help="join a scheduler as a worker node")
    pj.add_argument("--scheduler-url", required=True)
    pj.add_argument("--node-id", default=None)
    pj.add_argument("--host", default="127.0.0.1")
    pj.add_argument("--port", type=int, default=3010)
    pj.add_argument("--model-path", default=None)
    pj.add_argument("--model-name", default=None)
    pj.add_argument("--num-kv-blocks", type=int, default=4096)
    pj.add_argument("--block-size", type=int, default=32)
    pj.add_argument("--join-timeout", type=float, default=600.0)
    pj.set_defaults(fn=cmd_join)

    ps = sub.add_parser("serve", help="single-host OpenAI server")
    ps.add_argument("--host", default="0.0.0.0")
    ps.add_argument("--port", type=int, default=3000)
    ps.add_argument("--model-path", default=None)
    ps.add_argument("--model", default="deepseek-r1-distill-llama-8b")
    ps.add_argument("--model-name", default=None)
    add_engine_args(ps)
    ps.set_defaults(fn=cmd_serve)

    pc = sub.add_parser("chat", help="terminal chat client")
    pc.add_argument("--url", default="http://127.0.0.1:3000")
    pc.add_argument("--max-tokens", type=int, default=256)
    pc.set_defaults(fn=cmd_chat)

    args = p.parse_args(argv)
    args.fn(args)


if __name__ == "__main__":
    main()
