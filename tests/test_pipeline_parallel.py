"""Multi-process pipeline-parallel correctness on CPU (gloo, world_size 2/3).

The same synthetic full-model weights are loaded layer-range-filtered on each
stage; greedy PP output must equal the single-process output. This is the
CPU-side guarantee that the RCCL multi-GPU path is correct by construction
(same code path, different backend/device)."""

import os

import pytest
import torch
import torch.multiprocessing as mp

from parallax_amd.models import get_model_class
from parallax_amd.models.config import ModelConfig


def tiny_cfg():
    return ModelConfig(
        architecture="LlamaForCausalLM", vocab_size=211, hidden_size=64,
        num_layers=4, num_heads=4, num_kv_heads=2, head_dim=16,
        intermediate_size=128, max_position_embeddings=256, eos_token_ids=[],
        rope_theta=10000.0,
    )


def full_state_dict(cfg, seed=42):
    """Deterministic synthetic HF-style full-model state dict."""
    g = torch.Generator().manual_seed(seed)
    sd = {}

    def rand(*shape):
        return torch.randn(*shape, generator=g) * 0.05

    sd["model.embed_tokens.weight"] = rand(cfg.vocab_size, cfg.hidden_size)
    for i in range(cfg.num_layers):
        p = f"model.layers.{i}."
        hd = cfg.num_heads * cfg.head_dim
        kvd = cfg.num_kv_heads * cfg.head_dim
        sd[p + "self_attn.q_proj.weight"] = rand(hd, cfg.hidden_size)
        sd[p + "self_attn.k_proj.weight"] = rand(kvd, cfg.hidden_size)
        sd[p + "self_attn.v_proj.weight"] = rand(kvd, cfg.hidden_size)
        sd[p + "self_attn.o_proj.weight"] = rand(cfg.hidden_size, hd)
        sd[p + "mlp.gate_proj.weight"] = rand(cfg.intermediate_size, cfg.hidden_size)
        sd[p + "mlp.up_proj.weight"] = rand(cfg.intermediate_size, cfg.hidden_size)
        sd[p + "mlp.down_proj.weight"] = rand(cfg.hidden_size, cfg.intermediate_size)
        sd[p + "input_layernorm.weight"] = torch.ones(cfg.hidden_size)
        sd[p + "post_attention_layernorm.weight"] = torch.ones(cfg.hidden_size)
    sd["model.norm.weight"] = torch.ones(cfg.hidden_size)
    sd["lm_head.weight"] = rand(cfg.vocab_size, cfg.hidden_size)
    return sd


PROMPTS = [[5, 9, 13, 2, 7, 100, 42], [3, 3, 3, 99], [1] * 18]


def run_single_process():
    from parallax_amd.parallel import comm as comm_mod
    from parallax_amd.server.engine import Engine, EngineArgs
    from parallax_amd.server.sampling_params import SamplingParams

    cfg = tiny_cfg()
    comm_mod._CTX = None
    ctx = comm_mod.CommContext(
        world_size=1, rank=0, pp_size=1, tp_size=1, pp_rank=0, tp_rank=0,
        device=torch.device("cpu"),
    )
    eng = Engine(cfg, EngineArgs(block_size=8, num_kv_blocks=128,
                                 dtype=torch.float32), comm=ctx)
    for name, t in full_state_dict(cfg).items():
        eng.model.load_hf_weight(name, t)
    sp = [SamplingParams(temperature=0.0, max_new_tokens=6, ignore_eos=True)] * len(PROMPTS)
    return list(eng.generate(PROMPTS, sp).values())


def _pp_worker(rank, world, port, out_file):
    os.environ.update(
        MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
        RANK=str(rank), WORLD_SIZE=str(world),
    )
    import torch as _t

    from parallax_amd.parallel.comm import init_distributed
    from parallax_amd.server.engine import Engine, EngineArgs
    from parallax_amd.server.sampling_params import SamplingParams

    comm = init_distributed(pp_size=world, tp_size=1, backend="gloo",
                            device=_t.device("cpu"))
    cfg = tiny_cfg()
    eng = Engine(cfg, EngineArgs(block_size=8, num_kv_blocks=128,
                                 dtype=_t.float32, micro_batches=2), comm=comm)
    for name, t in full_state_dict(cfg).items():
        eng.model.load_hf_weight(name, t)
    sp = [SamplingParams(temperature=0.0, max_new_tokens=6, ignore_eos=True)] * len(PROMPTS)
    outs = list(eng.generate(PROMPTS, sp).values())
    if rank == 0:
        _t.save(outs, out_file)
    import torch.distributed as dist

    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.parametrize("world", [2, 3])
def test_pp_matches_single_process(tmp_path, world):
    expected = run_single_process()
    out_file = str(tmp_path / f"pp{world}.pt")
    port = 29600 + world
    mp.spawn(_pp_worker, args=(world, port, out_file), nprocs=world, join=True)
    got = torch.load(out_file)
    assert got == expected


MANY_PROMPTS = [[5, 9, 13, 2, 7], [3, 3, 3, 99], [1] * 9, [7, 8], [41, 2, 2, 2, 6, 6]]


def _pp4_counter_worker(rank, world, port, out_file):
    os.environ.update(
        MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
        RANK=str(rank), WORLD_SIZE=str(world),
    )
    import torch as _t

    from parallax_amd.parallel.comm import init_distributed
    from parallax_amd.server.engine import Engine, EngineArgs
    from parallax_amd.server.sampling_params import SamplingParams

    comm = init_distributed(pp_size=world, tp_size=1, backend="gloo",
                            device=_t.device("cpu"))
    cfg = tiny_cfg()
    eng = Engine(cfg, EngineArgs(block_size=8, num_kv_blocks=128,
                                 dtype=_t.float32, micro_batches=4), comm=comm)
    for name, t in full_state_dict(cfg).items():
        eng.model.load_hf_weight(name, t)
    sp = SamplingParams(temperature=0.0, max_new_tokens=6, ignore_eos=True)
    if rank == 0:
        for p in MANY_PROMPTS:
            eng.submit(p, sp)
    outs = {}
    # fixed step count on every rank: all ranks run the same iterations
    for _ in range(12):
        for out in eng.step():
            if out.token_id >= 0:
                outs.setdefault(out.rid, []).append(out.token_id)
    if rank == 0:
        _t.save({"object_sync_count": eng.object_sync_count,
                 "outs": list(outs.values())}, out_file)
    import torch.distributed as dist

    dist.barrier()
    dist.destroy_process_group()


def test_pp4_decode_path_is_object_free(tmp_path):
    """steady-state PP decode must not pickle host objects —
    the only payload broadcast is the one carrying the initial adds."""
    from parallax_amd.parallel import comm as comm_mod
    from parallax_amd.server.engine import Engine, EngineArgs
    from parallax_amd.server.sampling_params import SamplingParams

    cfg = tiny_cfg()
    comm_mod._CTX = None
    ctx = comm_mod.CommContext(
        world_size=1, rank=0, pp_size=1, tp_size=1, pp_rank=0, tp_rank=0,
        device=torch.device("cpu"),
    )
    eng = Engine(cfg, EngineArgs(block_size=8, num_kv_blocks=128,
                                 dtype=torch.float32), comm=ctx)
    for name, t in full_state_dict(cfg).items():
        eng.model.load_hf_weight(name, t)
    sp = [SamplingParams(temperature=0.0, max_new_tokens=6, ignore_eos=True)] * len(MANY_PROMPTS)
    expected = sorted(eng.generate(MANY_PROMPTS, sp).values())

    out_file = str(tmp_path / "pp4.pt")
    mp.spawn(_pp4_counter_worker, args=(4, 29640, out_file), nprocs=4, join=True)
    got = torch.load(out_file)
    assert got["object_sync_count"] == 1  # the initial adds payload, nothing else
    assert sorted(got["outs"]) == expected


def _pp_mid_stream_worker(rank, world, port, out_file):
    """Rank 0 submits requests from a SEPARATE THREAD while the step loop
    runs (the serving ingress pattern): the snapshot drain must replicate
    exactly the serialized adds to the other ranks. Every rank runs the SAME
    fixed number of steps (idle steps are just the flag broadcast), so the
    ranks stay in lockstep regardless of when the submits land."""
    os.environ.update(
        MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
        RANK=str(rank), WORLD_SIZE=str(world),
    )
    import threading
    import time as _time

    import torch as _t

    from parallax_amd.parallel.comm import init_distributed
    from parallax_amd.server.engine import Engine, EngineArgs
    from parallax_amd.server.sampling_params import SamplingParams

    comm = init_distributed(pp_size=world, tp_size=1, backend="gloo",
                            device=_t.device("cpu"))
    cfg = tiny_cfg()
    eng = Engine(cfg, EngineArgs(block_size=8, num_kv_blocks=128,
                                 dtype=_t.float32), comm=comm)
    for name, t in full_state_dict(cfg).items():
        eng.model.load_hf_weight(name, t)
    sp = SamplingParams(temperature=0.0, max_new_tokens=4, ignore_eos=True)
    N, STEPS = 12, 300

    th = None
    if rank == 0:
        def submitter():
            for i in range(N):
                eng.submit(PROMPTS[i % len(PROMPTS)], sp, rid=f"m{i}")
                _time.sleep(0.002)  # land mid-step on purpose

        th = threading.Thread(target=submitter)
        th.start()
    counts = {}
    for _ in range(STEPS):
        for out in eng.step():
            if out.token_id >= 0:
                counts[out.rid] = counts.get(out.rid, 0) + 1
    if th is not None:
        th.join()
    if rank == 0:
        _t.save(counts, out_file)
    import torch.distributed as dist

    dist.barrier()
    dist.destroy_process_group()


def test_pp_mid_stream_submission(tmp_path):
    """Serving ingress under PP: threaded submits racing the step loop."""
    out_file = str(tmp_path / "mid_stream.pt")
    mp.spawn(_pp_mid_stream_worker, args=(2, 29655, out_file), nprocs=2,
             join=True)
    counts = torch.load(out_file)
    assert len(counts) == 12 and all(v == 4 for v in counts.values())
