"""Tensor-parallel correctness on CPU (gloo, tp_size=2): greedy output of a
TP=2 engine matches the single-process engine bit-for-bit (all-reduce in
row-parallel layers, fused-QKV sharding, lm_head gather)."""

import os

import pytest
import torch
import torch.multiprocessing as mp

from tests.test_pipeline_parallel import PROMPTS, full_state_dict, run_single_process, tiny_cfg


def _tp_worker(rank, world, port, out_file):
    os.environ.update(
        MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
        RANK=str(rank), WORLD_SIZE=str(world),
    )
    import torch as _t

    from parallax_amd.parallel.comm import init_distributed
    from parallax_amd.server.engine import Engine, EngineArgs
    from parallax_amd.server.sampling_params import SamplingParams

    comm = init_distributed(pp_size=1, tp_size=world, backend="gloo",
                            device=_t.device("cpu"))
    cfg = tiny_cfg()
    eng = Engine(cfg, EngineArgs(block_size=8, num_kv_blocks=128,
                                 dtype=_t.float32), comm=comm)
    for name, t in full_state_dict(cfg).items():
        eng.model.load_hf_weight(name, t)
    sp = [SamplingParams(temperature=0.0, max_new_tokens=6, ignore_eos=True)] * len(PROMPTS)
    outs = list(eng.generate(PROMPTS, sp).values())
    if rank == 0:
        _t.save(outs, out_file)
    import torch.distributed as dist

    dist.barrier()
    dist.destroy_process_group()


def test_tp2_matches_single(tmp_path):
    expected = run_single_process()
    out_file = str(tmp_path / "tp2.pt")
    mp.spawn(_tp_worker, args=(2, 29711, out_file), nprocs=2, join=True)
    got = torch.load(out_file)
    assert got == expected


def _pptp_worker(rank, world, port, out_file):
    os.environ.update(
        MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
        RANK=str(rank), WORLD_SIZE=str(world),
    )
    import torch as _t

    from parallax_amd.parallel.comm import init_distributed
    from parallax_amd.server.engine import Engine, EngineArgs
    from parallax_amd.server.sampling_params import SamplingParams

    comm = init_distributed(pp_size=2, tp_size=2, backend="gloo",
                            device=_t.device("cpu"))
    cfg = tiny_cfg()
    eng = Engine(cfg, EngineArgs(block_size=8, num_kv_blocks=128,
                                 dtype=_t.float32, micro_batches=1), comm=comm)
    for name, t in full_state_dict(cfg).items():
        eng.model.load_hf_weight(name, t)
    sp = [SamplingParams(temperature=0.0, max_new_tokens=6, ignore_eos=True)] * len(PROMPTS)
    outs = list(eng.generate(PROMPTS, sp).values())
    if rank == 0:
        _t.save(outs, out_file)
    import torch.distributed as dist

    dist.barrier()
    dist.destroy_process_group()


def test_pp2_tp2_matches_single(tmp_path):
    """Combined 2-stage pipeline x 2-way tensor parallel (world=4, gloo):
    hidden states cross stages per tp-rank; row-parallel all-reduce within
    each stage; output must match the single-process engine."""
    expected = run_single_process()
    out_file = str(tmp_path / "pptp.pt")
    mp.spawn(_pptp_worker, args=(4, 29717, out_file), nprocs=4, join=True)
    got = torch.load(out_file)
    assert got == expected
