"""Expert parallelism (gloo, tp_size=2): a TP=2 engine shards the MoE experts
across the group (E/2 per rank, partial outputs all-reduced) and must
reproduce the single-process greedy output bit-for-bit. Uses qwen3_moe
(softmax router, 4 experts -> 2 per rank)."""

import os

import pytest
import torch
import torch.multiprocessing as mp

transformers = pytest.importorskip("transformers")

PROMPTS = [[7, 42, 99, 5, 81, 23, 115, 3], [9, 8, 7, 6, 5]]


def _make_cfg_and_sd(tmpdir):
    torch.manual_seed(51)
    hf_cfg = transformers.Qwen3MoeConfig(
        vocab_size=128, hidden_size=64, intermediate_size=128,
        moe_intermediate_size=32, num_hidden_layers=2, num_attention_heads=4,
        num_key_value_heads=2, head_dim=16, num_experts=4,
        num_experts_per_tok=2, norm_topk_prob=True, decoder_sparse_step=1,
        mlp_only_layers=[], max_position_embeddings=512,
        tie_word_embeddings=False, rope_theta=10000.0,
    )
    hf = transformers.Qwen3MoeForCausalLM(hf_cfg).eval()
    sd_path = os.path.join(tmpdir, "sd.pt")
    torch.save(hf.state_dict(), sd_path)
    cfg_dict = hf_cfg.to_dict() | {"architectures": ["Qwen3MoeForCausalLM"]}
    return cfg_dict, sd_path


def _run_single(cfg_dict, sd_path):
    from parallax_amd.models import get_model_class
    from parallax_amd.models.config import ModelConfig
    from parallax_amd.server.engine import Engine, EngineArgs
    from parallax_amd.server.sampling_params import SamplingParams

    cfg = ModelConfig.from_hf_config(cfg_dict)
    eng = Engine(cfg, EngineArgs(block_size=8, num_kv_blocks=128,
                                 dtype=torch.float32))
    m = get_model_class(cfg.architecture)(cfg).eval()
    for name, t in torch.load(sd_path).items():
        m.load_hf_weight(name, t)
    eng.model = m.float()
    sp = [SamplingParams(temperature=0.0, max_new_tokens=5, ignore_eos=True)] * 2
    return list(eng.generate(PROMPTS, sp).values())


def _ep_worker(rank, world, port, cfg_dict, sd_path, out_file):
    os.environ.update(
        MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
        RANK=str(rank), WORLD_SIZE=str(world),
    )
    import torch as _t

    from parallax_amd.models import get_model_class
    from parallax_amd.models.config import ModelConfig
    from parallax_amd.parallel.comm import init_distributed
    from parallax_amd.server.engine import Engine, EngineArgs
    from parallax_amd.server.sampling_params import SamplingParams

    comm = init_distributed(pp_size=1, tp_size=world, backend="gloo",
                            device=_t.device("cpu"))
    cfg = ModelConfig.from_hf_config(cfg_dict)
    eng = Engine(cfg, EngineArgs(block_size=8, num_kv_blocks=128,
                                 dtype=_t.float32), comm=comm)
    # experts must actually be sharded
    for layer in eng.model.layers:
        assert layer.mlp.experts.ep_size == world
        assert layer.mlp.experts.num_local_experts == 4 // world
    for name, t in _t.load(sd_path).items():
        eng.model.load_hf_weight(name, t)
    sp = [SamplingParams(temperature=0.0, max_new_tokens=5, ignore_eos=True)] * 2
    outs = list(eng.generate(PROMPTS, sp).values())
    if rank == 0:
        _t.save(outs, out_file)
    import torch.distributed as dist

    dist.barrier()
    dist.destroy_process_group()


def test_ep2_matches_single(tmp_path):
    cfg_dict, sd_path = _make_cfg_and_sd(str(tmp_path))
    expected = _run_single(cfg_dict, sd_path)
    out_file = str(tmp_path / "ep2.pt")
    mp.spawn(_ep_worker, args=(2, 29721, cfg_dict, sd_path, out_file),
             nprocs=2, join=True)
    got = torch.load(out_file)
    assert got == expected


def test_ep_loader_helpers_route_local_slice():
    """FusedMoE EP loader helpers: per-expert and fused loads only touch the
    local expert slice (unit-level, fake 2-way comm)."""
    from parallax_amd.parallel import comm as comm_mod
    from parallax_amd.models.config import ModelConfig
    from parallax_amd.models.moe import FusedMoE

    cfg = ModelConfig(
        architecture="Qwen3MoeForCausalLM", model_type="qwen3_moe",
        vocab_size=64, hidden_size=16, num_layers=1, num_heads=2,
        num_kv_heads=2, head_dim=8, intermediate_size=32,
        moe_intermediate_size=8, num_experts=4, num_experts_per_tok=2,
    )
    saved = comm_mod._CTX
    try:
        comm_mod._CTX = comm_mod.CommContext(
            world_size=2, rank=1, pp_size=1, tp_size=2, pp_rank=0, tp_rank=1,
            device=torch.device("cpu"))
        moe = FusedMoE(cfg)
        assert moe.ep_size == 2 and moe.ep_rank == 1
        assert moe.num_local_experts == 2 and moe.expert_offset == 2
        assert moe.w_gate_up.shape[0] == 2
        # expert 0/1 are foreign: loads are no-ops; 2/3 land at local 0/1
        g = torch.randn(8, 16)
        moe.w_gate_up.data.zero_()
        moe.load_expert_gate(0, g)
        assert moe.w_gate_up.abs().sum() == 0
        moe.load_expert_gate(3, g)
        assert torch.equal(moe.w_gate_up.data[1, :8], g)
        full = torch.randn(4, 16, 8)
        moe.load_fused_down(full)
        assert torch.equal(moe.w_down.data, full[2:4])
    finally:
        comm_mod._CTX = saved
