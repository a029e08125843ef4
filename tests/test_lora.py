"""LoRA adapter fusion at load (PEFT format): fused weights == base + B@A."""

import json
import os

import pytest
import torch

safetensors = pytest.importorskip("safetensors")
from safetensors.torch import save_file

from parallax_amd.models import get_model_class
from parallax_amd.models.config import ModelConfig
from parallax_amd.server.shard_loader import fuse_lora


def test_fuse_lora(tmp_path):
    torch.manual_seed(5)
    cfg = ModelConfig(
        architecture="LlamaForCausalLM", vocab_size=64, hidden_size=32,
        num_layers=2, num_heads=2, num_kv_heads=2, head_dim=16,
        intermediate_size=64, max_position_embeddings=128,
    )
    m = get_model_class(cfg.architecture)(cfg)
    m.init_random()
    q_before = m.layers[0].self_attn.qkv_proj.weight.data[:32].clone().float()
    o_before = m.layers[1].self_attn.o_proj.weight.data.clone().float()

    r = 4
    a_q = torch.randn(r, 32) * 0.1
    b_q = torch.randn(32, r) * 0.1
    a_o = torch.randn(r, 32) * 0.1
    b_o = torch.randn(32, r) * 0.1
    lora_dir = tmp_path / "adapter"
    os.makedirs(lora_dir)
    with open(lora_dir / "adapter_config.json", "w") as f:
        json.dump({"r": r, "lora_alpha": 8}, f)
    save_file({
        "base_model.model.model.layers.0.self_attn.q_proj.lora_A.weight": a_q,
        "base_model.model.model.layers.0.self_attn.q_proj.lora_B.weight": b_q,
        "base_model.model.model.layers.1.self_attn.o_proj.lora_A.weight": a_o,
        "base_model.model.model.layers.1.self_attn.o_proj.lora_B.weight": b_o,
    }, str(lora_dir / "adapter_model.safetensors"))

    n = fuse_lora(m, str(lora_dir))
    assert n == 2
    scale = 8 / r
    q_after = m.layers[0].self_attn.qkv_proj.weight.data[:32].float()
    torch.testing.assert_close(q_after, q_before + scale * (b_q @ a_q),
                               atol=2e-2, rtol=2e-2)
    o_after = m.layers[1].self_attn.o_proj.weight.data.float()
    torch.testing.assert_close(o_after, o_before + scale * (b_o @ a_o),
                               atol=2e-2, rtol=2e-2)
