"""GPU end-to-end smokes for the sparse/hybrid model families: tiny
random-init engines in bf16 run greedy decode through the HIP kernel paths
(DSA sparse MLA decode, MSA per-KV-head block-sparse decode, lightning/
DeltaNet recurrences, clamped grouped MoE) and must be deterministic."""

import pytest
import torch

from parallax_amd.models.config import ModelConfig
from parallax_amd.server.engine import Engine, EngineArgs
from parallax_amd.server.sampling_params import SamplingParams

pytestmark = pytest.mark.gpu

PROMPTS = [[1, 2, 3, 4, 5, 6, 7, 8, 9, 10, 11, 12], [9, 8, 7, 6, 5]]


def _run_twice(cfg, n_new=6, **eargs):
    def run():
        eng = Engine(cfg, EngineArgs(num_kv_blocks=128, max_batch_size=8,
                                     **eargs), random_weights=True)
        return eng.generate(
            PROMPTS,
            [SamplingParams(temperature=0.0, max_new_tokens=n_new,
                            ignore_eos=True)] * len(PROMPTS),
        )
    o1, o2 = run(), run()
    assert [len(v) for v in o1.values()] == [n_new] * len(PROMPTS)
    assert list(o1.values()) == list(o2.values())
    return o1


def test_deepseek_v32_engine_gpu():
    cfg = ModelConfig.from_hf_config({
        "architectures": ["DeepseekV32ForCausalLM"],
        "model_type": "deepseek_v32", "vocab_size": 512, "hidden_size": 256,
        "num_hidden_layers": 2, "num_attention_heads": 8,
        "num_key_value_heads": 8, "intermediate_size": 512,
        "moe_intermediate_size": 128, "n_routed_experts": 8,
        "num_experts_per_tok": 2, "n_shared_experts": 1, "n_group": 2,
        "topk_group": 1, "first_k_dense_replace": 1, "q_lora_rank": 96,
        # the MLA HIP kernel is compiled for the production dims (R=512 DR=64)
        "kv_lora_rank": 512, "qk_nope_head_dim": 64, "qk_rope_head_dim": 64,
        "v_head_dim": 64, "index_n_heads": 4, "index_head_dim": 64,
        "index_topk": 8, "max_position_embeddings": 2048,
        "rope_theta": 10000.0, "eos_token_id": None,
    })
    assert cfg.is_dsa
    _run_twice(cfg)


def test_minimax_m3_engine_gpu():
    cfg = ModelConfig.from_hf_config({
        "architectures": ["MiniMaxM3ForCausalLM"],
        "model_type": "minimax_m3", "vocab_size": 512, "hidden_size": 256,
        "num_hidden_layers": 2, "num_attention_heads": 8,
        "num_key_value_heads": 2, "head_dim": 64, "intermediate_size": 128,
        "dense_intermediate_size": 256, "shared_intermediate_size": 128,
        "num_local_experts": 8, "num_experts_per_tok": 2,
        "routed_scaling_factor": 1.5,
        "layer_types": ["minimax_m3_sparse", "full_attention"],
        "mlp_layer_types": ["sparse", "dense"],
        "index_n_heads": 2, "index_head_dim": 64, "index_block_size": 4,
        "index_topk_blocks": 2, "index_local_blocks": 1,
        "max_position_embeddings": 2048, "eos_token_id": None,
        "rope_parameters": {"rope_type": "default", "rope_theta": 10000.0},
    })
    assert cfg.is_msa
    _run_twice(cfg)


def test_minimax_lightning_engine_gpu():
    cfg = ModelConfig.from_hf_config({
        "architectures": ["MiniMaxForCausalLM"],
        "model_type": "minimax", "vocab_size": 512, "hidden_size": 256,
        "num_hidden_layers": 2, "num_attention_heads": 4,
        "num_key_value_heads": 2, "intermediate_size": 256,
        "num_local_experts": 4, "num_experts_per_tok": 2, "block_size": 8,
        "layer_types": ["linear_attention", "full_attention"],
        "max_position_embeddings": 2048, "eos_token_id": None,
        "rope_parameters": {"rope_type": "default", "rope_theta": 10000.0},
    })
    assert cfg.has_linear_layers
    _run_twice(cfg)


def test_qwen3_5_engine_gpu():
    cfg = ModelConfig.from_hf_config({
        "architectures": ["Qwen3_5ForCausalLM"],
        "model_type": "qwen3_5_text", "vocab_size": 512, "hidden_size": 256,
        "num_hidden_layers": 2, "num_attention_heads": 4,
        "num_key_value_heads": 2, "head_dim": 64, "intermediate_size": 512,
        "linear_num_key_heads": 2, "linear_num_value_heads": 4,
        "linear_key_head_dim": 32, "linear_value_head_dim": 32,
        "linear_conv_kernel_dim": 3,
        "layer_types": ["linear_attention", "full_attention"],
        "partial_rotary_factor": 0.25, "max_position_embeddings": 2048,
        "rope_theta": 10000.0, "eos_token_id": None,
    })
    assert cfg.has_linear_layers and cfg.qk_norm
    _run_twice(cfg)


def test_minimax_m2_engine_gpu():
    cfg = ModelConfig.from_hf_config({
        "architectures": ["MiniMaxM2ForCausalLM"],
        "model_type": "minimax_m2", "vocab_size": 512, "hidden_size": 256,
        "num_hidden_layers": 2, "num_attention_heads": 8,
        "num_key_value_heads": 2, "head_dim": 64, "intermediate_size": 128,
        "num_local_experts": 8, "num_experts_per_tok": 2,
        "max_position_embeddings": 2048, "rope_theta": 10000.0,
        "eos_token_id": None,
    })
    assert cfg.qk_norm_full
    _run_twice(cfg)


def test_constrained_decoding_gpu():
    """json_schema masks compose with the fused gumbel sampler on GPU."""
    import json

    from parallax_amd.models.config import ModelConfig
    from parallax_amd.server.engine import Engine, EngineArgs
    from parallax_amd.server.sampling_params import SamplingParams
    from parallax_amd.server.tokenizer_util import TokenizerWrapper

    tok = TokenizerWrapper(vocab_size=512)
    vocab = tok.vocab_strings()
    cfg = ModelConfig(
        # head_dim 64: the HIP prefill kernel supports head_dim 64/128 only
        architecture="LlamaForCausalLM", vocab_size=512, hidden_size=256,
        num_layers=2, num_heads=4, num_kv_heads=2, head_dim=64,
        intermediate_size=256, max_position_embeddings=512,
        eos_token_ids=[tok.eos_token_id],
    )
    eng = Engine(cfg, EngineArgs(block_size=16, num_kv_blocks=64),
                 random_weights=True)
    eng.set_grammar_vocab(vocab)
    schema = json.dumps({"type": "object", "properties": {
        "a": {"type": "integer"}, "b": {"type": "boolean"}}})
    sp = SamplingParams(temperature=1.0, max_new_tokens=150,
                        json_schema=schema)
    out = eng.generate([[5, 9, 13]], [sp])
    toks = list(out.values())[0]
    text = "".join(
        vocab[t] for t in toks if t != tok.eos_token_id and t < len(vocab)
    )
    obj = json.loads(text)
    assert isinstance(obj.get("a"), int) and isinstance(obj.get("b"), bool)
