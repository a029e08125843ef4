"""Model configuration parsed from a HF config.json dict.

Covers the fields used by the supported families (Llama/Qwen2/Qwen3/Qwen3-MoE/
DeepSeek-V3/GPT-OSS/GLM4-MoE) plus layer-type derivation for hybrid stacks —
parity with the reference's utils/layer_types.py:5-20 and config normalization
(utils/utils.py:343)."""

from __future__ import annotations

import json
import os
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional


@dataclass
class ModelConfig:
    architecture: str = "LlamaForCausalLM"
    model_type: str = "llama"
    vocab_size: int = 32000
    hidden_size: int = 4096
    num_layers: int = 32
    num_heads: int = 32
    num_kv_heads: int = 8
    head_dim: int = 128
    intermediate_size: int = 14336
    rms_norm_eps: float = 1e-5
    rope_theta: float = 500000.0
    rope_scaling: Optional[Dict[str, Any]] = None
    max_position_embeddings: int = 131072
    tie_word_embeddings: bool = False
    attention_bias: bool = False
    o_proj_bias: bool = False
    mlp_bias: bool = False
    qk_norm: bool = False            # qwen3-style per-head q/k RMSNorm
    qk_norm_full: bool = False       # minimax-m2: RMSNorm over the FULL q/k width
    partial_rotary_factor: float = 1.0
    sliding_window: int = -1         # -1 = full attention
    # which layers use the sliding window ("full" layers interleave, gpt-oss)
    layer_types: Optional[List[str]] = None
    eos_token_ids: List[int] = field(default_factory=lambda: [2])
    bos_token_id: Optional[int] = 1
    torch_dtype: str = "bfloat16"

    # -- MoE ------------------------------------------------------------------
    num_experts: int = 0
    num_experts_per_tok: int = 0
    moe_intermediate_size: int = 0
    num_shared_experts: int = 0
    shared_expert_intermediate_size: int = 0
    first_k_dense_layers: int = 0     # deepseek: dense layers before MoE starts
    moe_layer_freq: int = 1
    norm_topk_prob: bool = True
    routed_scaling_factor: float = 1.0
    scoring_func: str = "softmax"     # or "sigmoid" (deepseek v3)
    topk_group: int = 0
    n_group: int = 0
    moe_router_bias: bool = False     # deepseek v3 e_score_correction_bias

    # -- MLA (deepseek) ----------------------------------------------------------
    rope_interleave: bool = False    # GPT-J pairwise rope (deepseek default)
    q_lora_rank: int = 0
    kv_lora_rank: int = 0
    qk_nope_head_dim: int = 0
    qk_rope_head_dim: int = 0
    v_head_dim: int = 0

    # -- DSA indexer (deepseek v3.2 sparse attention) ---------------------------
    index_n_heads: int = 0
    index_head_dim: int = 0
    index_topk: int = 0

    # -- MSA indexer (minimax-m3 block-sparse attention) ------------------------
    index_block_size: int = 0
    index_topk_blocks: int = 0
    index_local_blocks: int = 0

    # -- attention sinks (gpt-oss) --------------------------------------------------
    attention_sinks: bool = False
    use_attn_gate: bool = False    # step3p5 head-wise sigmoid output gate

    # -- hybrid linear attention (qwen3-next gated deltanet) --------------------------
    linear_num_key_heads: int = 0
    linear_num_value_heads: int = 0
    linear_key_head_dim: int = 0
    linear_value_head_dim: int = 0
    linear_conv_kernel_dim: int = 4

    raw: Dict[str, Any] = field(default_factory=dict, repr=False)

    @property
    def rot_dim(self) -> int:
        return int(self.head_dim * self.partial_rotary_factor)

    @property
    def is_moe(self) -> bool:
        return self.num_experts > 0

    @property
    def is_mla(self) -> bool:
        return self.kv_lora_rank > 0

    @property
    def is_dsa(self) -> bool:
        return self.index_topk > 0 and self.index_head_dim > 0

    @property
    def is_msa(self) -> bool:
        return self.index_topk_blocks > 0 and self.index_block_size > 0

    @property
    def has_linear_layers(self) -> bool:
        return bool(self.layer_types) and "linear_attention" in self.layer_types

    def layer_type(self, layer_idx: int) -> str:
        """'attention' | 'sliding_attention' | 'linear_attention' | 'mla_attention'."""
        if self.is_mla:
            return "mla_attention"
        if self.layer_types is not None:
            return self.layer_types[layer_idx]
        if self.sliding_window > 0:
            return "sliding_attention"
        return "attention"

    def is_moe_layer(self, layer_idx: int) -> bool:
        if not self.is_moe:
            return False
        if layer_idx < self.first_k_dense_layers:
            return False
        return (layer_idx % max(1, self.moe_layer_freq)) == 0

    @classmethod
    def from_hf_config(cls, cfg: Dict[str, Any]) -> "ModelConfig":
        arch = (cfg.get("architectures") or ["LlamaForCausalLM"])[0]
        num_heads = cfg.get("num_attention_heads", 32)
        hidden = cfg.get("hidden_size", 4096)
        head_dim = cfg.get("head_dim") or hidden // num_heads
        rope_params = cfg.get("rope_parameters") or {}
        eos = cfg.get("eos_token_id", 2)
        eos_ids = eos if isinstance(eos, list) else [eos]
        sliding = cfg.get("sliding_window") or -1
        if cfg.get("use_sliding_window") is False:
            sliding = -1
        layer_types = cfg.get("layer_types")
        mc = cls(
            architecture=arch,
            model_type=cfg.get("model_type", "llama"),
            vocab_size=cfg.get("vocab_size", 32000),
            hidden_size=hidden,
            num_layers=cfg.get("num_hidden_layers", 32),
            num_heads=num_heads,
            num_kv_heads=cfg.get("num_key_value_heads", num_heads),
            head_dim=head_dim,
            intermediate_size=cfg.get("intermediate_size", 4 * hidden),
            rms_norm_eps=cfg.get("rms_norm_eps", 1e-5),
            rope_theta=cfg.get("rope_theta") or rope_params.get("rope_theta", 10000.0),
            rope_scaling=cfg.get("rope_scaling") or (rope_params or None),
            partial_rotary_factor=cfg.get("partial_rotary_factor")
            or rope_params.get("partial_rotary_factor", 1.0),
            max_position_embeddings=cfg.get("max_position_embeddings", 32768),
            tie_word_embeddings=cfg.get("tie_word_embeddings", False),
            attention_bias=cfg.get("attention_bias", cfg.get("qkv_bias", False)),
            o_proj_bias=cfg.get("attention_bias", False)
            and cfg.get("model_type") == "gpt_oss",
            attention_sinks=cfg.get("model_type") == "gpt_oss",
            use_attn_gate=cfg.get("use_head_wise_attn_gate",
                                  cfg.get("model_type") == "step3p5"),
            qk_norm=cfg.get("use_qk_norm", False)
            or cfg.get("model_type", "") in ("qwen3", "qwen3_moe", "qwen3_next",
                                             "minimax_m2", "qwen3_5_text",
                                             "qwen3_5_moe_text",
                                             "minimax_m3_vl_text", "minimax_m3",
                                             "step3p5"),
            qk_norm_full=cfg.get("model_type", "") == "minimax_m2",
            sliding_window=sliding,
            layer_types=layer_types,
            eos_token_ids=[e for e in eos_ids if e is not None],
            bos_token_id=cfg.get("bos_token_id"),
            torch_dtype=cfg.get("torch_dtype", "bfloat16"),
            # MoE
            num_experts=cfg.get("n_routed_experts") or cfg.get("num_experts")
            or cfg.get("num_local_experts") or 0,
            num_experts_per_tok=cfg.get("num_experts_per_tok")
            or cfg.get("experts_per_token") or 0,
            moe_intermediate_size=cfg.get("moe_intermediate_size", 0),
            num_shared_experts=cfg.get("n_shared_experts", 0),
            shared_expert_intermediate_size=cfg.get("shared_expert_intermediate_size", 0),
            first_k_dense_layers=cfg.get("first_k_dense_replace", 0),
            moe_layer_freq=cfg.get("moe_layer_freq", 1) if not isinstance(
                cfg.get("moe_layer_freq"), list) else 1,
            norm_topk_prob=cfg.get("norm_topk_prob", True),
            routed_scaling_factor=cfg.get("routed_scaling_factor", 1.0),
            scoring_func=cfg.get("scoring_func")
            or ("sigmoid" if cfg.get("model_type") in ("deepseek_v3", "deepseek_v32",
                                                       "kimi_k2", "glm4_moe",
                                                       "glm4v_moe", "minimax_m2",
                                                       "minimax_m3_vl_text",
                                                       "minimax_m3")
                else "softmax"),
            topk_group=cfg.get("topk_group", 0),
            n_group=cfg.get("n_group", 0),
            # MLA
            rope_interleave=cfg.get(
                "rope_interleave",
                cfg.get("model_type", "") in (
                    "deepseek_v2", "deepseek_v3", "deepseek_v32", "kimi_k2"),
            ),
            q_lora_rank=cfg.get("q_lora_rank") or 0,
            kv_lora_rank=cfg.get("kv_lora_rank") or 0,
            qk_nope_head_dim=cfg.get("qk_nope_head_dim") or 0,
            qk_rope_head_dim=cfg.get("qk_rope_head_dim") or 0,
            v_head_dim=cfg.get("v_head_dim") or 0,
            # DSA / MSA indexer
            index_n_heads=cfg.get("index_n_heads") or 0,
            index_head_dim=cfg.get("index_head_dim") or 0,
            index_topk=cfg.get("index_topk") or 0,
            index_block_size=cfg.get("index_block_size") or 0,
            index_topk_blocks=cfg.get("index_topk_blocks") or 0,
            index_local_blocks=cfg.get("index_local_blocks")
            if cfg.get("index_local_blocks") is not None else 0,
            # hybrid linear attention
            linear_num_key_heads=cfg.get("linear_num_key_heads") or (
                num_heads if cfg.get("model_type") in ("minimax", "minimax_m1")
                else 0),
            linear_num_value_heads=cfg.get("linear_num_value_heads") or (
                num_heads if cfg.get("model_type") in ("minimax", "minimax_m1")
                else 0),
            linear_key_head_dim=cfg.get("linear_key_head_dim") or (
                head_dim if cfg.get("model_type") in ("minimax", "minimax_m1")
                else 0),
            linear_value_head_dim=cfg.get("linear_value_head_dim") or (
                head_dim if cfg.get("model_type") in ("minimax", "minimax_m1")
                else 0),
            linear_conv_kernel_dim=cfg.get("linear_conv_kernel_dim") or (
                1 if cfg.get("model_type") in ("minimax", "minimax_m1") else 4),
            raw=cfg,
        )
        return mc

    @classmethod
    def from_pretrained(cls, model_path: str) -> "ModelConfig":
        with open(os.path.join(model_path, "config.json")) as f:
            return cls.from_hf_config(json.load(f))

    # -- sizing helpers used by the scheduling brain (scheduling/model_info.py) ---

    def dtype_bytes(self) -> int:
        return {"bfloat16": 2, "float16": 2, "float32": 4, "float8": 1}.get(
            self.torch_dtype, 2
        )

    def embedding_bytes(self) -> int:
        return self.vocab_size * self.hidden_size * self.dtype_bytes()

    def lm_head_bytes(self) -> int:
        return 0 if self.tie_word_embeddings else self.embedding_bytes()

    def per_layer_param_bytes(self, layer_idx: int = -1) -> int:
        h, d = self.hidden_size, self.dtype_bytes()
        if self.is_mla:
            attn = (
                (self.q_lora_rank or h) * h
                + self.num_heads * (self.qk_nope_head_dim + self.qk_rope_head_dim)
                * (self.q_lora_rank or h)
                + h * (self.kv_lora_rank + self.qk_rope_head_dim)
                + self.num_heads * (self.qk_nope_head_dim + self.v_head_dim)
                * self.kv_lora_rank
                + self.num_heads * self.v_head_dim * h
            )
        else:
            attn = (
                h * self.num_heads * self.head_dim
                + 2 * h * self.num_kv_heads * self.head_dim
                + self.num_heads * self.head_dim * h
            )
        if self.is_moe and (layer_idx < 0 or self.is_moe_layer(layer_idx)):
            inter = self.moe_intermediate_size or self.intermediate_size
            mlp = 3 * h * inter * self.num_experts
            if self.num_shared_experts:
                mlp += 3 * h * inter * self.num_shared_experts
            if self.shared_expert_intermediate_size:
                mlp += 3 * h * self.shared_expert_intermediate_size
        else:
            mlp = 3 * h * self.intermediate_size
        return int((attn + mlp) * d)
