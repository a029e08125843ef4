"""Qwen3: Llama-family block with per-head q/k RMSNorm (reference analogue:
src/parallax/models/qwen3.py). The qk_norm flag is derived from model_type in
ModelConfig.from_hf_config, so this is the Llama shard with the right registry
entries."""

from .llama import LlamaShardModel
from .registry import register_model


@register_model("Qwen3ForCausalLM")
class Qwen3ShardModel(LlamaShardModel):
    pass
