"""Model zoo: one module per architecture family, auto-registered for the shard
loader (parity with the reference's EntryClass autodiscovery, shard_loader.py:79)."""

from .registry import MODEL_REGISTRY, get_model_class, register_model

from . import llama  # noqa: F401  (registers LlamaForCausalLM / Qwen2ForCausalLM)
from . import qwen3  # noqa: F401
from . import qwen3_moe  # noqa: F401
from . import deepseek_v3  # noqa: F401
from . import deepseek_v32  # noqa: F401
from . import gpt_oss  # noqa: F401
from . import glm4_moe  # noqa: F401
from . import qwen3_next  # noqa: F401
from . import minimax_m2  # noqa: F401
from . import step3p5  # noqa: F401
from . import qwen3_5  # noqa: F401
from . import minimax  # noqa: F401
from . import minimax_m3  # noqa: F401
