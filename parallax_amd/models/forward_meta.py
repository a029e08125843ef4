"""Per-iteration forward metadata threaded through every decoder block.

One ForwardMeta describes either a (chunked-)prefill batch or a decode batch —
the equivalent of the reference's batch formation output (sglang/batch_info.py)
re-designed around our paged layout. All tensors live on the model's device.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import TYPE_CHECKING, Optional

import torch

if TYPE_CHECKING:
    from ..server.kv_cache import LinearStateCache, MLAKVCache, PagedKVCache


@dataclass
class ForwardMeta:
    is_prefill: bool
    positions: torch.Tensor          # [T] int32 — absolute position of each token
    slot_mapping: torch.Tensor       # [T] int64 — flat KV slot per token
    block_tables: torch.Tensor       # [B, max_blocks] int32
    seq_lens: torch.Tensor           # [B] int32 — total ctx len incl. this step
    query_lens: Optional[torch.Tensor] = None   # [B] int32, prefill only
    kv_cache: Optional["PagedKVCache"] = None
    mla_cache: Optional["MLAKVCache"] = None
    linear_cache: Optional["LinearStateCache"] = None
    linear_slots: Optional[torch.Tensor] = None  # [B] int32 slot ids (hybrid stacks)
    # indices of the last token of each request within the packed token dim
    # (where logits are needed; for decode this is arange(B))
    logits_indices: Optional[torch.Tensor] = None
    # host-side max of seq_lens (avoids a device sync when picking the
    # flash-decoding partition count)
    max_seq_len: int = 0

    @property
    def num_tokens(self) -> int:
        return self.positions.shape[0]

    @property
    def batch_size(self) -> int:
        return self.seq_lens.shape[0]
