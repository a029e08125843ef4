"""KV cache bookkeeping: block tables, prefix reuse, decode growth, budgets.

Capability parity with the reference's src/parallax/server/cache_manager.py:25
(budget calc from cache_memory_fraction, allocate_request with prefix reuse,
append_slot for decode growth, radix insert after prefill). Fresh design:
bookkeeping (this class) is separated from storage (kv_cache.py tensors), so the
same manager drives full-KV, compressed-MLA and hybrid linear stacks. Sizing
assumes 288 GB HBM3E per GPU — block counts come from a memory fraction of the
free device memory, not a fixed pool size.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional

import torch

from ..utils.logging_config import get_logger
from .allocator import BlockAllocator, OutOfBlocksError, SlotAllocator
from .kv_cache import KVCacheSpec
from .radix_cache import BlockRadixCache

logger = get_logger("server.cache_manager")


@dataclass
class RequestCacheState:
    block_table: List[int] = field(default_factory=list)
    num_cached_tokens: int = 0     # prefix-cache hit length (KV already present)
    num_allocated_tokens: int = 0  # capacity covered by block_table
    linear_slot: Optional[int] = None
    published_prefill: bool = False


class CacheManager:
    def __init__(
        self,
        block_size: int,
        num_blocks: int,
        enable_prefix_cache: bool = True,
        num_linear_slots: int = 0,
    ):
        self.block_size = block_size
        self.allocator = BlockAllocator(num_blocks, block_size)
        self.radix = BlockRadixCache(self.allocator, block_size, enable=enable_prefix_cache)
        self.slot_allocator = SlotAllocator(num_linear_slots) if num_linear_slots else None
        self._requests: Dict[str, RequestCacheState] = {}

    # -- sizing --------------------------------------------------------------

    @staticmethod
    def num_blocks_from_memory(
        spec: KVCacheSpec,
        device: torch.device,
        cache_memory_fraction: float = 0.80,
        reserve_bytes: int = 2 << 30,
    ) -> int:
        """Size the block pool from what is actually free on the device after
        weights are loaded (MI355X: 288 GB HBM3E — be generous)."""
        if device.type == "cuda":
            free_b, _total = torch.cuda.mem_get_info(device)
            budget = int(free_b * cache_memory_fraction) - reserve_bytes
        else:
            budget = 1 << 30  # CPU test default: 1 GiB
        return max(16, budget // spec.bytes_per_block())

    # -- admission -----------------------------------------------------------

    def blocks_needed(self, num_tokens: int) -> int:
        return (num_tokens + self.block_size - 1) // self.block_size

    def can_allocate(self, num_tokens: int, token_ids: Optional[List[int]] = None) -> bool:
        need = self.blocks_needed(num_tokens)
        if token_ids is not None:
            cached, _ = self.radix.match_prefix(token_ids)
            need -= len(cached)
        free_incl_evictable = self.allocator.num_free_blocks + self.radix.num_cached_blocks
        return free_incl_evictable >= need

    def allocate_request(
        self, rid: str, token_ids: List[int], max_new_tokens: int = 0
    ) -> RequestCacheState:
        """Allocate blocks covering the prompt (+1 slack block), reusing any
        cached prefix. Raises OutOfBlocksError if eviction cannot cover it."""
        assert rid not in self._requests, f"request {rid} already has cache state"
        cached_blocks, cached_tokens = self.radix.match_prefix(token_ids)
        # never reuse the *entire* prompt: the last token must be recomputed so
        # this step produces hidden states (standard prefix-cache guard)
        if cached_tokens >= len(token_ids) and cached_blocks:
            cached_blocks = cached_blocks[:-1]
            cached_tokens -= self.block_size
        total_tokens = len(token_ids)
        need = self.blocks_needed(total_tokens) - len(cached_blocks)
        if need > self.allocator.num_free_blocks:
            self.radix.evict_for(need)
        new_blocks = self.allocator.allocate(max(0, need))
        for b in cached_blocks:
            self.allocator.incref(b)
        state = RequestCacheState(
            block_table=cached_blocks + new_blocks,
            num_cached_tokens=cached_tokens,
            num_allocated_tokens=(len(cached_blocks) + len(new_blocks)) * self.block_size,
        )
        if self.slot_allocator is not None:
            state.linear_slot = self.slot_allocator.allocate()
        self._requests[rid] = state
        return state

    def append_tokens(self, rid: str, new_total_len: int) -> RequestCacheState:
        """Grow a request's block table to cover `new_total_len` tokens (decode)."""
        state = self._requests[rid]
        while state.num_allocated_tokens < new_total_len:
            if self.allocator.num_free_blocks == 0:
                self.radix.evict_for(1)
            state.block_table.extend(self.allocator.allocate(1))
            state.num_allocated_tokens += self.block_size
        return state

    # -- prefix publication ----------------------------------------------------

    def publish_prefill(self, rid: str, prompt_token_ids: List[int]) -> None:
        """After prefill completes, publish the prompt's full blocks so
        concurrent same-prefix requests hit the cache."""
        state = self._requests.get(rid)
        if state is None or state.published_prefill:
            return
        self.radix.insert(prompt_token_ids, state.block_table)
        state.published_prefill = True

    # -- release ----------------------------------------------------------------

    def free_request(self, rid: str, all_token_ids: Optional[List[int]] = None) -> None:
        state = self._requests.pop(rid, None)
        if state is None:
            return
        if all_token_ids is not None:
            self.radix.insert(all_token_ids, state.block_table)
        self.allocator.free(state.block_table)
        if state.linear_slot is not None and self.slot_allocator is not None:
            self.slot_allocator.free(state.linear_slot)

    # -- introspection ------------------------------------------------------------

    def get(self, rid: str) -> RequestCacheState:
        return self._requests[rid]

    @property
    def num_free_blocks(self) -> int:
        return self.allocator.num_free_blocks

    @property
    def token_capacity_free(self) -> int:
        return self.allocator.num_free_blocks * self.block_size

    def reset_prefix_cache(self) -> None:
        self.radix.reset()
