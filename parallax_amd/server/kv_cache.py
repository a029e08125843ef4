"""Paged KV cache tensors, MI355X-first layout.

Layout choice (differs from the reference's vLLM-packed Metal layout,
src/parallax/server/cache/kv_cache.py:84-193): K uses
``[num_blocks, num_kv_heads, block_size, head_dim]`` (token-major rows: a
QK^T A-fragment is 8 consecutive d of one token) and V is stored TRANSPOSED,
``[num_blocks, num_kv_heads, head_dim, block_size]`` (d-major rows: a PV
A-fragment is 8 consecutive tokens of one d) — so BOTH decode-attention MFMA
operands are direct 16 B HBM loads with no LDS transpose staging, and the
prefill kernel stages V^T as plain vectorized row copies. The cache write
pays a per-element transpose scatter once per token instead. 288 GB HBM3E
means block count is sized generously from a memory fraction rather than
packed tightly.

Also here: ``MLAKVCache`` — DeepSeek-style compressed latent cache (kv_lora_rank
latent + rope dims per token, no per-head expansion), and ``LinearStateCache`` —
slot-based conv/recurrent state for hybrid linear-attention stacks.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional

import torch


@dataclass
class KVCacheSpec:
    num_layers: int
    num_kv_heads: int
    head_dim: int
    block_size: int
    dtype: torch.dtype = torch.bfloat16
    # sparse-attention indexer keys: one index_dim vector per token per layer
    # (minimax-m3 MSA; dtype follows the main cache but never fp8)
    index_dim: int = 0

    @property
    def bytes_per_token_per_layer(self) -> int:
        return (2 * self.num_kv_heads * self.head_dim + self.index_dim) \
            * self.dtype.itemsize

    def bytes_per_block(self) -> int:
        return self.block_size * self.bytes_per_token_per_layer * self.num_layers


class PagedKVCache:
    """Per-shard paged KV storage: one K and one V tensor per local layer."""

    def __init__(self, spec: KVCacheSpec, num_blocks: int, device: torch.device):
        self.spec = spec
        self.num_blocks = num_blocks
        self.device = device
        # K rows are tokens (QK^T fragments read 8 consecutive d per token);
        # V is stored TRANSPOSED [.., D, block] so PV fragments read 8
        # consecutive tokens per d straight from HBM — the decode/prefill
        # kernels consume V^T without any LDS transpose staging
        k_shape = (num_blocks, spec.num_kv_heads, spec.block_size, spec.head_dim)
        v_shape = (num_blocks, spec.num_kv_heads, spec.head_dim, spec.block_size)
        self.k_caches: List[torch.Tensor] = [
            torch.zeros(k_shape, dtype=spec.dtype, device=device)
            for _ in range(spec.num_layers)
        ]
        self.v_caches: List[torch.Tensor] = [
            torch.zeros(v_shape, dtype=spec.dtype, device=device)
            for _ in range(spec.num_layers)
        ]
        idx_dtype = spec.dtype if spec.dtype != torch.float8_e4m3fn \
            else torch.bfloat16
        # +1 trash block for graph-safe pad-token stores (see MLAKVCache)
        self.index_caches: List[torch.Tensor] = (
            [
                torch.zeros((num_blocks + 1, spec.block_size, spec.index_dim),
                            dtype=idx_dtype, device=device)
                for _ in range(spec.num_layers)
            ]
            if spec.index_dim > 0 else []
        )

    def layer(self, idx: int):
        return self.k_caches[idx], self.v_caches[idx]

    def index_layer(self, idx: int) -> torch.Tensor:
        return self.index_caches[idx]

    @staticmethod
    def num_blocks_for_bytes(spec: KVCacheSpec, budget_bytes: int) -> int:
        return max(0, budget_bytes // spec.bytes_per_block())


class MLAKVCache:
    """Compressed MLA cache: per token one latent vector (kv_lora_rank) plus the
    decoupled rope key (qk_rope_head_dim) — reference dsa_cache.py:8 behavior,
    laid out ``[num_blocks, block_size, kv_lora_rank + rope_dim]`` so decode
    reads one contiguous row per token."""

    def __init__(
        self,
        num_layers: int,
        kv_lora_rank: int,
        rope_dim: int,
        block_size: int,
        num_blocks: int,
        device: torch.device,
        dtype: torch.dtype = torch.bfloat16,
        index_dim: int = 0,
    ):
        self.kv_lora_rank = kv_lora_rank
        self.rope_dim = rope_dim
        self.block_size = block_size
        self.num_blocks = num_blocks
        self.entry_dim = kv_lora_rank + rope_dim
        self.caches: List[torch.Tensor] = [
            torch.zeros((num_blocks, block_size, self.entry_dim), dtype=dtype, device=device)
            for _ in range(num_layers)
        ]
        # DSA (DeepSeek-V3.2): one indexer key per token per layer, shared
        # across indexer query heads (reference dsa_cache.py)
        self.index_dim = index_dim
        idx_dtype = dtype if dtype != torch.float8_e4m3fn else torch.bfloat16
        # +1 trash block at the end: graph-safe indexer-cache stores write
        # pad tokens (slot -1) there instead of branching (see
        # ops.reference.store_indexer_cache)
        self.index_caches: List[torch.Tensor] = (
            [
                torch.zeros((num_blocks + 1, block_size, index_dim),
                            dtype=idx_dtype, device=device)
                for _ in range(num_layers)
            ]
            if index_dim > 0
            else []
        )

    def layer(self, idx: int) -> torch.Tensor:
        return self.caches[idx]

    def index_layer(self, idx: int) -> torch.Tensor:
        return self.index_caches[idx]

    @staticmethod
    def bytes_per_block(
        num_layers: int, kv_lora_rank: int, rope_dim: int, block_size: int,
        dtype: torch.dtype = torch.bfloat16, index_dim: int = 0,
    ) -> int:
        return num_layers * block_size * (kv_lora_rank + rope_dim + index_dim) * dtype.itemsize


class LinearStateCache:
    """Slot-based state for linear-attention / SSM hybrid layers (reference
    cache/linear_cache.py): per slot a conv window state and a recurrent state,
    with snapshot/restore for prefix reuse."""

    def __init__(
        self,
        num_layers: int,
        conv_state_shape: tuple,
        recurrent_state_shape: tuple,
        num_slots: int,
        device: torch.device,
        dtype: torch.dtype = torch.bfloat16,
    ):
        self.num_slots = num_slots
        self.conv_states: List[torch.Tensor] = [
            torch.zeros((num_slots, *conv_state_shape), dtype=dtype, device=device)
            for _ in range(num_layers)
        ]
        self.recurrent_states: List[torch.Tensor] = [
            torch.zeros((num_slots, *recurrent_state_shape), dtype=torch.float32, device=device)
            for _ in range(num_layers)
        ]

    def reset_slot(self, slot: int) -> None:
        for c in self.conv_states:
            c[slot].zero_()
        for r in self.recurrent_states:
            r[slot].zero_()

    def snapshot(self, slot: int):
        return (
            [c[slot].clone() for c in self.conv_states],
            [r[slot].clone() for r in self.recurrent_states],
        )

    def restore(self, slot: int, snap) -> None:
        conv, rec = snap
        for dst, src in zip(self.conv_states, conv):
            dst[slot].copy_(src)
        for dst, src in zip(self.recurrent_states, rec):
            dst[slot].copy_(src)


def build_block_table_tensor(
    block_tables: List[List[int]], device: torch.device, pad: int = 0
) -> torch.Tensor:
    """Pad ragged per-request block tables into an int32 [batch, max_blocks] tensor."""
    max_len = max((len(bt) for bt in block_tables), default=1)
    max_len = max(max_len, 1)
    out = torch.full((len(block_tables), max_len), pad, dtype=torch.int32)
    for i, bt in enumerate(block_tables):
        if bt:
            out[i, : len(bt)] = torch.tensor(bt, dtype=torch.int32)
    return out.to(device)


def slot_mapping_for_positions(
    block_table: List[int], start_pos: int, num_tokens: int, block_size: int
) -> List[int]:
    """Flat cache-slot index (block_id * block_size + offset) for each new token."""
    out = []
    for pos in range(start_pos, start_pos + num_tokens):
        out.append(block_table[pos // block_size] * block_size + pos % block_size)
    return out
