"""Block-granularity radix prefix cache.

Capability parity with the reference's src/parallax/server/block_radix_cache.py
(block-granularity radix tree over token-id chunks with LRU eviction returning
blocks to the allocator); fresh design. One tree node per full KV block; a node
is evictable while only the tree holds its block (allocator refcount == 1).
"""

from __future__ import annotations

import time
from typing import Dict, List, Optional, Tuple

from .allocator import BlockAllocator


class _RadixNode:
    __slots__ = ("key", "block_id", "children", "parent", "last_access")

    def __init__(self, key: Optional[Tuple[int, ...]], block_id: Optional[int], parent):
        self.key = key              # token-id chunk of exactly block_size tokens
        self.block_id = block_id    # KV block holding this chunk
        self.children: Dict[Tuple[int, ...], _RadixNode] = {}
        self.parent: Optional[_RadixNode] = parent
        self.last_access = time.monotonic()


class BlockRadixCache:
    def __init__(self, allocator: BlockAllocator, block_size: int, enable: bool = True):
        self.allocator = allocator
        self.block_size = block_size
        self.enable = enable
        self.root = _RadixNode(None, None, None)
        self._num_nodes = 0
        # metrics
        self.hit_tokens = 0
        self.query_tokens = 0

    # -- lookup ------------------------------------------------------------

    def match_prefix(self, token_ids: List[int]) -> Tuple[List[int], int]:
        """Longest cached prefix of `token_ids` in full blocks.
        Returns (block_ids, num_matched_tokens). Does NOT change refcounts —
        the caller increfs blocks it attaches to a request."""
        self.query_tokens += len(token_ids)
        if not self.enable:
            return [], 0
        blocks: List[int] = []
        node = self.root
        now = time.monotonic()
        n_full = len(token_ids) // self.block_size
        for i in range(n_full):
            chunk = tuple(token_ids[i * self.block_size : (i + 1) * self.block_size])
            child = node.children.get(chunk)
            if child is None:
                break
            child.last_access = now
            blocks.append(child.block_id)
            node = child
        self.hit_tokens += len(blocks) * self.block_size
        return blocks, len(blocks) * self.block_size

    # -- insert ------------------------------------------------------------

    def insert(self, token_ids: List[int], block_table: List[int]) -> None:
        """Publish a finished/prefilled request's full blocks into the tree.
        Takes an extra refcount on each newly published block (the tree's hold).
        Blocks already present under the same prefix are left as-is."""
        if not self.enable:
            return
        node = self.root
        now = time.monotonic()
        n_full = min(len(token_ids) // self.block_size, len(block_table))
        for i in range(n_full):
            chunk = tuple(token_ids[i * self.block_size : (i + 1) * self.block_size])
            child = node.children.get(chunk)
            if child is None:
                block_id = block_table[i]
                self.allocator.incref(block_id)
                child = _RadixNode(chunk, block_id, node)
                node.children[chunk] = child
                self._num_nodes += 1
            child.last_access = now
            node = child

    # -- eviction ----------------------------------------------------------

    def evict(self, num_blocks: int) -> int:
        """Evict up to `num_blocks` least-recently-used evictable leaves,
        releasing the tree's refcount (block returns to the free list when no
        request still uses it). Returns the number evicted."""
        if not self.enable or num_blocks <= 0:
            return 0
        evicted = 0
        while evicted < num_blocks:
            victim = self._lru_evictable_leaf()
            if victim is None:
                break
            self.allocator.decref(victim.block_id)
            del victim.parent.children[victim.key]
            self._num_nodes -= 1
            evicted += 1
        return evicted

    def evict_for(self, needed_blocks: int) -> bool:
        """Ensure at least `needed_blocks` are free, evicting if necessary."""
        deficit = needed_blocks - self.allocator.num_free_blocks
        if deficit <= 0:
            return True
        self.evict(deficit)
        return self.allocator.num_free_blocks >= needed_blocks

    def _lru_evictable_leaf(self) -> Optional[_RadixNode]:
        best: Optional[_RadixNode] = None
        stack = list(self.root.children.values())
        while stack:
            node = stack.pop()
            if node.children:
                stack.extend(node.children.values())
            else:
                # leaf; evictable if only the tree holds the block
                if self.allocator.refcount(node.block_id) == 1:
                    if best is None or node.last_access < best.last_access:
                        best = node
        return best

    def reset(self) -> None:
        """Drop the whole tree (weight refit invalidates cached KV)."""
        stack = list(self.root.children.values())
        while stack:
            node = stack.pop()
            stack.extend(node.children.values())
            self.allocator.decref(node.block_id)
        self.root = _RadixNode(None, None, None)
        self._num_nodes = 0

    @property
    def num_cached_blocks(self) -> int:
        return self._num_nodes

    @property
    def hit_rate(self) -> float:
        return self.hit_tokens / self.query_tokens if self.query_tokens else 0.0
