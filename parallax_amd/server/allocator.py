"""Free-list allocators for paged KV blocks and linear-state slots.

Capability parity with the reference's src/parallax/server/cache/allocator.py:8,46
(`BlockAllocator` / `SlotAllocator`); fresh design with reference counting so
block-radix prefix sharing can hold blocks without copy.
"""

from __future__ import annotations

from typing import Dict, List


class OutOfBlocksError(RuntimeError):
    pass


class BlockAllocator:
    """Fixed pool of KV-cache blocks with refcounts (prefix-shared blocks are
    held by multiple owners; a block returns to the free list at refcount 0)."""

    def __init__(self, num_blocks: int, block_size: int):
        self.num_blocks = num_blocks
        self.block_size = block_size
        self._free: List[int] = list(range(num_blocks - 1, -1, -1))
        self._refcount: Dict[int, int] = {}

    @property
    def num_free_blocks(self) -> int:
        return len(self._free)

    @property
    def num_used_blocks(self) -> int:
        return self.num_blocks - len(self._free)

    def can_allocate(self, n: int) -> bool:
        return len(self._free) >= n

    def allocate(self, n: int) -> List[int]:
        if n > len(self._free):
            raise OutOfBlocksError(
                f"requested {n} blocks, {len(self._free)} free of {self.num_blocks}"
            )
        blocks = [self._free.pop() for _ in range(n)]
        for b in blocks:
            self._refcount[b] = 1
        return blocks

    def incref(self, block_id: int) -> None:
        self._refcount[block_id] += 1

    def decref(self, block_id: int) -> int:
        rc = self._refcount[block_id] - 1
        if rc < 0:
            raise RuntimeError(f"block {block_id} refcount underflow")
        if rc == 0:
            del self._refcount[block_id]
            self._free.append(block_id)
        else:
            self._refcount[block_id] = rc
        return rc

    def refcount(self, block_id: int) -> int:
        return self._refcount.get(block_id, 0)

    def free(self, blocks: List[int]) -> None:
        for b in blocks:
            self.decref(b)


class SlotAllocator:
    """Slot pool for linear-attention recurrent/conv state (hybrid layer stacks,
    reference cache/linear_cache.py). One slot per running request."""

    def __init__(self, num_slots: int):
        self.num_slots = num_slots
        self._free: List[int] = list(range(num_slots - 1, -1, -1))

    @property
    def num_free_slots(self) -> int:
        return len(self._free)

    def can_allocate(self, n: int = 1) -> bool:
        return len(self._free) >= n

    def allocate(self) -> int:
        if not self._free:
            raise OutOfBlocksError("no free linear-state slots")
        return self._free.pop()

    def free(self, slot: int) -> None:
        self._free.append(slot)
