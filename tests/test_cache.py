"""Allocator / radix prefix cache / cache-manager unit tests (CPU).

Test strategy mirrors the reference's cache tests: pure
bookkeeping, no tensors needed.
"""

import pytest
import torch

from parallax_amd.server.allocator import BlockAllocator, OutOfBlocksError
from parallax_amd.server.cache_manager import CacheManager
from parallax_amd.server.radix_cache import BlockRadixCache


def test_block_allocator_basic():
    a = BlockAllocator(8, 16)
    blocks = a.allocate(3)
    assert len(blocks) == 3 and a.num_free_blocks == 5
    a.free(blocks)
    assert a.num_free_blocks == 8
    with pytest.raises(OutOfBlocksError):
        a.allocate(9)


def test_block_allocator_refcount():
    a = BlockAllocator(4, 16)
    (b,) = a.allocate(1)
    a.incref(b)
    assert a.decref(b) == 1
    assert a.num_free_blocks == 3
    assert a.decref(b) == 0
    assert a.num_free_blocks == 4


def test_radix_match_and_insert():
    a = BlockAllocator(32, 4)
    r = BlockRadixCache(a, 4)
    toks = list(range(10))  # 2 full blocks + 2 spare
    blocks = a.allocate(3)
    r.insert(toks, blocks)
    hit_blocks, n = r.match_prefix(toks)
    assert n == 8 and hit_blocks == blocks[:2]
    # diverging suffix only matches the shared prefix
    hit_blocks, n = r.match_prefix(list(range(4)) + [99, 98, 97, 96])
    assert n == 4 and hit_blocks == blocks[:1]


def test_radix_eviction_returns_blocks():
    a = BlockAllocator(4, 4)
    r = BlockRadixCache(a, 4)
    blocks = a.allocate(2)
    r.insert(list(range(8)), blocks)
    a.free(blocks)  # request done; only the tree holds them
    assert a.num_free_blocks == 2
    assert r.evict(2) == 2
    assert a.num_free_blocks == 4


def test_cache_manager_prefix_reuse():
    cm = CacheManager(block_size=4, num_blocks=16)
    s1 = cm.allocate_request("r1", list(range(10)))
    assert s1.num_cached_tokens == 0
    cm.publish_prefill("r1", list(range(10)))
    s2 = cm.allocate_request("r2", list(range(10)) + [50, 51])
    assert s2.num_cached_tokens == 8  # two full blocks reused
    assert s2.block_table[:2] == s1.block_table[:2]
    cm.free_request("r1")
    cm.free_request("r2")
    # all blocks recoverable after eviction
    cm.radix.evict(100)
    assert cm.num_free_blocks == 16


def test_cache_manager_whole_prompt_cached_recomputes_last_block():
    cm = CacheManager(block_size=4, num_blocks=16)
    cm.allocate_request("r1", list(range(8)))
    cm.publish_prefill("r1", list(range(8)))
    s2 = cm.allocate_request("r2", list(range(8)))
    # must leave at least the last block to recompute (hidden states needed)
    assert s2.num_cached_tokens == 4


def test_cache_manager_decode_growth():
    cm = CacheManager(block_size=4, num_blocks=8)
    s = cm.allocate_request("r1", [1, 2, 3])
    assert len(s.block_table) == 1
    cm.append_tokens("r1", 5)
    assert len(cm.get("r1").block_table) == 2
    cm.append_tokens("r1", 5)  # idempotent
    assert len(cm.get("r1").block_table) == 2


def test_linear_state_cache_snapshot_restore():
    from parallax_amd.server.kv_cache import LinearStateCache

    c = LinearStateCache(num_layers=2, conv_state_shape=(6, 3),
                         recurrent_state_shape=(2, 4, 4), num_slots=4,
                         device=torch.device("cpu"), dtype=torch.float32)
    for l in range(2):
        c.conv_states[l][1].fill_(l + 1.0)
        c.recurrent_states[l][1].fill_(l + 10.0)
    snap = c.snapshot(1)
    c.reset_slot(1)
    assert c.conv_states[0][1].abs().sum() == 0
    c.restore(1, snap)
    assert torch.all(c.conv_states[1][1] == 2.0)
    assert torch.all(c.recurrent_states[0][1] == 10.0)
    # snapshot is a deep copy: later mutation doesn't corrupt it
    c.conv_states[0][1].fill_(99.0)
    c.restore(1, snap)
    assert torch.all(c.conv_states[0][1] == 1.0)


def test_mla_cache_index_trash_block():
    """The +1 trash block convention: store with slot -1 lands in the last
    block and never corrupts addressable blocks."""
    from parallax_amd.server.kv_cache import MLAKVCache
    from parallax_amd.ops import reference as ref

    c = MLAKVCache(num_layers=1, kv_lora_rank=8, rope_dim=4, block_size=4,
                   num_blocks=3, device=torch.device("cpu"),
                   dtype=torch.float32, index_dim=8)
    idx = c.index_layer(0)
    assert idx.shape[0] == 4  # 3 + trash
    keys = torch.randn(2, 8)
    ref.store_indexer_cache(keys, idx, torch.tensor([5, -1]))
    assert torch.equal(idx[1, 1], keys[0])
    # the pad token went to the trash block, not block 0..2
    assert idx[:3].abs().sum() == keys[0].abs().sum()
    assert idx[3].abs().sum() > 0


def test_radix_does_not_evict_referenced_blocks():
    """Blocks still referenced by a live request must survive LRU pressure:
    eviction only drops the TREE's refcount; the block returns to the free
    list only when the owning request also releases it."""
    alloc = BlockAllocator(8, block_size=4)
    radix = BlockRadixCache(block_size=4, allocator=alloc)
    a_blocks = alloc.allocate(2)          # live request's hold (rc=1)
    radix.insert(list(range(8)), a_blocks)  # tree's hold (rc=2)
    # blocks still referenced by the request are NOT evictable at all
    assert radix.evict(10) == 0
    assert alloc.num_free_blocks == 6
    # request releases its hold -> the leaf becomes evictable, LRU order
    for b in a_blocks:
        alloc.decref(b)
    assert radix.evict(10) == 2
    assert alloc.num_free_blocks == 8


try:
    from hypothesis import HealthCheck, given, settings
    from hypothesis import strategies as st
    _HYP = True
except ImportError:  # pragma: no cover
    _HYP = False

import pytest as _pytest


@_pytest.mark.skipif(not _HYP, reason="hypothesis not installed")
@settings(max_examples=50, deadline=None,
          suppress_health_check=[HealthCheck.too_slow])
@given(ops=st.lists(
    st.tuples(st.sampled_from(["insert", "match", "evict"]),
              st.integers(min_value=0, max_value=5),   # sequence family
              st.integers(min_value=1, max_value=40)), # length / evict count
    min_size=1, max_size=40))
def test_radix_model_based(ops):
    """Model-based check of the block-radix prefix cache against a naive
    prefix dictionary: matches never exceed what was inserted (and not
    evicted), matched blocks agree with the inserting table, and refcounts
    keep cached + free = pool."""
    from parallax_amd.server.allocator import BlockAllocator
    from parallax_amd.server.radix_cache import BlockRadixCache

    BS, POOL = 4, 64
    alloc = BlockAllocator(POOL, BS)
    radix = BlockRadixCache(alloc, BS)
    families = {i: list(range(10 * i, 10 * i + 64)) for i in range(6)}
    model = {}  # tuple(prefix tokens, block-aligned) -> block id list

    for op, fam, ln in ops:
        toks = families[fam][:ln]
        if op == "insert":
            nblocks = len(toks) // BS
            if nblocks == 0 or alloc.num_free_blocks < nblocks:
                continue
            table = alloc.allocate(nblocks)
            radix.insert(toks, table)
            for b in range(nblocks):
                model[tuple(toks[: (b + 1) * BS])] = None  # presence only
            # caller's reference is dropped (the request released)
            alloc.free(table)
        elif op == "match":
            blocks, n_cached = radix.match_prefix(toks)
            assert n_cached % BS == 0 and n_cached <= len(toks)
            assert len(blocks) == n_cached // BS
            # everything the radix claims must have been inserted some time
            if n_cached:
                assert tuple(toks[:n_cached]) in model
            # match does not incref; nothing to release
        else:
            radix.evict(ln % 8 + 1)
        assert alloc.num_free_blocks + radix.num_cached_blocks == POOL
