"""BlockManager: allocation, prefix reuse, eviction, ref counting.

Mirrors the engine-contract requirement SURVEY.md §2.16-bis item 6 (prefix
caching that CHWBL routing benefits from).
"""
import pytest

from kubeai_amd.engine.kvcache import BlockManager, NoFreeBlocks


def test_allocate_and_free():
    bm = BlockManager(num_blocks=8, block_size=16)
    table, cached = bm.allocate(list(range(40)))  # 3 blocks
    assert len(table) == 3 and cached == 0
    assert bm.num_free_blocks == 5
    bm.free(table)
    assert bm.num_free_blocks == 8


def test_prefix_reuse_after_seal():
    bm = BlockManager(num_blocks=16, block_size=16)
    toks = list(range(64))
    t1, cached1 = bm.allocate(toks)
    assert cached1 == 0
    # seal the first three blocks (KV computed)
    parent = None
    for i in range(3):
        parent = bm.seal_block(t1, i, tuple(toks[i * 16 : (i + 1) * 16]), parent)
    t2, cached2 = bm.allocate(toks + [999])
    assert cached2 == 48
    assert t2[:3] == t1[:3]  # physically shared
    assert bm.blocks[t1[0]].ref_count == 2
    bm.free(t1)
    bm.free(t2)


def test_prefix_survives_free_until_evicted():
    bm = BlockManager(num_blocks=8, block_size=16)
    toks = list(range(32))
    t1, _ = bm.allocate(toks)
    parent = None
    for i in range(2):
        parent = bm.seal_block(t1, i, tuple(toks[i * 16 : (i + 1) * 16]), parent)
    bm.free(t1)
    assert bm.num_free_blocks == 8
    t2, cached = bm.allocate(toks)
    assert cached == 32  # full 2-block hit
    bm.free(t2)
    # exhaust the pool with other content -> eviction drops the hashes
    t3, _ = bm.allocate(list(range(1000, 1000 + 8 * 16)))
    assert bm.num_free_blocks == 0
    bm.free(t3)
    t4, cached4 = bm.allocate(toks)
    assert cached4 == 0  # evicted
    bm.free(t4)


def test_max_cached_cap():
    bm = BlockManager(num_blocks=8, block_size=16)
    toks = list(range(32))
    t1, _ = bm.allocate(toks)
    parent = None
    for i in range(2):
        parent = bm.seal_block(t1, i, tuple(toks[i * 16 : (i + 1) * 16]), parent)
    bm.free(t1)
    # cap forces at least one uncached token (scheduler passes P-1)
    t2, cached = bm.allocate(toks, max_cached=len(toks) - 1)
    assert cached == 16
    bm.free(t2)


def test_no_free_blocks():
    bm = BlockManager(num_blocks=2, block_size=16)
    t1, _ = bm.allocate(list(range(32)))
    with pytest.raises(NoFreeBlocks):
        bm.allocate(list(range(100, 116)))
    bm.free(t1)


def test_salt_isolates_adapters():
    bm = BlockManager(num_blocks=8, block_size=16)
    toks = list(range(16))
    t1, _ = bm.allocate(toks, salt=0)
    bm.seal_block(t1, 0, tuple(toks), None, salt=0)
    bm.free(t1)
    _, cached_same = bm.allocate(toks, salt=0)
    assert cached_same == 16
    _, cached_other = bm.allocate(toks, salt=7)
    assert cached_other == 0


def test_append_and_lru_order():
    bm = BlockManager(num_blocks=4, block_size=16)
    t, _ = bm.allocate(list(range(16)))
    bm.append_block(t)
    assert len(t) == 2
    assert bm.num_free_blocks == 2
    bm.free(t)
    assert bm.num_free_blocks == 4
