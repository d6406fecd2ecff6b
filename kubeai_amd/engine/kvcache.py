"""Paged KV-cache block manager with prefix caching.

Re-designed for the engine contract in SURVEY.md §2.16-bis item 6: "paged KV
cache with prefix reuse" is what makes the control plane's CHWBL prefix-affine
routing pay off (the reference relies on vLLM for this; here it is in-house).

Design:
  - fixed-size token blocks (16) over a pool sized for the GPU's free HBM
    (288 GB per MI355X — pool sizing happens in the runner);
  - content hashing: a FULL block's identity is hash(parent_hash, tokens);
    identical prefixes across requests map to the same chain of hashes;
  - blocks with ref_count 0 go to an LRU free structure but keep their hash,
    so a later request with the same prefix resurrects them (cache hit);
  - eviction pops the least-recently-used free block and drops its hash.

All bookkeeping is O(1) per block operation.
"""
from __future__ import annotations

import dataclasses
from collections import OrderedDict
from typing import Optional


@dataclasses.dataclass
class Block:
    block_id: int
    ref_count: int = 0
    block_hash: Optional[int] = None  # set only for FULL, hashed blocks


def hash_block(parent_hash: Optional[int], tokens: tuple[int, ...], salt: int = 0) -> int:
    # Python's tuple hash is stable within a process; salt isolates models /
    # LoRA adapters sharing a pool (adapter-specific KV must not collide).
    return hash((parent_hash, salt, tokens))


class BlockManager:
    def __init__(self, num_blocks: int, block_size: int = 16):
        assert num_blocks > 0
        self.num_blocks = num_blocks
        self.block_size = block_size
        self.blocks = [Block(i) for i in range(num_blocks)]
        # free blocks in LRU order (front = evict first); all start free
        self._free: OrderedDict[int, None] = OrderedDict(
            (i, None) for i in range(num_blocks)
        )
        # content hash -> block_id (may point at a free-but-cached block)
        self._cache: dict[int, int] = {}
        # metrics
        self.cache_hit_tokens = 0
        self.cache_query_tokens = 0

    # ------------------------------------------------------------------
    @property
    def num_free_blocks(self) -> int:
        return len(self._free)

    def usage(self) -> float:
        return 1.0 - len(self._free) / self.num_blocks

    def hit_rate(self) -> float:
        if self.cache_query_tokens == 0:
            return 0.0
        return self.cache_hit_tokens / self.cache_query_tokens

    # ------------------------------------------------------------------
    def _take(self, block_id: int) -> None:
        """Remove a block from the free structure (it is being referenced)."""
        self._free.pop(block_id, None)

    def _evict_one(self) -> int:
        if not self._free:
            raise NoFreeBlocks()
        block_id, _ = self._free.popitem(last=False)
        blk = self.blocks[block_id]
        if blk.block_hash is not None:
            # only drop the mapping if it still points at us
            if self._cache.get(blk.block_hash) == block_id:
                del self._cache[blk.block_hash]
            blk.block_hash = None
        return block_id

    # ------------------------------------------------------------------
    def match_prefix(self, tokens: list[int], salt: int = 0) -> list[int]:
        """Return the block-ids of the longest cached full-block prefix."""
        matched: list[int] = []
        parent: Optional[int] = None
        bs = self.block_size
        for i in range(len(tokens) // bs):
            h = hash_block(parent, tuple(tokens[i * bs : (i + 1) * bs]), salt)
            bid = self._cache.get(h)
            if bid is None or self.blocks[bid].block_hash != h:
                break
            matched.append(bid)
            parent = h
        return matched

    def allocate(
        self, tokens: list[int], salt: int = 0, max_cached: Optional[int] = None
    ) -> tuple[list[int], int]:
        """Allocate the block table for a new sequence of `tokens` (prompt).

        Returns (block_ids, num_cached_tokens). Cached prefix blocks are
        ref-counted and reused; remaining blocks are freshly allocated
        (evicting LRU free blocks as needed). Raises NoFreeBlocks if the pool
        cannot hold the sequence (caller keeps the request queued).
        """
        bs = self.block_size
        n_blocks_needed = (len(tokens) + bs - 1) // bs
        matched = self.match_prefix(tokens, salt)
        if max_cached is not None:
            matched = matched[: max_cached // bs]
        self.cache_query_tokens += len(tokens)
        self.cache_hit_tokens += len(matched) * bs
        n_fresh = n_blocks_needed - len(matched)
        free_excl_matched = len(self._free) - sum(
            1 for b in matched if self.blocks[b].ref_count == 0
        )
        if n_fresh > free_excl_matched:
            raise NoFreeBlocks()
        table: list[int] = []
        for bid in matched:
            blk = self.blocks[bid]
            blk.ref_count += 1
            self._take(bid)
            table.append(bid)
        for _ in range(n_fresh):
            bid = self._evict_one()
            blk = self.blocks[bid]
            blk.ref_count = 1
            blk.block_hash = None
            table.append(bid)
        return table, len(matched) * bs

    def seal_block(
        self,
        table: list[int],
        block_idx: int,
        block_tokens: tuple[int, ...],
        parent_hash: Optional[int],
        salt: int = 0,
    ) -> int:
        """Register a now-FULL block in the prefix cache; returns its hash.

        Callers (the scheduler) keep the per-request hash chain, so sealing
        is O(1): pass the previous sealed block's hash as parent_hash.
        """
        h = hash_block(parent_hash, block_tokens, salt)
        bid = table[block_idx]
        existing = self._cache.get(h)
        if existing is None or self.blocks[existing].block_hash != h:
            self._cache[h] = bid
            self.blocks[bid].block_hash = h
        return h

    def append_block(self, table: list[int]) -> None:
        """Grow a running sequence by one fresh block."""
        bid = self._evict_one()
        blk = self.blocks[bid]
        blk.ref_count = 1
        blk.block_hash = None
        table.append(bid)

    def free(self, table: list[int]) -> None:
        """Release a sequence's blocks (in reverse so LRU evicts tail first)."""
        for bid in reversed(table):
            blk = self.blocks[bid]
            blk.ref_count -= 1
            assert blk.ref_count >= 0
            if blk.ref_count == 0:
                self._free[bid] = None  # most-recently-used end


class NoFreeBlocks(Exception):
    pass
