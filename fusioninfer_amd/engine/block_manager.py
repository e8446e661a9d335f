"""Paged-KV block manager: allocation, ref-counting, prefix-cache reuse.

Owns the mapping sequence -> physical cache blocks. Hash-based prefix
caching mirrors the reuse the reference's EPP prefix-cache scorer assumes
exists engine-side (reference pkg/router/strategy.go:51-68): full prompt
blocks are content-hashed (chained) and reusable across sequences.

Cache hits are consumed by the context-attention prefill path: the runner
skips cached tokens entirely and the paged prefill kernel attends the new
tokens over the cached blocks (ops/csrc/prefill_attention.hip, PAGED).
"""

from __future__ import annotations

from collections import OrderedDict
from typing import Dict, List

from fusioninfer_amd.engine.sequence import Sequence


class BlockManager:
    def __init__(self, num_blocks: int, block_size: int,
                 enable_prefix_caching: bool = False):
        self.num_blocks = num_blocks
        self.block_size = block_size
        self.enable_prefix_caching = enable_prefix_caching
        self.free_blocks: List[int] = list(range(num_blocks))
        self.ref_count: Dict[int, int] = {}
        # content hash -> block id, for FULL blocks only
        self.hash_to_block: Dict[int, int] = {}
        self.block_hash: Dict[int, int] = {}
        # blocks with ref 0 that still hold reusable content (LRU)
        self.cached_free: "OrderedDict[int, None]" = OrderedDict()
        # prefix-cache effectiveness counters (EPP's prefix scorer assumes
        # engine-side reuse; these make it observable)
        self.cache_query_tokens = 0
        self.cache_hit_tokens = 0

    # ------------------------------------------------------------ helpers
    def num_free(self) -> int:
        return len(self.free_blocks) + len(self.cached_free)

    def _pop_free_block(self) -> int:
        if self.free_blocks:
            return self.free_blocks.pop()
        # evict LRU cached block
        blk, _ = self.cached_free.popitem(last=False)
        h = self.block_hash.pop(blk, None)
        if h is not None:
            self.hash_to_block.pop(h, None)
        return blk

    def blocks_needed(self, num_tokens: int) -> int:
        return (num_tokens + self.block_size - 1) // self.block_size

    # ------------------------------------------------------------ alloc
    def can_allocate(self, num_tokens: int) -> bool:
        return self.num_free() >= self.blocks_needed(num_tokens)

    def allocate(self, seq: Sequence) -> None:
        """Allocate blocks for the whole prompt; reuse prefix-cache hits."""
        assert not seq.block_ids
        n = self.blocks_needed(seq.num_prompt_tokens)
        cached = 0
        if self.enable_prefix_caching:
            h = 0
            for b in range(n):
                start, end = b * self.block_size, (b + 1) * self.block_size
                # `>=`: at least ONE prompt token must be recomputed so the
                # sequence produces logits — and that token then lands in a
                # FRESH block, so shared (cached) blocks are never written.
                if end >= seq.num_prompt_tokens:
                    break
                h = hash((h, tuple(seq.prompt_token_ids[start:end])))
                blk = self.hash_to_block.get(h)
                if blk is None:
                    break
                self._take(blk)
                seq.block_ids.append(blk)
                cached += self.block_size
        seq.num_cached_tokens = cached
        if self.enable_prefix_caching:
            self.cache_query_tokens += seq.num_prompt_tokens
            self.cache_hit_tokens += cached
        while len(seq.block_ids) < n:
            blk = self._pop_free_block()
            self.ref_count[blk] = 1
            seq.block_ids.append(blk)
        if self.enable_prefix_caching:
            self._register_hashes(seq)

    def allocate_raw(self, seq: Sequence, num_blocks: int) -> None:
        """Allocate exactly num_blocks fresh blocks, bypassing the prefix
        cache (swap-in restore: the blocks will be overwritten with the
        sequence's saved KV, so cache sharing would corrupt shared data)."""
        assert not seq.block_ids
        for _ in range(num_blocks):
            blk = self._pop_free_block()
            self.ref_count[blk] = 1
            seq.block_ids.append(blk)
        seq.num_cached_tokens = 0

    def _take(self, blk: int) -> None:
        if blk in self.cached_free:
            del self.cached_free[blk]
            self.ref_count[blk] = 1
        else:
            self.ref_count[blk] += 1

    def _register_hashes(self, seq: Sequence) -> None:
        h = 0
        for b, blk in enumerate(seq.block_ids):
            end = (b + 1) * self.block_size
            if end > seq.num_prompt_tokens:
                break
            start = b * self.block_size
            h = hash((h, tuple(seq.prompt_token_ids[start:end])))
            if blk not in self.block_hash:
                self.block_hash[blk] = h
                self.hash_to_block.setdefault(h, blk)

    def _blocks_missing(self, seq: Sequence) -> int:
        """Blocks needed so the NEXT decode position (num_tokens-1) has a slot."""
        position = seq.num_tokens - 1
        have = len(seq.block_ids)
        need = position // self.block_size + 1
        return max(need - have, 0)

    def can_append_slot(self, seq: Sequence) -> bool:
        return self.num_free() >= self._blocks_missing(seq)

    def append_slot(self, seq: Sequence) -> None:
        """Ensure a block exists for position seq.num_tokens - 1."""
        for _ in range(self._blocks_missing(seq)):
            blk = self._pop_free_block()
            self.ref_count[blk] = 1
            seq.block_ids.append(blk)

    def extra_blocks_for(self, seq: Sequence, position: int) -> int:
        """Blocks missing for `position` to have a slot (spec-decode draft
        tails extend past the single decode slot the scheduler reserved)."""
        need = position // self.block_size + 1
        return max(need - len(seq.block_ids), 0)

    def append_slots_upto(self, seq: Sequence, position: int) -> None:
        for _ in range(self.extra_blocks_for(seq, position)):
            blk = self._pop_free_block()
            self.ref_count[blk] = 1
            seq.block_ids.append(blk)

    def free(self, seq: Sequence) -> None:
        for blk in seq.block_ids:
            self.ref_count[blk] -= 1
            if self.ref_count[blk] == 0:
                del self.ref_count[blk]
                if self.enable_prefix_caching and blk in self.block_hash:
                    self.cached_free[blk] = None
                else:
                    self.free_blocks.append(blk)
        seq.block_ids = []

    def slot_for(self, seq: Sequence, position: int) -> int:
        blk = seq.block_ids[position // self.block_size]
        return blk * self.block_size + position % self.block_size
