"""Paged KV-cache block manager with prefix caching.

Block accounting only (the tensors live in ModelRunner). Design:
  * fixed pool of `num_blocks` blocks of `block_size` tokens
  * ref-counted blocks; full blocks are content-addressed by chained
    XXH3-128 hashes (utils/hashing.py) for prefix-cache reuse
  * freed cached blocks go to an LRU of evictable blocks and are only
    reclaimed when the free list runs dry (so recent prefixes stay warm
    in the 288 GB HBM3E pool)

Emits KvCacheEvents (stored/removed hashes) for the service-level global
prefix index (reference: GlobalKVCacheMgr heartbeat updates).
"""
from __future__ import annotations

from collections import OrderedDict
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Set

from xllm_service_amd.utils.hashing import chain_block_hashes

from .sequence import Sequence


@dataclass
class KvCacheEvents:
    stored: Set[bytes] = field(default_factory=set)
    removed: Set[bytes] = field(default_factory=set)

    def drain(self) -> "KvCacheEvents":
        out = KvCacheEvents(set(self.stored), set(self.removed))
        self.stored.clear()
        self.removed.clear()
        return out


class BlockManager:
    def __init__(self, num_blocks: int, block_size: int,
                 enable_prefix_caching: bool = True):
        self.num_blocks = num_blocks
        self.block_size = block_size
        self.enable_prefix_caching = enable_prefix_caching
        self.free_blocks: List[int] = list(range(num_blocks - 1, -1, -1))
        self.ref_count = [0] * num_blocks
        self.block_hash: List[Optional[bytes]] = [None] * num_blocks
        self.hash_to_block: Dict[bytes, int] = {}
        # blocks with ref 0 but still holding cached content (LRU order)
        self.evictable: "OrderedDict[int, None]" = OrderedDict()
        self.events = KvCacheEvents()

    # ---- low-level ---------------------------------------------------------
    @property
    def num_free(self) -> int:
        return len(self.free_blocks) + len(self.evictable)

    def usage(self) -> float:
        return 1.0 - self.num_free / max(self.num_blocks, 1)

    def _pop_free(self) -> int:
        if self.free_blocks:
            return self.free_blocks.pop()
        # evict LRU cached block
        blk, _ = self.evictable.popitem(last=False)
        h = self.block_hash[blk]
        if h is not None:
            del self.hash_to_block[h]
            self.block_hash[blk] = None
            self.events.removed.add(h)
        return blk

    def _incref(self, blk: int):
        if self.ref_count[blk] == 0 and blk in self.evictable:
            del self.evictable[blk]
        self.ref_count[blk] += 1

    def _decref(self, blk: int):
        self.ref_count[blk] -= 1
        assert self.ref_count[blk] >= 0
        if self.ref_count[blk] == 0:
            if self.block_hash[blk] is not None and self.enable_prefix_caching:
                self.evictable[blk] = None
            else:
                self.block_hash[blk] = None
                self.free_blocks.append(blk)

    # ---- sequence-level API ------------------------------------------------
    def match_prefix(self, seq: Sequence) -> int:
        """Longest cached prefix of the prompt, in tokens (multiple of bs)."""
        if not self.enable_prefix_caching:
            return 0
        hashes = chain_block_hashes(seq.prompt_token_ids, self.block_size)
        n = 0
        for h in hashes:
            if h in self.hash_to_block:
                n += 1
            else:
                break
        # never match the whole prompt (need >= 1 uncomputed token)
        max_match = (seq.prompt_len - 1) // self.block_size
        return min(n, max_match) * self.block_size

    def can_allocate(self, seq: Sequence, num_tokens: int,
                     reserve_blocks: int = 0) -> bool:
        """Can we hold the first num_tokens of this sequence while leaving
        reserve_blocks free (blocks already promised to scheduled decodes)?"""
        cached = self.match_prefix(seq) if not seq.block_table else 0
        need = (num_tokens + self.block_size - 1) // self.block_size
        need -= cached // self.block_size
        return need <= self.num_free - reserve_blocks

    def allocate_prefill(self, seq: Sequence) -> int:
        """Allocate blocks for the whole prompt; returns cached-token count.

        Cached prefix blocks are shared (ref++); the rest come from the free
        pool. Must be preceded by can_allocate().
        """
        assert not seq.block_table
        cached_tokens = self.match_prefix(seq)
        hashes = chain_block_hashes(seq.prompt_token_ids, self.block_size)
        n_cached_blocks = cached_tokens // self.block_size
        for i in range(n_cached_blocks):
            blk = self.hash_to_block[hashes[i]]
            self._incref(blk)
            seq.block_table.append(blk)
        total_blocks = (seq.prompt_len + self.block_size - 1) // self.block_size
        for i in range(n_cached_blocks, total_blocks):
            blk = self._pop_free()
            self.ref_count[blk] = 1
            seq.block_table.append(blk)
        seq.num_computed_tokens = cached_tokens
        return cached_tokens

    def register_full_blocks(self, seq: Sequence):
        """Content-address prompt blocks that are now fully computed."""
        if not self.enable_prefix_caching:
            return
        hashes = chain_block_hashes(seq.prompt_token_ids, self.block_size)
        full = min(seq.num_computed_tokens // self.block_size, len(hashes))
        for i in range(full):
            blk = seq.block_table[i]
            if self.block_hash[blk] is None and hashes[i] not in self.hash_to_block:
                self.block_hash[blk] = hashes[i]
                self.hash_to_block[hashes[i]] = blk
                self.events.stored.add(hashes[i])

    def needs_append_block(self, seq: Sequence) -> bool:
        # decode writes KV for position total_len - 1 (the freshly-fed token)
        return (seq.total_len - 1) // self.block_size >= len(seq.block_table)

    def can_append(self, seq: Sequence, reserve_blocks: int = 0) -> bool:
        need = 1 if self.needs_append_block(seq) else 0
        return need <= self.num_free - reserve_blocks

    def append_slot(self, seq: Sequence) -> int:
        """Slot index for the token being decoded (block alloc on boundary)."""
        pos = seq.total_len - 1
        if pos // self.block_size >= len(seq.block_table):
            blk = self._pop_free()
            self.ref_count[blk] = 1
            seq.block_table.append(blk)
        blk = seq.block_table[pos // self.block_size]
        return blk * self.block_size + pos % self.block_size

    def free(self, seq: Sequence):
        for blk in seq.block_table:
            self._decref(blk)
        seq.block_table = []

    def reset_prefix_cache(self) -> int:
        """Drop every evictable cached block (role flip: a decode role
        wants its pool for long-lived decode KV, not prefill prefix reuse).
        Returns the number of blocks released to the free list."""
        n = len(self.evictable)
        for blk in list(self.evictable):
            h = self.block_hash[blk]
            if h is not None:
                self.hash_to_block.pop(h, None)
                self.block_hash[blk] = None
                self.events.removed.add(h)
        self.free_blocks.extend(self.evictable)
        self.evictable.clear()
        return n

    # ---- migration support (PD disaggregation) -----------------------------
    def allocate_raw(self, n: int) -> List[int]:
        """Allocate n unhashed blocks (decode side of a KV migration)."""
        assert n <= self.num_free
        out = []
        for _ in range(n):
            blk = self._pop_free()
            self.ref_count[blk] = 1
            out.append(blk)
        return out
