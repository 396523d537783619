"""GlobalKVCacheMgr: the cluster-wide prefix-cache index.

Chained XXH3-128 block hashes (utils/hashing.py, same scheme the worker
block manager uses) map to the set of instances holding that block, per
memory tier (hbm/dram/ssd). Heartbeat KvCacheEvents move instances between
tiers; the master batches dirty keys to the registry under XLLM:CACHE: so
replica masters share the index via watch.
(reference: scheduler/managers/global_kvcache_mgr.{h,cpp})
"""
from __future__ import annotations

import json
from collections import defaultdict
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional, Set

from xllm_service_amd.registry.server import RegistryClient
from xllm_service_amd.registry.store import WatchEvent
from xllm_service_amd.utils.hashing import chain_block_hashes

from .types import KEY_CACHE

TIERS = ("hbm", "dram", "ssd")
# score weight per tier: hbm hits are worth more than ssd hits
TIER_WEIGHT = {"hbm": 1.0, "dram": 0.7, "ssd": 0.4}


@dataclass
class CacheLocations:
    hbm: Set[str] = field(default_factory=set)
    dram: Set[str] = field(default_factory=set)
    ssd: Set[str] = field(default_factory=set)

    def empty(self) -> bool:
        return not (self.hbm or self.dram or self.ssd)

    def to_dict(self):
        return {"hbm": sorted(self.hbm), "dram": sorted(self.dram),
                "ssd": sorted(self.ssd)}

    @classmethod
    def from_dict(cls, d):
        return cls(set(d.get("hbm", [])), set(d.get("dram", [])),
                   set(d.get("ssd", [])))


@dataclass
class OverlapScores:
    """instance -> weighted matched-block score, plus total match length."""
    scores: Dict[str, float] = field(default_factory=dict)
    matched_blocks: int = 0


class GlobalKVCacheMgr:
    def __init__(self, registry: Optional[RegistryClient] = None,
                 block_size: int = 16,
                 is_master: Callable[[], bool] = lambda: True):
        self.registry = registry
        self.block_size = block_size
        self.is_master = is_master
        self.index: Dict[bytes, CacheLocations] = defaultdict(CacheLocations)
        self._dirty: Set[bytes] = set()

    async def start(self):
        if self.registry is None:
            return
        await self.registry.watch(KEY_CACHE, self._on_cache_event)
        for key, value in await self.registry.range(KEY_CACHE):
            h = bytes.fromhex(key[len(KEY_CACHE):])
            self.index[h] = CacheLocations.from_dict(json.loads(value))

    async def _on_cache_event(self, ev: WatchEvent):
        if self.is_master():
            return  # master is the writer; replicas sync via watch
        h = bytes.fromhex(ev.key[len(KEY_CACHE):])
        if ev.type == "put":
            self.index[h] = CacheLocations.from_dict(json.loads(ev.value))
        else:
            self.index.pop(h, None)

    # ---- heartbeat ingestion ------------------------------------------------
    def record_updated_kvcaches(self, instance: str, stored: List[bytes],
                                removed: List[bytes],
                                offloaded: Optional[List[bytes]] = None):
        for h in stored:
            loc = self.index[h]
            loc.hbm.add(instance)
            loc.dram.discard(instance)
            loc.ssd.discard(instance)
            self._dirty.add(h)
        for h in offloaded or []:
            loc = self.index[h]
            loc.hbm.discard(instance)
            loc.dram.add(instance)
            self._dirty.add(h)
        for h in removed:
            loc = self.index.get(h)
            if loc is None:
                continue
            loc.hbm.discard(instance)
            loc.dram.discard(instance)
            loc.ssd.discard(instance)
            self._dirty.add(h)
            if loc.empty():
                del self.index[h]

    def remove_instance(self, instance: str):
        for h, loc in list(self.index.items()):
            if instance in loc.hbm or instance in loc.dram or instance in loc.ssd:
                loc.hbm.discard(instance)
                loc.dram.discard(instance)
                loc.ssd.discard(instance)
                self._dirty.add(h)
                if loc.empty():
                    del self.index[h]

    # ---- match --------------------------------------------------------------
    def match(self, token_ids: List[int]) -> OverlapScores:
        """Walk chained block hashes until the first global miss, scoring
        each instance by tier-weighted matched blocks."""
        out = OverlapScores()
        hashes = chain_block_hashes(token_ids, self.block_size)
        for h in hashes:
            loc = self.index.get(h)
            if loc is None or loc.empty():
                break
            out.matched_blocks += 1
            for tier in TIERS:
                w = TIER_WEIGHT[tier]
                for inst in getattr(loc, tier):
                    out.scores[inst] = out.scores.get(inst, 0.0) + w
        return out

    # ---- registry sync (master, 3 s cadence) --------------------------------
    async def upload_kvcache(self):
        if self.registry is None:
            return
        dirty, self._dirty = self._dirty, set()
        for h in dirty:
            key = KEY_CACHE + h.hex()
            loc = self.index.get(h)
            if loc is None or loc.empty():
                await self.registry.delete(key)
            else:
                await self.registry.put_json(key, loc.to_dict())
