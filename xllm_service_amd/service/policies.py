"""Load-balance policies: RR, cache-aware routing (CAR), SLO-aware with
adaptive P<->D role reassignment.
(reference: scheduler/loadbalance_policy/ + select_instance_pair_on_slo in
instance_mgr.cpp:905-1021 — SURVEY.md 2.5/2.7)
"""
from __future__ import annotations

import logging
import time
from dataclasses import dataclass
from typing import Dict, List, Optional

from .instance_mgr import Instance, InstanceMgr
from .kvcache_mgr import GlobalKVCacheMgr
from .time_predictor import TPOTPredictor, TTFTPredictor
from .types import InstanceType

log = logging.getLogger("xllm.policy")


@dataclass
class SelectedPair:
    prefill: Optional[Instance]
    decode: Optional[Instance]

    @property
    def ok(self) -> bool:
        return self.prefill is not None or self.decode is not None


class LoadBalancePolicy:
    name = "base"

    def __init__(self, mgr: InstanceMgr, kv: GlobalKVCacheMgr):
        self.mgr = mgr
        self.kv = kv

    def select_instances_pair(self, token_ids: List[int]) -> SelectedPair:
        raise NotImplementedError


class RoundRobinPolicy(LoadBalancePolicy):
    name = "RR"

    def select_instances_pair(self, token_ids):
        p, d = self.mgr.next_rr_pair()
        return SelectedPair(p, d)


class CacheAwarePolicy(LoadBalancePolicy):
    """score = matched/max_blocks - gpu_cache_usage - waiting/max_waiting,
    evaluated independently for the prefill and decode pools; falls back to
    least-loaded when no prefix overlaps.
    (reference: cache_aware_routing.cpp:22-85)

    On top of the reference's heartbeat-reported load, the score includes
    the scheduler's OPTIMISTIC in-flight counters (num_decoding /
    pending_prefill_tokens, incremented at dispatch): heartbeats arrive
    every ~1-3 s, so within a request burst the reported load is stale and
    score ties would send the whole burst to one instance."""
    name = "CAR"
    MAX_WAITING = 64.0

    def _score(self, inst: Instance, overlap: Dict[str, float],
               max_blocks: float) -> float:
        s = 0.0
        if max_blocks > 0:
            s += overlap.get(inst.name, 0.0) / max_blocks
        s -= inst.load.gpu_cache_usage_perc
        inflight = (inst.num_decoding +
                    inst.pending_prefill_tokens / 1024.0)
        s -= (inst.load.waiting_requests_num + inflight) / self.MAX_WAITING
        return s

    def select_instances_pair(self, token_ids):
        prefills = self.mgr.schedulable_prefills()
        decodes = self.mgr.schedulable_decodes()
        if not prefills:
            return SelectedPair(None, None)
        ov = self.kv.match(token_ids or [])
        nblocks = max(len(token_ids or []) // self.kv.block_size, 1)
        prefill = max(prefills, key=lambda i: self._score(i, ov.scores, nblocks))
        decode = None
        if decodes:
            decode = max(decodes, key=lambda i: self._score(i, ov.scores, nblocks))
        elif prefill.itype != InstanceType.DEFAULT:
            return SelectedPair(None, None)
        return SelectedPair(prefill, decode)


class SloAwarePolicy(LoadBalancePolicy):
    """Pick the first decode whose predicted TPOT meets target_tpot (else
    min-TPOT), prefill with min predicted finish time; spill prefill onto an
    underloaded decode when prefill misses target_ttft; flip P<->D roles
    under sustained pressure.
    (reference: select_instance_pair_on_slo, instance_mgr.cpp:905-1063)"""
    name = "SLO_AWARE"

    def __init__(self, mgr, kv, target_ttft_ms: float = 1000.0,
                 target_tpot_ms: float = 50.0, flip_cooldown_s: float = 10.0):
        super().__init__(mgr, kv)
        self.target_ttft_ms = target_ttft_ms
        self.target_tpot_ms = target_tpot_ms
        self.flip_cooldown_s = flip_cooldown_s
        self._last_flip = 0.0
        self.ttft: Dict[str, TTFTPredictor] = {}
        self.tpot: Dict[str, TPOTPredictor] = {}

    # ---- observation ingestion ---------------------------------------------
    def seed_from_meta(self, instance: str, ttft_profile, tpot_profile):
        """Registration-time profiling samples (InstanceMetaInfo
        ttft_profile/tpot_profile) pre-seed the predictors so SLO-aware
        selection works before any runtime observations accrue."""
        for row in ttft_profile or []:
            if len(row) == 2:
                self.observe_ttft(instance, int(row[0]), float(row[1]))
        for row in tpot_profile or []:
            if len(row) == 3:
                self.observe_tpot(instance, int(row[0]), int(row[1]),
                                  float(row[2]))

    def observe_ttft(self, instance: str, num_tokens: int, ttft_ms: float):
        self.ttft.setdefault(instance, TTFTPredictor()).add_sample(
            num_tokens, ttft_ms)

    def observe_tpot(self, instance: str, batch: int, tokens: int,
                     tpot_ms: float):
        self.tpot.setdefault(instance, TPOTPredictor()).add_sample(
            batch, tokens, tpot_ms)

    # ---- selection ----------------------------------------------------------
    def _predicted_tpot(self, inst: Instance, extra_tokens: int) -> float:
        pred = self.tpot.get(inst.name)
        batch = inst.num_decoding + 1
        if pred is None:
            return 5.0 * batch  # cold start: prefer empty decodes
        return pred.predict(batch, extra_tokens)

    def _predicted_prefill_finish(self, inst: Instance,
                                  num_tokens: int) -> float:
        pred = self.ttft.get(inst.name)
        queued = inst.pending_prefill_tokens + num_tokens
        if pred is None:
            return 0.05 * queued
        return pred.predict(queued)

    def select_instances_pair(self, token_ids):
        n_tokens = len(token_ids or [])
        prefills = self.mgr.schedulable_prefills()
        decodes = self.mgr.schedulable_decodes()
        if not prefills and not decodes:
            return SelectedPair(None, None)
        if not decodes:
            if prefills and prefills[0].itype == InstanceType.DEFAULT:
                return SelectedPair(min(
                    prefills,
                    key=lambda i: self._predicted_prefill_finish(i, n_tokens)),
                    None)
            return SelectedPair(None, None)

        # decode choice: first meeting target, else min predicted TPOT
        scored = [(self._predicted_tpot(d, n_tokens), d) for d in decodes]
        scored.sort(key=lambda x: x[0])
        best_tpot, decode = scored[0]
        for tp, d in scored:
            if tp <= self.target_tpot_ms:
                best_tpot, decode = tp, d
                break

        # prefill choice: min predicted finish
        prefill = None
        if prefills:
            pf = [(self._predicted_prefill_finish(p, n_tokens), p)
                  for p in prefills]
            pf.sort(key=lambda x: x[0])
            best_ttft, prefill = pf[0]
            # prefill overloaded + an underloaded decode exists:
            # prefill-on-decode spillover
            if best_ttft > self.target_ttft_ms:
                idle = [d for d in decodes if d.num_decoding == 0
                        and d is not decode]
                if idle:
                    prefill = idle[0]
                elif best_tpot < self.target_tpot_ms * 0.5:
                    self._maybe_flip("decode_to_prefill")
        # no decode meets target & prefill pool underloaded -> flip P->D
        if best_tpot > self.target_tpot_ms and prefill is not None:
            if self._predicted_prefill_finish(prefill, 0) < \
                    self.target_ttft_ms * 0.3:
                self._maybe_flip("prefill_to_decode")
        return SelectedPair(prefill, decode)

    def _maybe_flip(self, direction: str):
        now = time.monotonic()
        if now - self._last_flip < self.flip_cooldown_s:
            return
        if direction == "prefill_to_decode":
            cands = [p for p in self.mgr.schedulable_prefills()
                     if p.itype in (InstanceType.MIX, InstanceType.PREFILL)]
            if cands and self.mgr.flip_instance_role(cands[-1].name, "decode"):
                self._last_flip = now
                log.info("SLO flip: %s prefill -> decode", cands[-1].name)
        else:
            # a decode drained to zero requests may flip back to prefill
            cands = [d for d in self.mgr.schedulable_decodes()
                     if d.num_decoding == 0
                     and d.itype in (InstanceType.MIX, InstanceType.DECODE)]
            if cands and self.mgr.flip_instance_role(cands[0].name, "prefill"):
                self._last_flip = now
                log.info("SLO flip: %s decode -> prefill", cands[0].name)


def create_policy(name: str, mgr: InstanceMgr, kv: GlobalKVCacheMgr,
                  **kwargs) -> LoadBalancePolicy:
    name = name.upper()
    if name == "RR":
        return RoundRobinPolicy(mgr, kv)
    if name == "CAR":
        return CacheAwarePolicy(mgr, kv)
    if name == "SLO_AWARE":
        return SloAwarePolicy(mgr, kv, **kwargs)
    raise ValueError(f"unknown load_balance_policy {name!r}")
