"""Worker-level continuous-batching scheduler.

Each call to schedule() builds one engine step:
  * running sequences decode one token each (preempting — offline first,
    then newest-online-first — when the block pool is exhausted)
  * waiting sequences are admitted with CHUNKED prefill under a per-step
    token budget, online (priority 0) ahead of offline (priority 1)

This is the per-instance half of the hybrid online/offline scheduling the
service layer drives (reference: Request::offline request/request.h:41 and
the SLO policy, SURVEY.md 2.5/2.7 — implemented for real here).
"""
from __future__ import annotations

from collections import deque
from dataclasses import dataclass, field
from typing import Deque, List, Optional

from .block_manager import BlockManager
from .sequence import Sequence, SeqStatus


@dataclass
class ScheduledPrefill:
    seq: Sequence
    chunk_start: int     # first prompt token this step
    chunk_len: int


@dataclass
class StepPlan:
    prefills: List[ScheduledPrefill] = field(default_factory=list)
    decodes: List[Sequence] = field(default_factory=list)
    preempted: List[Sequence] = field(default_factory=list)

    @property
    def empty(self) -> bool:
        return not self.prefills and not self.decodes


class EngineScheduler:
    def __init__(self, block_manager: BlockManager,
                 max_num_seqs: int = 256,
                 max_batched_tokens: int = 8192,
                 enable_chunked_prefill: bool = True,
                 prefill_hold_ms: float = 0.0,
                 swap_out=None, swap_in=None):
        self.bm = block_manager
        self.max_num_seqs = max_num_seqs
        self.max_batched_tokens = max_batched_tokens
        self.enable_chunked_prefill = enable_chunked_prefill
        # batch scattered arrivals into fewer (eager) prefill steps: hold
        # new admissions up to this long while decodes keep the hipGraph
        # fast path (serving arrivals otherwise pay one full eager step
        # per prompt; TTFT cost is bounded by the hold)
        self.prefill_hold_ms = prefill_hold_ms
        self._last_admit = 0.0
        # swap_out(seq) -> cpu_blocks | None; swap_in(seq) -> None
        # (engine wires these to the host-DRAM KV tier; None = recompute)
        self.swap_out = swap_out
        self.swap_in = swap_in
        self.waiting: Deque[Sequence] = deque()
        self.swapped: List[Sequence] = []
        self.running: List[Sequence] = []
        self.num_swap_outs = 0
        self.num_swap_ins = 0
        self.num_preempts = 0

    # ---- API ----------------------------------------------------------------
    def add(self, seq: Sequence):
        if seq.priority == 0:
            # online requests queue ahead of every offline request
            idx = len(self.waiting)
            for i, s in enumerate(self.waiting):
                if s.priority > 0:
                    idx = i
                    break
            self.waiting.insert(idx, seq)
        else:
            self.waiting.append(seq)

    def abort(self, request_id: str) -> Optional[Sequence]:
        for i, s in enumerate(self.running):
            if s.request_id == request_id:
                s.status = SeqStatus.FINISHED_ABORT
                self.bm.free(s)
                self.running.pop(i)
                return s
        for i, s in enumerate(self.waiting):
            if s.request_id == request_id:
                s.status = SeqStatus.FINISHED_ABORT
                if s.block_table:
                    self.bm.free(s)
                del self.waiting[i]
                return s
        for i, s in enumerate(self.swapped):
            if s.request_id == request_id:
                s.status = SeqStatus.FINISHED_ABORT
                if self.swap_in is not None:
                    # free the dram blocks without copying back
                    self.free_cpu_blocks(s)
                del self.swapped[i]
                return s
        return None

    # engine injects this (frees dram blocks of an aborted swapped seq)
    free_cpu_blocks = staticmethod(lambda seq: None)

    def has_work(self) -> bool:
        return bool(self.waiting or self.running or self.swapped)

    @property
    def num_waiting(self) -> int:
        return len(self.waiting)

    # ---- core ---------------------------------------------------------------
    def _do_preempt(self, victim: Sequence):
        """Swap the victim's KV to the dram tier when possible, else free +
        recompute (reference behaviour is client-side retry; we keep the
        request and restore it)."""
        self.running.remove(victim)
        victim.preempt_count += 1
        self.num_preempts += 1
        if self.swap_out is not None and victim.prefill_done:
            cpu_blocks = self.swap_out(victim)
            if cpu_blocks is not None:
                victim.cpu_block_table = cpu_blocks
                self.bm.free(victim)
                victim.status = SeqStatus.SWAPPED
                self.swapped.append(victim)
                self.num_swap_outs += 1
                return
        self.bm.free(victim)
        victim.status = SeqStatus.PREEMPTED
        victim.num_computed_tokens = 0
        n_out = len(victim.output_token_ids)
        if n_out:
            # fold emitted tokens into the recompute prompt, keeping the
            # emitted-token count so max_tokens/min_tokens/usage stay exact
            if victim.mrope_pos is not None:
                # extend the 3-D M-RoPE table for the folded decode tokens:
                # a text token at overall index i decodes at position
                # i + mrope_delta on all three rows (model_runner decode path)
                import numpy as np
                cur = victim.mrope_pos.shape[1]
                cols = (np.arange(cur, cur + n_out,
                                  dtype=victim.mrope_pos.dtype)
                        + victim.mrope_delta)
                victim.mrope_pos = np.concatenate(
                    [victim.mrope_pos, np.tile(cols, (3, 1))], axis=1)
            victim.prompt_token_ids = victim.all_token_ids()
            victim.output_token_ids = []
            victim.num_folded_output_tokens += n_out
        self.waiting.appendleft(victim)

    def _preempt_one(self) -> bool:
        """Preempt the lowest-priority, most recent running sequence."""
        if not self.running:
            return False
        victim_idx = None
        # offline victims first, most recent first
        for i in range(len(self.running) - 1, -1, -1):
            if self.running[i].priority > 0:
                victim_idx = i
                break
        if victim_idx is None:
            victim_idx = len(self.running) - 1
        victim = self.running[victim_idx]
        self._do_preempt(victim)
        return True

    def schedule(self) -> StepPlan:
        plan = StepPlan()
        budget = self.max_batched_tokens

        # 1. decodes for all running seqs (preempt on OOM). `reserved`
        # counts free blocks already promised to earlier decodes this step.
        reserved = 0
        for seq in list(self.running):
            if seq not in self.running:      # became a preemption victim
                continue
            if not seq.prefill_done:
                continue                     # mid-chunked-prefill: step 2
            while (not self.bm.can_append(seq, reserved)
                   and self._preempt_victim_excluding(seq, plan)):
                pass
            if self.bm.can_append(seq, reserved):
                plan.decodes.append(seq)
                if self.bm.needs_append_block(seq):
                    reserved += 1
                budget -= 1
            else:                            # pool exhausted: preempt self
                self._do_preempt(seq)
                plan.preempted.append(seq)

        # 2. continue chunked prefills already running
        for seq in self.running:
            if seq.prefill_done or budget <= 0:
                continue
            chunk = min(seq.prompt_len - seq.num_computed_tokens, budget)
            if chunk > 0:
                plan.prefills.append(
                    ScheduledPrefill(seq, seq.num_computed_tokens, chunk))
                budget -= chunk

        # 2.6 resume swapped sequences (dram -> hbm) before new admissions
        for seq in list(self.swapped):
            need = len(seq.cpu_block_table) + 1   # +1: next decode block
            if len(self.running) >= self.max_num_seqs or \
                    self.bm.num_free - reserved < need:
                break
            reserved += 1
            self.swap_in(seq)
            self.num_swap_ins += 1
            seq.status = SeqStatus.RUNNING
            self.swapped.remove(seq)
            self.running.append(seq)
            if seq.prefill_done:
                budget -= 1
                plan.decodes.append(seq)

        # 3. admit waiting sequences (leaving the blocks this step's decodes
        # will take in append_slot untouched)
        if (self.prefill_hold_ms > 0 and self.waiting and plan.decodes
                and not plan.prefills and len(self.waiting) < 4):
            import time as _t
            now = _t.monotonic()
            if (now - self._last_admit) * 1000.0 < self.prefill_hold_ms:
                return plan              # hold: decode-only graph step
            self._last_admit = now
        while self.waiting and budget > 0 and len(self.running) < self.max_num_seqs:
            seq = self.waiting[0]
            first_alloc = not seq.block_table
            if first_alloc:
                if not self.bm.can_allocate(seq, seq.prompt_len,
                                            reserve_blocks=reserved):
                    break
                self.bm.allocate_prefill(seq)
            remaining = seq.prompt_len - seq.num_computed_tokens
            chunk = min(remaining, budget)
            if not self.enable_chunked_prefill and chunk < remaining:
                if first_alloc:
                    self.bm.free(seq)
                    seq.num_computed_tokens = 0
                break
            if chunk <= 0:
                break
            self.waiting.popleft()
            seq.status = SeqStatus.RUNNING
            self.running.append(seq)
            plan.prefills.append(
                ScheduledPrefill(seq, seq.num_computed_tokens, chunk))
            budget -= chunk

        return plan

    def _preempt_victim_excluding(self, protected: Sequence,
                                  plan: StepPlan) -> bool:
        candidates = [s for s in self.running
                      if s is not protected and s not in plan.decodes]
        if not candidates:
            return False
        # offline first, then most recent online
        victim = None
        for s in reversed(candidates):
            if s.priority > 0:
                victim = s
                break
        if victim is None:
            victim = candidates[-1]
        self._do_preempt(victim)
        plan.preempted.append(victim)
        return True

    # ---- bookkeeping after a step -------------------------------------------
    def on_step_done(self, plan: StepPlan):
        """Advance prefill progress; register full prefix blocks; retire
        finished sequences."""
        for sp in plan.prefills:
            sp.seq.num_computed_tokens += sp.chunk_len
            if sp.seq.prefill_done:
                self.bm.register_full_blocks(sp.seq)
        done = [s for s in self.running if s.status.finished]
        for s in done:
            if not s.hold_blocks:
                self.bm.free(s)
            self.running.remove(s)
