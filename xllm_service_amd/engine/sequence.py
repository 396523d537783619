"""Sequence (request) state inside a worker engine."""
from __future__ import annotations

import enum
import time
from dataclasses import dataclass, field
from typing import List, Optional

from .sampling import SamplingParams


class SeqStatus(enum.Enum):
    WAITING = "waiting"
    RUNNING = "running"
    PREEMPTED = "preempted"
    SWAPPED = "swapped"         # KV offloaded to the host-DRAM tier
    FINISHED_STOP = "finished_stop"      # stop/eos token
    FINISHED_LENGTH = "finished_length"  # max_tokens reached
    FINISHED_ABORT = "finished_abort"    # cancelled (client disconnect etc.)

    @property
    def finished(self) -> bool:
        return self in (SeqStatus.FINISHED_STOP, SeqStatus.FINISHED_LENGTH,
                        SeqStatus.FINISHED_ABORT)


@dataclass(eq=False)   # identity eq/hash: `seq in running` must not compare
class Sequence:        # token lists element-wise (O(len) per membership test)
    request_id: str
    prompt_token_ids: List[int]
    params: SamplingParams
    eos_token_id: Optional[int] = None
    priority: int = 0           # 0 = online, 1 = offline (online preempts offline)
    arrival_time: float = field(default_factory=time.monotonic)

    status: SeqStatus = SeqStatus.WAITING
    output_token_ids: List[int] = field(default_factory=list)
    # recompute preemption folds already-emitted output tokens into
    # prompt_token_ids; this counter keeps max_tokens/min_tokens/usage
    # accounting correct across the fold
    num_folded_output_tokens: int = 0
    block_table: List[int] = field(default_factory=list)
    cpu_block_table: List[int] = field(default_factory=list)  # dram tier
    num_computed_tokens: int = 0          # prompt tokens already prefilled
    preempt_count: int = 0
    # multimodal: pre-computed image embeddings substituted at placeholder
    # token positions during prefill (EPD E->P handoff)
    mm_embeds: Optional["object"] = None      # torch.Tensor [n, hidden]
    mm_placeholder: Optional[int] = None
    # Qwen2-VL M-RoPE: [3, prompt_len] numpy position ids (None = 1-D rope)
    # and the text-position offset applied to every decode step
    mrope_pos: Optional["object"] = None
    mrope_delta: int = 0
    # PD-disaggregation: set on a decode instance receiving a migrated prefill
    migrated_in: bool = False
    # PD-disaggregation: keep KV blocks alive after finish (prefill side
    # holds them until the decode instance has pulled the blocks)
    hold_blocks: bool = False
    first_token_time: Optional[float] = None
    last_token_time: Optional[float] = None
    last_tbt_ms: float = 0.0     # inter-token gap of the latest decode step
    cumulative_logprob: float = 0.0

    @property
    def prompt_len(self) -> int:
        return len(self.prompt_token_ids)

    @property
    def num_emitted(self) -> int:
        """Total output tokens produced so far, including ones folded into
        the prompt by a recompute preemption."""
        return len(self.output_token_ids) + self.num_folded_output_tokens

    @property
    def orig_prompt_len(self) -> int:
        """Prompt length as submitted (excludes folded output tokens)."""
        return len(self.prompt_token_ids) - self.num_folded_output_tokens

    @property
    def total_len(self) -> int:
        return self.prompt_len + len(self.output_token_ids)

    @property
    def prefill_done(self) -> bool:
        return self.num_computed_tokens >= self.prompt_len

    def all_token_ids(self) -> List[int]:
        return self.prompt_token_ids + self.output_token_ids

    def append_token(self, token_id: int):
        self.output_token_ids.append(token_id)
        now = time.monotonic()
        if self.first_token_time is None:
            self.first_token_time = now
        elif self.last_token_time is not None:
            self.last_tbt_ms = (now - self.last_token_time) * 1000.0
        self.last_token_time = now

    def check_finish(self) -> bool:
        out = self.output_token_ids
        emitted = self.num_emitted
        if emitted >= self.params.max_tokens:
            self.status = SeqStatus.FINISHED_LENGTH
            return True
        if emitted >= self.params.min_tokens and out:
            last = out[-1]
            if (not self.params.ignore_eos and self.eos_token_id is not None
                    and last == self.eos_token_id):
                self.status = SeqStatus.FINISHED_STOP
                return True
            if last in self.params.stop_token_ids:
                self.status = SeqStatus.FINISHED_STOP
                return True
            for seq in self.params.stop_sequences:
                n = len(seq)
                if n and emitted >= n:
                    # a recompute fold may have moved part of the tail into
                    # prompt_token_ids; match across the fold boundary
                    tail = (out[-n:] if len(out) >= n
                            else self.all_token_ids()[-n:])
                    if tail == seq:
                        self.status = SeqStatus.FINISHED_STOP
                        return True
        return False
