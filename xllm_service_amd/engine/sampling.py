"""Per-request sampling parameters."""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import List, Optional


@dataclass
class SamplingParams:
    temperature: float = 0.0       # 0 => greedy
    top_p: float = 1.0
    top_k: int = -1                # -1 => disabled
    max_tokens: int = 128
    min_tokens: int = 0
    stop_token_ids: List[int] = field(default_factory=list)
    # multi-token stop sequences (token-suffix match on the output)
    stop_sequences: List[List[int]] = field(default_factory=list)
    ignore_eos: bool = False
    seed: Optional[int] = None
    logprobs: Optional[int] = None  # return top-N logprobs per token

    @property
    def greedy(self) -> bool:
        return self.temperature == 0.0
