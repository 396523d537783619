"""Per-forward attention metadata shared by models and the model runner.

Batch layout: [all prefill-chunk tokens (varlen, seq-major)] ++ [one decode
token per decoding sequence]. A single model forward covers both phases;
each attention layer dispatches the two halves to the prefill / decode
kernels.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch


@dataclass
class AttnMetadata:
    num_prefill_tokens: int
    num_decode_tokens: int
    slot_mapping: torch.Tensor           # [T] int64, cache slot per token (-1 skip)

    # prefill half (None when no prefill this step)
    cu_q: Optional[torch.Tensor] = None          # [n_prefill+1] int32
    prefill_seq_lens: Optional[torch.Tensor] = None   # [n_prefill] int32 (ctx+chunk)
    prefill_block_tables: Optional[torch.Tensor] = None  # [n_prefill, max_blk] int32

    # precomputed prefill tile decomposition (tile_seq, tile_q0) — set by
    # the prefill graph runner so capture never calls cu_q.cpu() (a sync
    # inside hipGraph capture aborts it); None = derive from cu_q
    prefill_tiles: Optional[tuple] = None

    # decode half
    decode_seq_lens: Optional[torch.Tensor] = None    # [n_decode] int32
    decode_block_tables: Optional[torch.Tensor] = None

    @property
    def num_tokens(self) -> int:
        return self.num_prefill_tokens + self.num_decode_tokens
