"""hipGraph capture of the decode step (torch.cuda.CUDAGraph == hipGraph on
ROCm).

Decode steps are launch-bound: ~400 small kernel launches per step left the
GPU ~78% idle pre-capture (profiles/kernel_stats_r01_pregraph.txt). The
whole 32-layer decode forward + logits is captured once per batch-size
bucket and replayed with inputs copied into static buffers; padding rows
point at block 0 with seq_len=16 so replays are always well-defined.
"""
from __future__ import annotations

import logging
from typing import Dict, List, Optional, Tuple

import numpy as np
import torch

from .metadata import AttnMetadata

log = logging.getLogger("xllm.graph")

BUCKETS = (1, 2, 4, 8, 16, 24, 32, 48, 64, 96, 128, 192, 256)


class DecodeGraphRunner:
    def __init__(self, model, kv_caches, device, max_model_len: int = 4096,
                 max_batch: int = 256):
        self.model = model
        self.kv_caches = kv_caches
        self.device = device
        self.max_blocks = (max_model_len + 15) // 16
        self.buckets = [b for b in BUCKETS if b <= max_batch]
        self.graphs: Dict[int, Tuple[torch.cuda.CUDAGraph, dict,
                                     torch.Tensor]] = {}
        self.pool = None
        # pinned host staging (one per bucket) so the per-replay H2D copies
        # are truly async; unpinned numpy->GPU copies serialize the stream
        self._pinned: Dict[int, dict] = {}
        # per-request padded block-table rows (numpy, cheap incremental update)
        self._bt_rows: Dict[str, Tuple[np.ndarray, int]] = {}

    def capture_all(self):
        for bs in sorted(self.buckets, reverse=True):  # largest first: pool
            self._capture(bs)
        log.info("captured %d decode graphs (max_blocks=%d)",
                 len(self.graphs), self.max_blocks)

    @torch.inference_mode()
    def _capture(self, bs: int):
        dev = self.device
        static = dict(
            input_ids=torch.zeros(bs, dtype=torch.long, device=dev),
            positions=torch.zeros(bs, dtype=torch.long, device=dev),
            slot_mapping=torch.full((bs,), -1, dtype=torch.long, device=dev),
            seq_lens=torch.full((bs,), 16, dtype=torch.int32, device=dev),
            block_tables=torch.zeros(bs, self.max_blocks, dtype=torch.int32,
                                     device=dev),
        )
        meta = AttnMetadata(
            num_prefill_tokens=0, num_decode_tokens=bs,
            slot_mapping=static["slot_mapping"],
            decode_seq_lens=static["seq_lens"],
            decode_block_tables=static["block_tables"])

        def fwd():
            hidden = self.model(static["input_ids"], static["positions"],
                                self.kv_caches, meta)
            return self.model.compute_logits(hidden)

        # warm up twice outside capture (allocator + lazy inits settle)
        for _ in range(2):
            fwd()
        torch.cuda.synchronize()
        graph = torch.cuda.CUDAGraph()
        ctx = (torch.cuda.graph(graph, pool=self.pool) if self.pool is not None
               else torch.cuda.graph(graph))
        with ctx:
            logits = fwd()
        if self.pool is None:
            self.pool = graph.pool()
        self.graphs[bs] = (graph, static, logits)
        # zero-init: padding rows of the FULL-width copies must hold valid
        # token ids / block ids (stale entries from a larger batch are fine
        # — they were valid once)
        self._pinned[bs] = dict(
            i64=torch.zeros(3, bs, dtype=torch.long, pin_memory=True),
            seq_lens=torch.full((bs,), 16, dtype=torch.int32,
                                pin_memory=True),
            block_tables=torch.zeros(bs, self.max_blocks, dtype=torch.int32,
                                     pin_memory=True),
        )

    # ------------------------------------------------------------------ run
    def bucket_for(self, n: int) -> Optional[int]:
        for b in self.buckets:
            if b >= n:
                return b
        return None

    def block_row(self, seq) -> np.ndarray:
        """Cached padded block-table row; invalidated by growth AND by
        preemption (preempt_count is part of the cache signature, since a
        re-admitted sequence may land on different blocks at equal length)."""
        sig = (len(seq.block_table), seq.preempt_count)
        row, cached_sig = self._bt_rows.get(seq.request_id, (None, None))
        if row is None:
            row = np.zeros(self.max_blocks, dtype=np.int32)
        if cached_sig != sig:
            bt = np.asarray(seq.block_table, dtype=np.int32)
            row[: len(bt)] = bt
            self._bt_rows[seq.request_id] = (row, sig)
        return row

    def forget(self, request_id: str):
        self._bt_rows.pop(request_id, None)

    @torch.inference_mode()
    def run(self, input_ids: np.ndarray, positions: np.ndarray,
            slots: np.ndarray, seq_lens: np.ndarray,
            bt_rows: List[np.ndarray]) -> torch.Tensor:
        n = len(input_ids)
        bs = self.bucket_for(n)
        graph, static, logits = self.graphs[bs]
        pin = self._pinned[bs]
        i64 = pin["i64"].numpy()
        i64[0, :n] = input_ids
        i64[1, :n] = positions
        i64[2, :n] = slots
        if n < bs:  # neutralize padding rows
            i64[2, n:bs] = -1
        pin["seq_lens"].numpy()[:n] = seq_lens
        if n < bs:
            pin["seq_lens"].numpy()[n:bs] = 16
        np.stack(bt_rows, out=pin["block_tables"].numpy()[:n])
        static["input_ids"].copy_(pin["i64"][0], non_blocking=True)
        static["positions"].copy_(pin["i64"][1], non_blocking=True)
        static["slot_mapping"].copy_(pin["i64"][2], non_blocking=True)
        static["seq_lens"].copy_(pin["seq_lens"], non_blocking=True)
        static["block_tables"][:n].copy_(pin["block_tables"][:n],
                                         non_blocking=True)
        graph.replay()
        return logits[:n]


PREFILL_BUCKETS = (128, 256, 512, 1024, 2048)


class PrefillGraphRunner:
    """hipGraph capture of single-sequence prefill chunks (the serving
    arrival shape: one Poisson prompt per step). An eager 1024-token
    prefill pays ~300 kernel launches of dispatch overhead per arrival;
    the captured forward replays it as one graph.

    Padding scheme (chunk of L real tokens in a B-token bucket):
      * pad slots = -1 (cache scatter skips them);
      * static seq_len = chunk_start + B, so pad q rows sit at HIGHER
        causal positions than every real row — real rows never attend a
        pad key, pad rows' outputs are discarded;
      * pad block-table entries = 0 (reads stale-but-valid memory that
        only pad rows can see);
      * positions run chunk_start..chunk_start+B-1, so the graph is only
        used when chunk_start + B <= max_model_len (rope table bound).
    Tile decomposition is precomputed per bucket (meta.prefill_tiles) —
    deriving it from cu_q calls .cpu(), and a sync inside capture aborts
    the capture."""

    def __init__(self, model, kv_caches, device, max_model_len: int = 4096,
                 max_tokens: int = 8192, pool=None):
        self.model = model
        self.kv_caches = kv_caches
        self.device = device
        self.max_model_len = max_model_len
        self.max_blocks = (max_model_len + 15) // 16
        self.buckets = [b for b in PREFILL_BUCKETS
                        if b <= min(max_tokens, max_model_len)]
        self.graphs: Dict[int, Tuple[torch.cuda.CUDAGraph, dict,
                                     torch.Tensor]] = {}
        self.pool = pool
        self._pinned: Dict[int, dict] = {}

    def capture_all(self):
        for b in sorted(self.buckets, reverse=True):
            self._capture(b)
        log.info("captured %d prefill graphs (buckets %s)",
                 len(self.graphs), self.buckets)

    @torch.inference_mode()
    def _capture(self, B: int):
        dev = self.device
        static = dict(
            input_ids=torch.zeros(B, dtype=torch.long, device=dev),
            positions=torch.arange(B, dtype=torch.long, device=dev),
            slot_mapping=torch.full((B,), -1, dtype=torch.long, device=dev),
            seq_lens=torch.full((1,), B, dtype=torch.int32, device=dev),
            block_tables=torch.zeros(1, self.max_blocks, dtype=torch.int32,
                                     device=dev),
        )
        cu_q = torch.tensor([0, B], dtype=torch.int32, device=dev)
        tiles = (torch.zeros((B + 127) // 128, dtype=torch.int32, device=dev),
                 torch.arange(0, B, 128, dtype=torch.int32, device=dev))
        meta = AttnMetadata(
            num_prefill_tokens=B, num_decode_tokens=0,
            slot_mapping=static["slot_mapping"],
            cu_q=cu_q,
            prefill_seq_lens=static["seq_lens"],
            prefill_block_tables=static["block_tables"],
            prefill_tiles=tiles)

        def fwd():
            return self.model(static["input_ids"], static["positions"],
                              self.kv_caches, meta)

        for _ in range(2):
            fwd()
        torch.cuda.synchronize()
        graph = torch.cuda.CUDAGraph()
        ctx = (torch.cuda.graph(graph, pool=self.pool) if self.pool is not None
               else torch.cuda.graph(graph))
        with ctx:
            hidden = fwd()
        if self.pool is None:
            self.pool = graph.pool()
        self.graphs[B] = (graph, static, hidden)
        self._pinned[B] = dict(
            i64=torch.zeros(3, B, dtype=torch.long, pin_memory=True),
            seq_len=torch.full((1,), B, dtype=torch.int32, pin_memory=True),
            block_tables=torch.zeros(1, self.max_blocks, dtype=torch.int32,
                                     pin_memory=True),
        )

    def bucket_for(self, n: int) -> Optional[int]:
        for b in self.buckets:
            if b >= n:
                return b
        return None

    @torch.inference_mode()
    def run(self, B: int, tokens: np.ndarray, positions: np.ndarray,
            slots: np.ndarray, chunk_start: int,
            block_table: List[int]) -> torch.Tensor:
        """Replay the B-bucket graph over a chunk of len(tokens) real
        tokens; returns the static hidden buffer (read rows < len(tokens)
        before the next replay)."""
        L = len(tokens)
        graph, static, hidden = self.graphs[B]
        pin = self._pinned[B]
        i64 = pin["i64"].numpy()
        i64[0, :L] = tokens
        i64[1, :L] = positions
        i64[1, L:B] = np.arange(chunk_start + L, chunk_start + B)
        i64[2, :L] = slots
        i64[2, L:B] = -1
        pin["seq_len"].numpy()[0] = chunk_start + B
        bt = pin["block_tables"].numpy()
        bt[0, :] = 0
        bt[0, :len(block_table)] = block_table
        static["input_ids"].copy_(pin["i64"][0], non_blocking=True)
        static["positions"].copy_(pin["i64"][1], non_blocking=True)
        static["slot_mapping"].copy_(pin["i64"][2], non_blocking=True)
        static["seq_lens"].copy_(pin["seq_len"], non_blocking=True)
        static["block_tables"].copy_(pin["block_tables"], non_blocking=True)
        graph.replay()
        return hidden
