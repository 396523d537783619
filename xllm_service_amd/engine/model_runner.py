"""Model runner: turns a StepPlan into one fused model forward + sampling.

Builds the mixed prefill+decode batch (prefill tokens first, then one token
per decoding sequence), drives the HIP kernels through the model, and samples
next tokens (greedy via the HIP argmax kernel; stochastic via torch GPU ops).
"""
from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import torch

from xllm_service_amd import ops
from xllm_service_amd.models.config import ModelConfig

from .metadata import AttnMetadata
from .scheduler import StepPlan
from .sequence import Sequence

BLOCK_SIZE = 16


class ModelRunner:
    def __init__(self, model, cfg: ModelConfig, device: torch.device,
                 num_blocks: int, dtype=torch.bfloat16,
                 enable_graphs: bool = True, max_model_len: int = 4096,
                 max_graph_batch: int = 256):
        self.model = model
        self.cfg = cfg
        self.device = device
        self.dtype = dtype
        self.num_blocks = num_blocks
        self.graph_runner = None
        self._graph_opts = (enable_graphs and device.type == "cuda",
                            max_model_len, max_graph_batch)
        n_kv = model.local_kv_heads if hasattr(model, "local_kv_heads") else cfg.num_kv_heads
        self.kv_caches: List[Tuple[torch.Tensor, torch.Tensor]] = [
            (torch.zeros(num_blocks, n_kv, BLOCK_SIZE, cfg.head_dim,
                         dtype=dtype, device=device),
             torch.zeros(num_blocks, n_kv, BLOCK_SIZE, cfg.head_dim,
                         dtype=dtype, device=device))
            for _ in range(cfg.num_layers)
        ]
        self.cpu_kv_caches: List[Tuple[torch.Tensor, torch.Tensor]] = []

    def alloc_cpu_caches(self, num_cpu_blocks: int):
        """Pinned host-DRAM tier for KV swapping."""
        n_kv = self.kv_caches[0][0].shape[1]
        pin = self.device.type == "cuda"
        self.cpu_kv_caches = [
            (torch.zeros(num_cpu_blocks, n_kv, BLOCK_SIZE, self.cfg.head_dim,
                         dtype=self.dtype, pin_memory=pin),
             torch.zeros(num_cpu_blocks, n_kv, BLOCK_SIZE, self.cfg.head_dim,
                         dtype=self.dtype, pin_memory=pin))
            for _ in range(self.cfg.num_layers)
        ]

    @staticmethod
    def kv_cache_blocks_for(cfg: ModelConfig, device: torch.device,
                            n_kv_local: int,
                            gpu_memory_utilization: float = 0.85,
                            dtype_bytes: int = 2,
                            max_blocks: Optional[int] = None) -> int:
        """Size the block pool from actually-free HBM after weights load."""
        if device.type == "cuda":
            free, _total = torch.cuda.mem_get_info(device)
            budget = int(free * gpu_memory_utilization)
        else:
            budget = 2 << 30  # CPU tests: 2 GiB worth of blocks
        per_block = (2 * n_kv_local * BLOCK_SIZE * cfg.head_dim *
                     dtype_bytes * cfg.num_layers)
        n = max(budget // per_block, 16)
        if max_blocks:
            n = min(n, max_blocks)
        return int(n)

    def capture_graphs(self):
        enable, max_len, max_bs = self._graph_opts
        if not enable:
            return
        from .graph_runner import DecodeGraphRunner, PrefillGraphRunner
        self.graph_runner = DecodeGraphRunner(
            self.model, self.kv_caches, self.device,
            max_model_len=max_len, max_batch=max_bs)
        self.graph_runner.capture_all()
        from xllm_service_amd.distributed import parallel_state as ps
        import os
        # OPT-IN (XLLM_PREFILL_GRAPHS=1): replaying padded single-seq
        # prefill chunks aborts with a device fault on the debug-model
        # chunked-prefill shapes (L < bucket; tests/test_gpu_engine.py
        # test_gpu_prefix_cache_and_chunked_prefill) while the full-length
        # serving shape (L == 1024 == bucket) runs and saves ~2.5 ms per
        # arrival. Until the padded-replay fault is isolated this stays
        # off the default path — docs/TODO_ROUND3.md.
        if (ps.tp_size() == 1
                and os.environ.get("XLLM_PREFILL_GRAPHS") == "1"):
            shared = os.environ.get("XLLM_PREFILL_GRAPH_SHARED_POOL") == "1"
            self.prefill_graph = PrefillGraphRunner(
                self.model, self.kv_caches, self.device,
                max_model_len=max_len,
                pool=self.graph_runner.pool if shared else None)
            self.prefill_graph.capture_all()

    # ---- batch construction -------------------------------------------------
    def _build_batch(self, plan: StepPlan, bm):
        tokens: List[int] = []
        positions: List[int] = []
        slots: List[int] = []
        # M-RoPE: (row_range, seq) pairs whose prefill chunks need 3-D ids
        mrope_spans = []

        cu_q = [0]
        p_seq_lens: List[int] = []
        p_tables: List[List[int]] = []
        for sp in plan.prefills:
            seq = sp.seq
            chunk_toks = seq.prompt_token_ids[sp.chunk_start:
                                              sp.chunk_start + sp.chunk_len]
            if seq.mrope_pos is not None:
                mrope_spans.append((len(tokens), sp))
            tokens.extend(chunk_toks)
            positions.extend(range(sp.chunk_start, sp.chunk_start + sp.chunk_len))
            for pos in range(sp.chunk_start, sp.chunk_start + sp.chunk_len):
                blk = seq.block_table[pos // BLOCK_SIZE]
                slots.append(blk * BLOCK_SIZE + pos % BLOCK_SIZE)
            cu_q.append(cu_q[-1] + sp.chunk_len)
            p_seq_lens.append(sp.chunk_start + sp.chunk_len)
            p_tables.append(seq.block_table)

        d_seq_lens: List[int] = []
        d_tables: List[List[int]] = []
        for seq in plan.decodes:
            last_tok = (seq.output_token_ids[-1] if seq.output_token_ids
                        else seq.prompt_token_ids[-1])
            # decode rope positions shift by the M-RoPE delta (text position
            # after images compresses below total_len); attn metadata keeps
            # the un-shifted lengths
            pos = seq.total_len - 1 + seq.mrope_delta
            tokens.append(last_tok)
            positions.append(pos)
            slots.append(bm.append_slot(seq))
            d_seq_lens.append(seq.total_len)
            d_tables.append(seq.block_table)

        import numpy as _np
        if mrope_spans:
            pos_np = _np.tile(_np.asarray(positions, dtype=_np.int64), (3, 1))
            for row0, sp in mrope_spans:
                pos_np[:, row0:row0 + sp.chunk_len] = \
                    sp.seq.mrope_pos[:, sp.chunk_start:
                                     sp.chunk_start + sp.chunk_len]
        else:
            pos_np = _np.asarray(positions, dtype=_np.int64)

        def pad_tables(tables: List[List[int]]) -> torch.Tensor:
            if not tables:
                return torch.zeros(0, 1, dtype=torch.int32, device=self.device)
            w = max(len(t) for t in tables)
            out = torch.zeros(len(tables), w, dtype=torch.int32)
            for i, t in enumerate(tables):
                out[i, :len(t)] = torch.tensor(t, dtype=torch.int32)
            return out.to(self.device)

        dev = self.device
        np_ = cu_q[-1]
        from xllm_service_amd.distributed import parallel_state as ps
        if ps.tp_size() > 1 and ps.tp_rank() == 0:
            def tables_np(tables):
                if not tables:
                    return None
                w = max(len(t) for t in tables)
                out = _np.zeros((len(tables), w), dtype=_np.int32)
                for i, t in enumerate(tables):
                    out[i, :len(t)] = t
                return out
            self._tp_batch = dict(
                kind="eager", np=np_, nd=len(plan.decodes),
                input_ids=_np.asarray(tokens, dtype=_np.int64),
                positions=pos_np,
                slots=_np.asarray(slots, dtype=_np.int64),
                cu_q=_np.asarray(cu_q, dtype=_np.int32)
                if plan.prefills else None,
                p_seq_lens=_np.asarray(p_seq_lens, dtype=_np.int32)
                if plan.prefills else None,
                p_tables=tables_np(p_tables) if plan.prefills else None,
                d_seq_lens=_np.asarray(d_seq_lens, dtype=_np.int32)
                if plan.decodes else None,
                d_tables=tables_np(d_tables) if plan.decodes else None,
            )
        meta = AttnMetadata(
            num_prefill_tokens=np_,
            num_decode_tokens=len(plan.decodes),
            slot_mapping=torch.tensor(slots, dtype=torch.long, device=dev),
            cu_q=(torch.tensor(cu_q, dtype=torch.int32, device=dev)
                  if plan.prefills else None),
            prefill_seq_lens=(torch.tensor(p_seq_lens, dtype=torch.int32,
                                           device=dev) if plan.prefills else None),
            prefill_block_tables=pad_tables(p_tables) if plan.prefills else None,
            decode_seq_lens=(torch.tensor(d_seq_lens, dtype=torch.int32,
                                          device=dev) if plan.decodes else None),
            decode_block_tables=pad_tables(d_tables) if plan.decodes else None,
        )
        input_ids = torch.tensor(tokens, dtype=torch.long, device=dev)
        pos_t = torch.from_numpy(pos_np).to(dev)
        return input_ids, pos_t, meta

    # ---- sampling -----------------------------------------------------------
    def _logprobs(self, logits: torch.Tensor, seqs: List[Sequence],
                  tokens: List[int]):
        """Per-token logprob + top-N alternatives for requesting seqs."""
        if not any(s.params.logprobs is not None for s in seqs):
            return {}
        lp = torch.log_softmax(logits.float(), dim=-1)
        out = {}
        for i, seq in enumerate(seqs):
            n = seq.params.logprobs
            if n is None:
                continue
            row = lp[i]
            chosen = float(row[tokens[i]])
            top = {}
            if n > 0:
                vals, idx = torch.topk(row, min(n, row.numel()))
                top = {int(t): float(v) for t, v in zip(idx, vals)}
            out[seq.request_id] = {"token_logprob": chosen, "top": top}
        return out

    def _sample(self, logits: torch.Tensor, seqs: List[Sequence]) -> List[int]:
        greedy = all(s.params.greedy for s in seqs)
        if greedy:
            if logits.is_cuda and logits.dtype == torch.bfloat16:
                return ops.greedy_sample(logits.contiguous()).tolist()
            return logits.float().argmax(-1).tolist()
        out: List[int] = []
        probs_all = None
        for i, seq in enumerate(seqs):
            p = seq.params
            if p.greedy:
                out.append(int(logits[i].float().argmax()))
                continue
            lg = logits[i].float() / max(p.temperature, 1e-5)
            if p.top_k > 0:
                kth = torch.topk(lg, min(p.top_k, lg.numel())).values[-1]
                lg[lg < kth] = -float("inf")
            if p.top_p < 1.0:
                sorted_lg, idx = lg.sort(descending=True)
                probs = sorted_lg.softmax(-1)
                cum = probs.cumsum(-1)
                cut = (cum - probs) > p.top_p
                sorted_lg[cut] = -float("inf")
                lg = torch.full_like(lg, -float("inf")).scatter(0, idx, sorted_lg)
            probs = lg.softmax(-1)
            gen = None
            if p.seed is not None:
                gen = torch.Generator(device=probs.device).manual_seed(
                    p.seed + seq.total_len)
            out.append(int(torch.multinomial(probs, 1, generator=gen)))
        return out

    # ---- tensor-parallel batch broadcast ------------------------------------
    def _tp_bcast(self, obj):
        from xllm_service_amd.distributed import parallel_state as ps
        if ps.tp_size() <= 1:
            return
        import torch.distributed as dist
        box = [obj]
        dist.broadcast_object_list(box, src=0, group=ps.tp_group())

    def follower_step(self) -> bool:
        """TP rank > 0: receive one batch from rank 0 and run the forward
        (participating in the TP collectives). Returns False on stop."""
        from xllm_service_amd.distributed import parallel_state as ps
        import torch.distributed as dist
        box = [None]
        dist.broadcast_object_list(box, src=0, group=ps.tp_group())
        obj = box[0]
        if obj is None or obj.get("kind") == "stop":
            return False
        with torch.inference_mode():
            if obj["kind"] == "graph":
                logits = self.graph_runner.run(
                    obj["input_ids"], obj["positions"], obj["slots"],
                    obj["seq_lens"], list(obj["bt_rows"]))
            else:
                dev = self.device
                meta = AttnMetadata(
                    num_prefill_tokens=obj["np"],
                    num_decode_tokens=obj["nd"],
                    slot_mapping=torch.from_numpy(obj["slots"]).to(dev),
                    cu_q=(torch.from_numpy(obj["cu_q"]).to(dev)
                          if obj["cu_q"] is not None else None),
                    prefill_seq_lens=(
                        torch.from_numpy(obj["p_seq_lens"]).to(dev)
                        if obj["p_seq_lens"] is not None else None),
                    prefill_block_tables=(
                        torch.from_numpy(obj["p_tables"]).to(dev)
                        if obj["p_tables"] is not None else None),
                    decode_seq_lens=(
                        torch.from_numpy(obj["d_seq_lens"]).to(dev)
                        if obj["d_seq_lens"] is not None else None),
                    decode_block_tables=(
                        torch.from_numpy(obj["d_tables"]).to(dev)
                        if obj["d_tables"] is not None else None),
                )
                input_ids = torch.from_numpy(obj["input_ids"]).to(dev)
                positions = torch.from_numpy(obj["positions"]).to(dev)
                hidden = self.model(input_ids, positions, self.kv_caches,
                                    meta)
                if obj["rows"]:
                    sel = hidden[torch.tensor(obj["rows"], dtype=torch.long,
                                              device=dev)]
                    self.model.compute_logits(sel)  # joins the all-gather
        return True

    def stop_followers(self):
        self._tp_bcast({"kind": "stop"})

    prefill_graph = None

    # ---- one step -----------------------------------------------------------
    @torch.inference_mode()
    def execute(self, plan: StepPlan, bm) -> Dict[str, int]:
        """Run one step; returns {request_id: sampled_token} for sequences
        that produced a token this step (completed prefills + decodes)."""
        pg = self.prefill_graph
        if pg is not None and not plan.decodes and len(plan.prefills) == 1:
            sp = plan.prefills[0]
            seq = sp.seq
            from xllm_service_amd.distributed import parallel_state as ps
            if (seq.mm_embeds is None and seq.mrope_pos is None
                    and ps.tp_size() == 1):
                b = pg.bucket_for(sp.chunk_len)
                if (b is not None
                        and sp.chunk_start + b <= pg.max_model_len):
                    return self._execute_prefill_graph(sp, b)
        gr = self.graph_runner
        if (gr is not None and not plan.prefills and plan.decodes
                and gr.bucket_for(len(plan.decodes)) is not None
                and all(s.total_len <= gr.max_blocks * BLOCK_SIZE
                        for s in plan.decodes)):
            return self._execute_decode_graph(plan, bm)
        input_ids, positions, meta = self._build_batch(plan, bm)
        tp_batch = getattr(self, "_tp_batch", None)
        self._tp_batch = None
        if tp_batch is not None:
            # rows that will need logits (followers must join the logits
            # all-gather with the same selection size)
            rows_pre = []
            for i, sp in enumerate(plan.prefills):
                if sp.chunk_start + sp.chunk_len >= sp.seq.prompt_len:
                    rows_pre.append(int(meta.cu_q[i + 1]) - 1)
            rows_pre.extend(meta.num_prefill_tokens + j
                            for j in range(len(plan.decodes)))
            tp_batch["rows"] = rows_pre
            self._tp_bcast(tp_batch)
        inputs_embeds = None
        if any(sp.seq.mm_embeds is not None for sp in plan.prefills):
            inputs_embeds = self._merge_mm_embeds(plan, input_ids, meta)
        hidden = self.model(input_ids, positions, self.kv_caches, meta,
                            inputs_embeds=inputs_embeds)

        # rows that need logits: last token of each COMPLETED prefill chunk,
        # plus every decode row
        rows: List[int] = []
        seqs: List[Sequence] = []
        for i, sp in enumerate(plan.prefills):
            if sp.chunk_start + sp.chunk_len >= sp.seq.prompt_len:
                rows.append(int(meta.cu_q[i + 1]) - 1)
                seqs.append(sp.seq)
        np_ = meta.num_prefill_tokens
        for j, seq in enumerate(plan.decodes):
            rows.append(np_ + j)
            seqs.append(seq)
        if not rows:
            return {}
        sel = hidden[torch.tensor(rows, dtype=torch.long, device=hidden.device)]
        logits = self.model.compute_logits(sel)
        tokens = self._sample(logits, seqs)
        lps = self._logprobs(logits, seqs, tokens)
        return {seq.request_id: (tok, lps.get(seq.request_id))
                for seq, tok in zip(seqs, tokens)}

    def _merge_mm_embeds(self, plan: StepPlan, input_ids, meta):
        """Replace image-placeholder rows of the token embeddings with the
        per-sequence vision embeddings (EPD E->P handoff). Chunked prefill
        is handled by counting placeholders before the chunk start."""
        embeds = self.model.embed(input_ids)
        off = 0
        for i, sp in enumerate(plan.prefills):
            seq = sp.seq
            chunk = slice(int(meta.cu_q[i]), int(meta.cu_q[i + 1]))
            if seq.mm_embeds is None:
                continue
            ph = seq.mm_placeholder
            mm = seq.mm_embeds.to(embeds.device, embeds.dtype)
            n_before = sum(1 for t in seq.prompt_token_ids[:sp.chunk_start]
                           if t == ph)
            chunk_ids = input_ids[chunk]
            mask = chunk_ids == ph
            n_here = int(mask.sum())
            if n_here:
                embeds[chunk][mask] = mm[n_before:n_before + n_here]
        return embeds

    def _execute_prefill_graph(self, sp, b) -> Dict[str, int]:
        """Single-sequence prefill chunk via the captured graph (the
        serving arrival fast path; see PrefillGraphRunner padding notes)."""
        import numpy as np
        seq = sp.seq
        cs, L = sp.chunk_start, sp.chunk_len
        tokens = np.asarray(seq.prompt_token_ids[cs:cs + L], dtype=np.int64)
        positions = np.arange(cs, cs + L, dtype=np.int64)
        bt = np.asarray(seq.block_table, dtype=np.int64)
        slots = bt[positions // BLOCK_SIZE] * BLOCK_SIZE \
            + positions % BLOCK_SIZE
        hidden = self.prefill_graph.run(b, tokens, positions, slots, cs,
                                        seq.block_table)
        if cs + L < seq.prompt_len:
            return {}                     # mid-chunk: no token yet
        # clone the row OUT of the graph's private pool before the eager
        # lm_head GEMM reads it (hipBLASLt reads past a tightly-packed
        # pool allocation were the suspected device fault)
        sel = hidden[L - 1:L].clone()
        logits = self.model.compute_logits(sel)
        toks = self._sample(logits, [seq])
        lps = self._logprobs(logits, [seq], toks)
        return {seq.request_id: (toks[0], lps.get(seq.request_id))}

    def _execute_decode_graph(self, plan: StepPlan, bm) -> Dict[str, int]:
        import numpy as np
        seqs = plan.decodes
        n = len(seqs)
        input_ids = np.empty(n, dtype=np.int64)
        positions = np.empty(n, dtype=np.int64)
        slots = np.empty(n, dtype=np.int64)
        seq_lens = np.empty(n, dtype=np.int32)
        gr = self.graph_runner
        bt_rows = []
        for i, seq in enumerate(seqs):
            input_ids[i] = (seq.output_token_ids[-1] if seq.output_token_ids
                            else seq.prompt_token_ids[-1])
            positions[i] = seq.total_len - 1 + seq.mrope_delta
            slots[i] = bm.append_slot(seq)
            seq_lens[i] = seq.total_len
            bt_rows.append(gr.block_row(seq))
        from xllm_service_amd.distributed import parallel_state as ps
        if ps.tp_size() > 1 and ps.tp_rank() == 0:
            self._tp_bcast(dict(kind="graph", input_ids=input_ids,
                                positions=positions, slots=slots,
                                seq_lens=seq_lens, bt_rows=bt_rows))
        logits = gr.run(input_ids, positions, slots, seq_lens, bt_rows)
        tokens = self._sample(logits, seqs)
        lps = self._logprobs(logits, seqs, tokens)
        return {seq.request_id: (tok, lps.get(seq.request_id))
                for seq, tok in zip(seqs, tokens)}
