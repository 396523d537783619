"""Llama-family decoder (Llama-3 / Qwen2 text tower) on the paged-KV engine.

GEMMs go through hipBLASLt (torch linear); everything else on the hot path is
a hand-written gfx950 HIP op: fused(add+)RMSNorm, fused RoPE, paged-attention
prefill/decode, fused SwiGLU. Covers BASELINE.md configs 2-4.
"""
from __future__ import annotations

import math
from typing import List, Optional, Tuple

import torch
import torch.nn as nn

from xllm_service_amd import ops
from xllm_service_amd.distributed import parallel_state as ps
from xllm_service_amd.distributed.layers import (ColumnParallelLinear,
                                                 MergedColumnParallelLinear,
                                                 RowParallelLinear)
from xllm_service_amd.engine.metadata import AttnMetadata
from xllm_service_amd.models.config import ModelConfig
from xllm_service_amd.ops import ref as op_ref


class LlamaAttention(nn.Module):
    def __init__(self, cfg: ModelConfig, dtype):
        super().__init__()
        tp = ps.tp_size()
        self.cfg = cfg
        self.n_heads = cfg.num_heads // tp
        self.n_kv_heads = max(cfg.num_kv_heads // tp, 1)
        self.head_dim = cfg.head_dim
        self.scale = 1.0 / math.sqrt(cfg.head_dim)
        self.mrope_sections = tuple(getattr(cfg, "mrope_section", ()) or ())
        self.qkv_proj = MergedColumnParallelLinear(
            cfg.hidden_size,
            [cfg.q_size, cfg.kv_size, cfg.kv_size],
            dtype=dtype)
        self.o_proj = RowParallelLinear(cfg.q_size, cfg.hidden_size, dtype=dtype)
        # NOTE: the split-K weight-streaming GEMM (csrc/ops/skinny_gemm.hip
        # v2) beats hipBLASLt warm on the o-projection microbench (23.1us vs
        # 30.4us at M=64) but REGRESSES the end-to-end decode bench in-graph
        # (9.75k -> 9.23k tok/s: the partial-reduce second launch and ws
        # round-trip cost more than the GEMM saves), so library GEMMs keep
        # the model path; use_skinny stays available per-layer.

    def forward(self, x, positions, kv_cache, meta: AttnMetadata, cos_sin):
        qkv = self.qkv_proj(x)
        q_sz = self.n_heads * self.head_dim
        kv_sz = self.n_kv_heads * self.head_dim
        T = x.shape[0]
        # strided head views into the fused qkv output — the HIP ops are
        # stride-aware, so no .contiguous() copies on the hot path
        qh = qkv[:, :q_sz].unflatten(-1, (self.n_heads, self.head_dim))
        kh = qkv[:, q_sz:q_sz + kv_sz].unflatten(
            -1, (self.n_kv_heads, self.head_dim))
        vh = qkv[:, q_sz + kv_sz:].unflatten(
            -1, (self.n_kv_heads, self.head_dim))
        k_cache, v_cache = kv_cache
        qh = ops.fused_rope_cache(positions, qh, kh, vh, k_cache, v_cache,
                                  meta.slot_mapping, cos_sin, self.head_dim,
                                  mrope_sections=self.mrope_sections)

        np_, nd = meta.num_prefill_tokens, meta.num_decode_tokens
        out = torch.empty(T, q_sz, dtype=qkv.dtype, device=qkv.device)
        out3 = out.unflatten(-1, (self.n_heads, self.head_dim))
        if np_:
            ops.paged_attn_prefill(
                qh[:np_], k_cache, v_cache, meta.prefill_block_tables,
                meta.cu_q, meta.prefill_seq_lens, self.scale,
                out=out3[:np_], tiles=meta.prefill_tiles)
        if nd:
            ops.paged_attn_decode(
                qh[np_:], k_cache, v_cache, meta.decode_block_tables,
                meta.decode_seq_lens, self.scale, out=out3[np_:])
        return self.o_proj(out)


class LlamaMLP(nn.Module):
    def __init__(self, cfg: ModelConfig, dtype):
        super().__init__()
        self.gate_up = MergedColumnParallelLinear(
            cfg.hidden_size, [cfg.intermediate_size, cfg.intermediate_size],
            dtype=dtype)
        self.down = RowParallelLinear(cfg.intermediate_size, cfg.hidden_size,
                                      dtype=dtype)

    def forward(self, x):
        return self.down(ops.silu_and_mul(self.gate_up(x)))


class LlamaDecoderLayer(nn.Module):
    def __init__(self, cfg: ModelConfig, dtype):
        super().__init__()
        self.input_norm = nn.Parameter(torch.ones(cfg.hidden_size, dtype=dtype))
        self.post_norm = nn.Parameter(torch.ones(cfg.hidden_size, dtype=dtype))
        self.attn = LlamaAttention(cfg, dtype)
        self.mlp = LlamaMLP(cfg, dtype)
        self.eps = cfg.rms_eps

    def forward(self, x, residual, positions, kv_cache, meta, cos_sin):
        if residual is None:
            residual = x
            x = ops.rmsnorm(x, self.input_norm, self.eps)
        else:
            x, residual = ops.fused_add_rmsnorm(x, residual, self.input_norm,
                                                self.eps)
        x = self.attn(x, positions, kv_cache, meta, cos_sin)
        x, residual = ops.fused_add_rmsnorm(x, residual, self.post_norm, self.eps)
        x = self.mlp(x)
        return x, residual


class LlamaForCausalLM(nn.Module):
    def __init__(self, cfg: ModelConfig, dtype=torch.bfloat16):
        super().__init__()
        self.cfg = cfg
        self.dtype = dtype
        self.embed = nn.Embedding(cfg.vocab_size, cfg.hidden_size, dtype=dtype)
        self.layers = nn.ModuleList(
            [LlamaDecoderLayer(cfg, dtype) for _ in range(cfg.num_layers)])
        self.final_norm = nn.Parameter(torch.ones(cfg.hidden_size, dtype=dtype))
        if cfg.tie_word_embeddings:
            self.lm_head = None
        else:
            self.lm_head = ColumnParallelLinear(cfg.hidden_size, cfg.vocab_size,
                                                dtype=dtype)
        cs = op_ref.rope_table(cfg.head_dim, cfg.max_position, cfg.rope_theta)
        self.register_buffer("cos_sin", cs, persistent=False)

    # number of kv heads actually stored on this rank (TP-sharded)
    @property
    def local_kv_heads(self) -> int:
        return max(self.cfg.num_kv_heads // ps.tp_size(), 1)

    def random_init(self, seed: int = 0):
        """Deterministic random init (no checkpoints available offline).
        Generates on the parameters' device (fast path for the 8B model)."""
        dev = next(self.parameters()).device
        gen = torch.Generator(device=dev).manual_seed(seed)
        std = 1.0 / math.sqrt(self.cfg.hidden_size)
        for name, p in self.named_parameters():
            if "norm" in name:
                continue
            with torch.no_grad():
                vals = torch.randn(p.shape, generator=gen, device=dev,
                                   dtype=torch.float32)
                p.copy_((vals * std).to(p.dtype))

    def forward(self, input_ids, positions, kv_caches: List[Tuple],
                meta: AttnMetadata,
                inputs_embeds: Optional[torch.Tensor] = None):
        x = inputs_embeds if inputs_embeds is not None else self.embed(input_ids)
        residual = None
        for i, layer in enumerate(self.layers):
            x, residual = layer(x, residual, positions, kv_caches[i], meta,
                                self.cos_sin)
        x, _ = ops.fused_add_rmsnorm(x, residual, self.final_norm, self.cfg.rms_eps)
        return x

    def compute_logits(self, hidden: torch.Tensor) -> torch.Tensor:
        if self.lm_head is not None:
            logits = self.lm_head(hidden)
        else:
            logits = hidden @ self.embed.weight.t()
        return ps.tp_all_gather(logits, dim=-1)
