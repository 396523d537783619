"""Tensor-parallel linear layers (Megatron-style sharding over RCCL/xGMI).

At tp_size()==1 these are plain linears with zero overhead. Sharding follows
the standard column(QKV, gate/up) -> row(o_proj, down) pattern so each
transformer block needs exactly ONE all-reduce per sublayer — sized for the
xGMI point-to-point topology (few, large collectives).
"""
from __future__ import annotations

import torch
import torch.nn as nn

from . import parallel_state as ps


def _maybe_skinny(x, weight, bias):
    """Decode batches (rows <= 64, where the kernel measures at or ahead
    of hipBLASLt) route to the split-K weight-streaming kernel when the
    layer opted in (ops/skinny_gemm.hip); else hipBLASLt."""
    from xllm_service_amd import ops as xops
    if (x.is_cuda and 0 < x.shape[0] <= 64 and xops.HAS_EXT
            and weight.shape[1] % 64 == 0 and weight.shape[0] % 4 == 0):
        return xops.skinny_gemm(x, weight, bias)
    return torch.nn.functional.linear(x, weight, bias)


class ColumnParallelLinear(nn.Module):
    """Y = X W^T with W sharded along the output dim."""

    def __init__(self, in_features: int, out_features: int, bias: bool = False,
                 dtype=None):
        super().__init__()
        tp = ps.tp_size()
        assert out_features % tp == 0, (out_features, tp)
        self.in_features = in_features
        self.out_features = out_features
        self.shard_out = out_features // tp
        self.weight = nn.Parameter(
            torch.empty(self.shard_out, in_features, dtype=dtype))
        self.bias = nn.Parameter(torch.zeros(self.shard_out, dtype=dtype)) if bias else None
        self.use_skinny = False  # opt-in decode fast path (see _maybe_skinny)

    def forward(self, x):
        if self.use_skinny:
            return _maybe_skinny(x, self.weight, self.bias)
        return torch.nn.functional.linear(x, self.weight, self.bias)


class MergedColumnParallelLinear(ColumnParallelLinear):
    """Several column-parallel projections fused into one GEMM (e.g. QKV,
    gate+up). Each constituent is sharded independently so the shard layout
    is [sum of per-part shards, in]."""

    def __init__(self, in_features: int, out_sizes: list[int], bias: bool = False,
                 dtype=None):
        tp = ps.tp_size()
        for s in out_sizes:
            assert s % tp == 0, (s, tp)
        self.out_sizes = out_sizes
        super().__init__(in_features, sum(out_sizes), bias=bias, dtype=dtype)

    def shard_split(self, y: torch.Tensor):
        tp = ps.tp_size()
        return torch.split(y, [s // tp for s in self.out_sizes], dim=-1)


class RowParallelLinear(nn.Module):
    """Y = X W^T with W sharded along the input dim; output all-reduced."""

    def __init__(self, in_features: int, out_features: int, bias: bool = False,
                 dtype=None):
        super().__init__()
        tp = ps.tp_size()
        assert in_features % tp == 0, (in_features, tp)
        self.in_features = in_features
        self.out_features = out_features
        self.shard_in = in_features // tp
        self.weight = nn.Parameter(
            torch.empty(out_features, self.shard_in, dtype=dtype))
        # bias added once (rank 0's contribution) to keep the sum correct
        self.bias = nn.Parameter(torch.zeros(out_features, dtype=dtype)) if bias else None
        self.use_skinny = False

    def forward(self, x):
        if self.use_skinny:
            y = _maybe_skinny(x, self.weight, None)
        else:
            y = torch.nn.functional.linear(x, self.weight)
        y = ps.tp_all_reduce(y)
        if self.bias is not None:
            y = y + self.bias
        return y


def shard_llama_state_dict(full_sd, cfg, tp: int, rank: int):
    """Shard a full (tp=1) Llama-family state dict for TP rank `rank`.

    Column-parallel weights (qkv, gate_up, lm_head) split by output rows per
    constituent; row-parallel weights (o_proj, down) split by input columns.
    Used by tests (TP == single-GPU equivalence) and by checkpoint loading.
    """
    import torch
    q_sz = cfg.q_size
    kv_sz = cfg.kv_size
    out = {}
    for k, v in full_sd.items():
        if "qkv_proj" in k:  # weight or bias: rows [q | k | v]
            q, kk, vv = torch.split(v, [q_sz, kv_sz, kv_sz], dim=0)
            parts = []
            for t, sz in ((q, q_sz), (kk, kv_sz), (vv, kv_sz)):
                s = sz // tp
                parts.append(t[rank * s:(rank + 1) * s])
            out[k] = torch.cat(parts, dim=0)
        elif "gate_up" in k:
            g, u = v.chunk(2, dim=0)
            s = g.shape[0] // tp
            out[k] = torch.cat([g[rank * s:(rank + 1) * s],
                                u[rank * s:(rank + 1) * s]], dim=0)
        elif "lm_head" in k:
            s = v.shape[0] // tp
            out[k] = v[rank * s:(rank + 1) * s]
        elif ("o_proj.weight" in k) or ("down.weight" in k):
            s = v.shape[1] // tp
            out[k] = v[:, rank * s:(rank + 1) * s]
        else:
            out[k] = v
    return out
