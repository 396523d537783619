"""Op dispatch: hand-written CDNA4 HIP kernels on GPU, torch references on CPU.

Policy (enforced, not silent): when running on a GPU, the HIP extension MUST
be present and is the only path taken — a missing extension raises instead of
falling back to eager torch, so a GPU run can never silently measure the
reference implementation.
"""
from __future__ import annotations

import torch

from . import ref

try:
    from xllm_service_amd import _ops  # built in-tree by setup.py
    HAS_EXT = True
except ImportError:  # CPU-only environments without a built extension
    _ops = None
    HAS_EXT = False


def _require_ext():
    if not HAS_EXT:
        raise RuntimeError(
            "xllm_service_amd._ops (gfx950 HIP extension) is not built but a "
            "GPU op was requested. Build it with: "
            "PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace"
        )


def rmsnorm(x: torch.Tensor, w: torch.Tensor, eps: float) -> torch.Tensor:
    if x.is_cuda:
        _require_ext()
        out = torch.empty_like(x)
        _ops.rmsnorm(out, x, w, eps)
        return out
    return ref.rmsnorm(x, w, eps)


def fused_add_rmsnorm(x, residual, w, eps):
    """In-place on GPU: residual += x; x = rmsnorm(residual)*w. Returns (x, residual)."""
    if x.is_cuda:
        _require_ext()
        _ops.fused_add_rmsnorm(x, residual, w, eps)
        return x, residual
    out, new_res = ref.fused_add_rmsnorm(x, residual, w, eps)
    return out, new_res


def rope(positions, q, k, cos_sin, head_dim, rot_dim):
    """In-place on GPU; out-of-place on CPU. Returns (q, k)."""
    if q.is_cuda:
        _require_ext()
        _ops.rope(positions, q, k, cos_sin, head_dim, rot_dim)
        return q, k
    return ref.rope(positions, q, k, cos_sin, head_dim, rot_dim)


def fused_rope_cache(positions, q, k, v, k_cache, v_cache, slot_mapping,
                     cos_sin, rot_dim, mrope_sections=None):
    """GPU: one kernel ropes q in place and scatters roped-k + v into the
    paged cache. CPU: composed from the reference ops. With mrope_sections
    (Qwen2-VL M-RoPE) positions is [3, T]."""
    if positions.dim() == 1:
        mrope_sections = None
    if q.is_cuda:
        _require_ext()
        ms0 = ms1 = 0
        if mrope_sections:
            ms0 = int(mrope_sections[0])
            ms1 = ms0 + int(mrope_sections[1])
        _ops.fused_rope_cache(positions, q, k, v, k_cache, v_cache,
                              slot_mapping, cos_sin, rot_dim, ms0, ms1)
        return q
    D = k_cache.shape[3]
    q2, k2 = ref.rope(positions, q, k, cos_sin, D, rot_dim,
                      mrope_sections=mrope_sections)
    ref.reshape_and_cache(k2, v, k_cache, v_cache, slot_mapping)
    return q2


def silu_and_mul(x: torch.Tensor) -> torch.Tensor:
    if x.is_cuda:
        _require_ext()
        d = x.shape[-1] // 2
        out = torch.empty(x.shape[:-1] + (d,), dtype=x.dtype, device=x.device)
        _ops.silu_and_mul(out, x)
        return out
    return ref.silu_and_mul(x)


def gelu_and_mul(x: torch.Tensor) -> torch.Tensor:
    if x.is_cuda:
        _require_ext()
        d = x.shape[-1] // 2
        out = torch.empty(x.shape[:-1] + (d,), dtype=x.dtype, device=x.device)
        _ops.gelu_and_mul(out, x)
        return out
    return ref.gelu_and_mul(x)


def reshape_and_cache(k, v, k_cache, v_cache, slot_mapping):
    if k.is_cuda:
        _require_ext()
        _ops.reshape_and_cache(k, v, k_cache, v_cache, slot_mapping)
        return
    ref.reshape_and_cache(k, v, k_cache, v_cache, slot_mapping)


def copy_blocks(k_cache, v_cache, pairs):
    if k_cache.is_cuda:
        _require_ext()
        _ops.copy_blocks(k_cache, v_cache, pairs)
        return
    for s, d in pairs.tolist():
        k_cache[d].copy_(k_cache[s])
        v_cache[d].copy_(v_cache[s])


def paged_attn_decode(q, k_cache, v_cache, block_tables, seq_lens, scale,
                      out=None):
    if q.is_cuda:
        _require_ext()
        if out is None:
            out = torch.empty(q.shape, dtype=q.dtype, device=q.device)
        _ops.paged_attn_decode(out, q, k_cache, v_cache, block_tables,
                               seq_lens, scale)
        return out
    res = ref.paged_attn_decode(q, k_cache, v_cache, block_tables, seq_lens,
                                scale)
    if out is not None:
        out.copy_(res)
        return out
    return res


def _prefill_tiles(cu_q):
    """Host-side tile decomposition for the prefill kernel (128 q rows/WG)."""
    tile_seq, tile_q0 = [], []
    for s in range(len(cu_q) - 1):
        qlen = int(cu_q[s + 1]) - int(cu_q[s])
        for q0 in range(0, qlen, 128):
            tile_seq.append(s)
            tile_q0.append(q0)
    return tile_seq, tile_q0


def paged_attn_prefill(q, k_cache, v_cache, block_tables, cu_q, seq_lens,
                       scale, out=None, tiles=None):
    if q.is_cuda:
        _require_ext()
        if k_cache.shape[-1] != 128:
            # the MFMA prefill kernel is head_dim-128 only; other head dims
            # (e.g. OPT/tiny-VL 64) run an eager torch path ON GPU — loud
            # and documented, not a silent bypass of the native 128 path
            res = ref.paged_attn_prefill(q, k_cache, v_cache, block_tables,
                                         cu_q, seq_lens, scale)
            if out is not None:
                out.copy_(res)
                return out
            return res
        if out is None:
            out = torch.empty(q.shape, dtype=q.dtype, device=q.device)
        dev = q.device
        if tiles is None:
            tile_seq, tile_q0 = _prefill_tiles(cu_q.cpu())
            tiles = (torch.tensor(tile_seq, dtype=torch.int32, device=dev),
                     torch.tensor(tile_q0, dtype=torch.int32, device=dev))
        _ops.paged_attn_prefill(
            out, q, k_cache, v_cache, block_tables, cu_q, seq_lens,
            tiles[0], tiles[1], scale)
        return out
    res = ref.paged_attn_prefill(q, k_cache, v_cache, block_tables, cu_q,
                                 seq_lens, scale)
    if out is not None:
        out.copy_(res)
        return out
    return res


def greedy_sample(logits: torch.Tensor) -> torch.Tensor:
    if logits.is_cuda:
        _require_ext()
        return _ops.greedy_sample(logits)
    return ref.greedy_sample(logits)


def swap_blocks(dst_cache, src_cache, src_blocks, dst_blocks):
    """Move whole KV blocks between the GPU cache and the host-DRAM tier
    (or CPU<->CPU on the test path)."""
    if HAS_EXT and (src_cache.is_cuda or dst_cache.is_cuda):
        _ops.swap_blocks(dst_cache, src_cache, list(src_blocks),
                         list(dst_blocks))
        return
    for s_, d_ in zip(src_blocks, dst_blocks):
        dst_cache[d_].copy_(src_cache[s_])


def migrate_blocks_peer(dst_cache, dst_device, src_cache, src_device,
                        src_blocks, dst_blocks):
    _require_ext()
    _ops.migrate_blocks_peer(dst_cache, dst_device, src_cache, src_device,
                             list(src_blocks), list(dst_blocks))


def enable_peer_access(device: int, peer: int):
    _require_ext()
    _ops.enable_peer_access(device, peer)


def mfma_gemm(a, b, bias=None):
    """C = a @ b.T (+ bias): the hand-written vision-encoder GEMM on GPU;
    torch on CPU."""
    if a.is_cuda:
        _require_ext()
        return _ops.mfma_gemm(a, b, bias)
    return torch.nn.functional.linear(a.float(), b.float(),
                                      bias.float() if bias is not None
                                      else None).to(a.dtype)


def skinny_gemm(a, b, bias=None):
    """Split-K weight-streaming GEMM for decode batches (M <= 64).
    GPU-only; callers fall back to F.linear elsewhere."""
    _require_ext()
    return _ops.skinny_gemm(a, b, bias)


def pack_gemm_weight(w: torch.Tensor) -> torch.Tensor:
    """Re-order a [N, K] bf16 weight into the per-wave contiguous stream
    layout csrc/ops/packed_gemm.hip consumes:
    P[N/16][K/32][kgroup 4][row 16][8 k]. Pure reshuffle (done once at
    load time); the original tensor stays for prefill (hipBLASLt)."""
    N, K = w.shape
    assert N % 16 == 0 and K % 32 == 0, (N, K)
    return (w.view(N // 16, 16, K // 32, 4, 8)
             .permute(0, 2, 3, 1, 4).contiguous().view(N // 16, K * 16))


def packed_gemm(a, w_packed, n: int, bias=None, s_override: int = 0):
    """C[M, n] = a @ W^T with W pre-packed by pack_gemm_weight (decode
    batches, M <= 128). GPU-only; callers fall back to F.linear."""
    _require_ext()
    return _ops.packed_gemm(a, w_packed, n, bias, s_override)


def mfma_probe_16x16x32(a, b):
    _require_ext()
    return _ops.mfma_probe_16x16x32(a, b)
