"""Pure-PyTorch fp32 reference implementations of every HIP op.

These serve two purposes:
  * numerics oracles for the GPU kernels (tests/test_gpu_ops.py compares the
    HIP kernels against these at fp32)
  * the CPU execution path (OPT-125m control-plane config)

They are intentionally simple and readable, not fast.
"""
from __future__ import annotations

import torch


def rmsnorm(x: torch.Tensor, w: torch.Tensor, eps: float) -> torch.Tensor:
    xf = x.float()
    var = xf.pow(2).mean(dim=-1, keepdim=True)
    return (xf * torch.rsqrt(var + eps) * w.float()).to(x.dtype)


def fused_add_rmsnorm(x, residual, w, eps):
    """Returns (normed, new_residual)."""
    summed = (x.float() + residual.float())
    out = rmsnorm(summed, w, eps).to(x.dtype)
    return out, summed.to(residual.dtype)


def rope_table(rot_dim: int, max_pos: int, base: float = 10000.0,
               scaling: float = 1.0) -> torch.Tensor:
    """cos/sin table [max_pos, rot_dim]: first half cos, second half sin."""
    inv_freq = 1.0 / (base ** (torch.arange(0, rot_dim, 2).float() / rot_dim))
    t = torch.arange(max_pos).float() / scaling
    freqs = torch.outer(t, inv_freq)  # [max_pos, rot_dim/2]
    return torch.cat([freqs.cos(), freqs.sin()], dim=-1)


def rope(positions, q, k, cos_sin, head_dim, rot_dim, mrope_sections=None):
    """NeoX rotate-half RoPE, applied out-of-place (reference). With
    mrope_sections (t, h, w summing to rot_dim//2) positions is [3, T] and
    frequency i takes its position from the section containing i."""
    def _apply(x):
        if mrope_sections:
            pos3 = positions
            T = pos3.shape[1]
            half = rot_dim // 2
            sec = torch.empty(half, dtype=torch.long)
            s0, s1, _ = mrope_sections
            sec[:s0] = 0
            sec[s0:s0 + s1] = 1
            sec[s0 + s1:] = 2
            # per (token, freq) position -> gather cos/sin rows
            pf = pos3[sec, :].T                      # [T, half]
            cos_t = cos_sin[:, :half][pf, torch.arange(half)]   # [T, half]
            sin_t = cos_sin[:, half:rot_dim][pf, torch.arange(half)]
            xs = x.reshape(T, -1, head_dim).float()
            cos = cos_t.unsqueeze(1)
            sin = sin_t.unsqueeze(1)
            x1 = xs[..., :half]
            x2 = xs[..., half:rot_dim]
            o1 = x1 * cos - x2 * sin
            o2 = x2 * cos + x1 * sin
            out = torch.cat([o1, o2, xs[..., rot_dim:]], dim=-1)
            return out.to(x.dtype).reshape(x.shape)
        T = positions.shape[0]
        xs = x.reshape(T, -1, head_dim).float()
        cs = cos_sin[positions]  # [T, rot_dim]
        cos = cs[:, : rot_dim // 2].unsqueeze(1)
        sin = cs[:, rot_dim // 2:].unsqueeze(1)
        x1 = xs[..., : rot_dim // 2]
        x2 = xs[..., rot_dim // 2: rot_dim]
        o1 = x1 * cos - x2 * sin
        o2 = x2 * cos + x1 * sin
        out = torch.cat([o1, o2, xs[..., rot_dim:]], dim=-1)
        return out.to(x.dtype).reshape(x.shape)
    return _apply(q), _apply(k)


def silu_and_mul(x: torch.Tensor) -> torch.Tensor:
    d = x.shape[-1] // 2
    g = x[..., :d].float()
    return (torch.nn.functional.silu(g) * x[..., d:].float()).to(x.dtype)


def gelu_and_mul(x: torch.Tensor) -> torch.Tensor:
    d = x.shape[-1] // 2
    g = x[..., :d].float()
    return (torch.nn.functional.gelu(g, approximate="tanh") * x[..., d:].float()).to(x.dtype)


def reshape_and_cache(k, v, k_cache, v_cache, slot_mapping):
    """k/v: [T, n_kv, D]; caches [blocks, n_kv, bs, D]."""
    bs = k_cache.shape[2]
    for t in range(k.shape[0]):
        slot = int(slot_mapping[t])
        if slot < 0:
            continue
        b, off = divmod(slot, bs)
        k_cache[b, :, off, :] = k[t]
        v_cache[b, :, off, :] = v[t]


def gather_kv(seq_len, block_table, k_cache, v_cache):
    """Return contiguous K/V [seq_len, n_kv, D] for one sequence."""
    bs = k_cache.shape[2]
    nblk = (seq_len + bs - 1) // bs
    ks, vs = [], []
    for i in range(nblk):
        blk = int(block_table[i])
        take = min(bs, seq_len - i * bs)
        ks.append(k_cache[blk, :, :take, :].transpose(0, 1))
        vs.append(v_cache[blk, :, :take, :].transpose(0, 1))
    return torch.cat(ks, dim=0), torch.cat(vs, dim=0)


def paged_attn_decode(q, k_cache, v_cache, block_tables, seq_lens, scale):
    """q: [S, Hq, D] -> out [S, Hq, D]."""
    S, Hq, D = q.shape
    n_kv = k_cache.shape[1]
    G = Hq // n_kv
    out = torch.empty_like(q)
    for s in range(S):
        L = int(seq_lens[s])
        K, V = gather_kv(L, block_tables[s], k_cache, v_cache)  # [L, n_kv, D]
        for h in range(Hq):
            kv = h // G
            attn = (q[s, h].float() @ K[:, kv].float().T) * scale  # [L]
            p = torch.softmax(attn, dim=-1)
            out[s, h] = (p @ V[:, kv].float()).to(q.dtype)
    return out


def paged_attn_prefill(q, k_cache, v_cache, block_tables, cu_q, seq_lens, scale):
    """Varlen causal prefill over the paged cache.

    q: [total_q, Hq, D]; seq i has queries cu_q[i]:cu_q[i+1] and seq_lens[i]
    total keys (ctx = seq_lens[i] - q_len keys precede the new chunk).
    """
    total_q, Hq, D = q.shape
    n_kv = k_cache.shape[1]
    G = Hq // n_kv
    out = torch.empty_like(q)
    nseq = len(seq_lens)
    for s in range(nseq):
        q0, q1 = int(cu_q[s]), int(cu_q[s + 1])
        qlen = q1 - q0
        L = int(seq_lens[s])
        ctx = L - qlen
        K, V = gather_kv(L, block_tables[s], k_cache, v_cache)
        for h in range(Hq):
            kv = h // G
            attn = (q[q0:q1, h].float() @ K[:, kv].float().T) * scale  # [qlen, L]
            # causal: query local i (global ctx+i) sees keys <= ctx+i
            dev = attn.device
            mask = torch.arange(L, device=dev)[None, :] > \
                (ctx + torch.arange(qlen, device=dev))[:, None]
            attn.masked_fill_(mask, float("-inf"))
            p = torch.softmax(attn, dim=-1)
            out[q0:q1, h] = (p @ V[:, kv].float()).to(q.dtype)
    return out


def greedy_sample(logits: torch.Tensor) -> torch.Tensor:
    return logits.float().argmax(dim=-1)
