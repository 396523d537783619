"""CPU sanity tests for the torch reference ops (the GPU numerics oracles).

The paged-attention references are cross-checked against plain dense SDPA so
the oracle itself is trustworthy before HIP kernels are compared against it.
"""
import math

import pytest
import torch

from xllm_service_amd.ops import ref

torch.manual_seed(0)


def make_cache(num_blocks=32, n_kv=2, bs=16, D=64, dtype=torch.float32):
    k = torch.randn(num_blocks, n_kv, bs, D, dtype=dtype)
    v = torch.randn(num_blocks, n_kv, bs, D, dtype=dtype)
    return k, v


def test_rmsnorm_matches_manual():
    x = torch.randn(4, 64)
    w = torch.randn(64)
    out = ref.rmsnorm(x, w, 1e-6)
    rms = torch.sqrt((x ** 2).mean(-1, keepdim=True) + 1e-6)
    assert torch.allclose(out, x / rms * w, atol=1e-5)


def test_fused_add_rmsnorm():
    x, r = torch.randn(4, 64), torch.randn(4, 64)
    w = torch.randn(64)
    out, new_r = ref.fused_add_rmsnorm(x, r, w, 1e-6)
    assert torch.allclose(new_r, x + r, atol=1e-6)
    assert torch.allclose(out, ref.rmsnorm(x + r, w, 1e-6), atol=1e-6)


def test_rope_matches_hf_style():
    """Rotate-half formulation must equal HF's cos/sin duplication form."""
    D, rot = 64, 64
    T, H = 5, 3
    pos = torch.tensor([0, 1, 2, 7, 11])
    q = torch.randn(T, H * D)
    k = torch.randn(T, 2 * D)
    table = ref.rope_table(rot, 32)
    qo, ko = ref.rope(pos, q, k, table, D, rot)

    # HF formulation
    inv_freq = 1.0 / (10000.0 ** (torch.arange(0, rot, 2).float() / rot))
    freqs = torch.outer(pos.float(), inv_freq)
    emb = torch.cat([freqs, freqs], dim=-1)
    cos, sin = emb.cos()[:, None, :], emb.sin()[:, None, :]

    def rotate_half(x):
        x1, x2 = x[..., : x.shape[-1] // 2], x[..., x.shape[-1] // 2:]
        return torch.cat([-x2, x1], dim=-1)

    qh = q.view(T, H, D)
    expect = qh * cos + rotate_half(qh) * sin
    assert torch.allclose(qo.view(T, H, D), expect, atol=1e-5)


def test_paged_attn_decode_vs_sdpa():
    S, Hq, n_kv, D, bs = 3, 4, 2, 64, 16
    k_cache, v_cache = make_cache(D=D, n_kv=n_kv, bs=bs)
    seq_lens = torch.tensor([5, 16, 37], dtype=torch.int32)
    max_blocks = 4
    block_tables = torch.arange(S * max_blocks, dtype=torch.int32).reshape(S, max_blocks)
    q = torch.randn(S, Hq, D)
    scale = 1.0 / math.sqrt(D)
    out = ref.paged_attn_decode(q, k_cache, v_cache, block_tables, seq_lens, scale)

    for s in range(S):
        L = int(seq_lens[s])
        K, V = ref.gather_kv(L, block_tables[s], k_cache, v_cache)
        expect = torch.nn.functional.scaled_dot_product_attention(
            q[s].unsqueeze(1),                       # [Hq, 1, D]
            K.transpose(0, 1).repeat_interleave(Hq // n_kv, 0),
            V.transpose(0, 1).repeat_interleave(Hq // n_kv, 0),
            scale=scale,
        ).squeeze(1)
        assert torch.allclose(out[s], expect, atol=1e-4), f"seq {s}"


@pytest.mark.parametrize("ctx", [0, 16])
def test_paged_attn_prefill_vs_sdpa(ctx):
    Hq, n_kv, D, bs = 4, 2, 64, 16
    k_cache, v_cache = make_cache(D=D, n_kv=n_kv, bs=bs)
    q_lens = [7, 20]
    seq_lens = torch.tensor([ctx + n for n in q_lens], dtype=torch.int32)
    cu_q = torch.tensor([0, 7, 27], dtype=torch.int32)
    block_tables = torch.arange(2 * 4, dtype=torch.int32).reshape(2, 4)
    q = torch.randn(27, Hq, D)
    scale = 1.0 / math.sqrt(D)
    out = ref.paged_attn_prefill(q, k_cache, v_cache, block_tables, cu_q,
                                 seq_lens, scale)

    for s, qlen in enumerate(q_lens):
        q0 = int(cu_q[s])
        L = int(seq_lens[s])
        K, V = ref.gather_kv(L, block_tables[s], k_cache, v_cache)
        Ke = K.transpose(0, 1).repeat_interleave(Hq // n_kv, 0)
        Ve = V.transpose(0, 1).repeat_interleave(Hq // n_kv, 0)
        mask = torch.zeros(qlen, L, dtype=torch.bool)
        for i in range(qlen):
            mask[i, : ctx + i + 1] = True
        expect = torch.nn.functional.scaled_dot_product_attention(
            q[q0:q0 + qlen].transpose(0, 1), Ke, Ve,
            attn_mask=mask[None], scale=scale)
        assert torch.allclose(out[q0:q0 + qlen].transpose(0, 1), expect,
                              atol=1e-4), f"seq {s}"


def test_reshape_and_cache_roundtrip():
    n_kv, bs, D = 2, 16, 64
    k_cache = torch.zeros(4, n_kv, bs, D)
    v_cache = torch.zeros(4, n_kv, bs, D)
    T = 10
    k = torch.randn(T, n_kv, D)
    v = torch.randn(T, n_kv, D)
    slots = torch.tensor([0, 1, 2, 17, 18, 33, 34, 35, 63, 40])
    ref.reshape_and_cache(k, v, k_cache, v_cache, slots)
    for t, slot in enumerate(slots.tolist()):
        b, off = divmod(slot, bs)
        assert torch.equal(k_cache[b, :, off, :], k[t])
        assert torch.equal(v_cache[b, :, off, :], v[t])


def test_greedy_sample():
    logits = torch.randn(5, 100)
    assert torch.equal(ref.greedy_sample(logits), logits.argmax(-1))


def test_silu_and_mul():
    x = torch.randn(3, 32)
    out = ref.silu_and_mul(x)
    assert torch.allclose(
        out, torch.nn.functional.silu(x[:, :16]) * x[:, 16:], atol=1e-6)
