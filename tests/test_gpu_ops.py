"""GPU numerics tests: each HIP kernel vs the plain-PyTorch fp32 reference.

Run on an MI355X via:  gpurun -- 'python -m pytest tests/test_gpu_ops.py -x -q -m gpu'
"""
import math

import pytest
import torch

pytestmark = pytest.mark.gpu

from xllm_service_amd import ops
from xllm_service_amd.ops import ref


def _assert_close(got, want, atol, rtol=1.6e-2, label=""):
    got = got.float().cpu()
    want = want.float().cpu()
    diff = (got - want).abs()
    tol = atol + rtol * want.abs()
    bad = diff > tol
    assert not bad.any(), (
        f"{label}: {bad.sum().item()}/{bad.numel()} mismatches, "
        f"max abs diff {diff.max().item():.5f}")


@pytest.fixture(scope="module")
def dev():
    assert torch.cuda.is_available()
    if not ops.HAS_EXT:
        raise RuntimeError("HIP extension missing on a GPU box — build failed")
    return torch.device("cuda:0")


def test_mfma_fragment_layout(dev):
    """Verifies the A/B/C lane mappings assumed by the prefill kernel."""
    torch.manual_seed(1)
    # asymmetric inputs so a transposed mapping cannot pass (guide §3)
    a = torch.randn(16, 32, device=dev).to(torch.bfloat16)
    b = torch.randn(32, 16, device=dev).to(torch.bfloat16)
    c = ops.mfma_probe_16x16x32(a, b)
    expect = a.float() @ b.float()
    _assert_close(c, expect, atol=5e-2, label="mfma 16x16x32 bf16")


def test_rmsnorm(dev):
    torch.manual_seed(0)
    for T, H in [(1, 4096), (17, 4096), (256, 4096), (64, 1024), (8, 14336)]:
        x = torch.randn(T, H, device=dev, dtype=torch.bfloat16)
        w = torch.randn(H, device=dev, dtype=torch.bfloat16)
        got = ops.rmsnorm(x, w, 1e-5)
        want = ref.rmsnorm(x.float().cpu(), w.float().cpu(), 1e-5)
        _assert_close(got, want, atol=3e-2, label=f"rmsnorm {T}x{H}")


def test_fused_add_rmsnorm(dev):
    torch.manual_seed(0)
    T, H = 33, 4096
    x = torch.randn(T, H, device=dev, dtype=torch.bfloat16)
    r = torch.randn(T, H, device=dev, dtype=torch.bfloat16)
    w = torch.randn(H, device=dev, dtype=torch.bfloat16)
    want_out, want_res = ref.fused_add_rmsnorm(
        x.float().cpu(), r.float().cpu(), w.float().cpu(), 1e-5)
    got_out, got_res = ops.fused_add_rmsnorm(x, r, w, 1e-5)
    _assert_close(got_res, want_res, atol=3e-2, label="residual")
    _assert_close(got_out, want_out, atol=3e-2, label="normed")


def test_rope(dev):
    torch.manual_seed(0)
    T, Hq, Hk, D = 9, 32, 8, 128
    pos = torch.randint(0, 4096, (T,), device=dev)
    q = torch.randn(T, Hq * D, device=dev, dtype=torch.bfloat16)
    k = torch.randn(T, Hk * D, device=dev, dtype=torch.bfloat16)
    table = ref.rope_table(D, 8192).to(dev)
    want_q, want_k = ref.rope(pos.cpu(), q.float().cpu(), k.float().cpu(),
                              table.cpu(), D, D)
    got_q, got_k = ops.rope(pos, q, k, table, D, D)
    _assert_close(got_q, want_q, atol=3e-2, label="rope q")
    _assert_close(got_k, want_k, atol=3e-2, label="rope k")


def test_fused_rope_cache(dev):
    torch.manual_seed(4)
    T, Hq, Hkv, D, bs, blocks = 11, 32, 8, 128, 16, 8
    qkv = torch.randn(T, (Hq + 2 * Hkv) * D, device=dev, dtype=torch.bfloat16)
    q = qkv[:, :Hq * D].unflatten(-1, (Hq, D))
    k = qkv[:, Hq * D:(Hq + Hkv) * D].unflatten(-1, (Hkv, D))
    v = qkv[:, (Hq + Hkv) * D:].unflatten(-1, (Hkv, D))
    kc = torch.zeros(blocks, Hkv, bs, D, device=dev, dtype=torch.bfloat16)
    vc = torch.zeros_like(kc)
    pos = torch.randint(0, 2048, (T,), device=dev)
    slots = torch.randperm(blocks * bs, device=dev)[:T]
    table = ref.rope_table(D, 4096).to(dev)
    # reference on CPU fp32
    q_ref, k_ref = ref.rope(pos.cpu(), q.float().cpu().reshape(T, -1),
                            k.float().cpu().reshape(T, -1), table.cpu(), D, D)
    kc_ref = torch.zeros(blocks, Hkv, bs, D)
    vc_ref = torch.zeros_like(kc_ref)
    ref.reshape_and_cache(k_ref.reshape(T, Hkv, D), v.float().cpu(),
                          kc_ref, vc_ref, slots.cpu())
    ops.fused_rope_cache(pos, q, k, v, kc, vc, slots, table, D)
    _assert_close(q.reshape(T, -1), q_ref, atol=3e-2, label="fused q")
    _assert_close(kc, kc_ref, atol=3e-2, label="fused k_cache")
    _assert_close(vc, vc_ref, atol=3e-2, label="fused v_cache")


def test_fused_rope_cache_mrope(dev):
    """Sectioned M-RoPE (Qwen2-VL) path of the fused kernel vs the CPU
    reference with [3, T] positions."""
    torch.manual_seed(5)
    T, Hq, Hkv, D, bs, blocks = 9, 8, 2, 128, 16, 8
    sections = (16, 24, 24)
    qkv = torch.randn(T, (Hq + 2 * Hkv) * D, device=dev, dtype=torch.bfloat16)
    q = qkv[:, :Hq * D].unflatten(-1, (Hq, D))
    k = qkv[:, Hq * D:(Hq + Hkv) * D].unflatten(-1, (Hkv, D))
    v = qkv[:, (Hq + Hkv) * D:].unflatten(-1, (Hkv, D))
    kc = torch.zeros(blocks, Hkv, bs, D, device=dev, dtype=torch.bfloat16)
    vc = torch.zeros_like(kc)
    pos3 = torch.randint(0, 2048, (3, T), device=dev)
    slots = torch.randperm(blocks * bs, device=dev)[:T]
    table = ref.rope_table(D, 4096).to(dev)
    q_ref, k_ref = ref.rope(pos3.cpu(), q.float().cpu().reshape(T, -1),
                            k.float().cpu().reshape(T, -1), table.cpu(), D, D,
                            mrope_sections=sections)
    kc_ref = torch.zeros(blocks, Hkv, bs, D)
    vc_ref = torch.zeros_like(kc_ref)
    ref.reshape_and_cache(k_ref.reshape(T, Hkv, D), v.float().cpu(),
                          kc_ref, vc_ref, slots.cpu())
    ops.fused_rope_cache(pos3, q, k, v, kc, vc, slots, table, D,
                         mrope_sections=sections)
    _assert_close(q.reshape(T, -1), q_ref, atol=3e-2, label="mrope q")
    _assert_close(kc, kc_ref, atol=3e-2, label="mrope k_cache")
    _assert_close(vc, vc_ref, atol=3e-2, label="mrope v_cache")


def test_silu_and_mul(dev):
    x = torch.randn(37, 2 * 14336, device=dev, dtype=torch.bfloat16)
    got = ops.silu_and_mul(x)
    want = ref.silu_and_mul(x.float().cpu())
    _assert_close(got, want, atol=3e-2, label="silu_and_mul")


def test_reshape_and_cache(dev):
    torch.manual_seed(0)
    n_kv, bs, D, blocks = 8, 16, 128, 16
    k_cache = torch.zeros(blocks, n_kv, bs, D, device=dev, dtype=torch.bfloat16)
    v_cache = torch.zeros_like(k_cache)
    T = 40
    k = torch.randn(T, n_kv, D, device=dev, dtype=torch.bfloat16)
    v = torch.randn(T, n_kv, D, device=dev, dtype=torch.bfloat16)
    slots = torch.randperm(blocks * bs, device=dev)[:T]
    ops.reshape_and_cache(k, v, k_cache, v_cache, slots)
    kc = torch.zeros_like(k_cache).cpu().float()
    vc = torch.zeros_like(v_cache).cpu().float()
    ref.reshape_and_cache(k.float().cpu(), v.float().cpu(), kc, vc, slots.cpu())
    assert torch.equal(k_cache.float().cpu(), kc)
    assert torch.equal(v_cache.float().cpu(), vc)


@pytest.mark.parametrize("G,D", [(1, 128), (4, 128), (7, 128), (8, 128),
                                 (4, 64)])
def test_paged_attn_decode(dev, G, D):
    torch.manual_seed(G)
    n_kv, bs = 4, 16
    Hq = n_kv * G
    seq_lens_list = [1, 16, 100, 1023]
    S = len(seq_lens_list)
    max_blocks = (max(seq_lens_list) + bs - 1) // bs
    blocks = S * max_blocks + 1
    k_cache = torch.randn(blocks, n_kv, bs, D, device=dev, dtype=torch.bfloat16)
    v_cache = torch.randn_like(k_cache)
    perm = torch.randperm(blocks - 1, device=dev)[: S * max_blocks].to(torch.int32)
    block_tables = perm.reshape(S, max_blocks).contiguous()
    seq_lens = torch.tensor(seq_lens_list, dtype=torch.int32, device=dev)
    q = torch.randn(S, Hq, D, device=dev, dtype=torch.bfloat16)
    scale = 1.0 / math.sqrt(D)
    got = ops.paged_attn_decode(q, k_cache, v_cache, block_tables, seq_lens, scale)
    want = ref.paged_attn_decode(
        q.float().cpu(), k_cache.float().cpu(), v_cache.float().cpu(),
        block_tables.cpu(), seq_lens.cpu(), scale)
    _assert_close(got, want, atol=3e-2, label=f"decode G={G}")


@pytest.mark.parametrize("ctx", [0, 64, 333])
def test_paged_attn_prefill(dev, ctx):
    torch.manual_seed(ctx)
    n_kv, D, bs = 2, 128, 16
    Hq = 8
    q_lens = [1, 64, 200]
    seq_lens_list = [ctx + n for n in q_lens]
    S = len(q_lens)
    max_blocks = (max(seq_lens_list) + bs - 1) // bs
    blocks = S * max_blocks + 1
    k_cache = torch.randn(blocks, n_kv, bs, D, device=dev, dtype=torch.bfloat16)
    v_cache = torch.randn_like(k_cache)
    perm = torch.randperm(blocks - 1, device=dev)[: S * max_blocks].to(torch.int32)
    block_tables = perm.reshape(S, max_blocks).contiguous()
    seq_lens = torch.tensor(seq_lens_list, dtype=torch.int32, device=dev)
    cu = [0]
    for n in q_lens:
        cu.append(cu[-1] + n)
    cu_q = torch.tensor(cu, dtype=torch.int32, device=dev)
    total_q = cu[-1]
    q = torch.randn(total_q, Hq, D, device=dev, dtype=torch.bfloat16)
    scale = 1.0 / math.sqrt(D)
    got = ops.paged_attn_prefill(q, k_cache, v_cache, block_tables, cu_q,
                                 seq_lens, scale)
    want = ref.paged_attn_prefill(
        q.float().cpu(), k_cache.float().cpu(), v_cache.float().cpu(),
        block_tables.cpu(), cu_q.cpu(), seq_lens.cpu(), scale)
    _assert_close(got, want, atol=3e-2, label=f"prefill ctx={ctx}")


@pytest.mark.parametrize("M,N,K", [(128, 128, 64), (100, 1280, 1184),
                                   (256, 512, 1024), (64, 3584, 5120)])
def test_mfma_gemm(dev, M, N, K):
    torch.manual_seed(M + N)
    a = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
    b = torch.randn(N, K, device=dev, dtype=torch.bfloat16)
    bias = torch.randn(N, device=dev, dtype=torch.bfloat16)
    got = ops.mfma_gemm(a, b, bias)
    want = (a.float() @ b.float().t()) + bias.float()
    _assert_close(got, want, atol=0.05 + 0.02 * (K / 1024),
                  label=f"mfma_gemm {M}x{N}x{K}")


@pytest.mark.parametrize("M,N,K", [(1, 6144, 4096), (64, 6144, 4096),
                                   (64, 4096, 4096), (100, 4096, 14336),
                                   (33, 512, 1536), (128, 1024, 4160)])
def test_skinny_gemm(dev, M, N, K):
    torch.manual_seed(M)
    a = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
    b = torch.randn(N, K, device=dev, dtype=torch.bfloat16)
    bias = torch.randn(N, device=dev, dtype=torch.bfloat16)
    got = ops.skinny_gemm(a, b, bias)
    want = torch.nn.functional.linear(a.float(), b.float(), bias.float())
    _assert_close(got, want, atol=0.05 + 0.02 * (K / 1024),
                  label=f"skinny {M}x{N}x{K}")


@pytest.mark.parametrize("M,N,K", [(1, 6144, 4096), (64, 6144, 4096),
                                   (64, 4096, 4096), (64, 28672, 4096),
                                   (64, 4096, 14336), (33, 512, 1536),
                                   (128, 1024, 4352)])
def test_packed_gemm(dev, M, N, K):
    torch.manual_seed(M + 1)
    a = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
    b = torch.randn(N, K, device=dev, dtype=torch.bfloat16)
    bias = torch.randn(N, device=dev, dtype=torch.bfloat16)
    wp = ops.pack_gemm_weight(b)
    got = ops.packed_gemm(a, wp, N, bias)
    want = torch.nn.functional.linear(a.float(), b.float(), bias.float())
    _assert_close(got, want, atol=0.05 + 0.02 * (K / 1024),
                  label=f"packed {M}x{N}x{K}")


def test_greedy_sample(dev):
    torch.manual_seed(0)
    logits = torch.randn(33, 128256, device=dev, dtype=torch.bfloat16)
    got = ops.greedy_sample(logits)
    want = logits.float().argmax(-1)
    assert torch.equal(got.cpu(), want.cpu())


def test_copy_blocks(dev):
    n_kv, bs, D = 2, 16, 128
    k_cache = torch.randn(8, n_kv, bs, D, device=dev, dtype=torch.bfloat16)
    v_cache = torch.randn_like(k_cache)
    k0, v0 = k_cache.clone(), v_cache.clone()
    pairs = torch.tensor([[0, 5], [2, 7]], dtype=torch.long, device=dev)
    ops.copy_blocks(k_cache, v_cache, pairs)
    assert torch.equal(k_cache[5], k0[0]) and torch.equal(v_cache[7], v0[2])
    assert torch.equal(k_cache[1], k0[1])
