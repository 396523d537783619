#!/usr/bin/env python3
"""Kernel microbenchmarks on MI355X: decode-shape GEMMs (L3-cold, the
steady-state serving condition) and the paged-attention kernels.

Run: gpurun -- 'python scripts/bench_kernels.py'
"""
import os
import sys
import time

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)

import torch  # noqa: E402


def evt_time(fn, flush=None, reps=20, warmup=5):
    """Median CUDA-event time of fn() with optional L3 flush between reps."""
    times = []
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    for i in range(warmup + reps):
        if flush is not None:
            flush()
        s.record()
        fn()
        e.record()
        torch.cuda.synchronize()
        if i >= warmup:
            times.append(s.elapsed_time(e) * 1000)  # us
    times.sort()
    return times[len(times) // 2]


def main():
    dev = "cuda:0"
    torch.manual_seed(0)
    # L3 flusher: stream 512 MB so no GEMM operand stays L3-resident
    flt = torch.zeros(256 << 20, dtype=torch.uint8, device=dev)

    def flush():
        flt.add_(1)

    print("== decode GEMMs (M=64), L3-warm vs L3-cold ==")
    for (M, N, K, tag) in [(64, 6144, 4096, "qkv"), (64, 4096, 4096, "o"),
                           (64, 28672, 4096, "gate_up"),
                           (64, 4096, 14336, "down"),
                           (64, 128256, 4096, "lm_head")]:
        x = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
        W = torch.randn(N, K, device=dev, dtype=torch.bfloat16)
        fn = lambda: torch.nn.functional.linear(x, W)  # noqa: E731
        warm = evt_time(fn)
        cold = evt_time(fn, flush=flush)
        mb = N * K * 2 / 1e6
        print(f"{tag:8s} {M}x{N}x{K}: warm {warm:7.1f}us  cold {cold:7.1f}us "
              f" (W {mb:.0f}MB -> cold {mb/cold*1e3/1e3:.2f} TB/s)")

    print("== skinny_gemm vs library (L3-cold) ==")
    from xllm_service_amd import ops as xops
    for (M, N, K, tag) in [(64, 6144, 4096, "qkv"), (64, 4096, 4096, "o"),
                           (64, 28672, 4096, "gate_up"),
                           (64, 4096, 14336, "down")]:
        x = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
        W = torch.randn(N, K, device=dev, dtype=torch.bfloat16)
        t = evt_time(lambda: xops.skinny_gemm(x, W), flush=flush)
        tw = evt_time(lambda: xops.skinny_gemm(x, W))
        mb = N * K * 2 / 1e6
        print(f"skinny {tag:8s}: warm {tw:7.1f}us  cold {t:7.1f}us "
              f"({mb/t*1e3/1e3:.2f} TB/s cold)")

    print("== paged attention decode (batch 64, seq 1024, 8 kv heads, G=4) ==")
    from xllm_service_amd import ops
    n_kv, D, bs, G, S, L = 8, 128, 16, 4, 64, 1024
    blocks = S * (L // bs) + 1
    kc = torch.randn(blocks, n_kv, bs, D, device=dev, dtype=torch.bfloat16)
    vc = torch.randn_like(kc)
    bt = torch.arange(S * (L // bs), dtype=torch.int32,
                      device=dev).reshape(S, L // bs).contiguous()
    sl = torch.full((S,), L, dtype=torch.int32, device=dev)
    q = torch.randn(S, n_kv * G, D, device=dev, dtype=torch.bfloat16)
    out = torch.empty_like(q)
    kv_mb = S * L * n_kv * D * 2 * 2 / 1e6

    def attn():
        ops.paged_attn_decode(q, kc, vc, bt, sl, 0.088, out=out)

    t = evt_time(attn, flush=flush)
    print(f"decode attn: {t:.1f}us  KV {kv_mb:.0f}MB -> {kv_mb/t*1e3/1e3:.2f} TB/s")

    for S2, L2 in [(16, 1024), (128, 1024), (64, 4000)]:
        blocks2 = S2 * ((L2 + 15) // bs) + 1
        kc2 = torch.randn(blocks2, n_kv, bs, D, device=dev, dtype=torch.bfloat16)
        vc2 = torch.randn_like(kc2)
        bt2 = torch.arange(S2 * ((L2 + 15) // bs), dtype=torch.int32,
                           device=dev).reshape(S2, -1).contiguous()
        sl2 = torch.full((S2,), L2, dtype=torch.int32, device=dev)
        q2 = torch.randn(S2, n_kv * G, D, device=dev, dtype=torch.bfloat16)
        o2 = torch.empty_like(q2)
        mb2 = S2 * L2 * n_kv * D * 2 * 2 / 1e6

        def attn2():
            ops.paged_attn_decode(q2, kc2, vc2, bt2, sl2, 0.088, out=o2)

        t2 = evt_time(attn2, flush=flush)
        print(f"decode attn S={S2} L={L2}: {t2:.1f}us  {mb2/t2*1e3/1e3:.2f} TB/s")

    print("== prefill attention (16 seqs x 1024 tokens) ==")
    S3, Lp = 16, 1024
    cu = torch.arange(0, (S3 + 1) * Lp, Lp, dtype=torch.int32, device=dev)
    q3 = torch.randn(S3 * Lp, n_kv * G, D, device=dev, dtype=torch.bfloat16)
    o3 = torch.empty_like(q3)
    blocks3 = S3 * (Lp // bs) + 1
    kc3 = torch.randn(blocks3, n_kv, bs, D, device=dev, dtype=torch.bfloat16)
    vc3 = torch.randn_like(kc3)
    bt3 = torch.arange(S3 * (Lp // bs), dtype=torch.int32,
                       device=dev).reshape(S3, -1).contiguous()
    sl3 = torch.full((S3,), Lp, dtype=torch.int32, device=dev)

    def prefill():
        ops.paged_attn_prefill(q3, kc3, vc3, bt3, cu, sl3, 0.088, out=o3)

    t3 = evt_time(prefill, flush=flush, reps=10)
    # causal FLOPs: per seq 2 * 2 * L^2/2 * D * Hq
    fl = S3 * 2 * 2 * (Lp * Lp / 2) * D * (n_kv * G) / 1e12
    print(f"prefill attn: {t3:.1f}us  {fl/(t3/1e6):.0f} TFLOP/s")

    print("== small ops (decode shapes, T=64) ==")
    x = torch.randn(64, 4096, device=dev, dtype=torch.bfloat16)
    w = torch.randn(4096, device=dev, dtype=torch.bfloat16)
    r = torch.randn_like(x)
    print(f"rmsnorm: {evt_time(lambda: ops.rmsnorm(x, w, 1e-5)):.1f}us")
    print(f"fused_add_rms: "
          f"{evt_time(lambda: ops.fused_add_rmsnorm(x, r, w, 1e-5)):.1f}us")
    g = torch.randn(64, 28672, device=dev, dtype=torch.bfloat16)
    print(f"silu_mul: {evt_time(lambda: ops.silu_and_mul(g)):.1f}us")


if __name__ == "__main__":
    main()
