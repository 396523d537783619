#!/usr/bin/env python3
"""Cold-stream decode-GEMM microbench: hipBLASLt (TunableOp table) vs the
packed-weight streaming kernel (csrc/ops/packed_gemm.hip) vs skinny v2.

Decode graphs re-read every layer's weights each step, so L2 is always cold
for weights — the bench round-robins over independent weight copies sized
past L2 to reproduce that. Shapes are Llama-3-8B TP1 decode projections
plus lm_head. Reports us/op and effective weight-stream TB/s.

  gpurun -- 'python scripts/bench_gemm.py > gpurun_out/gemm_bench.log 2>&1'
"""
import os
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)

_T = os.path.join(ROOT, "configs", "tunableop_gfx950.csv")
if os.path.exists(_T):
    os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
    os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "0")
    os.environ.setdefault("PYTORCH_TUNABLEOP_FILENAME", _T)

import torch

from xllm_service_amd import ops

SHAPES = [  # (name, N, K)
    ("qkv", 6144, 4096),
    ("o", 4096, 4096),
    ("gate_up", 28672, 4096),
    ("down", 4096, 14336),
    ("lm_head", 128256, 4096),
]
M = int(os.environ.get("BM", "64"))
COPIES = 8
ITERS = 48


def timed(fn, n=ITERS):
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    fn(0)  # warm compile
    torch.cuda.synchronize()
    s.record()
    for i in range(1, n + 1):
        fn(i)
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / n * 1000.0  # us


def main():
    dev = "cuda:0"
    torch.manual_seed(0)
    print(f"M={M}, {COPIES} weight copies, {ITERS} iters (cold stream)")
    for name, N, K in SHAPES:
        copies = min(COPIES, max(2, (512 << 20) // (N * K * 2)))
        a = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
        ws = [torch.randn(N, K, device=dev, dtype=torch.bfloat16)
              for _ in range(copies)]
        wps = [ops.pack_gemm_weight(w) for w in ws]
        gb = N * K * 2 / 1e9

        t_lib = timed(lambda i: torch.nn.functional.linear(a, ws[i % copies]))
        t_s = {}
        for s_try in (1, 2, 4):
            if N % (64 * s_try):
                continue
            try:
                t_s[s_try] = timed(
                    lambda i: ops.packed_gemm(a, wps[i % copies], N,
                                              s_override=s_try))
            except Exception:
                pass
        s_best = min(t_s, key=t_s.get)
        t_pk = t_s[s_best]
        sweep = "/".join(f"S{k}:{v:.0f}" for k, v in t_s.items())
        from xllm_service_amd.ops import _ops
        t_pr = timed(lambda i: _ops.packed_gemm_probe(a, wps[i % copies], N))
        try:
            t_sk = timed(lambda i: ops.skinny_gemm(a, ws[i % copies])) \
                if M <= 64 else float("nan")
        except Exception:
            t_sk = float("nan")

        # numerics spot check
        got = ops.packed_gemm(a, wps[0], N).float()
        want = torch.nn.functional.linear(a.float(), ws[0].float())
        err = (got - want).abs().max().item()
        print(f"{name:8s} N={N:6d} K={K:6d} ({gb*1000:6.1f} MB): "
              f"lib {t_lib:7.1f}us ({gb/(t_lib/1e6):5.2f} TB/s) | "
              f"packed[{sweep}] {t_pk:7.1f}us ({gb/(t_pk/1e6):5.2f} TB/s) | "
              f"noA {t_pr:7.1f}us ({gb/(t_pr/1e6):5.2f} TB/s) | "
              f"skinny {t_sk:7.1f}us | maxerr {err:.3f}")
        del ws, wps
        torch.cuda.empty_cache()


if __name__ == "__main__":
    main()
