#!/usr/bin/env python3
"""Prefill-shape GEMM rates (M=1024..4096) with/without the TunableOp
table: is ~700 TF/s the library ceiling here or a bad algo pick?"""
import os, sys
ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)
if os.environ.get("USE_TUNABLE", "1") == "1":
    _T = os.path.join(ROOT, "configs", "tunableop_gfx950.csv")
    if os.path.exists(_T):
        os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
        os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "0")
        os.environ.setdefault("PYTORCH_TUNABLEOP_FILENAME", _T)
import torch

def t_us(fn, n=30):
    s = torch.cuda.Event(enable_timing=True); e = torch.cuda.Event(enable_timing=True)
    fn(); torch.cuda.synchronize(); s.record()
    for _ in range(n): fn()
    e.record(); torch.cuda.synchronize()
    return s.elapsed_time(e) / n * 1000

print("tunable:", os.environ.get("PYTORCH_TUNABLEOP_ENABLED", "0"))
dev = "cuda:0"
for M in (1024, 2048, 4096):
    for name, N, K in (("qkv",6144,4096),("o",4096,4096),
                       ("gate_up",28672,4096),("down",4096,14336)):
        a = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
        w = torch.randn(N, K, device=dev, dtype=torch.bfloat16)
        t = t_us(lambda: torch.nn.functional.linear(a, w))
        tf = 2*M*N*K/1e12/(t/1e6)
        print(f"M={M:5d} {name:8s}: {t:8.1f}us {tf:7.0f} TF/s")
