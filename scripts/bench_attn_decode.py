#!/usr/bin/env python3
"""Decode-attention microbench across batch/seq shapes (timed, TB/s)."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from xllm_service_amd import ops

dev = "cuda:0"
torch.manual_seed(0)
n_kv, D, bs, G = 8, 128, 16, 4
for S, L in ((64, 2048), (128, 1100), (128, 2048), (32, 4000)):
    blocks = S * ((L + bs - 1) // bs) + 1
    kc = torch.randn(blocks, n_kv, bs, D, device=dev, dtype=torch.bfloat16)
    vc = torch.randn_like(kc)
    bt = torch.arange(S * (L // bs), dtype=torch.int32,
                      device=dev).reshape(S, -1).contiguous()
    sl = torch.full((S,), L, dtype=torch.int32, device=dev)
    q = torch.randn(S, n_kv * G, D, device=dev, dtype=torch.bfloat16)
    out = torch.empty_like(q)
    st = torch.cuda.Event(enable_timing=True); en = torch.cuda.Event(enable_timing=True)
    for _ in range(10):
        ops.paged_attn_decode(q, kc, vc, bt, sl, 0.088, out=out)
    torch.cuda.synchronize(); st.record()
    for _ in range(50):
        ops.paged_attn_decode(q, kc, vc, bt, sl, 0.088, out=out)
    en.record(); torch.cuda.synchronize()
    us = st.elapsed_time(en) / 50 * 1000
    gb = S * L * n_kv * D * 2 * 2 / 1e9
    print(f"S={S:4d} L={L:5d}: {us:7.1f} us  {gb/(us/1e6):5.2f} TB/s")
    del kc, vc
    torch.cuda.empty_cache()
