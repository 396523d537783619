#!/usr/bin/env python3
"""Decode-attention kernel in isolation (for rocprofv3 --pmc runs)."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from xllm_service_amd import ops

dev = "cuda:0"
torch.manual_seed(0)
n_kv, D, bs, G, S, L = 8, 128, 16, 4, 64, 1024
blocks = S * (L // bs) + 1
kc = torch.randn(blocks, n_kv, bs, D, device=dev, dtype=torch.bfloat16)
vc = torch.randn_like(kc)
bt = torch.arange(S * (L // bs), dtype=torch.int32, device=dev).reshape(S, -1).contiguous()
sl = torch.full((S,), L, dtype=torch.int32, device=dev)
q = torch.randn(S, n_kv * G, D, device=dev, dtype=torch.bfloat16)
out = torch.empty_like(q)
for _ in range(30):
    ops.paged_attn_decode(q, kc, vc, bt, sl, 0.088, out=out)
torch.cuda.synchronize()
print("done")
