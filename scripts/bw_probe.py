#!/usr/bin/env python3
"""HBM read-bandwidth vs working-set size: is the ~2 TB/s cold-GEMM cap a
kernel problem or a machine characteristic at small (32-256 MB) streams?"""
import os, sys, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

def main():
    dev = "cuda:0"
    flt = torch.zeros(256 << 20, dtype=torch.uint8, device=dev)
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    for mb in (16, 34, 64, 117, 235, 512, 1024):
        n = mb * (1 << 20) // 2
        x = torch.randn(n, dtype=torch.bfloat16, device=dev)
        ts = []
        for i in range(8):
            flt.float().sum()                   # L3 flush
            torch.cuda.synchronize()
            s.record()
            y = x.view(-1, 8192).float().sum(dim=0)  # row-streaming read
            e.record()
            torch.cuda.synchronize()
            ts.append(s.elapsed_time(e) * 1e3)
        ts.sort()
        t = ts[len(ts)//2]
        print(f"{mb:5d} MB: {t:8.1f} us  {mb/t*1e3/1e3:5.2f} TB/s")
        del x

if __name__ == "__main__":
    main()
