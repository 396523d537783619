#!/usr/bin/env python3
"""Does warming the LLC ahead of a decode GEMM pay?

Per decode step the engine streams ~14 GB of weights but averages only
~2.1 TB/s of DRAM — individual GEMMs cap at 1.6-5 TB/s. If a side-stream
"toucher" can pull the NEXT op's weights into the 256 MB LLC while the
current op computes, every GEMM reads warm. This probe measures:
  1. warm vs cold library GEMM rates (does LLC-resident W actually help?)
  2. serial [touch W2; gemm(W1)] vs overlapped on two streams
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

M = 64


def t_ms(fn, n=32, sync_each=False):
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    fn(0)
    torch.cuda.synchronize()
    s.record()
    for i in range(1, n + 1):
        fn(i)
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / n


def main():
    dev = "cuda:0"
    torch.manual_seed(0)
    flusher = torch.zeros(512 << 20, dtype=torch.uint8, device=dev)

    for name, N, K in (("o", 4096, 4096), ("gate_up", 28672, 4096),
                       ("down", 4096, 14336)):
        gb = N * K * 2 / 1e9
        a = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
        w = torch.randn(N, K, device=dev, dtype=torch.bfloat16)
        copies = [torch.randn(N, K, device=dev, dtype=torch.bfloat16)
                  for _ in range(8)]

        # cold: rotate over 8 copies (past LLC)
        t_cold = t_ms(lambda i: torch.nn.functional.linear(a, copies[i % 8]))
        # warm: same copy every time (LLC-resident if it fits)
        t_warm = t_ms(lambda i: torch.nn.functional.linear(a, w))
        print(f"{name:8s} ({gb*1000:6.1f} MB): cold {t_cold*1000:7.1f}us "
              f"({gb/(t_cold/1e3):5.2f} TB/s) | warm {t_warm*1000:7.1f}us "
              f"({gb/(t_warm/1e3):5.2f} TB/s)")

        # explicit touch then gemm, serial on one stream
        def touch(wt):
            # read-sum in big rows: allocates into cache hierarchy
            return wt.view(-1, 16384).float().sum()

        def serial(i):
            touch(copies[(i + 1) % 8])
            torch.nn.functional.linear(a, copies[i % 8])

        t_serial = t_ms(serial)

        side = torch.cuda.Stream()

        def overlap(i):
            with torch.cuda.stream(side):
                touch(copies[(i + 1) % 8])
            torch.nn.functional.linear(a, copies[i % 8])
            torch.cuda.current_stream().wait_stream(side)

        t_ov = t_ms(overlap)
        # touched-then-used: does a prior touch make the NEXT gemm warm?
        def touched_gemm(i):
            touch(copies[i % 8])
            torch.nn.functional.linear(a, copies[i % 8])

        t_tg = t_ms(touched_gemm)
        print(f"         serial touch+gemm {t_serial*1000:7.1f}us | "
              f"overlap {t_ov*1000:7.1f}us | touch-then-use "
              f"{t_tg*1000:7.1f}us")
        del copies
        torch.cuda.empty_cache()
    del flusher


if __name__ == "__main__":
    main()
