#!/usr/bin/env python3
"""PD-disaggregated SLO-goodput bench (VERDICT round-1 item 1).

This is a thin launcher: the PD measurement lives in the driver-contract
bench (bench.py), which at N>1 runs the real topology — rank 0 hosts the
master process + load generators, every rank hosts one worker process
(PREFILL/DECODE split 1P+1D at N=2, 1P+3D at N=4, 2P+6D at N=8), open-loop
Poisson arrivals through uvicorn TCP -> master -> msgrpc -> workers, KV
blocks migrating prefill->decode over IPC/xGMI, goodput gated on
p50 TTFT <= 1 s.

  python scripts/bench_pd.py --gpus 8 --steps 120 --warmup 10
"""
import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def main():
    args = sys.argv[1:]
    n = 8
    if "--gpus" in args:
        n = int(args[args.index("--gpus") + 1])
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           f"--nproc-per-node={n}", "--master-addr", "127.0.0.1",
           os.path.join(ROOT, "bench.py")] + args
    if "--gpus" not in args:
        cmd += ["--gpus", str(n)]
    raise SystemExit(subprocess.run(cmd, cwd=ROOT).returncode)


if __name__ == "__main__":
    main()
