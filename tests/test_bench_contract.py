"""The driver contract for bench.py: single JSON line on rank 0, whole-job
aggregate value, works under torch.distributed.run with N ranks (the driver
launches N=2,4,8 for the scaling bench).

Default mode is the full serving stack (SLO-goodput through HTTP -> master
-> RPC -> worker processes); --mode engine is the closed-loop engine-step
microbench."""
import json
import os
import socket
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _free_port():
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _last_json(out: str) -> dict:
    for line in reversed(out.strip().splitlines()):
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError(f"no JSON line in output:\n{out[-2000:]}")


def _check(rec: dict, n: int, parallelism: str):
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in rec, f"missing {key}"
    assert rec["n_gpus"] == n
    assert rec["value"] > 0 and rec["ms_per_step"] > 0
    assert rec["config"]["parallelism"] == parallelism
    assert rec["data"] == "synthetic"


def test_bench_serving_single_rank_cpu():
    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "3", "--warmup", "1",
         "--device", "cpu"],
        cwd=ROOT, capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stderr[-2000:]
    rec = _last_json(out.stdout)
    _check(rec, 1, "colocated-1gpu")
    # SLO-goodput extras are part of the record (self-describing metric)
    for key in ("p50_ttft_ms", "slo_ttft_ms", "arrival_rate_req_s",
                "total_tok_per_s"):
        assert key in rec, f"missing {key}"
    assert rec["metric"].startswith("SLO-goodput")


def test_bench_serving_torchrun_world2_cpu():
    port = _free_port()
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(port), "bench.py", "--gpus", "2",
         "--steps", "3", "--warmup", "1", "--device", "cpu"],
        cwd=ROOT, capture_output=True, text=True, timeout=600)
    assert out.returncode == 0, out.stderr[-2000:]
    rec = _last_json(out.stdout)
    _check(rec, 2, "pd-1p1d")       # PD-disaggregated topology at N=2


def test_bench_engine_mode_cpu():
    out = subprocess.run(
        [sys.executable, "bench.py", "--mode", "engine", "--steps", "2",
         "--warmup", "1", "--device", "cpu", "--model", "llama-tiny"],
        cwd=ROOT, capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stderr[-2000:]
    rec = _last_json(out.stdout)
    _check(rec, 1, "dp1")
    assert rec["metric"].startswith("engine-step")
