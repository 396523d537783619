#!/usr/bin/env python3
"""Flagship benchmark — the driver contract (see repo instructions).

Default mode measures the BASELINE.json headline the honest way: SLO-goodput
(output tok/s from requests whose TTFT meets the 1 s SLO) through the FULL
serving stack — HTTP (uvicorn, real TCP) -> master scheduler -> msgrpc ->
per-GPU worker processes -> SSE streaming back — under open-loop Poisson
arrivals whose rate is auto-calibrated to ~0.9x the measured capacity.

Topology scales with --gpus N (one worker process per GPU):
  N=1            one DEFAULT (colocated prefill+decode) worker
  N>=2           PD-disaggregated: max(1, N//4) PREFILL + rest DECODE
                 (8 GPUs -> 2P+6D, BASELINE config 3), KV blocks migrating
                 prefill->decode over xGMI (hipIpc + peer copies)

A "step" is one COMPLETED request in steady state: after --warmup completed
requests under the Poisson load, the timed region spans exactly --steps
further completions (barrier + torch.cuda.synchronize on both sides, MAX
elapsed over ranks). value = sum of output tokens of SLO-passing requests
completed in the window / elapsed — the whole-job aggregate.

  python bench.py --gpus 1 --steps 20 --warmup 5
  python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
      --master-addr 127.0.0.1 bench.py --gpus 8 --steps 120 --warmup 10

--mode engine keeps the round-1 closed-loop engine-step microbench (kernel
iteration tool; its metric string says so).
"""
from __future__ import annotations

import argparse
import asyncio
import json
import os
import random
import socket
import subprocess
import sys
import time

ROOT = os.path.dirname(os.path.abspath(__file__))

# hipBLASLt/rocBLAS algo selection tuned offline on MI355X (TunableOp);
# must be configured before the first torch import in the process.
_TUNABLE = os.path.join(ROOT, "configs", "tunableop_gfx950.csv")
if os.path.exists(_TUNABLE):
    os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
    os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "0")
    os.environ.setdefault("PYTORCH_TUNABLEOP_FILENAME", _TUNABLE)

if os.environ.get("BENCH_LOADGEN") != "1":
    import torch


def parse_args():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=20,
                    help="serving: timed completed requests; "
                         "engine: timed engine steps")
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--mode", choices=("serving", "engine"),
                    default="serving")
    ap.add_argument("--model", default="llama-3-8b")
    ap.add_argument("--concurrency", type=int, default=128,
                    help="ramp/calibration concurrency per decode GPU "
                         "(batch sweep on MI355X: 64->9.7k, 96->11.2k, "
                         "128->13.9k, 192->13.8k tok/s engine-step)")
    ap.add_argument("--input-len", type=int, default=1024)
    ap.add_argument("--output-len", type=int, default=1024)
    ap.add_argument("--max-batched-tokens", type=int, default=8192,
                    help="worker chunked-prefill budget per step")
    ap.add_argument("--device", default=None)
    ap.add_argument("--policy", default="CAR")
    ap.add_argument("--slo-ttft-ms", type=float, default=1000.0)
    ap.add_argument("--pace", type=float, default=0.95,
                    help="Poisson arrival rate as a fraction of the "
                         "ramp-measured capacity")
    ap.add_argument("--arrival-rate", type=float, default=0.0,
                    help="req/s override (0 = auto-calibrate)")
    ap.add_argument("--ramp-s", type=float, default=0.0,
                    help="capacity-calibration window (0 = auto)")
    ap.add_argument("--push-interval-ms", type=float, default=0.0,
                    help="worker->master token push coalescing window "
                         "(0 = auto: scales with decode GPU count so the "
                         "master's SSE fan-out stays off the hot path)")
    ap.add_argument("--max-kv-blocks", type=int, default=0)
    ap.add_argument("--startup-timeout", type=float, default=900.0)
    return ap.parse_args()


# --------------------------------------------------------------------------
# engine mode (round-1 closed-loop engine-step microbench)
# --------------------------------------------------------------------------
def run_engine_mode(args, rank, world, local_rank, use_gpu, dist, device,
                    model_name):
    from xllm_service_amd.engine.engine import LLMEngine
    from xllm_service_amd.engine.sampling import SamplingParams
    from xllm_service_amd.models.config import get_config

    cfg = get_config(model_name)
    eng = LLMEngine(model_name, device=device,
                    max_num_seqs=args.concurrency * 2,
                    max_batched_tokens=args.max_batched_tokens,
                    max_kv_blocks=None if use_gpu else 4096,
                    seed=0)
    torch.manual_seed(1234 + rank)
    rid = 0

    def feed(n):
        nonlocal rid
        for _ in range(n):
            prompt = torch.randint(0, cfg.vocab_size,
                                   (args.input_len,)).tolist()
            eng.add_request(f"r{rank}-{rid}", prompt,
                            SamplingParams(max_tokens=args.output_len,
                                           ignore_eos=True))
            rid += 1

    feed(args.concurrency)

    def one_step():
        outs = eng.step()
        done = sum(1 for o in outs if o.finished)
        if done:
            feed(done)
        return sum(len(o.new_token_ids) for o in outs)

    for _ in range(args.warmup):
        one_step()
    if dist:
        dist.barrier()
    if use_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    generated = 0
    for _ in range(args.steps):
        generated += one_step()
    if use_gpu:
        torch.cuda.synchronize()
    if dist:
        dist.barrier()
    t1 = time.perf_counter()
    elapsed = t1 - t0
    if dist:
        te = torch.tensor([elapsed], dtype=torch.float64)
        dist.all_reduce(te, op=dist.ReduceOp.MAX)
        elapsed = float(te.item())
        tg = torch.tensor([float(generated)], dtype=torch.float64)
        dist.all_reduce(tg, op=dist.ReduceOp.SUM)
        generated = int(tg.item())
    n_gpus = world if world > 1 else (1 if use_gpu else args.gpus)
    if rank == 0:
        print(json.dumps({
            "metric": "engine-step throughput (out tok/s, closed loop, "
                      "no serving stack), Llama-3-8B",
            "value": round(generated / elapsed, 2),
            "unit": "tok/s", "n_gpus": n_gpus,
            "steps": args.steps, "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000, 3),
            "higher_is_better": True, "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if use_gpu else "fp32", "data": "synthetic",
            "config": {"model": model_name,
                       "global_batch": args.concurrency * n_gpus,
                       "seq_len": args.input_len + args.output_len,
                       "input_len": args.input_len,
                       "output_len": args.output_len,
                       "parallelism": f"dp{n_gpus}"},
        }))


# --------------------------------------------------------------------------
# serving mode helpers
# --------------------------------------------------------------------------
def free_port():
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def topology(n: int):
    """Instance role per rank."""
    if n <= 1:
        return ["DEFAULT"]
    n_p = max(1, n // 4)
    return ["PREFILL"] * n_p + ["DECODE"] * (n - n_p)


def spawn_worker(args, name, itype, device, registry_port, log_path,
                 model_name):
    cmd = [sys.executable, "-m", "xllm_service_amd.engine.worker",
           "--name", name, "--type", itype, "--model", model_name,
           "--registry-host", "127.0.0.1",
           "--registry-port", str(registry_port),
           "--max-batched-tokens", str(args.max_batched_tokens),
           "--push-interval-ms", str(args.push_interval_ms),
           "--seed", "0"]
    if device:
        cmd += ["--device", device]
    if args.max_kv_blocks:
        cmd += ["--max-kv-blocks", str(args.max_kv_blocks)]
    logf = open(log_path, "wb")
    return subprocess.Popen(cmd, cwd=ROOT, stdout=logf,
                            stderr=subprocess.STDOUT), logf


class Completions:
    """Completion log + exact K-step timing. Only open-loop ("poisson")
    requests count as steps — the closed-loop ramp burst has queued-up
    TTFTs by construction and its stragglers complete in bursts."""

    def __init__(self):
        self.records = []          # (t_end, ttft_s, out_tokens, tag, hdr)
        self.durations = []        # completed request wall time
        self.phase_count = 0
        self.k = None
        self.t1 = None
        self.event = asyncio.Event()

    def arm(self, k):
        self.phase_count, self.k, self.t1 = 0, k, None
        self.event = asyncio.Event()
        if k <= 0:
            self.t1 = time.monotonic()
            self.event.set()

    def on_complete(self, t, ttft, ntok, tag, t_hdr=0.0, dur=0.0):
        self.records.append((t, ttft, ntok, tag, t_hdr))
        if dur > 0:
            self.durations.append(dur)
        if self.k is not None and self.t1 is None and tag == "poisson":
            self.phase_count += 1
            if self.phase_count >= self.k:
                self.t1 = t
                self.event.set()



# --------------------------------------------------------------------------
# load-generator child process (no torch; driven over stdin/stdout).
# A single asyncio loop cannot parse tens of thousands of SSE chunks per
# second AND issue new requests on time — measured client-side TTFT then
# includes loadgen queueing, not serving latency. So the load is spread
# over M light processes; completions stream back as DONE lines
# (CLOCK_MONOTONIC is system-wide on Linux, so child timestamps are
# directly comparable in the parent).
# --------------------------------------------------------------------------
def loadgen_child_main():
    import httpx
    cfg = json.loads(os.environ["BENCH_LOADGEN_CFG"])

    async def run():
        client = httpx.AsyncClient(
            base_url=f"http://127.0.0.1:{cfg['http_port']}", timeout=600.0,
            limits=httpx.Limits(max_connections=2048,
                                max_keepalive_connections=2048))
        rnd = random.Random(cfg["seed"])
        ramping = {"on": False}
        quit_ev = asyncio.Event()
        tasks = []

        async def one_request(tag):
            prompt = [rnd.randrange(10, cfg["vocab_hi"])
                      for _ in range(cfg["input_len"])]
            t0 = time.monotonic()
            ttft = None
            t_hdr = None
            try:
                async with client.stream("POST", "/v1/completions", json={
                        "model": cfg["model"], "prompt": prompt,
                        "max_tokens": cfg["output_len"],
                        "temperature": 0.0, "ignore_eos": True,
                        "stream": True}) as resp:
                    t_hdr = time.monotonic() - t0
                    if resp.status_code != 200:
                        return
                    async for line in resp.aiter_lines():
                        if ttft is None and line.startswith("data: ") \
                                and "[DONE]" not in line:
                            ttft = time.monotonic() - t0
            except (httpx.HTTPError, OSError):
                return
            if ttft is not None:
                t1 = time.monotonic()
                print(f"DONE {t1:.4f} {ttft:.4f} "
                      f"{cfg['output_len']} {tag} {t_hdr:.4f} "
                      f"{t1 - t0:.4f}", flush=True)

        async def ramp_client(stagger):
            await asyncio.sleep(rnd.uniform(0, stagger))
            while ramping["on"]:
                await one_request("ramp")

        async def poisson(rate):
            while not quit_ev.is_set():
                tasks.append(asyncio.create_task(one_request("poisson")))
                await asyncio.sleep(rnd.expovariate(rate))

        loop = asyncio.get_event_loop()
        while True:
            line = await loop.run_in_executor(None, sys.stdin.readline)
            if not line:
                break
            parts = line.split()
            if not parts:
                continue
            if parts[0] == "ramp":
                ramping["on"] = True
                stag = float(parts[2]) if len(parts) > 2 else 0.0
                for _ in range(int(parts[1])):
                    tasks.append(asyncio.create_task(ramp_client(stag)))
            elif parts[0] == "stopramp":
                ramping["on"] = False
            elif parts[0] == "poisson":
                tasks.append(asyncio.create_task(poisson(float(parts[1]))))
            elif parts[0] == "quit":
                break
        quit_ev.set()
        ramping["on"] = False
        for t in tasks:
            t.cancel()
        await asyncio.gather(*tasks, return_exceptions=True)
        await client.aclose()

    asyncio.run(run())


def summarize_window(records, t0, t1, slo_s, rate, output_len):
    """Pure window math for the serving record (unit-tested): SLO-gated
    goodput capped at the offered load, raw window rate, TTFT percentiles.
    records: (t_end, ttft_s, out_tokens, tag, hdr_s)."""
    window = [r for r in records if t0 <= r[0] <= t1 and r[3] == "poisson"]
    elapsed = max(t1 - t0, 1e-3)
    offered = rate * output_len
    good = sum(r[2] for r in window if r[1] <= slo_s) / elapsed
    alltok = sum(r[2] for r in window) / elapsed
    ttfts = sorted(r[1] for r in window)
    p50 = ttfts[len(ttfts) // 2] * 1000 if ttfts else 0.0
    p99 = ttfts[int(len(ttfts) * 0.99)] * 1000 if ttfts else 0.0
    hdrs = sorted(r[4] for r in window)
    p50h = hdrs[len(hdrs) // 2] * 1000 if hdrs else 0.0
    return dict(value=min(good, offered), total=min(alltok, offered),
                window_raw=alltok, offered=offered, p50_ttft_ms=p50,
                p99_ttft_ms=p99, p50_headers_ms=p50h, n=len(window))


def _parse_prom(text: str) -> dict:
    vals = {}
    for line in text.splitlines():
        if line.startswith("#") or " " not in line:
            continue
        name, _, v = line.rpartition(" ")
        try:
            vals[name] = float(v)
        except ValueError:
            pass
    return vals


async def run_serving_rank0(args, world, dist, use_gpu, model_name,
                            barrier, my_worker_device):
    import httpx

    http_port = free_port()
    rpc_port = free_port()
    registry_port = free_port()
    os.makedirs("gpurun_out", exist_ok=True)
    # the master runs in its own process (real deployment shape; keeps the
    # load generator's event loop out of the serving path)
    mlog = open("gpurun_out/bench_master.log", "wb")
    mproc = subprocess.Popen(
        [sys.executable, "-m", "xllm_service_amd.service.master",
         "--http-host", "127.0.0.1", "--http-port", str(http_port),
         "--rpc-port", str(rpc_port), "--registry-port", str(registry_port),
         "--model-id", model_name, "--policy", args.policy],
        cwd=ROOT, stdout=mlog, stderr=subprocess.STDOUT)

    roles = topology(max(world, 1))
    n_total = len(roles)
    if world > 1:
        # ship the registry address to the other ranks (they spawn their
        # own worker subprocess against it)
        await asyncio.get_event_loop().run_in_executor(
            None, dist.broadcast_object_list,
            [("127.0.0.1", registry_port)], 0)

    proc = logf = None
    client = httpx.AsyncClient(
        base_url=f"http://127.0.0.1:{http_port}", timeout=600.0,
        limits=httpx.Limits(max_connections=4096,
                            max_keepalive_connections=4096))

    async def prom() -> dict:
        try:
            r = await client.get("/metrics")
            return _parse_prom(r.text)
        except (httpx.HTTPError, OSError):
            return {}

    try:
        deadline = time.monotonic() + args.startup_timeout
        while not await prom():            # master HTTP up
            if mproc.poll() is not None or time.monotonic() > deadline:
                raise RuntimeError("master process did not come up; tail:\n"
                                   + open("gpurun_out/bench_master.log",
                                          "rb").read()[-4000:].decode(
                                              errors="replace"))
            await asyncio.sleep(0.5)
        proc, logf = spawn_worker(
            args, f"{roles[0].lower()}-0", roles[0], my_worker_device,
            registry_port, "gpurun_out/bench_worker_r0.log", model_name)

        def n_instances(m):
            return sum(v for k, v in m.items()
                       if k.startswith("cluster_schedulable_instances"))

        while True:
            m = await prom()
            if n_instances(m) >= n_total:
                break
            if proc.poll() is not None:
                raise RuntimeError(
                    "worker died; tail:\n" + open(
                        "gpurun_out/bench_worker_r0.log",
                        "rb").read()[-4000:].decode(errors="replace"))
            if time.monotonic() > deadline:
                raise TimeoutError(
                    f"{n_instances(m)}/{n_total} workers registered "
                    "before timeout")
            await asyncio.sleep(0.5)
        await asyncio.sleep(1.0)   # let P<->D links settle
        await barrier()            # B1: cluster up

        comp = Completions()
        vocab_hi = 120000 if "8b" in model_name else 400
        n_decode = sum(1 for r in roles if r != "PREFILL") or 1
        conc = args.concurrency * n_decode

        # spawn the load-generator children (see loadgen_child_main)
        n_lg = max(2, 2 * n_decode)
        lg_env = dict(os.environ, BENCH_LOADGEN="1")
        children = []
        lg_logs = []
        for i in range(n_lg):
            lg_env["BENCH_LOADGEN_CFG"] = json.dumps(dict(
                http_port=http_port, model=model_name,
                input_len=args.input_len, output_len=args.output_len,
                vocab_hi=vocab_hi, seed=1000 + i))
            lf = open(f"gpurun_out/bench_loadgen_{i}.log", "wb")
            lg_logs.append(lf)
            children.append(subprocess.Popen(
                [sys.executable, os.path.abspath(__file__)],
                cwd=ROOT, env=dict(lg_env), stdin=subprocess.PIPE,
                stdout=subprocess.PIPE, stderr=lf))

        def cmd_all(line_fmt, *per_child):
            for i, ch in enumerate(children):
                vals = [pc[i] for pc in per_child]
                ch.stdin.write((line_fmt.format(*vals) + "\n").encode())
                ch.stdin.flush()

        async def reader(ch):
            loop = asyncio.get_event_loop()
            while True:
                line = await loop.run_in_executor(None, ch.stdout.readline)
                if not line:
                    return
                if line.startswith(b"DONE"):
                    parts = line.split()
                    comp.on_complete(float(parts[1]), float(parts[2]),
                                     int(parts[3]), parts[4].decode(),
                                     float(parts[5]) if len(parts) > 5
                                     else 0.0,
                                     float(parts[6]) if len(parts) > 6
                                     else 0.0)

        from concurrent.futures import ThreadPoolExecutor
        asyncio.get_event_loop().set_default_executor(
            ThreadPoolExecutor(max_workers=n_lg + 8))
        readers = [asyncio.create_task(reader(ch)) for ch in children]

        # ---- phase A: closed-loop ramp + capacity calibration ----------
        ramp_s = args.ramp_s or (8.0 if use_gpu else 2.0)
        shares = [conc // n_lg + (1 if i < conc % n_lg else 0)
                  for i in range(n_lg)]
        # stagger ramp starts so the first generation of closed-loop
        # requests doesn't complete as one synchronized bunch (which
        # makes the ramp->Poisson handover oscillate for many request
        # durations)
        cmd_all("ramp {} {}", shares, [ramp_s] * n_lg)

        async def gen_tokens():
            return (await prom()).get("generated_tokens_total", 0.0)

        async def active_now():
            return (await prom()).get("server_active_requests", 0.0)

        while await active_now() < 0.9 * conc:    # ramp fully started
            await asyncio.sleep(0.5)
            if time.monotonic() > deadline:
                raise TimeoutError("ramp never filled")
        await asyncio.sleep(2.0)
        c0, tA = await gen_tokens(), time.monotonic()
        await asyncio.sleep(ramp_s)
        cap_tok_s = (await gen_tokens() - c0) / (time.monotonic() - tA)

        rate = args.arrival_rate or max(
            args.pace * cap_tok_s / args.output_len, 0.2)

        # ---- phase B: open-loop Poisson ---------------------------------
        # hand over WITHOUT draining: ramp re-issues stop as Poisson
        # arrivals start, so in-flight stays near steady (~R x D) instead
        # of refilling from empty (a from-empty window under-measures the
        # steady rate for its whole first request-duration)
        cmd_all("stopramp", )
        poisson_t0 = time.monotonic()
        cmd_all("poisson {}", [rate / n_lg] * n_lg)
        d_est = args.output_len * conc / max(cap_tok_s, 1.0)
        settle = poisson_t0 + 1.5 * d_est - time.monotonic()
        if settle > 0:
            await asyncio.sleep(settle)   # ramp residuals gone, B steady
        comp.arm(args.warmup)

        async def wait_or_die(ev):
            while not ev.is_set():
                if any(ch.poll() is not None for ch in children):
                    raise RuntimeError("a load-generator child died")
                if mproc.poll() is not None:
                    raise RuntimeError("master process died")
                if proc.poll() is not None:
                    raise RuntimeError("worker process died")
                try:
                    await asyncio.wait_for(ev.wait(), 5.0)
                except asyncio.TimeoutError:
                    pass

        await wait_or_die(comp.event)    # W completions at steady state

        await barrier()                  # B2
        if use_gpu:
            torch.cuda.synchronize()
        comp.arm(args.steps)
        t0 = time.monotonic()
        await wait_or_die(comp.event)
        t1 = comp.t1
        if use_gpu:
            torch.cuda.synchronize()
        await barrier()                  # B3

        elapsed = max(t1 - t0, 1e-3)   # K completions can share a tick
        if world > 1:
            te = torch.tensor([elapsed], dtype=torch.float64)
            await asyncio.get_event_loop().run_in_executor(
                None, dist.all_reduce, te, dist.ReduceOp.MAX)
            elapsed = float(te.item())

        sm = summarize_window(comp.records, t0, t0 + elapsed,
                              args.slo_ttft_ms / 1000.0, rate,
                              args.output_len)

        n_p = sum(1 for r in roles if r == "PREFILL")
        n_d = len(roles) - n_p
        par = ("colocated-1gpu" if n_total == 1
               else f"pd-{n_p}p{n_d}d")
        print(json.dumps({
            "metric": "SLO-goodput (out tok/s under p50 TTFT SLO), "
                      "Llama-3-8B PD-disagg on 8 MI355X",
            "value": round(sm["value"], 2),
            "unit": "tok/s",
            "n_gpus": n_total,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if use_gpu else "fp32",
            "data": "synthetic",
            "total_tok_per_s": round(sm["total"], 2),
            "window_tok_per_s": round(sm["window_raw"], 2),
            "offered_tok_per_s": round(sm["offered"], 2),
            "p50_ttft_ms": round(sm["p50_ttft_ms"], 1),
            "p99_ttft_ms": round(sm["p99_ttft_ms"], 1),
            "slo_ttft_ms": args.slo_ttft_ms,
            "arrival_rate_req_s": round(rate, 2),
            "p50_headers_ms": round(sm["p50_headers_ms"], 1),
            "p50_request_s": round(sorted(comp.durations)[
                len(comp.durations) // 2], 2) if comp.durations else 0.0,
            "calibrated_capacity_tok_s": round(cap_tok_s, 1),
            "requests_timed": sm["n"],
            "config": {
                "model": model_name,
                "global_batch": conc,
                "seq_len": args.input_len + args.output_len,
                "input_len": args.input_len,
                "output_len": args.output_len,
                "parallelism": par,
                "stack": "http(uvicorn tcp)+sse -> master proc -> msgrpc "
                         f"-> {n_total} worker proc(s)",
                "policy": args.policy,
                "push_interval_ms": args.push_interval_ms,
            },
        }))
        cmd_all("quit", )
        for rt in readers:
            rt.cancel()
        await asyncio.gather(*readers, return_exceptions=True)
        # debugging artifacts: per-request completions + final master metrics
        try:
            with open("gpurun_out/bench_requests.csv", "w") as f:
                f.write("t_end,ttft_s,tokens,tag\n")
                for r in comp.records:
                    f.write(f"{r[0]:.3f},{r[1]:.3f},{r[2]},{r[3]}\n")
            mtxt = (await client.get("/metrics")).text
            with open("gpurun_out/bench_metrics_after.txt", "w") as f:
                f.write(mtxt)
        except Exception:
            pass
        await barrier()                  # B4: teardown
    finally:
        await client.aclose()
        procs = [proc, mproc] + [c for c in locals().get("children", [])]
        for p in procs:
            if p is not None:
                p.terminate()
        for p in procs:
            if p is not None:
                try:
                    p.wait(timeout=15)
                except subprocess.TimeoutExpired:
                    p.kill()
        for lf in locals().get("lg_logs", []):
            lf.close()
        if logf:
            logf.close()
        mlog.close()


def run_serving_follower(args, rank, dist, use_gpu, model_name):
    """Ranks > 0: host one worker subprocess on this rank's GPU and follow
    the barrier protocol (B1 cluster-up, B2/B3 timing, B4 teardown)."""
    box = [None]
    dist.broadcast_object_list(box, src=0)
    host, registry_port = box[0]
    device = f"cuda:{int(os.environ.get('LOCAL_RANK', rank))}" if use_gpu \
        else "cpu"
    roles = topology(int(os.environ["WORLD_SIZE"]))
    os.makedirs("gpurun_out", exist_ok=True)
    proc, logf = spawn_worker(
        args, f"{roles[rank].lower()}-{rank}", roles[rank], device,
        registry_port, f"gpurun_out/bench_worker_r{rank}.log", model_name)
    try:
        dist.barrier()                   # B1
        dist.barrier()                   # B2
        if use_gpu:
            torch.cuda.synchronize()
        t0 = time.monotonic()
        dist.barrier()                   # B3
        t1 = time.monotonic()
        if use_gpu:
            torch.cuda.synchronize()
        te = torch.tensor([t1 - t0], dtype=torch.float64)
        dist.all_reduce(te, op=dist.ReduceOp.MAX)
        dist.barrier()                   # B4
    finally:
        proc.terminate()
        try:
            proc.wait(timeout=15)
        except subprocess.TimeoutExpired:
            proc.kill()
        logf.close()


def main():
    args = parse_args()
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))

    use_gpu = torch.cuda.is_available() if args.device is None else (
        args.device.startswith("cuda"))
    dist = None
    if world > 1:
        import torch.distributed as dist_mod
        dist = dist_mod
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        # gloo for the benchmark's own barriers/reductions (host-side);
        # the workers' data plane (RCCL / xGMI) is their own. Bounded
        # timeout: a failing rank must not hang the others for 30 min.
        import datetime
        dist.init_process_group(backend="gloo",
                                timeout=datetime.timedelta(seconds=900))
        if use_gpu:
            torch.cuda.set_device(local_rank)

    model_name = args.model
    if not use_gpu:
        # CPU fallback so `python bench.py` runs anywhere; the measured
        # config on GPU is the flagship model.
        if model_name == "llama-3-8b":
            model_name = "llama-tiny"
        if args.input_len == 1024:
            args.input_len = 48
        if args.output_len == 1024:
            args.output_len = 16
        if args.concurrency == 128:
            args.concurrency = 8

    if args.push_interval_ms <= 0:
        r = topology(max(world, 1))
        nd = sum(1 for x in r if x != "PREFILL") or 1
        args.push_interval_ms = 25.0 * max(1.0, nd / 2.0)

    if args.mode == "engine":
        device = args.device or (f"cuda:{local_rank}" if use_gpu else "cpu")
        run_engine_mode(args, rank, world, local_rank, use_gpu, dist,
                        device, model_name)
    elif rank == 0:
        my_dev = (args.device or
                  (f"cuda:{local_rank}" if use_gpu else "cpu"))

        async def barrier():
            if dist:
                await asyncio.get_event_loop().run_in_executor(
                    None, dist.barrier)

        try:
            asyncio.run(run_serving_rank0(args, world, dist, use_gpu,
                                          model_name, barrier, my_dev))
        except Exception:
            if world > 1:
                raise          # multi-rank: peers are barrier-synced
            # N=1 safety net: a serving-stack failure still yields an
            # HONESTLY-LABELLED record (engine metric string, not the
            # SLO-goodput headline)
            import traceback
            traceback.print_exc(file=sys.stderr)
            print("serving bench failed; falling back to engine mode",
                  file=sys.stderr)
            run_engine_mode(args, rank, world, local_rank, use_gpu, dist,
                            my_dev, model_name)
    else:
        run_serving_follower(args, rank, dist, use_gpu, model_name)

    if dist:
        dist.destroy_process_group()


if __name__ == "__main__":
    if os.environ.get("BENCH_LOADGEN") == "1":
        loadgen_child_main()
    else:
        main()
