#!/usr/bin/env python3
"""GPU PD-disaggregation check: real multi-process workers on one MI355X.

Starts the master (embedded registry) in-process, spawns PREFILL and DECODE
workers as SEPARATE processes sharing cuda:0 (so the hipIpc + xGMI migration
path is exercised across real process boundaries — on an 8-GPU node the same
code runs across GPUs), then verifies PD output == colocated output.

Run via: gpurun -- 'python scripts/pd_gpu_check.py'
"""
import asyncio
import os
import subprocess
import sys
import time

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)

MODEL = "llama-debug-128"


async def run_requests(master, prompts, max_tokens=8):
    import httpx
    from xllm_service_amd.service.http_api import build_app
    app = build_app(master)
    out = []
    async with httpx.AsyncClient(transport=httpx.ASGITransport(app=app),
                                 base_url="http://t", timeout=120.0) as client:
        for p in prompts:
            r = await client.post("/v1/completions", json={
                "model": MODEL, "prompt": p, "max_tokens": max_tokens,
                "temperature": 0.0, "ignore_eos": True})
            assert r.status_code == 200, r.text
            out.append(r.json()["choices"][0]["text"])
    return out


async def scenario(worker_specs, prompts):
    from xllm_service_amd.service.master import Master, MasterOptions
    master = Master(MasterOptions(load_balance_policy="RR", model_id=MODEL,
                                  host_registry=True, registry_port=0,
                                  rpc_port=0))
    await master.start(serve_http=False)
    procs = []
    try:
        for name, itype in worker_specs:
            procs.append(subprocess.Popen([
                sys.executable, "-m", "xllm_service_amd.engine.worker",
                "--name", name, "--type", itype, "--model", MODEL,
                "--device", "cuda:0", "--registry-port",
                str(master.opts.registry_port), "--max-kv-blocks", "256",
                "--seed", "11", "--no-graphs"],
                cwd=ROOT, stdout=subprocess.PIPE, stderr=subprocess.STDOUT))
        for _ in range(600):
            if master.scheduler.has_available_instances():
                break
            await asyncio.sleep(0.5)
            for p in procs:
                if p.poll() is not None:
                    print(p.communicate()[0].decode()[-3000:])
                    raise RuntimeError("worker died during startup")
        else:
            raise TimeoutError("workers never became available")
        return await run_requests(master, prompts)
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
    prompts = [list(range(40, 76)), list(range(5, 25))]
    colo = asyncio.run(scenario([("w0", "DEFAULT")], prompts))
    print("colocated:", colo)
    pd = asyncio.run(scenario([("p0", "PREFILL"), ("d0", "DECODE")], prompts))
    print("pd       :", pd)
    assert pd == colo, f"PD output differs!\n{pd}\nvs\n{colo}"
    print("PD_GPU_CHECK_OK: xGMI/IPC migration produced identical output")


if __name__ == "__main__":
    main()
