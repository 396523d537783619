#!/usr/bin/env python3
"""GPU multimodal EPD check: E + P + D as three separate processes on one
MI355X (BASELINE config 5 shape, 1-device edition — the same code paths run
across GPUs on an 8-GPU node).

The ENCODE worker runs the vision tower and ships embeddings to the
PREFILL worker (E->P handoff); prefill computes the prompt + first token
and migrates KV to the DECODE worker over the IPC/xGMI path. Output must
equal the colocated (single DEFAULT worker, in-process vision) run.

Run via: gpurun -- 'python scripts/epd_gpu_check.py'
"""
import asyncio
import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)

MODEL = "qwen2-vl-tiny128"

MESSAGES = [
    [{"role": "user", "content": [
        {"type": "image", "grid": [28, 28], "seed": 3},
        {"type": "text", "text": "describe"}]}],
    [{"role": "user", "content": [
        {"type": "image", "grid": [14, 28], "seed": 5},
        {"type": "text", "text": "what is this"}]}],
]


async def run_requests(master):
    import httpx
    from xllm_service_amd.service.http_api import build_app
    app = build_app(master)
    out = []
    async with httpx.AsyncClient(transport=httpx.ASGITransport(app=app),
                                 base_url="http://t", timeout=180.0) as client:
        for msgs in MESSAGES:
            r = await client.post("/v1/chat/completions", json={
                "model": MODEL, "messages": msgs, "max_tokens": 8,
                "temperature": 0.0, "ignore_eos": True})
            assert r.status_code == 200, r.text
            out.append(r.json()["choices"][0]["message"]["content"])
    return out


async def scenario(worker_specs):
    from xllm_service_amd.service.master import Master, MasterOptions
    master = Master(MasterOptions(load_balance_policy="RR", model_id=MODEL,
                                  host_registry=True, registry_port=0,
                                  rpc_port=0))
    await master.start(serve_http=False)
    procs = []
    try:
        os.makedirs("gpurun_out", exist_ok=True)
        logs = []
        for name, itype in worker_specs:
            lf = open(f"gpurun_out/epd_{name}.log", "wb")
            logs.append(lf)
            procs.append(subprocess.Popen([
                sys.executable, "-m", "xllm_service_amd.engine.worker",
                "--name", name, "--type", itype, "--model", MODEL,
                "--device", "cuda:0", "--registry-port",
                str(master.opts.registry_port), "--max-kv-blocks", "256",
                "--seed", "11", "--no-graphs"],
                cwd=ROOT, stdout=lf, stderr=subprocess.STDOUT))
        want_encode = any(t == "ENCODE" for _, t in worker_specs)
        for _ in range(600):
            ready = master.scheduler.has_available_instances() and (
                not want_encode or master.instance_mgr.encode_index)
            if ready:
                break
            await asyncio.sleep(0.5)
            for p in procs:
                if p.poll() is not None:
                    raise RuntimeError("worker died during startup")
        else:
            raise TimeoutError("workers never became available")
        await asyncio.sleep(1.0)
        return await run_requests(master)
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
    colo = asyncio.run(scenario([("w0", "DEFAULT")]))
    print("colocated:", colo)
    epd = asyncio.run(scenario([("e0", "ENCODE"), ("p0", "PREFILL"),
                                ("d0", "DECODE")]))
    print("epd      :", epd)
    assert epd == colo, f"EPD output differs!\n{epd}\nvs\n{colo}"
    print("EPD_GPU_CHECK_OK: three-stage E/P/D (separate processes, "
          "embedding handoff + KV migration) matches colocated output")


if __name__ == "__main__":
    main()
