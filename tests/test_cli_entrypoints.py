"""The README deployment commands work as-is: `python -m ...master` and
`python -m ...worker` as real subprocesses, speaking over the loopback
(CPU, llama-tiny)."""
import json
import os
import socket
import subprocess
import sys
import time
import urllib.request

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _free_port():
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


@pytest.mark.timeout(180)
def test_master_and_worker_cli():
    http_port, rpc_port, reg_port = _free_port(), _free_port(), _free_port()
    env = dict(os.environ)
    procs = []
    try:
        procs.append(subprocess.Popen(
            [sys.executable, "-m", "xllm_service_amd.service.master",
             "--http-host", "127.0.0.1", "--http-port", str(http_port),
             "--rpc-port", str(rpc_port), "--registry-port", str(reg_port),
             "--model-id", "llama-tiny", "--policy", "RR"],
            cwd=ROOT, env=env, stdout=subprocess.PIPE,
            stderr=subprocess.STDOUT))
        procs.append(subprocess.Popen(
            [sys.executable, "-m", "xllm_service_amd.engine.worker",
             "--name", "w0", "--type", "DEFAULT", "--model", "llama-tiny",
             "--device", "cpu", "--registry-port", str(reg_port),
             "--seed", "0"],
            cwd=ROOT, env=env, stdout=subprocess.PIPE,
            stderr=subprocess.STDOUT))

        base = f"http://127.0.0.1:{http_port}"
        deadline = time.monotonic() + 120
        body = None
        while time.monotonic() < deadline:
            for p in procs:
                assert p.poll() is None, p.stdout.read().decode()[-2000:]
            try:
                req = urllib.request.Request(
                    base + "/v1/completions", method="POST",
                    headers={"Content-Type": "application/json"},
                    data=json.dumps({
                        "model": "llama-tiny", "prompt": [3, 4, 5],
                        "max_tokens": 4, "temperature": 0.0,
                        "ignore_eos": True}).encode())
                with urllib.request.urlopen(req, timeout=10) as r:
                    body = json.loads(r.read())
                break
            except Exception:
                time.sleep(0.5)
        assert body is not None, "service never became ready"
        assert body["usage"]["completion_tokens"] == 4
        with urllib.request.urlopen(base + "/v1/models", timeout=10) as r:
            models = json.loads(r.read())
        assert models["data"][0]["id"] == "llama-tiny"
        with urllib.request.urlopen(base + "/metrics", timeout=10) as r:
            assert b"server_request_in_total" in r.read()
    finally:
        for p in procs:
            p.terminate()
        for p in procs:
            try:
                p.wait(timeout=10)
            except subprocess.TimeoutExpired:
                p.kill()
