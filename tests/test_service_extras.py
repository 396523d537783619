"""Logprobs, request tracing, runtime flag reload, multi-master replicas."""
import asyncio
import json
import os

import pytest

from xllm_service_amd.engine.worker import Worker
from xllm_service_amd.service.master import Master, MasterOptions

from test_service_integration import (http_client, make_master, wait_for,
                                      worker_kwargs)


@pytest.fixture
def anyio_backend():
    return "asyncio"


@pytest.mark.anyio
async def test_logprobs_and_tracer_and_reload(tmp_path):
    trace_path = str(tmp_path / "trace.jsonl")
    master = make_master(policy="RR", enable_request_trace=True,
                         trace_path=trace_path)
    await master.start(serve_http=False)
    worker = Worker("w0", "DEFAULT", **worker_kwargs(master))
    try:
        await worker.start()
        await wait_for(lambda: master.instance_mgr.get("w0"))
        client = await http_client(master)

        r = await client.post("/v1/completions", json={
            "model": "llama-tiny", "prompt": [3, 4, 5], "max_tokens": 4,
            "temperature": 0.0, "ignore_eos": True, "logprobs": 3})
        assert r.status_code == 200, r.text
        lp = r.json()["choices"][0]["logprobs"]
        assert len(lp["token_logprobs"]) == 4
        assert all(isinstance(v, float) and v <= 0 for v in lp["token_logprobs"])
        # byte-fallback tokenizer maps some ids to the same text, so the
        # top-N dict may collapse; presence is what matters here
        assert all(1 <= len(t) <= 3 for t in lp["top_logprobs"])

        # streaming completion carries per-chunk logprobs
        async with client.stream("POST", "/v1/completions", json={
                "model": "llama-tiny", "prompt": [3, 4, 5], "max_tokens": 4,
                "temperature": 0.0, "ignore_eos": True, "logprobs": 2,
                "stream": True}) as resp:
            assert resp.status_code == 200
            chunks = []
            async for line in resp.aiter_lines():
                if line.startswith("data: ") and line != "data: [DONE]":
                    chunks.append(json.loads(line[6:]))
        stream_lps = [v for c in chunks
                      for v in (c["choices"][0].get("logprobs") or {})
                      .get("token_logprobs", [])]
        assert len(stream_lps) == 4
        assert all(isinstance(v, float) and v <= 0 for v in stream_lps)

        # streaming chat carries OpenAI chat-style logprobs content blocks
        async with client.stream("POST", "/v1/chat/completions", json={
                "model": "llama-tiny",
                "messages": [{"role": "user", "content": "hi"}],
                "max_tokens": 3, "temperature": 0.0, "ignore_eos": True,
                "logprobs": True, "top_logprobs": 2,
                "stream": True}) as resp:
            assert resp.status_code == 200
            items = []
            async for line in resp.aiter_lines():
                if line.startswith("data: ") and line != "data: [DONE]":
                    c = json.loads(line[6:])
                    if c.get("choices"):
                        items.extend((c["choices"][0].get("logprobs") or {})
                                     .get("content", []))
        assert len(items) == 3
        assert all("logprob" in it and "top_logprobs" in it for it in items)

        # multi-token stop sequences + echo (OpenAI completions params)
        r = await client.post("/v1/completions", json={
            "model": "llama-tiny", "prompt": "ab", "max_tokens": 4,
            "temperature": 0.0, "ignore_eos": True, "echo": True})
        assert r.status_code == 200, r.text
        assert r.json()["choices"][0]["text"].startswith("ab")

        # tracer wrote request records
        assert os.path.exists(trace_path)
        lines = [json.loads(x) for x in open(trace_path)]
        assert any(rec["direction"] == "dispatch" for rec in lines)

        # runtime flag reload
        r = await client.post("/admin/reload_flags",
                              json={"target_ttft_ms": 500,
                                    "target_tpot_ms": 30})
        assert r.json()["reloaded"]["target_ttft_ms"] == 500
        assert master.opts.target_tpot_ms == 30
        await client.aclose()
    finally:
        await worker.stop()
        await master.stop()


@pytest.mark.anyio
async def test_replica_sync_and_master_takeover():
    """Second service replica shares the cluster view via registry watches
    and takes over mastership when the master's lease lapses."""
    from xllm_service_amd.registry.server import RegistryService
    registry = RegistryService()          # standalone, outlives both masters
    reg_port = await registry.start()

    def replica():
        return Master(MasterOptions(
            load_balance_policy="RR", model_id="llama-tiny",
            host_registry=False, registry_host="127.0.0.1",
            registry_port=reg_port, rpc_port=0, service_lease_ttl_s=0.5))

    a, b = replica(), replica()
    await a.start(serve_http=False)
    await b.start(serve_http=False)
    assert a.is_master and not b.is_master

    wk = worker_kwargs(a)
    wk["registry_port"] = reg_port
    worker = Worker("w0", "DEFAULT", **wk)
    try:
        await worker.start()
        # both replicas discover the instance via registry watch
        await wait_for(lambda: a.instance_mgr.get("w0"))
        await wait_for(lambda: b.instance_mgr.get("w0"))
        assert b.scheduler.has_available_instances()

        # master A's 3s sync uploads load metrics; replica B follows them
        a.opts.heartbeat_sync_s = 0.2
        await wait_for(
            lambda: a.instance_mgr.instances["w0"].load is not None)

        # kill A; its service lease lapses; B must take over mastership
        await a.stop()
        await wait_for(lambda: b.is_master, timeout=15.0)
        # B can still serve requests through the same worker pool
        client = await http_client(b)
        r = await client.post("/v1/completions", json={
            "model": "llama-tiny", "prompt": [9, 9], "max_tokens": 2,
            "temperature": 0.0, "ignore_eos": True})
        assert r.status_code == 200, r.text
        await client.aclose()
    finally:
        await worker.stop()
        await b.stop()
        await registry.stop()


@pytest.mark.anyio
async def test_anthropic_messages_api():
    master = make_master(policy="RR")
    await master.start(serve_http=False)
    worker = Worker("w0", "DEFAULT", **worker_kwargs(master))
    try:
        await worker.start()
        await wait_for(lambda: master.instance_mgr.get("w0"))
        client = await http_client(master)
        r = await client.post("/v1/messages", json={
            "model": "llama-tiny", "max_tokens": 5,
            "system": "be brief",
            "messages": [{"role": "user", "content": "hi"}],
            "temperature": 0.0})
        assert r.status_code == 200, r.text
        body = r.json()
        assert body["type"] == "message" and body["role"] == "assistant"
        assert body["usage"]["output_tokens"] == 5
        assert body["stop_reason"] == "max_tokens"
        # streaming variant
        events = []
        async with client.stream("POST", "/v1/messages", json={
                "model": "llama-tiny", "max_tokens": 4, "stream": True,
                "messages": [{"role": "user", "content": "hi"}],
                "temperature": 0.0}) as resp:
            assert resp.status_code == 200
            async for line in resp.aiter_lines():
                if line.startswith("event: "):
                    events.append(line[7:])
        assert events[0] == "message_start"
        assert "content_block_delta" in events
        assert events[-1] == "message_stop"
        await client.aclose()
    finally:
        await worker.stop()
        await master.stop()
