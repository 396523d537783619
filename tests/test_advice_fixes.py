"""Regression tests for the round-1 advisor findings:

  * recompute preemption must not inflate max_tokens or corrupt M-RoPE
  * migrated-in requests re-apply the max_tokens clamp on the decode side
  * heartbeat LatencyMetrics carry real TTFT/TBT samples
  * non-stream client disconnects cancel the request
  * text-level stop strings (OpenAI semantics) at the service layer
"""
import asyncio

import pytest
import torch

from xllm_service_amd.engine.engine import LLMEngine
from xllm_service_amd.engine.sampling import SamplingParams
from xllm_service_amd.models.config import get_config


# --------------------------------------------------------------- preemption
def _drain(eng):
    outs = {}
    while eng.has_work():
        for o in eng.step():
            outs.setdefault(o.request_id, []).extend(o.new_token_ids)
    return outs


def test_recompute_preemption_respects_max_tokens():
    """swap disabled -> preemption recomputes; the victim must still emit
    exactly max_tokens tokens, identical to the no-pressure run."""
    cfg = get_config("llama-tiny")
    torch.manual_seed(31)
    prompts = [torch.randint(0, cfg.vocab_size, (48,)).tolist()
               for _ in range(3)]

    def run(blocks):
        eng = LLMEngine("llama-tiny", device="cpu", max_kv_blocks=blocks,
                        seed=7, enable_prefix_caching=False, swap_space_mb=0)
        for i, p in enumerate(prompts):
            eng.add_request(f"r{i}", p,
                            SamplingParams(max_tokens=12, ignore_eos=True))
        return _drain(eng), eng.scheduler.num_preempts

    free, _ = run(256)
    tight, n_pre = run(10)
    assert n_pre > 0, "no preemption happened (pool too big?)"
    for rid, toks in tight.items():
        assert len(toks) == 12, f"{rid} emitted {len(toks)} != max_tokens"
    assert tight == free


def test_mm_recompute_preemption_swapless():
    """A multimodal (M-RoPE) victim of a recompute preemption must rebuild
    its position table for the folded output tokens instead of crashing."""
    cfg = get_config("qwen2-vl-tiny")
    torch.manual_seed(33)
    mm = torch.randn(8, cfg.hidden_size)
    ph = cfg.image_pad_token_id
    prompt_mm = [3, 4] + [ph] * 8 + [5, 6, 7]
    prompt_txt = [torch.randint(0, cfg.vocab_size, (40,)).tolist()
                  for _ in range(2)]

    def run(blocks):
        eng = LLMEngine("qwen2-vl-tiny", device="cpu", max_kv_blocks=blocks,
                        seed=0, enable_prefix_caching=False, swap_space_mb=0)
        eng.add_request("mm", prompt_mm,
                        SamplingParams(max_tokens=12, ignore_eos=True),
                        mm_embeds=mm, mm_grids=[(1, 2, 4)])
        for i, p in enumerate(prompt_txt):
            eng.add_request(f"t{i}", p,
                            SamplingParams(max_tokens=12, ignore_eos=True))
        return _drain(eng), eng.scheduler.num_preempts

    free, _ = run(64)
    tight, n_pre = run(9)
    assert n_pre > 0
    for rid, toks in tight.items():
        assert len(toks) == 12
    assert tight == free


def test_migrated_request_max_tokens_clamped():
    """Decode-side activation must re-clamp max_tokens so total_len never
    exceeds max_model_len (the prefill side ships un-clamped params)."""
    kw = dict(device="cpu", max_kv_blocks=64, seed=7,
              enable_prefix_caching=False, max_model_len=48)
    pre = LLMEngine("llama-tiny", **kw)
    dec = LLMEngine("llama-tiny", **kw)
    cfg = get_config("llama-tiny")
    torch.manual_seed(35)
    prompt = torch.randint(0, cfg.vocab_size, (40,)).tolist()

    # prefill: first token only, hold blocks (worker PREFILL role shape)
    pre.add_request("req", prompt,
                    SamplingParams(max_tokens=1, ignore_eos=True),
                    hold_blocks=True)
    first = _drain(pre)["req"]
    assert len(first) == 1
    blocks_src = pre.held_block_table("req")
    data = pre.export_block_bytes(blocks_src)

    blocks_dst = dec.alloc_migration_blocks(len(blocks_src))
    dec.import_block_bytes(blocks_dst, data)
    # un-clamped params straight off the wire
    fin = dec.activate_migrated_request(
        "req", prompt, first, blocks_dst,
        SamplingParams(max_tokens=500, ignore_eos=True))
    assert fin is None
    out = _drain(dec)["req"]
    seqs_done = len(first) + len(out)
    assert seqs_done == 48 - 40, f"emitted {seqs_done}, budget is 8"

    # a prompt that leaves no budget at all: activation reports the finish
    prompt2 = torch.randint(0, cfg.vocab_size, (47,)).tolist()
    pre2 = LLMEngine("llama-tiny", **kw)
    pre2.add_request("r2", prompt2,
                     SamplingParams(max_tokens=1, ignore_eos=True),
                     hold_blocks=True)
    first2 = _drain(pre2)["r2"]
    b2 = dec.alloc_migration_blocks(len(pre2.held_block_table("r2")))
    dec.import_block_bytes(b2, pre2.export_block_bytes(
        pre2.held_block_table("r2")))
    fin2 = dec.activate_migrated_request(
        "r2", prompt2, first2, b2,
        SamplingParams(max_tokens=500, ignore_eos=True))
    assert fin2 == "length"
    dec.free_blocks(b2)
    assert "r2" not in dec.seqs


# ----------------------------------------------------------- stop scanner
class _MapTok:
    """Toy tokenizer: id -> fixed string from a vocab list."""

    def __init__(self, vocab):
        self.vocab = vocab

    def decode(self, ids, skip_special_tokens=True):
        return "".join(self.vocab[i] for i in ids)


def _scan_text(scanner, token_lists, tok):
    """Feed batches; return (final_text, stopped)."""
    text = ""
    for toks in token_lists:
        out, override, stopped = scanner.feed(toks)
        text += tok.decode(out)
        if stopped:
            return text + override, True
    text += tok.decode(scanner.flush())
    return text, False


def test_stop_scanner_cross_token_boundary():
    from xllm_service_amd.service.stop_scanner import StopTextScanner
    tok = _MapTok(["ax", "y", "zb", "q"])
    sc = StopTextScanner(tok, ["xyz"])
    text, stopped = _scan_text(sc, [[0], [1], [2]], tok)
    assert stopped
    assert text == "a"          # everything from the stop on is trimmed


def test_stop_scanner_no_match_flush():
    from xllm_service_amd.service.stop_scanner import StopTextScanner
    tok = _MapTok(["ax", "y", "zb", "q"])
    sc = StopTextScanner(tok, ["xyzQ"])
    text, stopped = _scan_text(sc, [[0], [1], [2], [3]], tok)
    assert not stopped
    assert text == "axyzbq"


def test_stop_scanner_match_inside_one_token():
    from xllm_service_amd.service.stop_scanner import StopTextScanner
    tok = _MapTok(["hello STOP world"])
    sc = StopTextScanner(tok, ["STOP"])
    text, stopped = _scan_text(sc, [[0]], tok)
    assert stopped
    assert text == "hello "


def test_stop_scanner_streams_released_tokens_eagerly():
    from xllm_service_amd.service.stop_scanner import StopTextScanner
    tok = _MapTok(["aa", "bb", "cc"])
    sc = StopTextScanner(tok, ["XY"])   # window = 1 char
    out, override, stopped = sc.feed([0, 1])
    assert not stopped and override is None
    # all but the last char-covering token can be released immediately
    assert tok.decode(out).startswith("aa")


# ------------------------------------------- service-level stop + disconnect
@pytest.fixture
def anyio_backend():
    return "asyncio"


@pytest.mark.anyio
async def test_service_text_stop_trims_and_aborts():
    """Text-level stop through the real master scheduler: generations are
    injected as byte tokens whose text crosses BPE-style boundaries the
    engine token-matcher would never see."""
    from tests.test_service_integration import make_master
    from xllm_service_amd.service.request import ServiceRequest
    from xllm_service_amd.service.response_handler import ResponseHandler
    from xllm_service_amd.service.stop_scanner import StopTextScanner

    master = make_master(policy="RR")
    await master.start(serve_http=False)
    try:
        sch = master.scheduler
        req = ServiceRequest(service_request_id="r1", kind="completion",
                             model="llama-tiny", stream=False,
                             token_ids=[1, 2, 3],
                             stop_texts=["lo w"])
        req.stop_scanner = StopTextScanner(master.tokenizer, req.stop_texts)
        sch.requests["r1"] = req
        # "hell" ... "o wo" ... "rld": the stop "l w" spans two deltas
        for i, piece in enumerate((b"hell", b"o wo", b"rld")):
            await sch.handle_generation(dict(
                service_request_id="r1",
                token_ids=[b + 2 for b in piece],  # ByteTokenizer: byte+2
                finished=(i == 2), finish_reason="length" if i == 2 else None,
                prompt_tokens=3, completion_tokens=4 * (i + 1)))
            if "r1" not in sch.requests:
                break
        rh = ResponseHandler(master.tokenizer)
        toks, text, usage, finish, err = await rh._collect(req)
        assert err is None
        assert text == "hel"
        assert finish == "stop"
    finally:
        await master.stop()


@pytest.mark.anyio
async def test_nonstream_disconnect_cancels():
    """A non-stream client that disconnects mid-generation must abort the
    request (the round-1 is_disconnected hook was dead code)."""
    from tests.test_service_integration import make_master
    from xllm_service_amd.service.request import ServiceRequest

    master = make_master(policy="RR")
    await master.start(serve_http=False)
    try:
        sch = master.scheduler

        class GoneHttp:
            async def is_disconnected(self):
                return True

        req = ServiceRequest(service_request_id="r2", kind="completion",
                             model="llama-tiny", stream=False,
                             token_ids=[1, 2, 3])
        req.http_request = GoneHttp()
        req._last_disc_check = -10.0
        sch.requests["r2"] = req
        ok = await sch.handle_generation(dict(
            service_request_id="r2", token_ids=[65],
            finished=False, prompt_tokens=3, completion_tokens=1))
        assert ok is False          # worker is told to abort
        assert "r2" not in sch.requests
        # the pending collector is unblocked with an abort delta
        d = req.output_queue.get_nowait()
        assert d.finished and d.error
    finally:
        await master.stop()


@pytest.mark.anyio
async def test_latency_samples_reach_master():
    """recent_max_ttft/tbt in heartbeats must carry real measurements
    (they were always 0 in round 1)."""
    from tests.test_service_integration import (http_client, make_master,
                                                wait_for, worker_kwargs)
    from xllm_service_amd.engine.worker import Worker

    master = make_master(policy="RR")
    await master.start(serve_http=False)
    kw = worker_kwargs(master)
    kw["heartbeat_s"] = 0.3
    w = Worker("w0", "DEFAULT", **kw)
    try:
        await w.start()
        await wait_for(lambda: master.scheduler.has_available_instances())
        client = await http_client(master)
        r = await client.post("/v1/completions", json={
            "model": "llama-tiny", "prompt": list(range(40, 80)),
            "max_tokens": 24, "temperature": 0.0, "ignore_eos": True})
        assert r.status_code == 200

        def got_latency():
            inst = master.scheduler.mgr.get("w0")
            return (inst is not None
                    and inst.latency.recent_max_ttft_ms > 0
                    and inst.latency.recent_max_tbt_ms > 0)
        await wait_for(got_latency, timeout=8.0)
        await client.aclose()
    finally:
        await w.stop()
        await master.stop()


# ---------------------------------------------- migration overlap (PD)
class _FakeEvent:
    def __init__(self):
        self.done = False

    def query(self):
        return self.done


def test_decode_continues_during_migration_pull():
    """Pending migrated-in requests must not stall the decode loop: steps
    keep producing tokens while the (fake) copy event is in flight, and
    the sequence activates only once it fires."""
    eng = LLMEngine("llama-tiny", device="cpu", max_kv_blocks=128, seed=7,
                    enable_prefix_caching=False)
    cfg = get_config("llama-tiny")
    torch.manual_seed(41)
    eng.add_request("bg", torch.randint(0, cfg.vocab_size, (24,)).tolist(),
                    SamplingParams(max_tokens=64, ignore_eos=True))

    # prefill the migrating request's KV on a source engine, ship bytes
    src = LLMEngine("llama-tiny", device="cpu", max_kv_blocks=64, seed=7,
                    enable_prefix_caching=False)
    prompt = torch.randint(0, cfg.vocab_size, (20,)).tolist()
    src.add_request("mig", prompt, SamplingParams(max_tokens=1,
                                                  ignore_eos=True),
                    hold_blocks=True)
    while src.has_work():
        first = [o.new_token_ids[0] for o in src.step() if o.new_token_ids]
    blocks_src = src.held_block_table("mig")

    dst_blocks = eng.alloc_migration_blocks(len(blocks_src))
    eng.import_block_bytes(dst_blocks, src.export_block_bytes(blocks_src))
    ev = _FakeEvent()
    eng.enqueue_migrated_request("mig", prompt, first, dst_blocks,
                                 SamplingParams(max_tokens=6,
                                                ignore_eos=True), event=ev)
    bg_before = len(eng.seqs["bg"].output_token_ids)
    for _ in range(4):
        outs = eng.step()
        assert all(o.request_id == "bg" for o in outs)
    assert len(eng.seqs["bg"].output_token_ids) >= bg_before + 4
    assert "mig" not in eng.seqs          # still pending

    ev.done = True
    for _ in range(3):
        eng.step()
    assert "mig" in eng.seqs              # activated after the event
    # and it decodes to completion
    while eng.seqs.get("mig") is not None and eng.has_work():
        eng.step()


def test_abort_during_migration_pull_frees_blocks():
    eng = LLMEngine("llama-tiny", device="cpu", max_kv_blocks=64, seed=7,
                    enable_prefix_caching=False)
    free0 = eng.block_manager.num_free
    blocks = eng.alloc_migration_blocks(4)
    ev = _FakeEvent()
    eng.enqueue_migrated_request("m2", [1, 2, 3], [4], blocks,
                                 SamplingParams(max_tokens=8), event=ev)
    assert eng.abort_request("m2") is True
    eng.step()                            # copy still in flight: no free yet
    assert eng.block_manager.num_free == free0 - 4
    ev.done = True
    eng.step()                            # event fired: blocks released
    assert eng.block_manager.num_free == free0
    assert "m2" not in eng.seqs


@pytest.mark.anyio
async def test_xgmi_migration_falls_back_to_bytes():
    """If the xGMI/IPC migration path fails (unproven topologies), the
    prefill worker must retry with the serialized-bytes transport instead
    of erroring the request."""
    from tests.test_service_integration import (http_client, make_master,
                                                wait_for, worker_kwargs)
    from xllm_service_amd.engine.worker import Worker

    master = make_master(policy="RR")
    await master.start(serve_http=False)
    p0 = Worker("p0", "PREFILL", **worker_kwargs(master))
    d0 = Worker("d0", "DECODE", **worker_kwargs(master))
    try:
        await p0.start()
        await d0.start()
        await wait_for(lambda: master.scheduler.has_available_instances())
        # force the prefill side to attempt the xgmi transport on CPU —
        # the decode side's IPC open fails, the fallback must kick in
        p0._pick_transport = lambda peer: "xgmi"
        client = await http_client(master)
        r = await client.post("/v1/completions", json={
            "model": "llama-tiny", "prompt": list(range(50, 90)),
            "max_tokens": 6, "temperature": 0.0, "ignore_eos": True})
        assert r.status_code == 200, r.text
        assert r.json()["usage"]["completion_tokens"] == 6
        await client.aclose()
    finally:
        await p0.stop()
        await d0.stop()
        await master.stop()
