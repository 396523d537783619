"""Multimodal EPD tests on CPU: vision tower, embedding substitution, and
the full E/P/D three-stage split vs colocated (BASELINE.md config 5 shape).
"""
import asyncio

import httpx
import pytest
import torch

from xllm_service_amd.engine.engine import LLMEngine
from xllm_service_amd.engine.sampling import SamplingParams
from xllm_service_amd.engine.worker import VisionEncoder, Worker
from xllm_service_amd.models.config import get_config
from xllm_service_amd.service.http_api import build_app
from xllm_service_amd.service.master import Master, MasterOptions

from test_service_integration import (http_client, make_master, wait_for,
                                      worker_kwargs)

MODEL = "qwen2-vl-tiny"


@pytest.fixture
def anyio_backend():
    return "asyncio"


def test_vision_encoder_shapes_and_determinism():
    enc = VisionEncoder(MODEL, "cpu", seed=0)
    imgs = [dict(grid_h=4, grid_w=8, seed=3)]
    e1 = enc.encode(imgs)
    e2 = enc.encode(imgs)
    assert e1.shape == (2 * 4, 256)  # (4/2)*(8/2) tokens, hidden 256
    assert torch.equal(e1, e2)
    enc2 = VisionEncoder(MODEL, "cpu", seed=0)
    assert torch.allclose(enc2.encode(imgs), e1)


def test_engine_mm_embedding_substitution():
    cfg = get_config(MODEL)
    eng = LLMEngine(MODEL, device="cpu", max_kv_blocks=128, seed=5)
    pad = cfg.image_pad_token_id
    torch.manual_seed(2)
    mm = torch.randn(4, cfg.hidden_size)
    text = torch.randint(20, cfg.vocab_size, (10,)).tolist()
    prompt = [pad] * 4 + text
    eng.add_request("m1", prompt, SamplingParams(max_tokens=4,
                                                 ignore_eos=True),
                    mm_embeds=mm)
    out_mm = []
    while eng.has_work():
        for o in eng.step():
            out_mm.extend(o.new_token_ids)
    # different embeddings must change the output (substitution is real)
    # NOTE: a scaled copy (mm * 3) would be a no-op perturbation — the
    # first RMSNorm is scale-invariant per row — so use fresh random embeds
    eng2 = LLMEngine(MODEL, device="cpu", max_kv_blocks=128, seed=5)
    torch.manual_seed(99)
    eng2.add_request("m2", prompt, SamplingParams(max_tokens=4,
                                                  ignore_eos=True),
                     mm_embeds=torch.randn(4, cfg.hidden_size))
    out_mm2 = []
    while eng2.has_work():
        for o in eng2.step():
            out_mm2.extend(o.new_token_ids)
    assert len(out_mm) == 4 and len(out_mm2) == 4
    assert out_mm != out_mm2


@pytest.mark.anyio
async def test_epd_three_stage_matches_colocated():
    """1E+1P+1D must produce exactly what a colocated DEFAULT instance
    (running the vision tower in-process) produces."""
    chat_body = {
        "model": MODEL,
        "messages": [{"role": "user", "content": [
            {"type": "image", "grid": [4, 8], "seed": 7},
            {"type": "text", "text": "describe"},
        ]}],
        "max_tokens": 6, "temperature": 0.0, "ignore_eos": True,
    }
    outputs = {}
    for mode in ("colocated", "epd"):
        master = make_master(policy="RR", model_id=MODEL)
        await master.start(serve_http=False)
        specs = ([("w0", "DEFAULT")] if mode == "colocated" else
                 [("e0", "ENCODE"), ("p0", "PREFILL"), ("d0", "DECODE")])
        workers = [Worker(n, t, **worker_kwargs(master, model=MODEL))
                   for n, t in specs]
        try:
            for w in workers:
                await w.start()
            await wait_for(lambda: master.scheduler.has_available_instances())
            if mode == "epd":
                await wait_for(
                    lambda: master.instance_mgr.schedulable_encodes())
            client = await http_client(master)
            r = await client.post("/v1/chat/completions", json=chat_body)
            assert r.status_code == 200, r.text
            body = r.json()
            outputs[mode] = body["choices"][0]["message"]["content"]
            assert body["usage"]["completion_tokens"] == 6
            await client.aclose()
        finally:
            for w in workers:
                await w.stop()
            await master.stop()
    assert outputs["epd"] == outputs["colocated"]


# ---------------------------------------------------------------- M-RoPE
def test_mrope_positions_grid():
    """Hand-checked example of Qwen2-VL 3-D position assignment."""
    from xllm_service_amd.models.qwen2_vl import mrope_positions
    ids = [5, 6] + [9] * 6 + [7, 8]          # text, (1,2,3) image, text
    pos3, delta = mrope_positions(ids, 9, [(1, 2, 3)])
    assert pos3[:, 0].tolist() == [0, 0, 0]
    assert pos3[0, 2:8].tolist() == [2] * 6              # temporal
    assert pos3[1, 2:8].tolist() == [2, 2, 2, 3, 3, 3]   # height
    assert pos3[2, 2:8].tolist() == [2, 3, 4, 2, 3, 4]   # width
    assert pos3[:, 8].tolist() == [5, 5, 5]              # resumes at max+1
    assert delta == 7 - len(ids)


def test_mrope_engine_active_and_chunk_invariant():
    """3-D ids must change the output vs 1-D rope, and chunked prefill must
    slice per-chunk 3-D ids identically to single-shot prefill."""
    cfg = get_config(MODEL)

    def run(max_bt, with_grids):
        eng = LLMEngine(MODEL, device="cpu", max_kv_blocks=128, seed=0,
                        max_batched_tokens=max_bt)
        torch.manual_seed(11)
        mm = torch.randn(8, cfg.hidden_size)
        ph = cfg.image_pad_token_id
        prompt = [3, 4] + [ph] * 8 + [5, 6, 7]
        outs = eng.generate([prompt],
                            SamplingParams(max_tokens=8, ignore_eos=True),
                            mm_embeds=[mm],
                            mm_grids=[[(1, 2, 4)]] if with_grids else None)
        return outs[0]

    base = run(16384, True)
    assert run(16384, False) != base          # M-RoPE actually in effect
    assert run(4, True) == base               # chunked == single-shot


def test_mrope_delta_continues_decode():
    """Decode positions continue at max(pos)+1: a seq with an image must
    keep mrope_delta < 0 and still generate deterministically."""
    cfg = get_config(MODEL)
    eng = LLMEngine(MODEL, device="cpu", max_kv_blocks=128, seed=0)
    torch.manual_seed(11)
    mm = torch.randn(8, cfg.hidden_size)
    ph = cfg.image_pad_token_id
    prompt = [3, 4] + [ph] * 8 + [5, 6, 7]
    eng.add_request("m", prompt,
                    SamplingParams(max_tokens=4, ignore_eos=True),
                    mm_embeds=mm, mm_grids=[(1, 2, 4)])
    seq = eng.seqs["m"]
    assert seq.mrope_pos is not None
    # image of (1,2,4) occupies 4 slots instead of 8 -> delta = -4
    assert seq.mrope_delta == -4
    while not all(o.finished for o in eng.step()):
        pass


def test_mm_request_survives_preemption():
    """A multimodal sequence preempted under memory pressure recomputes or
    swaps with its image embeddings and M-RoPE state intact (outputs match
    the run without pressure)."""
    from xllm_service_amd.engine.engine import LLMEngine
    from xllm_service_amd.engine.sampling import SamplingParams
    cfg = get_config(MODEL)
    torch.manual_seed(21)
    mm = torch.randn(8, cfg.hidden_size)
    ph = cfg.image_pad_token_id
    prompt_mm = [3, 4] + [ph] * 8 + [5, 6, 7]
    prompt_txt = [torch.randint(0, cfg.vocab_size, (40,)).tolist()
                  for _ in range(2)]

    def run(max_kv_blocks):
        eng = LLMEngine(MODEL, device="cpu", max_kv_blocks=max_kv_blocks,
                        seed=0, enable_prefix_caching=False)
        eng.add_request("mm", prompt_mm,
                        SamplingParams(max_tokens=12, ignore_eos=True),
                        mm_embeds=mm, mm_grids=[(1, 2, 4)])
        for i, p in enumerate(prompt_txt):
            eng.add_request(f"t{i}", p,
                            SamplingParams(max_tokens=12, ignore_eos=True))
        outs = {}
        while eng.has_work():
            for o in eng.step():
                outs.setdefault(o.request_id, []).extend(o.new_token_ids)
        return outs, eng.scheduler.num_preempts

    free_outs, _ = run(64)                 # plenty of blocks: no pressure
    tight_outs, pressure = run(9)          # tiny pool: forces preemption
    assert pressure > 0, "no preemption happened (pool too big?)"
    assert tight_outs == free_outs
