"""Unit tests for service-layer components: hashing, global KV index,
policies, parsers, chat template, tokenizers, predictors."""
import pytest

from xllm_service_amd.chat_template import JinjaChatTemplate
from xllm_service_amd.service.instance_mgr import Instance
from xllm_service_amd.service.kvcache_mgr import GlobalKVCacheMgr
from xllm_service_amd.service.parsers import (infer_model_family,
                                              make_parsers,
                                              make_stream_parsers)
from xllm_service_amd.service.time_predictor import (TPOTPredictor,
                                                     TTFTPredictor)
from xllm_service_amd.service.types import InstanceMetaInfo, LoadMetrics
from xllm_service_amd.tokenizer import (ByteTokenizer, IncrementalDecoder,
                                        TokenizerFactory)
from xllm_service_amd.utils.hashing import chain_block_hashes


def test_chained_hash_prefix_property():
    toks = list(range(64))
    h1 = chain_block_hashes(toks, 16)
    h2 = chain_block_hashes(toks[:32] + [999] * 32, 16)
    assert len(h1) == 4
    assert h1[:2] == h2[:2]          # shared prefix -> same chain
    assert h1[2:] != h2[2:]          # divergence propagates
    # chain pins the WHOLE prefix: same block content, different prefix
    h3 = chain_block_hashes([7] * 16 + toks[16:32], 16)
    assert h3[1] != h1[1]


def test_global_kvcache_match_and_tiers():
    kv = GlobalKVCacheMgr(block_size=16)
    toks = list(range(48))
    hashes = chain_block_hashes(toks, 16)
    kv.record_updated_kvcaches("i1", stored=hashes[:2], removed=[])
    kv.record_updated_kvcaches("i2", stored=hashes[:3], removed=[])
    ov = kv.match(toks)
    assert ov.matched_blocks == 3
    assert ov.scores["i2"] > ov.scores["i1"]
    # removal shrinks the walk
    kv.record_updated_kvcaches("i2", stored=[], removed=[hashes[2]])
    ov = kv.match(toks)
    assert ov.matched_blocks == 2
    # offload moves tier and lowers weight
    kv.record_updated_kvcaches("i1", stored=[], removed=[],
                               offloaded=[hashes[0]])
    ov = kv.match(toks)
    assert 0 < ov.scores["i1"] < ov.scores["i2"]


class _FakeMgr:
    def __init__(self, prefills, decodes):
        self._p, self._d = prefills, decodes
        self.instances = {i.name: i for i in prefills + decodes}
        self.prefill_index = [i.name for i in prefills]
        self.decode_index = [i.name for i in decodes]

    def schedulable_prefills(self):
        return self._p

    def schedulable_decodes(self):
        return self._d

    def flip_instance_role(self, name, side):
        return False


def _inst(name, itype="PREFILL", waiting=0, cache=0.0):
    i = Instance(InstanceMetaInfo(name=name, itype=itype))
    i.load = LoadMetrics(waiting_requests_num=waiting,
                         gpu_cache_usage_perc=cache)
    return i


def test_cache_aware_policy_prefers_prefix_holder():
    from xllm_service_amd.service.policies import CacheAwarePolicy
    kv = GlobalKVCacheMgr(block_size=16)
    toks = list(range(64))
    hashes = chain_block_hashes(toks, 16)
    kv.record_updated_kvcaches("p1", stored=hashes, removed=[])
    p0, p1 = _inst("p0"), _inst("p1")
    d0 = _inst("d0", "DECODE")
    pol = CacheAwarePolicy(_FakeMgr([p0, p1], [d0]), kv)
    pair = pol.select_instances_pair(toks)
    assert pair.prefill.name == "p1"
    # but heavy load on the cache holder flips the choice
    p1.load = LoadMetrics(waiting_requests_num=64, gpu_cache_usage_perc=0.99)
    pair = pol.select_instances_pair(toks)
    assert pair.prefill.name == "p0"


def test_slo_policy_picks_fast_decode():
    from xllm_service_amd.service.policies import SloAwarePolicy
    kv = GlobalKVCacheMgr(block_size=16)
    p0 = _inst("p0")
    d_slow, d_fast = _inst("ds", "DECODE"), _inst("df", "DECODE")
    pol = SloAwarePolicy(_FakeMgr([p0], [d_slow, d_fast]), kv,
                         target_tpot_ms=50.0)
    for b in range(1, 20):
        pol.observe_tpot("ds", b, 128, 40.0 + 12.0 * b)   # slow instance
        pol.observe_tpot("df", b, 128, 10.0 + 1.0 * b)    # fast instance
    d_slow.num_decoding = 4
    d_fast.num_decoding = 4
    pair = pol.select_instances_pair(list(range(128)))
    assert pair.decode.name == "df"


def test_ttft_tpot_predictors_fit():
    tt = TTFTPredictor()
    for n in range(10, 200, 10):
        tt.add_sample(n, 5 + 0.1 * n + 0.001 * n * n)
    assert abs(tt.predict(100) - (5 + 10 + 10)) < 1.5
    tp = TPOTPredictor()
    for b in range(1, 20):
        tp.add_sample(b, 100, 8 + 2 * b + 0.01 * 100)
    assert abs(tp.predict(10, 100) - (8 + 20 + 1)) < 1.0


def test_model_family_inference():
    assert infer_model_family("Qwen3-32B") == "qwen3"
    assert infer_model_family("qwen2-vl-7b") == "qwen2"
    assert infer_model_family("DeepSeek-V3") == "deepseek_v3"
    assert infer_model_family("Kimi-K2-Instruct") == "kimi_k2"
    assert infer_model_family("unknown-model-7b") is None
    rp, tp = make_parsers("unknown-model-7b")
    assert rp is None and tp is None  # auto silently disables


def test_reasoning_and_toolcall_nonstream():
    rp, tp = make_parsers("Qwen3-8B")
    text = ("<think>let me think\nhard</think>\nThe answer.\n"
            '<tool_call>{"name": "get_weather", "arguments": {"city": "SF"}}'
            "</tool_call>")
    reasoning, rest = rp.extract(text)
    assert reasoning == "let me think\nhard"
    content, calls = tp.extract(rest)
    assert content == "The answer."
    assert len(calls) == 1 and calls[0].name == "get_weather"
    assert '"city"' in calls[0].arguments


def test_streaming_parsers_chunked():
    rp, tp = make_stream_parsers("Qwen3-8B")
    full = ('<think>abc</think>hello <tool_call>{"name": "f", '
            '"arguments": {}}</tool_call> bye')
    reason = content = ""
    calls = []
    for i in range(0, len(full), 3):   # ragged chunks
        r, c = rp.feed(full[i:i + 3])
        reason += r
        c2, cl = tp.feed(c)
        content += c2
        calls.extend(cl)
    calls.extend(tp.flush())
    assert reason == "abc"
    assert content == "hello  bye"
    assert len(calls) == 1 and calls[0].name == "f"


def test_chat_template_default_and_custom():
    ct = JinjaChatTemplate()
    out = ct.apply([{"role": "user", "content": "hi"}])
    assert "<|im_start|>user\nhi<|im_end|>" in out
    assert out.endswith("<|im_start|>assistant\n")
    # custom template with kwargs + tools suppression
    ct2 = JinjaChatTemplate(
        "{% if tools %}TOOLS:{{ tools|length }} {% endif %}"
        "{% for m in messages %}{{ m['content'] }}{% endfor %}"
        "{% if extra %}E={{ extra }}{% endif %}")
    msgs = [{"role": "user", "content": "x"}]
    tools = [{"type": "function", "function": {"name": "f"}}]
    assert ct2.apply(msgs, tools=tools) == "TOOLS:1 x"
    assert ct2.apply(msgs, tools=tools, tool_choice="none") == "x"
    assert ct2.apply(msgs, chat_template_kwargs={"extra": 1}) == "xE=1"


def test_byte_tokenizer_roundtrip_and_incremental():
    tk = ByteTokenizer()
    ids = tk.encode("hello é world")
    assert tk.decode(ids) == "hello é world"
    dec = IncrementalDecoder(tk)
    text = ""
    for i in ids:
        text += dec.push([i])
    assert text == "hello é world"


def test_tiktoken_tokenizer(tmp_path):
    import base64
    vocab = {}
    rank = 0
    for b in range(256):
        vocab[bytes([b])] = rank
        rank += 1
    for merged in [b"he", b"ll", b"llo", b"hello", b" wo", b"rld"]:
        vocab[merged] = rank
        rank += 1
    path = tmp_path / "test.tiktoken"
    with open(path, "wb") as f:
        for tok, r in vocab.items():
            f.write(base64.b64encode(tok) + b" " + str(r).encode() + b"\n")
    from xllm_service_amd.tokenizer.tiktoken_tok import TiktokenTokenizer
    tk = TiktokenTokenizer(str(path))
    ids = tk.encode("hello world")
    assert tk.decode(ids) == "hello world"
    assert len(ids) < len("hello world")  # merges happened
    factory_tk = TokenizerFactory.create(str(tmp_path))
    assert type(factory_tk).__name__ == "TiktokenTokenizer"


def test_slo_policy_seeded_from_registration_profiles():
    """TTFT/TPOT profiling samples in InstanceMetaInfo pre-seed the SLO
    predictors via InstanceMgr.profile_seed_cb before any runtime
    observations accrue."""
    from xllm_service_amd.service.policies import SloAwarePolicy

    class _Mgr:
        pass

    pol = SloAwarePolicy(_Mgr(), None)
    ttft = [[256 * i, 40.0 + 0.5 * (256 * i)] for i in range(1, 9)]
    tpot = [[b, b, 8.0 + 2.0 * b] for b in range(1, 9)]
    pol.seed_from_meta("w0", ttft, tpot)
    assert "w0" in pol.ttft and "w0" in pol.tpot
    # predictors fit: roughly linear TTFT and affine TPOT
    assert abs(pol.ttft["w0"].predict(1024) - (40 + 512)) < 60
    assert abs(pol.tpot["w0"].predict(4, 4) - 16.0) < 4.0


@pytest.mark.parametrize("model,family,rtag,ttags", [
    ("DeepSeek-V3-0324", "deepseek_v3", "<think>",
     ("<｜tool▁call▁begin｜>", "<｜tool▁call▁end｜>")),
    ("Kimi-K2-Instruct", "kimi_k2", "<think>",
     ("<|tool_call_begin|>", "<|tool_call_end|>")),
    ("GLM-4.5-Air", "glm4_moe", "<think>", ("<tool_call>", "</tool_call>")),
    ("step3-32k", "step3", "<think>", ("<tool_call>", "</tool_call>")),
    ("Qwen2.5-72B", "qwen2", None, ("<tool_call>", "</tool_call>")),
    ("Llama-3.1-70B", "llama", None, ("<|python_tag|>", "<|eom_id|>")),
])
def test_parser_families(model, family, rtag, ttags):
    """Every supported model family's tag set parses reasoning and tool
    calls, non-stream and streaming (reference parser bridge tag sets)."""
    from xllm_service_amd.service.parsers import infer_model_family
    assert infer_model_family(model) == family
    rp, tp = make_parsers(model)
    payload = '{"name": "f", "arguments": {"x": 1}}'
    text = ""
    if rtag:
        text += f"{rtag}thinking...{rtag.replace('<', '</', 1)}"
    text += f"hello {ttags[0]}{payload}{ttags[1]} bye"
    if rp is not None:
        reasoning, text2 = rp.extract(text)
        assert reasoning == "thinking..."
    else:
        assert rtag is None
        text2 = text
    assert tp is not None
    rest, calls = tp.extract(text2)
    assert len(calls) == 1 and calls[0].name == "f"
    assert "hello" in rest and "bye" in rest

    # streaming variant fed in awkward 3-char pieces
    srp, stp = make_stream_parsers(model)
    out_text, out_calls, reason_acc = "", [], ""
    buf = text
    for i in range(0, len(buf), 3):
        piece = buf[i:i + 3]
        if srp is not None:
            rdelta, piece = srp.feed(piece)
            if rdelta:
                reason_acc += rdelta
        if stp is not None and piece:
            piece, cs = stp.feed(piece)
            out_calls.extend(cs)
    if stp is not None:
        out_calls.extend(stp.flush())
    if srp is not None:
        assert reason_acc == "thinking..."
    assert len(out_calls) == 1 and out_calls[0].name == "f"


def test_sse_delta_coalescing():
    """_next_delta merges backlogged plain deltas, stops at finishes, and
    parks unmergeable deltas in the hold slot."""
    import asyncio

    from xllm_service_amd.service.request import GenerationDelta
    from xllm_service_amd.service.response_handler import _next_delta

    async def run():
        q = asyncio.Queue()
        hold = []
        q.put_nowait(GenerationDelta(token_ids=[1]))
        q.put_nowait(GenerationDelta(token_ids=[2]))
        q.put_nowait(GenerationDelta(token_ids=[3], finished=True,
                                     finish_reason="stop",
                                     usage_prompt_tokens=5,
                                     usage_completion_tokens=3))
        g = await _next_delta(q, hold)
        assert g.token_ids == [1, 2, 3] and g.finished
        assert g.usage_completion_tokens == 3

        # unmergeable (text override) delta is held and delivered next
        q.put_nowait(GenerationDelta(token_ids=[4]))
        q.put_nowait(GenerationDelta(token_ids=[], text="tail",
                                     finished=True, finish_reason="stop"))
        g1 = await _next_delta(q, hold)
        assert g1.token_ids == [4] and not g1.finished
        g2 = await _next_delta(q, hold)
        assert g2.text == "tail" and g2.finished

        # logprob-carrying deltas are never merged
        q.put_nowait(GenerationDelta(token_ids=[5],
                                     logprobs=[{"token_logprob": -0.5}]))
        q.put_nowait(GenerationDelta(token_ids=[6]))
        g3 = await _next_delta(q, hold)
        assert g3.token_ids == [5]
    asyncio.run(run())


def test_max_tokens_zero_rejected():
    import asyncio

    import httpx

    from tests.test_service_integration import (http_client, make_master,
                                                wait_for, worker_kwargs)
    from xllm_service_amd.engine.worker import Worker

    async def run():
        master = make_master(policy="RR")
        await master.start(serve_http=False)
        w = Worker("w0", "DEFAULT", **worker_kwargs(master))
        try:
            await w.start()
            await wait_for(
                lambda: master.scheduler.has_available_instances())
            client = await http_client(master)
            r = await client.post("/v1/completions", json={
                "model": "llama-tiny", "prompt": [1, 2, 3],
                "max_tokens": 0})
            assert r.status_code == 400
            r = await client.post("/v1/chat/completions", json={
                "model": "llama-tiny",
                "messages": [{"role": "user", "content": "x"}],
                "max_tokens": 0})
            assert r.status_code == 400
            await client.aclose()
        finally:
            await w.stop()
            await master.stop()
    asyncio.run(run())


def test_unsupported_shapes_rejected():
    import asyncio

    from tests.test_service_integration import (http_client, make_master,
                                                wait_for, worker_kwargs)
    from xllm_service_amd.engine.worker import Worker

    async def run():
        master = make_master(policy="RR")
        await master.start(serve_http=False)
        w = Worker("w0", "DEFAULT", **worker_kwargs(master))
        try:
            await w.start()
            await wait_for(
                lambda: master.scheduler.has_available_instances())
            client = await http_client(master)
            r = await client.post("/v1/completions", json={
                "model": "llama-tiny", "prompt": [1, 2, 3], "n": 2})
            assert r.status_code == 400
            r = await client.post("/v1/completions", json={
                "model": "llama-tiny", "prompt": ["two", "prompts"]})
            assert r.status_code == 400
            await client.aclose()
        finally:
            await w.stop()
            await master.stop()
    asyncio.run(run())


def test_engine_failure_surfaces_error_to_client():
    """A poisoned engine step aborts running sequences WITH an error
    message — clients must see a 500, not an empty 200 (round-2 fix)."""
    import asyncio

    from tests.test_service_integration import (http_client, make_master,
                                                wait_for, worker_kwargs)
    from xllm_service_amd.engine.worker import Worker

    async def run():
        master = make_master(policy="RR")
        await master.start(serve_http=False)
        w = Worker("w0", "DEFAULT", **worker_kwargs(master))
        try:
            await w.start()
            await wait_for(
                lambda: master.scheduler.has_available_instances())

            # poison the NEXT engine step only
            real_step = w.engine.step
            state = {"armed": False, "fired": False}

            def boom():
                if state["armed"] and not state["fired"]:
                    state["fired"] = True
                    raise RuntimeError("injected device fault")
                return real_step()
            w.engine.step = boom

            client = await http_client(master)
            state["armed"] = True
            r = await client.post("/v1/completions", json={
                "model": "llama-tiny", "prompt": list(range(20, 40)),
                "max_tokens": 4, "temperature": 0.0, "ignore_eos": True})
            assert r.status_code == 500, r.text
            assert "injected device fault" in r.json()["error"]["message"]

            # worker keeps serving afterwards
            r2 = await client.post("/v1/completions", json={
                "model": "llama-tiny", "prompt": list(range(20, 40)),
                "max_tokens": 4, "temperature": 0.0, "ignore_eos": True})
            assert r2.status_code == 200, r2.text
            await client.aclose()
        finally:
            await w.stop()
            await master.stop()
    asyncio.run(run())
