"""Property-based nets over the round-2 hot logic: the text-level stop
scanner, the windowed incremental detokenizer, and engine block-pool
accounting under adversarial add/abort/step interleavings."""
import hypothesis.strategies as st
import torch
from hypothesis import HealthCheck, given, settings

from xllm_service_amd.tokenizer import ByteTokenizer, IncrementalDecoder


# ---------------------------------------------------------------- decoder
@settings(max_examples=60, deadline=None)
@given(st.text(min_size=0, max_size=60),
       st.lists(st.integers(min_value=1, max_value=7), max_size=12))
def test_windowed_decoder_matches_full_decode(text, chunk_sizes):
    tk = ByteTokenizer()
    ids = tk.encode(text)
    dec = IncrementalDecoder(tk)
    out = ""
    i = 0
    for c in chunk_sizes:
        if i >= len(ids):
            break
        out += dec.push(ids[i:i + c])
        i += c
    out += dec.push(ids[i:])
    full = tk.decode(ids)
    # contract: a trailing U+FFFD may be withheld at stream end (it looks
    # like a partial UTF-8 sequence); everything before it must match
    assert full == text
    assert out == full or (full.startswith(out)
                           and set(full[len(out):]) == {"\ufffd"})


# ------------------------------------------------------------ stop scanner
def _scan_all(tokenizer, stops, token_chunks):
    from xllm_service_amd.service.stop_scanner import StopTextScanner
    sc = StopTextScanner(tokenizer, stops)
    text = ""
    for chunk in token_chunks:
        out, override, hit = sc.feed(chunk)
        text += tokenizer.decode(out)
        if hit:
            return text + override, True
    text += tokenizer.decode(sc.flush())
    return text, False


@settings(max_examples=80, deadline=None,
          suppress_health_check=[HealthCheck.filter_too_much])
@given(st.text(alphabet="abcX ", min_size=0, max_size=48),
       st.text(alphabet="abcX ", min_size=1, max_size=6),
       st.integers(min_value=1, max_value=9))
def test_stop_scanner_matches_oracle(text, stop, chunk):
    tk = ByteTokenizer()
    ids = tk.encode(text)
    chunks = [ids[i:i + chunk] for i in range(0, len(ids), chunk)]
    got, hit = _scan_all(tk, [stop], chunks)
    idx = text.find(stop)
    if idx >= 0:
        assert hit and got == text[:idx]
    else:
        assert not hit and got == text


# ----------------------------------------------------------- engine fuzz
@settings(max_examples=10, deadline=None,
          suppress_health_check=[HealthCheck.too_slow])
@given(st.lists(st.tuples(st.sampled_from(["add", "abort", "step"]),
                          st.integers(min_value=0, max_value=9)),
                min_size=6, max_size=30),
       st.integers(min_value=0, max_value=1000))
def test_engine_block_accounting_fuzz(ops_list, seed):
    """Random add/abort/step interleavings must never crash, and after
    draining, every KV block returns to the pool (prefix-cache evictables
    count as free)."""
    from xllm_service_amd.engine.engine import LLMEngine
    from xllm_service_amd.engine.sampling import SamplingParams
    eng = LLMEngine("llama-tiny", device="cpu", max_kv_blocks=24, seed=3,
                    swap_space_mb=0, max_num_seqs=6)
    free0 = eng.block_manager.num_free
    torch.manual_seed(seed)
    nid = 0
    live = []
    for op, arg in ops_list:
        if op == "add":
            n = 4 + (arg * 7) % 40
            prompt = torch.randint(0, eng.cfg.vocab_size, (n,)).tolist()
            eng.add_request(f"f{nid}", prompt,
                            SamplingParams(max_tokens=1 + arg % 5,
                                           ignore_eos=True))
            live.append(f"f{nid}")
            nid += 1
        elif op == "abort" and live:
            eng.abort_request(live.pop(arg % len(live)))
        elif op == "step":
            eng.step()
    while eng.has_work():
        eng.step()
    assert eng.block_manager.num_free == free0


# ------------------------------------------------------- gemm weight pack
@settings(max_examples=12, deadline=None)
@given(st.sampled_from([(64, 256), (128, 512), (192, 1024)]))
def test_pack_gemm_weight_is_a_permutation(shape):
    """The packed layout must be a pure reshuffle: inverting the permute
    recovers the original weight exactly."""
    from xllm_service_amd import ops
    n, k = shape
    w = torch.randn(n, k, dtype=torch.bfloat16)
    p = ops.pack_gemm_weight(w)
    back = (p.view(n // 16, k // 32, 4, 16, 8)
             .permute(0, 3, 1, 2, 4).reshape(n, k))
    assert torch.equal(back, w)


@settings(max_examples=8, deadline=None,
          suppress_health_check=[HealthCheck.too_slow])
@given(st.lists(st.tuples(st.sampled_from(["add", "addoff", "abort", "step"]),
                          st.integers(min_value=0, max_value=9)),
                min_size=8, max_size=26),
       st.integers(min_value=0, max_value=500))
def test_engine_fuzz_with_swap_and_priorities(ops_list, seed):
    """Same invariant with the DRAM swap tier enabled and mixed
    online/offline priorities: preemption may swap or recompute; all
    HBM and DRAM blocks return to their pools after draining."""
    from xllm_service_amd.engine.engine import LLMEngine
    from xllm_service_amd.engine.sampling import SamplingParams
    eng = LLMEngine("llama-tiny", device="cpu", max_kv_blocks=20, seed=3,
                    swap_space_mb=1, max_num_seqs=6)
    free0 = eng.block_manager.num_free
    cfree0 = eng.cpu_block_manager.num_free if eng.cpu_block_manager else 0
    torch.manual_seed(seed)
    nid = 0
    live = []
    for op, arg in ops_list:
        if op in ("add", "addoff"):
            n = 4 + (arg * 7) % 36
            prompt = torch.randint(0, eng.cfg.vocab_size, (n,)).tolist()
            eng.add_request(f"s{nid}", prompt,
                            SamplingParams(max_tokens=1 + arg % 6,
                                           ignore_eos=True),
                            priority=1 if op == "addoff" else 0)
            live.append(f"s{nid}")
            nid += 1
        elif op == "abort" and live:
            eng.abort_request(live.pop(arg % len(live)))
        elif op == "step":
            eng.step()
    while eng.has_work():
        eng.step()
    assert eng.block_manager.num_free == free0
    if eng.cpu_block_manager:
        assert eng.cpu_block_manager.num_free == cfree0
