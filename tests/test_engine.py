"""End-to-end engine tests on the CPU reference path.

The oracle is an independent dense-attention forward (torch SDPA) over the
same randomly-initialised weights — so paged KV, chunked prefill, prefix
caching and preemption must all reproduce exact dense-greedy outputs.
"""
import math

import pytest
import torch
import torch.nn.functional as F

from xllm_service_amd.engine.engine import LLMEngine
from xllm_service_amd.engine.sampling import SamplingParams
from xllm_service_amd.models.config import get_config
from xllm_service_amd.ops import ref as op_ref

torch.manual_seed(0)


def dense_greedy(model, cfg, prompt, n_tokens):
    """Independent dense reference: full forward each step, torch SDPA."""
    toks = list(prompt)
    D = cfg.head_dim
    for _ in range(n_tokens):
        T = len(toks)
        x = model.embed(torch.tensor(toks))
        pos = torch.arange(T)
        cos_sin = op_ref.rope_table(D, cfg.max_position, cfg.rope_theta)
        residual = None
        for layer in model.layers:
            if residual is None:
                residual = x
                h = op_ref.rmsnorm(x, layer.input_norm, cfg.rms_eps)
            else:
                h, residual = op_ref.fused_add_rmsnorm(
                    x, residual, layer.input_norm, cfg.rms_eps)
            qkv = F.linear(h, layer.attn.qkv_proj.weight,
                           layer.attn.qkv_proj.bias)  # qwen2 has qkv bias
            q, k, v = torch.split(
                qkv, [cfg.q_size, cfg.kv_size, cfg.kv_size], dim=-1)
            q, k = op_ref.rope(pos, q.contiguous(), k.contiguous(),
                               cos_sin, D, D)
            qh = q.view(T, cfg.num_heads, D).transpose(0, 1)
            kh = k.view(T, cfg.num_kv_heads, D).transpose(0, 1)
            vh = v.view(T, cfg.num_kv_heads, D).transpose(0, 1)
            rep = cfg.num_heads // cfg.num_kv_heads
            attn = F.scaled_dot_product_attention(
                qh.float(), kh.float().repeat_interleave(rep, 0),
                vh.float().repeat_interleave(rep, 0),
                is_causal=True, scale=1.0 / math.sqrt(D)).to(h.dtype)
            a = attn.transpose(0, 1).reshape(T, -1)
            o = F.linear(a, layer.attn.o_proj.weight)
            h, residual = op_ref.fused_add_rmsnorm(
                o, residual, layer.post_norm, cfg.rms_eps)
            gu = F.linear(h, layer.mlp.gate_up.weight)
            x = F.linear(op_ref.silu_and_mul(gu), layer.mlp.down.weight)
        h, _ = op_ref.fused_add_rmsnorm(x, residual, model.final_norm,
                                        cfg.rms_eps)
        logits = model.compute_logits(h[-1:])
        toks.append(int(logits.float().argmax()))
    return toks[len(prompt):]


@pytest.fixture(scope="module")
def tiny_engine():
    return LLMEngine("llama-tiny", device="cpu", max_kv_blocks=512, seed=7)


def test_engine_matches_dense_reference(tiny_engine):
    eng = tiny_engine
    cfg = get_config("llama-tiny")
    torch.manual_seed(3)
    prompts = [torch.randint(0, cfg.vocab_size, (n,)).tolist()
               for n in (5, 17, 33)]
    n_out = 8
    got = eng.generate(prompts, SamplingParams(max_tokens=n_out,
                                               ignore_eos=True))
    for p, g in zip(prompts, got):
        want = dense_greedy(eng.model, cfg, p, n_out)
        assert g == want, f"prompt len {len(p)}: {g} != {want}"


def test_chunked_prefill_same_output():
    eng = LLMEngine("llama-tiny", device="cpu", max_kv_blocks=512,
                    max_batched_tokens=8, seed=7)
    cfg = get_config("llama-tiny")
    torch.manual_seed(5)
    prompt = torch.randint(0, cfg.vocab_size, (30,)).tolist()
    got = eng.generate([prompt], SamplingParams(max_tokens=6, ignore_eos=True))
    want = dense_greedy(eng.model, cfg, prompt, 6)
    assert got[0] == want


def test_prefix_cache_reuse_and_correctness():
    eng = LLMEngine("llama-tiny", device="cpu", max_kv_blocks=512, seed=7)
    cfg = get_config("llama-tiny")
    torch.manual_seed(9)
    shared = torch.randint(0, cfg.vocab_size, (40,)).tolist()
    p1 = shared + [1, 2, 3]
    p2 = shared + [4, 5, 6, 7]
    out1 = eng.generate([p1], SamplingParams(max_tokens=4, ignore_eos=True))[0]
    # second request must hit the cached prefix
    from xllm_service_amd.engine.sequence import Sequence
    probe = Sequence("probe", p2, SamplingParams())
    assert eng.block_manager.match_prefix(probe) >= 32  # 2 full blocks
    out2 = eng.generate([p2], SamplingParams(max_tokens=4, ignore_eos=True))[0]
    assert out2 == dense_greedy(eng.model, cfg, p2, 4)
    assert out1 == dense_greedy(eng.model, cfg, p1, 4)


def test_preemption_under_tiny_pool():
    # pool too small for all three sequences at once -> must preempt + recompute
    eng = LLMEngine("llama-tiny", device="cpu", max_kv_blocks=14, seed=7,
                    enable_prefix_caching=False)
    cfg = get_config("llama-tiny")
    torch.manual_seed(11)
    prompts = [torch.randint(0, cfg.vocab_size, (48,)).tolist()
               for _ in range(3)]
    got = eng.generate(prompts, SamplingParams(max_tokens=5, ignore_eos=True))
    for p, g in zip(prompts, got):
        assert g == dense_greedy(eng.model, cfg, p, 5)


def test_online_preempts_offline():
    eng = LLMEngine("llama-tiny", device="cpu", max_kv_blocks=10, seed=7,
                    enable_prefix_caching=False)
    cfg = get_config("llama-tiny")
    torch.manual_seed(13)
    off_prompt = torch.randint(0, cfg.vocab_size, (64,)).tolist()
    on_prompt = torch.randint(0, cfg.vocab_size, (64,)).tolist()
    eng.add_request("offline", off_prompt,
                    SamplingParams(max_tokens=20, ignore_eos=True), priority=1)
    eng.step()
    eng.add_request("online", on_prompt,
                    SamplingParams(max_tokens=5, ignore_eos=True), priority=0)
    online_done_at = offline_done_at = None
    for i in range(400):
        for out in eng.step():
            if out.finished and out.request_id == "online":
                online_done_at = i
            if out.finished and out.request_id == "offline":
                offline_done_at = i
        if not eng.has_work():
            break
    assert online_done_at is not None and offline_done_at is not None
    assert online_done_at < offline_done_at
    # and the offline request still produced correct output despite preemption
    # (its sequence was recomputed; verify token count)


def test_abort():
    eng = LLMEngine("llama-tiny", device="cpu", max_kv_blocks=64, seed=7)
    eng.add_request("r1", [1, 2, 3, 4], SamplingParams(max_tokens=50,
                                                       ignore_eos=True))
    eng.step()
    assert eng.abort_request("r1")
    assert not eng.has_work()
    assert eng.block_manager.num_free == eng.block_manager.num_blocks


def test_opt_engine_runs():
    eng = LLMEngine("opt-125m", device="cpu", max_kv_blocks=64, seed=1)
    out = eng.generate([[10, 11, 12, 13, 14]],
                       SamplingParams(max_tokens=4, ignore_eos=True))
    assert len(out[0]) == 4


def test_swap_preemption_preserves_kv():
    """Preemption under memory pressure swaps KV to the host-DRAM tier and
    restores it exactly (outputs == dense oracle, no recompute)."""
    # 9-block pool, two 60-token prompts (4 blocks each): both cross a
    # block boundary at token 64 mid-decode -> one must swap out
    eng = LLMEngine("llama-tiny", device="cpu", max_kv_blocks=9, seed=7,
                    enable_prefix_caching=False)
    assert eng.cpu_block_manager is not None
    cfg = get_config("llama-tiny")
    torch.manual_seed(17)
    prompts = [torch.randint(0, cfg.vocab_size, (60,)).tolist()
               for _ in range(2)]
    got = eng.generate(prompts, SamplingParams(max_tokens=8, ignore_eos=True))
    assert eng.scheduler.num_swap_outs > 0, "no swap happened (pool too big?)"
    assert eng.scheduler.num_swap_ins > 0
    for p, g in zip(prompts, got):
        assert g == dense_greedy(eng.model, cfg, p, 8)
    # all tiers drained
    assert eng.block_manager.num_free == eng.block_manager.num_blocks
    assert (eng.cpu_block_manager.num_free ==
            eng.cpu_block_manager.num_blocks)


def test_ssd_swap_tier_preserves_kv(tmp_path):
    """With no DRAM tier (swap_space_mb=0), preemption spools KV blocks to
    SSD files and restores them exactly; spool files are cleaned up."""
    import os
    ssd = str(tmp_path / "kvspool")
    eng = LLMEngine("llama-tiny", device="cpu", max_kv_blocks=9, seed=7,
                    enable_prefix_caching=False, swap_space_mb=0,
                    ssd_swap_dir=ssd)
    assert eng.cpu_block_manager is None
    cfg = get_config("llama-tiny")
    torch.manual_seed(17)
    prompts = [torch.randint(0, cfg.vocab_size, (60,)).tolist()
               for _ in range(2)]
    got = eng.generate(prompts, SamplingParams(max_tokens=8, ignore_eos=True))
    assert eng.scheduler.num_swap_outs > 0, "no ssd swap happened"
    assert eng.scheduler.num_swap_ins > 0
    for p, g in zip(prompts, got):
        assert g == dense_greedy(eng.model, cfg, p, 8)
    assert eng.block_manager.num_free == eng.block_manager.num_blocks
    assert os.listdir(ssd) == []          # spools deleted after swap-in


def test_multi_token_stop_sequence():
    """A stop sequence longer than one token halts generation at the match
    (token-suffix matching)."""
    eng = LLMEngine("llama-tiny", device="cpu", max_kv_blocks=64, seed=3)
    cfg = get_config("llama-tiny")
    torch.manual_seed(5)
    prompt = torch.randint(0, cfg.vocab_size, (12,)).tolist()
    # discover what the model would generate unconstrained
    free = eng.generate([prompt], SamplingParams(max_tokens=8,
                                                 ignore_eos=True))[0]
    assert len(free) == 8
    # now stop at the 2-token sequence (free[2], free[3])
    eng2 = LLMEngine("llama-tiny", device="cpu", max_kv_blocks=64, seed=3)
    got = eng2.generate([prompt], SamplingParams(
        max_tokens=8, ignore_eos=True,
        stop_sequences=[[free[2], free[3]]]))[0]
    assert got == free[:4]          # stops right after the match


def test_qwen2_text_model():
    """Plain Qwen2 (llama stack + qkv bias) runs on the paged engine and
    matches the dense oracle."""
    eng = LLMEngine("qwen2-tiny", device="cpu", max_kv_blocks=64, seed=2)
    cfg = get_config("qwen2-tiny")
    torch.manual_seed(9)
    prompt = torch.randint(0, cfg.vocab_size, (20,)).tolist()
    got = eng.generate([prompt], SamplingParams(max_tokens=6,
                                                ignore_eos=True))[0]
    assert got == dense_greedy(eng.model, cfg, prompt, 6)


def test_max_model_len_enforced():
    """Prompts past max_model_len are rejected; max_tokens is clamped so
    total length never exceeds the graph/rope capacity."""
    eng = LLMEngine("llama-tiny", device="cpu", max_kv_blocks=512,
                    max_model_len=256)
    torch.manual_seed(0)
    with pytest.raises(ValueError):
        eng.add_request("too-long", list(range(30)) * 10,
                        SamplingParams(max_tokens=4))   # 300 tokens
    with pytest.raises(ValueError):
        eng.add_request("empty", [], SamplingParams(max_tokens=4))
    prompt = torch.randint(0, 100, (250,)).tolist()
    out = eng.generate([prompt], SamplingParams(max_tokens=64,
                                                ignore_eos=True))[0]
    assert len(out) == 256 - 250                        # clamped to capacity
