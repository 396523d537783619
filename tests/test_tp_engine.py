"""Tensor-parallel ENGINE group test (CPU, gloo world=2).

Rank 0 drives the full continuous-batching engine; rank 1 runs the follower
loop replaying broadcast batches. Outputs must exactly match a single
(tp=1) engine loaded from the same full checkpoint.
"""
import os
import socket

import pytest
import torch
import torch.multiprocessing as mp

from xllm_service_amd.models.config import get_config

MODEL = "llama-tiny"


def _make_checkpoint(path):
    from xllm_service_amd.models.llama import LlamaForCausalLM
    cfg = get_config(MODEL)
    torch.manual_seed(0)
    model = LlamaForCausalLM(cfg, dtype=torch.float32)
    model.random_init(5)
    torch.save(model.state_dict(), path)


def _prompts(cfg):
    torch.manual_seed(41)
    return [torch.randint(0, cfg.vocab_size, (n,)).tolist()
            for n in (7, 22, 40)]


def _single_outputs(ckpt):
    from xllm_service_amd.engine.engine import LLMEngine
    from xllm_service_amd.engine.sampling import SamplingParams
    cfg = get_config(MODEL)
    eng = LLMEngine(MODEL, device="cpu", max_kv_blocks=128,
                    load_state_path=ckpt)
    return eng.generate(_prompts(cfg),
                        SamplingParams(max_tokens=6, ignore_eos=True))


def _tp_rank(rank, world, port, ckpt, out_path):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    from xllm_service_amd.engine.engine import LLMEngine
    from xllm_service_amd.engine.sampling import SamplingParams
    cfg = get_config(MODEL)
    eng = LLMEngine(MODEL, device="cpu", max_kv_blocks=128, tp_size=world,
                    load_state_path=ckpt)
    from xllm_service_amd.distributed import parallel_state as ps
    if rank == 0:
        out = eng.generate(_prompts(cfg),
                           SamplingParams(max_tokens=6, ignore_eos=True))
        eng.runner.stop_followers()
        torch.save(out, out_path)
    else:
        eng.follower_loop()
    ps.shutdown()


def test_tp2_engine_group_matches_single(tmp_path):
    ckpt = str(tmp_path / "full.pt")
    out_path = str(tmp_path / "tp_out.pt")
    _make_checkpoint(ckpt)
    want = _single_outputs(ckpt)
    with socket.socket() as sock:
        sock.bind(("127.0.0.1", 0))
        port = sock.getsockname()[1]
    mp.spawn(_tp_rank, args=(2, port, ckpt, out_path), nprocs=2, join=True)
    got = torch.load(out_path)
    assert got == want, f"{got} != {want}"
