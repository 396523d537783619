"""Tensor-parallel equivalence on CPU (gloo, world_size=2).

Spawns 2 processes; each builds the TP=2 shard of a full Llama model (same
seed) and runs one paged prefill forward; rank 0 checks the TP hidden states
and logits match the single-process (tp=1) model.
"""
import os

import pytest
import torch
import torch.multiprocessing as mp

from xllm_service_amd.models.config import get_config

MODEL = "llama-tiny"


def _full_model_outputs():
    from xllm_service_amd.engine.metadata import AttnMetadata
    from xllm_service_amd.models.llama import LlamaForCausalLM
    cfg = get_config(MODEL)
    torch.manual_seed(0)
    model = LlamaForCausalLM(cfg, dtype=torch.float32)
    model.random_init(3)
    T = 12
    input_ids = torch.arange(10, 10 + T)
    positions = torch.arange(T)
    kv = [(torch.zeros(8, cfg.num_kv_heads, 16, cfg.head_dim),
           torch.zeros(8, cfg.num_kv_heads, 16, cfg.head_dim))
          for _ in range(cfg.num_layers)]
    meta = AttnMetadata(
        num_prefill_tokens=T, num_decode_tokens=0,
        slot_mapping=torch.arange(T),
        cu_q=torch.tensor([0, T], dtype=torch.int32),
        prefill_seq_lens=torch.tensor([T], dtype=torch.int32),
        prefill_block_tables=torch.tensor([[0]], dtype=torch.int32))
    with torch.inference_mode():
        hidden = model(input_ids, positions, kv, meta)
        logits = model.compute_logits(hidden[-1:])
    return model.state_dict(), hidden, logits


def _tp_worker(rank, world, port, sd_path, out_path):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank), MASTER_ADDR="127.0.0.1",
                      MASTER_PORT=str(port))
    import torch
    from xllm_service_amd.distributed import parallel_state as ps
    from xllm_service_amd.distributed.layers import shard_llama_state_dict
    from xllm_service_amd.engine.metadata import AttnMetadata
    from xllm_service_amd.models.llama import LlamaForCausalLM
    ps.init_distributed(backend="gloo")
    ps.init_tensor_parallel(world)
    cfg = get_config(MODEL)
    model = LlamaForCausalLM(cfg, dtype=torch.float32)
    full_sd = torch.load(sd_path)
    model.load_state_dict(shard_llama_state_dict(full_sd, cfg, world, rank))
    T = 12
    input_ids = torch.arange(10, 10 + T)
    positions = torch.arange(T)
    n_kv_local = cfg.num_kv_heads // world
    kv = [(torch.zeros(8, n_kv_local, 16, cfg.head_dim),
           torch.zeros(8, n_kv_local, 16, cfg.head_dim))
          for _ in range(cfg.num_layers)]
    meta = AttnMetadata(
        num_prefill_tokens=T, num_decode_tokens=0,
        slot_mapping=torch.arange(T),
        cu_q=torch.tensor([0, T], dtype=torch.int32),
        prefill_seq_lens=torch.tensor([T], dtype=torch.int32),
        prefill_block_tables=torch.tensor([[0]], dtype=torch.int32))
    with torch.inference_mode():
        hidden = model(input_ids, positions, kv, meta)
        logits = model.compute_logits(hidden[-1:])
    if rank == 0:
        torch.save({"hidden": hidden, "logits": logits}, out_path)
    ps.shutdown()


def test_tp2_matches_single(tmp_path):
    sd, hidden, logits = _full_model_outputs()
    sd_path = str(tmp_path / "full.pt")
    out_path = str(tmp_path / "tp_out.pt")
    torch.save(sd, sd_path)
    import socket
    with socket.socket() as sock:   # pick a free port (avoid suite clashes)
        sock.bind(("127.0.0.1", 0))
        port = sock.getsockname()[1]
    mp.spawn(_tp_worker, args=(2, port, sd_path, out_path), nprocs=2,
             join=True)
    got = torch.load(out_path)
    assert torch.allclose(got["hidden"], hidden, atol=1e-4), \
        (got["hidden"] - hidden).abs().max()
    assert torch.allclose(got["logits"], logits, atol=1e-4)
