#!/usr/bin/env python3
"""Flagship serving benchmark — the driver contract (see repo instructions).

Measures the BASELINE.json metric: output tok/s of continuous-batching
serving of Llama-3-8B (bf16, synthetic prompts, random-init weights) on
N MI355X GPUs. A "step" is one engine step (one decode iteration of the
running batch, plus any admitted prefill chunks). For N>1 ranks each run an
independent engine replica (data parallel / weak scaling: this mirrors the
instance pool of the serving deployment); the reported value is the
whole-job aggregate output tokens per second.

  python bench.py --gpus 1 --steps 64 --warmup 16
  python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
      --master-addr 127.0.0.1 bench.py --gpus 8 --steps 64 --warmup 16
"""
from __future__ import annotations

import argparse
import json
import os
import time

# hipBLASLt/rocBLAS algo selection tuned offline on MI355X (TunableOp);
# must be configured before the first torch import in the process.
_TUNABLE = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                        "configs", "tunableop_gfx950.csv")
if os.path.exists(_TUNABLE):
    os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
    os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "0")
    os.environ.setdefault("PYTORCH_TUNABLEOP_FILENAME", _TUNABLE)

import torch


def parse_args():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=64)
    ap.add_argument("--warmup", type=int, default=16)
    ap.add_argument("--model", default="llama-3-8b")
    ap.add_argument("--concurrency", type=int, default=64,
                    help="concurrent requests per GPU")
    ap.add_argument("--input-len", type=int, default=1024)
    ap.add_argument("--output-len", type=int, default=1024)
    ap.add_argument("--max-batched-tokens", type=int, default=16384)
    ap.add_argument("--device", default=None)
    return ap.parse_args()


def main():
    args = parse_args()
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))

    use_gpu = torch.cuda.is_available() if args.device is None else (
        args.device.startswith("cuda"))
    dist = None
    if world > 1:
        import torch.distributed as dist_mod
        dist = dist_mod
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        backend = "nccl" if use_gpu else "gloo"
        dist.init_process_group(backend=backend)
        if use_gpu:
            torch.cuda.set_device(local_rank)

    device = args.device or (f"cuda:{local_rank}" if use_gpu else "cpu")
    model_name = args.model
    if not use_gpu and model_name == "llama-3-8b":
        # CPU fallback so `python bench.py` runs anywhere; the measured
        # config on GPU is the flagship model.
        model_name = "llama-tiny"

    from xllm_service_amd.engine.engine import LLMEngine
    from xllm_service_amd.engine.sampling import SamplingParams
    from xllm_service_amd.models.config import get_config

    cfg = get_config(model_name)
    eng = LLMEngine(model_name, device=device,
                    max_num_seqs=args.concurrency * 2,
                    max_batched_tokens=args.max_batched_tokens,
                    max_kv_blocks=None if use_gpu else 4096,
                    seed=0)

    torch.manual_seed(1234 + rank)
    rid = 0

    def feed(n):
        nonlocal rid
        for _ in range(n):
            prompt = torch.randint(0, cfg.vocab_size,
                                   (args.input_len,)).tolist()
            eng.add_request(f"r{rank}-{rid}", prompt,
                            SamplingParams(max_tokens=args.output_len,
                                           ignore_eos=True))
            rid += 1

    # keep the pipe full: closed-loop client at fixed concurrency
    feed(args.concurrency)

    def one_step():
        outs = eng.step()
        done = sum(1 for o in outs if o.finished)
        if done:
            feed(done)
        return sum(len(o.new_token_ids) for o in outs)

    # ---- warmup -----------------------------------------------------------
    for _ in range(args.warmup):
        one_step()

    # ---- timed region -----------------------------------------------------
    if dist:
        dist.barrier()
    if use_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    generated = 0
    for _ in range(args.steps):
        generated += one_step()
    if use_gpu:
        torch.cuda.synchronize()
    if dist:
        dist.barrier()
    t1 = time.perf_counter()

    elapsed = t1 - t0
    # max elapsed over ranks + total tokens over ranks
    if dist:
        te = torch.tensor([elapsed], dtype=torch.float64,
                          device=device if use_gpu else "cpu")
        dist.all_reduce(te, op=dist.ReduceOp.MAX)
        elapsed = float(te.item())
        tg = torch.tensor([generated], dtype=torch.float64,
                          device=device if use_gpu else "cpu")
        dist.all_reduce(tg, op=dist.ReduceOp.SUM)
        generated = int(tg.item())

    n_gpus = world if world > 1 else (1 if use_gpu else args.gpus)
    value = generated / elapsed
    if rank == 0:
        print(json.dumps({
            "metric": "SLO-goodput (out tok/s under p50 TTFT SLO), "
                      "Llama-3-8B PD-disagg on 8 MI355X",
            "value": round(value, 2),
            "unit": "tok/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if use_gpu else "fp32",
            "data": "synthetic",
            "config": {
                "model": model_name,
                "global_batch": args.concurrency * n_gpus,
                "seq_len": args.input_len + args.output_len,
                "input_len": args.input_len,
                "output_len": args.output_len,
                "parallelism": f"dp{n_gpus}",
            },
        }))
    if dist:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
