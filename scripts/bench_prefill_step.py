#!/usr/bin/env python3
"""What does ONE serving arrival cost the decode loop?

Fills the engine with a steady decode batch, times pure-decode steps, then
injects single 1024-token prompts and times the steps that carry their
prefill. The difference is the per-arrival cost the serving mix pays
(engine-mode bench windows contain no completions, so they measure pure
decode and overstate serving capacity).
"""
import os
import sys
import time

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)
_T = os.path.join(ROOT, "configs", "tunableop_gfx950.csv")
if os.path.exists(_T):
    os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
    os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "0")
    os.environ.setdefault("PYTORCH_TUNABLEOP_FILENAME", _T)
import torch

from xllm_service_amd.engine.engine import LLMEngine
from xllm_service_amd.engine.sampling import SamplingParams

B = int(os.environ.get("BB", "128"))
IN_LEN = 1024


def timed_step(eng):
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    eng.step()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) * 1000.0


def main():
    eng = LLMEngine("llama-3-8b", device="cuda:0", seed=0,
                    max_num_seqs=512, max_batched_tokens=8192)
    cfg = eng.cfg
    torch.manual_seed(7)
    for i in range(B):
        eng.add_request(f"d{i}", torch.randint(
            0, cfg.vocab_size, (IN_LEN,)).tolist(),
            SamplingParams(max_tokens=4096, ignore_eos=True))
    # drain all initial prefills
    while any(not s.prefill_done for s in eng.scheduler.running) \
            or eng.scheduler.waiting:
        eng.step()
    for _ in range(10):
        eng.step()

    dec = sorted(timed_step(eng) for _ in range(30))
    d_med = dec[len(dec) // 2]
    print(f"batch {B}: pure-decode step {d_med:.2f} ms "
          f"({B / d_med * 1000:.0f} tok/s)")

    # inject single arrivals; time the prefill-carrying step
    costs = []
    for j in range(6):
        eng.add_request(f"a{j}", torch.randint(
            0, cfg.vocab_size, (IN_LEN,)).tolist(),
            SamplingParams(max_tokens=4096, ignore_eos=True))
        t = timed_step(eng)       # step with the 1024-token prefill
        costs.append(t)
        for _ in range(4):
            eng.step()
    costs.sort()
    p_med = costs[len(costs) // 2]
    print(f"prefill(1024)-carrying step: {p_med:.2f} ms "
          f"(arrival cost ~{p_med - d_med:.2f} ms)")

    # per-op timing of one eager prefill-only forward
    from xllm_service_amd.engine.scheduler import StepPlan
    evs = []
    prof = torch.profiler.profile(
        activities=[torch.profiler.ProfilerActivity.CUDA],
        record_shapes=False)
    eng.add_request("p0", torch.randint(
        0, cfg.vocab_size, (IN_LEN,)).tolist(),
        SamplingParams(max_tokens=8, ignore_eos=True))
    with prof:
        eng.step()
        torch.cuda.synchronize()
    ka = {}
    for e in prof.key_averages():
        if e.device_time_total > 0:
            ka[e.key] = e.device_time_total
    top = sorted(ka.items(), key=lambda kv: -kv[1])[:12]
    tot = sum(ka.values())
    print(f"prefill+decode step GPU time total {tot/1000:.2f} ms; top kernels:")
    for k, v in top:
        print(f"  {v/1000:8.3f} ms  {k[:90]}")


if __name__ == "__main__":
    main()
