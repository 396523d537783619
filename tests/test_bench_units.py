"""Unit coverage for bench.py's orchestration helpers (driver contract)."""
import importlib.util
import os
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
spec = importlib.util.spec_from_file_location(
    "bench_mod", os.path.join(ROOT, "bench.py"))
bench = importlib.util.module_from_spec(spec)
sys.modules["bench_mod"] = bench
spec.loader.exec_module(bench)


def test_topology_shapes():
    assert bench.topology(1) == ["DEFAULT"]
    assert bench.topology(2) == ["PREFILL", "DECODE"]
    assert bench.topology(4) == ["PREFILL", "DECODE", "DECODE", "DECODE"]
    assert bench.topology(8) == ["PREFILL"] * 2 + ["DECODE"] * 6


def test_parse_prom():
    txt = ("# HELP x y\n"
           "generated_tokens_total 123.0\n"
           'cluster_schedulable_instances{side="prefill"} 2.0\n'
           "garbage line without number x\n")
    v = bench._parse_prom(txt)
    assert v["generated_tokens_total"] == 123.0
    assert v['cluster_schedulable_instances{side="prefill"}'] == 2.0


def test_completions_exact_k():
    import asyncio

    async def run():
        c = bench.Completions()
        c.on_complete(1.0, 0.1, 16, "ramp")       # ramp never counts
        c.arm(2)
        c.on_complete(2.0, 0.1, 16, "poisson")
        assert not c.event.is_set()
        c.on_complete(3.0, 0.2, 16, "poisson", dur=1.5)
        assert c.event.is_set() and c.t1 == 3.0
        assert c.durations == [1.5]
        c.arm(0)                                   # k=0 fires immediately
        assert c.event.is_set()
    asyncio.run(run())


def test_summarize_window_gating_and_cap():
    recs = [
        (10.0, 0.1, 100, "poisson", 0.002),   # good
        (11.0, 2.0, 100, "poisson", 0.002),   # misses the 1s TTFT SLO
        (12.0, 0.2, 100, "poisson", 0.004),   # good
        (12.5, 0.2, 100, "ramp", 0.002),      # ramp never counted
        (99.0, 0.1, 100, "poisson", 0.002),   # outside window
    ]
    sm = bench.summarize_window(recs, t0=9.0, t1=13.0, slo_s=1.0,
                                rate=1.0, output_len=100)
    assert sm["n"] == 3
    assert sm["window_raw"] == 300 / 4.0
    assert sm["value"] == 200 / 4.0          # SLO-gated
    assert sm["offered"] == 100.0
    # over-offered window is capped
    burst = [(10.0 + i * 0.01, 0.1, 100, "poisson", 0.001)
             for i in range(50)]
    sm2 = bench.summarize_window(burst, 10.0, 10.5, 1.0, rate=2.0,
                                 output_len=100)
    assert sm2["value"] == 200.0             # capped at offered
    assert sm2["window_raw"] > 200.0
