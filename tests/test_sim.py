"""Simulation tests (BASELINE config #1: CPU-only plumbing harness).

Covers the reference simulation's core behaviors
(`simulations/llm_ig_simulation/src/`): the calibrated latency model,
admission caps, preemption-to-recompute, LoRA KV-budget accounting, policy
A/B (metric-aware routing beats random at overload), and determinism.
"""

import pytest

from llm_instance_gateway_amd.sim import (
    SimEngine,
    SimEnv,
    SimRequest,
    decode_delay,
    prefill_delay,
)
from llm_instance_gateway_amd.sim.core import (
    MAX_NUM_BATCH_TOKENS,
    MAX_NUM_SEQ,
    MAX_NUM_TOKENS_ALLOWED,
)
from llm_instance_gateway_amd.sim.main import run_once


def test_latency_model_constants():
    """Reference constants.py:1-9."""
    assert prefill_delay(0) == pytest.approx(0.04)  # floor
    assert prefill_delay(1000) == pytest.approx(1000 * 6.769375513e-5 + 0.01969)
    assert decode_delay(0, 0) == pytest.approx(0.014)
    assert decode_delay(10000, 32) == pytest.approx(
        10000 * 5.353485087e-7 + 0.014 + 32 * 1.026494433e-4
    )


def test_kv_capacity_constant():
    """2810 blocks x 16 tokens - 512 (constants.py:13-15)."""
    assert MAX_NUM_TOKENS_ALLOWED == 2810 * 16 - 512
    assert MAX_NUM_BATCH_TOKENS == 512
    assert MAX_NUM_SEQ == 256


def _mk_req(i, t, ins, outs, slo=0.5, lora=None):
    return SimRequest(
        id=i, arrival_time=t, input_size=ins, output_size=outs,
        target_latency_per_token=slo, lora=lora,
    )


def test_engine_completes_requests_and_times():
    env = SimEnv()
    e = SimEngine(env, 0)
    r = _mk_req(0, 0.0, 100, 10)
    e.submit(r)
    env.run(until=10.0)
    assert r.end_decode_time is not None
    # first token at end of the prefill batch; 9 decode steps after
    assert r.ttft == pytest.approx(prefill_delay(100))
    assert r.e2e_latency > r.ttft
    assert e.decoded == [r]
    assert e.tokens_in_decode() == 0


def test_engine_prefill_batch_cap():
    """More than 512 prompt tokens never prefill in one batch."""
    env = SimEnv()
    e = SimEngine(env, 0)
    for i in range(4):
        e.submit(_mk_req(i, 0.0, 200, 4))  # 4x200 > 512 -> at least 2 batches
    env.run(until=0.041)  # one prefill delay (floor 0.04)
    started = [r for r in (e.decode + e.decoded) if r.start_prefill_time is not None]
    assert 0 < len(started) <= 2  # 3rd would exceed 512
    env.run(until=10.0)
    assert len(e.decoded) == 4


def test_lora_load_shrinks_kv_budget():
    """continous_batching.py:94-97 + LORA_DICT."""
    env = SimEnv()
    e = SimEngine(env, 0, lora_cost={"tweet": 1600})
    before = e.max_num_tokens_allowed
    e.submit(_mk_req(0, 0.0, 50, 4, lora="tweet"))
    env.run(until=5.0)
    assert e.max_num_tokens_allowed == before - 1600
    assert "tweet" in e.lora_loaded
    # second use of the same adapter costs nothing
    e.submit(_mk_req(1, 0.0, 50, 4, lora="tweet"))
    env.run(until=10.0)
    assert e.max_num_tokens_allowed == before - 1600


def test_recompute_under_kv_pressure():
    """Preemption-by-recompute fires above 90% KV
    (continous_batching.py:117-131); preempted requests still finish."""
    env = SimEnv()
    e = SimEngine(env, 0)
    # shrink the pool so pressure is reachable quickly
    e.max_num_tokens_allowed = 2000
    reqs = [_mk_req(i, 0.0, 400, 600) for i in range(4)]
    for r in reqs:
        e.submit(r)
    env.run(until=300.0)
    assert sum(r.recompute_count for r in reqs) > 0
    assert len(e.decoded) == 4


def test_run_once_deterministic():
    a = run_once("least", 10, 10, num_servers=3, duration=10, seed=42)
    b = run_once("least", 10, 10, num_servers=3, duration=10, seed=42)
    assert a == b
    c = run_once("least", 10, 10, num_servers=3, duration=10, seed=43)
    assert c["completed"] != a["completed"] or c["latency_mean"] != a["latency_mean"]


def test_metric_aware_routing_beats_random_at_overload():
    """The reference's headline sim finding (loadbalancer.py policy A/B):
    KV/latency-aware placement outperforms random under pressure."""
    kw = dict(num_servers=6, duration=30, seed=0, num_loras=8)
    random_g = run_once("random", 35, 35, **kw)["goodput_rps"]
    least_g = run_once("least", 35, 35, **kw)["goodput_rps"]
    lat_g = run_once("leastlatency", 35, 35, **kw)["goodput_rps"]
    assert least_g > random_g * 1.02, (least_g, random_g)
    assert lat_g > random_g * 1.02, (lat_g, random_g)


def test_all_requests_complete_at_low_load():
    r = run_once("smart", 2, 2, num_servers=3, duration=20, seed=1)
    assert r["completed"] == r["offered"]
    assert r["pct_under_slo"] > 90.0  # uncontended: both tiers meet SLO


def test_smart_policy_wins_low_tier_under_congestion():
    """The smart policy's tier segregation (pack loose-SLO traffic, spread
    tight-SLO traffic over SLO-qualifying engines, affinity escape hatch)
    must beat random/least on LOW-tier SLO attainment in the congestion
    regime — the reference sim's design goal for `smart`."""
    import random as _r

    from llm_instance_gateway_amd.sim.core import SimEngine, SimEnv
    from llm_instance_gateway_amd.sim.main import SLO_TPOT_LOW, generate_workload
    from llm_instance_gateway_amd.sim.router import SimRouter

    def lo_good(policy):
        env = SimEnv()
        engines = [SimEngine(env, i) for i in range(6)]
        router = SimRouter(env, engines, routing_type=policy, rng=_r.Random(1))
        reqs = generate_workload(env, router, 20, 20, 30, _r.Random(0), 8)
        env.run(until=130)
        rs = [r for r in reqs if r.target_latency_per_token == SLO_TPOT_LOW]
        return sum(r.within_slo for r in rs if r.end_decode_time is not None)

    smart, least, rnd = lo_good("smart"), lo_good("least"), lo_good("random")
    assert smart > least * 1.5, (smart, least)
    assert smart > rnd * 1.3, (smart, rnd)
