"""Production filter tree over the simulated fleet (sim/tree_router.py).

Pins the harness behind profiles/r02_threshold_sweep.md: the real
router/filters.py tree routes sim traffic, sheds sheddable overload, and
keeps LoRA affinity when queues are shallow.
"""

import random

from llm_instance_gateway_amd.sim.core import SimEngine, SimEnv
from llm_instance_gateway_amd.sim.main import generate_workload
from llm_instance_gateway_amd.sim.tree_router import TreeSimRouter


def _run(rate: float, duration: float = 60.0, rate_high: float = None,
         **router_kw):
    rng = random.Random(0)
    env = SimEnv()
    engines = [SimEngine(env, i) for i in range(4)]
    router = TreeSimRouter(env, engines, rng=random.Random(1), **router_kw)
    hi = rate if rate_high is None else rate_high
    reqs = generate_workload(env, router, rate, hi, duration, rng, num_loras=4)
    env.run(until=duration + 60.0)
    done = [r for r in reqs if r.end_decode_time is not None]
    return router, reqs, done


def test_tree_routes_and_completes_under_light_load():
    router, reqs, done = _run(rate=3.0)
    assert len(done) == len(reqs)
    assert router.dropped == []
    assert router.routed == len(reqs)


def test_tree_sheds_sheddable_overload_only():
    router, reqs, done = _run(rate=30.0)
    assert router.dropped, "overload must shed"
    # only sheddable (non-critical) traffic is ever dropped
    assert all(not r.critical for r in router.dropped)
    # critical traffic still completes (queues drain in the 60 s tail)
    crit = [r for r in reqs if r.critical]
    crit_done = [r for r in crit if r.end_decode_time is not None]
    assert len(crit_done) >= 0.9 * len(crit)


def test_affinity_concentrates_adapters_when_queues_shallow():
    """CRITICAL traffic takes the affinity branch (low-queueing path): with
    shallow queues each adapter should live on FEW engines.  (Sheddable
    traffic deliberately uses the spread-tolerant lowLoRACost branch —
    reference scheduler.go:46-57 — so it is excluded here.)"""
    router, reqs, done = _run(rate=4.0, rate_high=0.0)
    resident = sum(len(e.lora_loaded) for e in router.engines)
    # 4 adapters, 4 engines: no-affinity worst case is 16 pairs
    assert resident <= 8, resident
