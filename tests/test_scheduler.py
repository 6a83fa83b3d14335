"""Router filter-chain tests.

Mirrors the reference's scheduler test scenarios
(pkg/ext-proc/scheduling/filter_test.go): whole-tree behavior on 3-engine
fixtures (critical pick, sheddable accept, sheddable drop) plus individual
band filters and predicates.
"""

import random

import pytest

from llm_instance_gateway_amd.router.filters import (
    Filter,
    FilterError,
    DropRequest,
    build_default_tree,
    least_queuing_filter_func,
    least_kv_cache_filter_func,
    low_lora_cost_predicate,
    lora_affinity_predicate,
    can_accept_new_lora_predicate,
    to_filter_func,
)
from llm_instance_gateway_amd.router.scheduler import Scheduler, ResourceExhausted
from llm_instance_gateway_amd.router.types import (
    Engine,
    EngineMetrics,
    EngineState,
    LLMRequest,
)


def mk(name, waiting=0, kv=0.0, active=(), max_active=2, running=0):
    return EngineState(
        engine=Engine(name=name),
        metrics=EngineMetrics(
            active_models={a: 1 for a in active},
            max_active_models=max_active,
            waiting_queue_size=waiting,
            running_queue_size=running,
            kv_cache_usage_percent=kv,
        ),
    )


def three_engines_healthy():
    return [
        mk("e1", waiting=0, kv=0.2, active=("foo", "bar")),
        mk("e2", waiting=3, kv=0.1, active=("foo", "critical")),
        mk("e3", waiting=10, kv=0.2, active=("foo",)),
    ]


class TestDefaultTree:
    def test_critical_request_picks_affinity_low_kv(self):
        # e2: low queue band fails (band = 0+10//3=3 → e1,e2 in band? min=0,
        # band=0+(10-0)//3=3 → waiting<=3 → e1,e2) … but affinity to
        # "critical" narrows to e2 before the band filters run.
        tree = build_default_tree()
        req = LLMRequest(model="critical", resolved_target_model="critical", critical=True)
        out = tree.apply(req, three_engines_healthy())
        assert [e.engine.name for e in out] == ["e2"]

    def test_sheddable_request_accepted(self):
        tree = build_default_tree()
        req = LLMRequest(model="sheddable", resolved_target_model="sheddable", critical=False)
        out = tree.apply(req, three_engines_healthy())
        assert [e.engine.name for e in out] == ["e1"]

    def test_sheddable_request_dropped(self):
        tree = build_default_tree()
        req = LLMRequest(model="sheddable", resolved_target_model="sheddable", critical=False)
        engines = [
            mk("e1", waiting=10, kv=0.9, active=("foo", "bar")),
            mk("e2", waiting=3, kv=0.85, active=("foo", "critical")),
            mk("e3", waiting=10, kv=0.85, active=("foo",)),
        ]
        with pytest.raises(DropRequest):
            tree.apply(req, engines)

    def test_error_filter_without_successor_raises(self):
        def boom(req, engines):
            raise FilterError("filter error")

        f = Filter(name="err", func=boom)
        with pytest.raises(FilterError):
            f.apply(LLMRequest(model="m"), [])

    def test_failure_passes_original_set_to_next(self):
        # A failing predicate must hand the ORIGINAL engines to
        # next_on_failure (reference filter.go:66-72).
        seen = {}

        def record(req, engines):
            seen["n"] = len(engines)
            return engines

        f = Filter(
            name="never",
            func=to_filter_func(lambda r, e: False),
            next_on_failure=Filter(name="rec", func=record),
        )
        out = f.apply(LLMRequest(model="m"), three_engines_healthy())
        assert seen["n"] == 3 and len(out) == 3


class TestBandFilters:
    def test_least_queuing_band_integer_division(self):
        # min=0, max=10, band = 0 + (10-0)//3 = 3 → keep waiting<=3.
        out = least_queuing_filter_func(LLMRequest(model="m"), three_engines_healthy())
        assert sorted(e.engine.name for e in out) == ["e1", "e2"]

    def test_least_queuing_all_equal(self):
        engines = [mk("a", waiting=5), mk("b", waiting=5)]
        out = least_queuing_filter_func(LLMRequest(model="m"), engines)
        assert len(out) == 2

    def test_least_kv_band_float_division(self):
        engines = [mk("a", kv=0.0), mk("b", kv=0.3), mk("c", kv=0.9)]
        # band = 0 + 0.9/3 = 0.3 → a, b
        out = least_kv_cache_filter_func(LLMRequest(model="m"), engines)
        assert sorted(e.engine.name for e in out) == ["a", "b"]


class TestPredicates:
    def test_low_lora_cost(self):
        req = LLMRequest(model="x", resolved_target_model="x")
        assert low_lora_cost_predicate(req, mk("a", active=("x",), max_active=1))
        assert low_lora_cost_predicate(req, mk("a", active=("y",), max_active=2))
        assert not low_lora_cost_predicate(req, mk("a", active=("y",), max_active=1))

    def test_affinity(self):
        req = LLMRequest(model="x", resolved_target_model="x")
        assert lora_affinity_predicate(req, mk("a", active=("x",)))
        assert not lora_affinity_predicate(req, mk("a", active=("y",)))

    def test_can_accept(self):
        req = LLMRequest(model="x")
        assert can_accept_new_lora_predicate(req, mk("a", active=("y",), max_active=2))
        assert not can_accept_new_lora_predicate(req, mk("a", active=("y", "z"), max_active=2))


class FixedProvider:
    def __init__(self, states):
        self.states = states

    def all_engine_metrics(self):
        return self.states


class TestScheduler:
    def test_schedule_returns_engine(self):
        sched = Scheduler(FixedProvider(three_engines_healthy()), rng=random.Random(1))
        req = LLMRequest(model="critical", resolved_target_model="critical", critical=True)
        assert sched.schedule(req).name == "e2"

    def test_schedule_shed_maps_to_resource_exhausted(self):
        engines = [mk("e1", waiting=10, kv=0.95)]
        sched = Scheduler(FixedProvider(engines))
        with pytest.raises(ResourceExhausted):
            sched.schedule(LLMRequest(model="m", critical=False))


def test_custom_thresholds_change_shed_behavior():
    """The reference's compile-time consts are config knobs here
    (scheduler.go:15-24 'make configurable' TODO)."""
    from llm_instance_gateway_amd.router.filters import DropRequest, build_default_tree
    from llm_instance_gateway_amd.router.types import EngineMetrics, EngineState, Engine, LLMRequest

    states = [
        EngineState(
            engine=Engine("e0", 0),
            metrics=EngineMetrics(waiting_queue_size=3, kv_cache_usage_percent=0.5),
        )
    ]
    req = LLMRequest(model="m", critical=False)
    # default thresholds (queue<=5, kv<=0.8): accepted
    assert build_default_tree().apply(req, states)
    # tightened queue threshold: dropped
    import pytest as _pytest

    with _pytest.raises(DropRequest):
        build_default_tree(queue_threshold_critical=2).apply(req, states)
    # tightened KV threshold: dropped
    with _pytest.raises(DropRequest):
        build_default_tree(kv_cache_threshold=0.4).apply(req, states)


def test_pool_ref_scoping_and_fallback():
    """pool_ref scopes candidates to tagged + untagged engines; when NOTHING
    matches, fall back to the full set rather than stranding the request."""
    tagged = [
        EngineState(engine=Engine("p1-a", pool="pool-1"), metrics=EngineMetrics()),
        EngineState(engine=Engine("p2-a", pool="pool-2"), metrics=EngineMetrics()),
    ]
    sched = Scheduler(FixedProvider(tagged), rng=random.Random(0))
    for _ in range(5):
        assert sched.schedule(
            LLMRequest(model="m", critical=True, pool_ref="pool-2")
        ).name == "p2-a"
    # no engine carries this pool tag -> full-set fallback
    eng = sched.schedule(
        LLMRequest(model="m", critical=True, pool_ref="nonexistent-pool")
    )
    assert eng.name in {"p1-a", "p2-a"}
