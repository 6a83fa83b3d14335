"""Property-based filter-tree invariants (hypothesis): for ANY fleet state,
the default tree returns a non-empty subset of the input, never drops
Critical traffic, and sheds non-critical traffic exactly when no engine has
sheddable headroom (reference scheduler.go:74-90 semantics)."""

import hypothesis.strategies as st
from hypothesis import given, settings

from llm_instance_gateway_amd.router.filters import DropRequest, build_default_tree
from llm_instance_gateway_amd.router.types import Engine, EngineMetrics, EngineState, LLMRequest

metrics_st = st.builds(
    EngineMetrics,
    active_models=st.dictionaries(
        st.sampled_from(["a1", "a2", "a3"]), st.just(1), max_size=3
    ),
    max_active_models=st.integers(0, 4),
    running_queue_size=st.integers(0, 300),
    waiting_queue_size=st.integers(0, 300),
    kv_cache_usage_percent=st.floats(0.0, 1.0, allow_nan=False),
)

fleet_st = st.lists(metrics_st, min_size=1, max_size=8).map(
    lambda ms: [
        EngineState(engine=Engine(name=f"e{i}"), metrics=m)
        for i, m in enumerate(ms)
    ]
)

req_st = st.builds(
    LLMRequest,
    model=st.just("m"),
    resolved_target_model=st.sampled_from(["a1", "a2", "zz"]),
    critical=st.booleans(),
)


@settings(max_examples=200, deadline=None)
@given(states=fleet_st, req=req_st)
def test_tree_returns_nonempty_subset_or_sheds(states, req):
    tree = build_default_tree()
    try:
        out = tree.apply(req, states)
    except DropRequest:
        # only non-critical traffic is sheddable, and only when NO engine
        # has headroom (queue <= 5 and kv <= 0.8)
        assert not req.critical
        assert not any(
            s.metrics.waiting_queue_size <= 5
            and s.metrics.kv_cache_usage_percent <= 0.8
            for s in states
        )
        return
    assert out, "tree returned empty without raising"
    names = {s.engine.name for s in states}
    assert all(s.engine.name in names for s in out)
    if not req.critical:
        # accepted sheddable traffic implies headroom existed somewhere
        assert any(
            s.metrics.waiting_queue_size <= 5
            and s.metrics.kv_cache_usage_percent <= 0.8
            for s in states
        )


@settings(max_examples=100, deadline=None)
@given(states=fleet_st)
def test_critical_never_dropped(states):
    tree = build_default_tree()
    out = tree.apply(
        LLMRequest(model="m", resolved_target_model="a1", critical=True), states
    )
    assert out
