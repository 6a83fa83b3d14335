"""Prometheus exposition/parse tests.

Mirrors reference `backend/vllm/metrics_test.go` (family -> PodMetrics incl.
latest-timestamp LoRA series selection, metrics_test.go:22-80) plus exposition
round-trip, which the reference cannot test because it exports nothing.
"""

import pytest

from llm_instance_gateway_amd.metrics.prometheus import (
    Counter,
    GatewayMetrics,
    Histogram,
    parse_engine_metrics,
    render_engine_metrics,
)
from llm_instance_gateway_amd.router.types import EngineMetrics


def test_render_parse_roundtrip():
    m = EngineMetrics(
        active_models={"lora-a": 1, "lora-b": 1},
        max_active_models=4,
        running_queue_size=7,
        waiting_queue_size=3,
        kv_cache_usage_percent=0.625,
        kv_cache_max_token_capacity=44448,
    )
    text = render_engine_metrics("engine-0", m, now=123.0)
    out = parse_engine_metrics(text, EngineMetrics())
    assert out.running_queue_size == 7
    assert out.waiting_queue_size == 3
    assert out.kv_cache_usage_percent == pytest.approx(0.625)
    assert sorted(out.active_models) == ["lora-a", "lora-b"]
    assert out.max_active_models == 4


def test_parse_keeps_existing_on_missing_families():
    """Reference provider.go:150-159: failed/partial scrape keeps last-known."""
    existing = EngineMetrics(
        active_models={"x": 1}, max_active_models=2,
        running_queue_size=5, waiting_queue_size=9,
        kv_cache_usage_percent=0.5,
    )
    out = parse_engine_metrics("vllm:num_requests_running 1\n", existing)
    assert out.running_queue_size == 1
    assert out.waiting_queue_size == 9  # retained
    assert out.kv_cache_usage_percent == 0.5  # retained
    assert out.active_models == {"x": 1}  # retained


def test_lora_latest_wins_by_gauge_value():
    """Reference getLatestLoraMetric (metrics.go:135-150): among multiple
    vllm:lora_requests_info series pick greatest value (timestamp)."""
    text = (
        'vllm:lora_requests_info{running_lora_adapters="old-a,old-b",max_lora="2"} 100\n'
        'vllm:lora_requests_info{running_lora_adapters="new-a",max_lora="8"} 200\n'
    )
    out = parse_engine_metrics(text, EngineMetrics())
    assert out.active_models == {"new-a": 1}
    assert out.max_active_models == 8


def test_lora_empty_adapter_set():
    text = 'vllm:lora_requests_info{running_lora_adapters="",max_lora="4"} 5\n'
    out = parse_engine_metrics(text, EngineMetrics(active_models={"stale": 1}))
    assert out.active_models == {}
    assert out.max_active_models == 4


def test_histogram_buckets_and_render():
    h = Histogram("t_seconds", "test", buckets=(0.1, 1.0, 10.0))
    for v in (0.05, 0.5, 5.0, 50.0):
        h.observe(v)
    text = h.render()
    assert 't_seconds_bucket{le="0.1"} 1' in text
    assert 't_seconds_bucket{le="1.0"} 2' in text
    assert 't_seconds_bucket{le="10.0"} 3' in text
    assert 't_seconds_bucket{le="+Inf"} 4' in text
    assert "t_seconds_count 4" in text


def test_counter_labels():
    c = Counter("reqs_total", "test")
    c.inc(model="m1", outcome="ok")
    c.inc(model="m1", outcome="ok")
    c.inc(model="m2", outcome="shed")
    assert c.value(model="m1", outcome="ok") == 2
    text = c.render()
    assert 'reqs_total{model="m1",outcome="ok"} 2' in text
    assert 'reqs_total{model="m2",outcome="shed"} 1' in text


def test_gateway_metrics_render_all():
    gm = GatewayMetrics()
    gm.requests_total.inc(model="m", outcome="ok")
    gm.ttft.observe(0.12)
    text = gm.render()
    for name in (
        "gateway_requests_total",
        "gateway_requests_shed_total",
        "gateway_ttft_seconds",
        "gateway_tpot_seconds",
        "gateway_e2e_latency_seconds",
    ):
        assert name in text


def test_render_parse_roundtrip_property():
    """Property: render -> parse is lossless for any gauge values and
    adapter names (incl. escaping-sensitive characters)."""
    import hypothesis.strategies as st
    from hypothesis import given, settings

    name_st = st.text(
        alphabet=st.characters(
            whitelist_categories=("L", "N"), whitelist_characters="-_./\\\""
        ),
        min_size=1,
        max_size=20,
    ).filter(lambda s: "," not in s)

    @settings(max_examples=80, deadline=None)
    @given(
        adapters=st.lists(name_st, max_size=4, unique=True),
        running=st.integers(0, 10**6),
        waiting=st.integers(0, 10**6),
        kv=st.floats(0, 1, allow_nan=False),
        max_active=st.integers(0, 64),
    )
    def check(adapters, running, waiting, kv, max_active):
        m = EngineMetrics(
            active_models={a: 1 for a in adapters},
            max_active_models=max_active,
            running_queue_size=running,
            waiting_queue_size=waiting,
            kv_cache_usage_percent=kv,
        )
        out = parse_engine_metrics(render_engine_metrics("e", m, now=1.0), EngineMetrics())
        assert out.running_queue_size == running
        assert out.waiting_queue_size == waiting
        assert abs(out.kv_cache_usage_percent - kv) < 1e-5
        assert sorted(out.active_models) == sorted(adapters)
        assert out.max_active_models == max_active

    check()


def test_preemption_counter_roundtrip():
    m = EngineMetrics(num_preemptions_total=17)
    out = parse_engine_metrics(render_engine_metrics("e", m, now=1.0), EngineMetrics())
    assert out.num_preemptions_total == 17
