"""Sanity for the shared fakes (reference backend/fake.go seams)."""

from llm_instance_gateway_amd.router.datastore import Datastore
from llm_instance_gateway_amd.router.provider import Provider
from llm_instance_gateway_amd.router.types import Engine, EngineMetrics, LLMRequest
from llm_instance_gateway_amd.testing import (
    FakeMetricsSource,
    fake_engine,
    fake_engine_states,
    make_datastore,
    make_scheduler,
)


def test_fake_metrics_source_with_provider():
    """Reference provider_test.go:61-87: fetch error leaves zero-value
    metrics; good fetch lands."""
    ds = Datastore(engines=[fake_engine(0), fake_engine(1)])
    src = FakeMetricsSource(
        res={"engine-0": EngineMetrics(waiting_queue_size=4)},
        err={"engine-1": RuntimeError("connection refused")},
    )
    p = Provider(src, ds)
    p.refresh_engines_once()
    errs = p.refresh_metrics_once(timeout_s=2)
    assert len(errs) == 1
    by = {s.engine.name: s.metrics for s in p.all_engine_metrics()}
    assert by["engine-0"].waiting_queue_size == 4
    assert by["engine-1"].waiting_queue_size == 0


def test_fake_states_and_scheduler():
    states = fake_engine_states(8, adapters_per_engine=3, seed=1)
    sched = make_scheduler(states, seed=2)
    eng = sched.schedule(LLMRequest(model="m", critical=True))
    assert eng.name.startswith("engine-")
    ds = make_datastore(3)
    assert ds.fetch_model("model-2") is not None
    assert ds.fetch_model("model-9") is None
