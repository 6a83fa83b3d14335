"""Provider tests (reference backend/provider_test.go): init populates the
map; per-engine fetch errors keep last-known metrics; engine set sync."""

import pytest

from llm_instance_gateway_amd.router.datastore import Datastore
from llm_instance_gateway_amd.router.provider import Provider
from llm_instance_gateway_amd.router.types import Engine, EngineMetrics


class FakeSource:
    """Reference FakePodMetricsClient (backend/fake.go:10-21)."""

    def __init__(self, res=None, err=None):
        self.res = res or {}
        self.err = err or {}

    def fetch_metrics(self, engine, existing):
        if engine.name in self.err:
            raise self.err[engine.name]
        return self.res[engine.name]


def test_init_populates_metrics():
    ds = Datastore(engines=[Engine("e0", 0), Engine("e1", 1)])
    src = FakeSource(
        res={
            "e0": EngineMetrics(waiting_queue_size=2, kv_cache_usage_percent=0.5),
            "e1": EngineMetrics(waiting_queue_size=7),
        }
    )
    p = Provider(src, ds)
    p.init(start_loops=False)
    states = {s.engine.name: s for s in p.all_engine_metrics()}
    assert states["e0"].metrics.waiting_queue_size == 2
    assert states["e0"].metrics.kv_cache_usage_percent == 0.5
    assert states["e1"].metrics.waiting_queue_size == 7


def test_fetch_error_keeps_last_known_metrics():
    ds = Datastore(engines=[Engine("e0", 0)])
    src = FakeSource(res={"e0": EngineMetrics(waiting_queue_size=4)})
    p = Provider(src, ds)
    p.init(start_loops=False)
    # Now the source starts failing: metrics must stay at last-known values
    # (reference provider.go:150-159, provider_test.go:61-87).
    src.err = {"e0": RuntimeError("connection refused")}
    errs = p.refresh_metrics_once()
    assert len(errs) == 1
    assert p.all_engine_metrics()[0].metrics.waiting_queue_size == 4


def test_engine_set_sync_add_remove():
    ds = Datastore(engines=[Engine("e0", 0)])
    p = Provider(None, ds)
    p.init(start_loops=False)
    assert [s.engine.name for s in p.all_engine_metrics()] == ["e0"]
    ds.add_engine(Engine("e1", 1))
    ds.remove_engine("e0")
    p.refresh_engines_once()
    assert [s.engine.name for s in p.all_engine_metrics()] == ["e1"]


def test_direct_mode_update():
    ds = Datastore(engines=[Engine("e0", 0)])
    p = Provider(None, ds)
    p.init(start_loops=False)
    p.update_engine_metrics(Engine("e0", 0), EngineMetrics(waiting_queue_size=9))
    assert p.all_engine_metrics()[0].metrics.waiting_queue_size == 9


def test_snapshot_isolated_from_updates():
    """Snapshots are shared read-only references; writers REPLACE the state
    object (atomic swap, provider.py all_engine_metrics), so a snapshot taken
    before an update is unaffected by it."""
    ds = Datastore(engines=[Engine("e0", 0)])
    p = Provider(None, ds)
    p.init(start_loops=False)
    snap = p.all_engine_metrics()[0]
    p.update_engine_metrics(Engine("e0", 0), EngineMetrics(waiting_queue_size=9))
    assert snap.metrics.waiting_queue_size == 0  # old snapshot untouched
    assert p.all_engine_metrics()[0].metrics.waiting_queue_size == 9
    # explicit clone() still available for callers that must mutate
    c = snap.clone()
    c.metrics.waiting_queue_size = 123
    assert snap.metrics.waiting_queue_size == 0


def test_refresh_loops_update_live():
    """The reference's two daemon refresh loops (provider.go:60-101): with a
    mutating source, the snapshot follows within a few intervals."""
    import time as _t

    from llm_instance_gateway_amd.testing import FakeMetricsSource

    ds = Datastore(engines=[Engine("e0", 0)])
    src = FakeMetricsSource(res={"e0": EngineMetrics(waiting_queue_size=1)})
    p = Provider(src, ds)
    p.init(refresh_engines_interval_s=0.05, refresh_metrics_interval_s=0.02)
    try:
        deadline = _t.monotonic() + 5
        while _t.monotonic() < deadline:
            states = p.all_engine_metrics()
            if states and states[0].metrics.waiting_queue_size == 1:
                break
            _t.sleep(0.02)
        assert states[0].metrics.waiting_queue_size == 1
        # source moves; snapshot follows
        src.res["e0"] = EngineMetrics(waiting_queue_size=9)
        deadline = _t.monotonic() + 5
        while _t.monotonic() < deadline:
            if p.all_engine_metrics()[0].metrics.waiting_queue_size == 9:
                break
            _t.sleep(0.02)
        assert p.all_engine_metrics()[0].metrics.waiting_queue_size == 9
        # engine-set loop follows the datastore
        ds.add_engine(Engine("e1", 1))
        deadline = _t.monotonic() + 5
        while _t.monotonic() < deadline:
            if len(p.all_engine_metrics()) == 2:
                break
            _t.sleep(0.02)
        assert len(p.all_engine_metrics()) == 2
    finally:
        p.stop()
