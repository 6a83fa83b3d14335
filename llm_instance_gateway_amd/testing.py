"""Test fakes: the seams every hermetic test plugs into.

Reference `pkg/ext-proc/backend/fake.go` (FakePodMetricsClient /
FakeDataStore) and `pkg/ext-proc/test/utils.go` (StartExtProc wiring, fake
request/pod generators).  These are importable product code, not test-local
helpers, for the same reason the reference ships them in the main tree: the
load benchmark (`benchmarks/bench_router.py`) and downstream users fake the
same seams.
"""

from __future__ import annotations

import random
from typing import Dict, List, Optional

from .api.types import InferenceModel, TargetModel
from .router.datastore import Datastore
from .router.scheduler import Scheduler
from .router.types import Engine, EngineMetrics, EngineState


class FakeMetricsSource:
    """Reference FakePodMetricsClient (fake.go:10-21): canned responses or
    errors per engine name."""

    def __init__(
        self,
        res: Optional[Dict[str, EngineMetrics]] = None,
        err: Optional[Dict[str, Exception]] = None,
    ):
        self.res = dict(res or {})
        self.err = dict(err or {})

    def fetch_metrics(self, engine: Engine, existing: EngineMetrics) -> EngineMetrics:
        if engine.name in self.err:
            raise self.err[engine.name]
        return self.res.get(engine.name, existing)


class StaticProvider:
    """EngineMetricsProvider over a fixed list of states."""

    def __init__(self, states: List[EngineState]):
        self.states = list(states)

    def all_engine_metrics(self) -> List[EngineState]:
        return list(self.states)


def fake_engine(i: int) -> Engine:
    """Reference FakePod (test/utils.go:73-80): engine i at a stable name."""
    return Engine(name=f"engine-{i}", gpu_id=i)


def fake_engine_states(
    n: int, adapters_per_engine: int = 0, seed: int = 0
) -> List[EngineState]:
    """Reference fakePods/fakeMetrics (benchmark.go:74-110)."""
    rng = random.Random(seed)
    out = []
    for i in range(n):
        out.append(
            EngineState(
                engine=fake_engine(i),
                metrics=EngineMetrics(
                    active_models={
                        f"adapter-{i}-{j}": 1 for j in range(adapters_per_engine)
                    },
                    max_active_models=max(4, adapters_per_engine + 2),
                    running_queue_size=rng.randrange(0, 8),
                    waiting_queue_size=rng.randrange(0, 10),
                    kv_cache_usage_percent=rng.random() * 0.7,
                    kv_cache_max_token_capacity=44448,
                ),
            )
        )
    return out


def make_datastore(n_models: int = 5, pool_ref: str = "") -> Datastore:
    """Reference fakeModels (benchmark.go:82-90)."""
    ds = Datastore()
    for m in range(n_models):
        ds.store_model(
            InferenceModel(
                model_name=f"model-{m}",
                target_models=[TargetModel(name=f"model-{m}", weight=100)],
                pool_ref=pool_ref,
            )
        )
    return ds


def make_scheduler(states: List[EngineState], seed: int = 0) -> Scheduler:
    """A router over fixed fake state (hermetic harness core,
    test/utils.go:21-51)."""
    return Scheduler(StaticProvider(states), rng=random.Random(seed))
