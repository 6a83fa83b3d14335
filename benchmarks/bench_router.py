#!/usr/bin/env python3
"""Router micro-benchmark: the ghz harness analogue.

Reference `pkg/ext-proc/test/benchmark/benchmark.go:22-62`: a local ext-proc
with 200 fake pods x 5 adapters each, 100k `Process` gRPC calls, ghz summary.
Here the gRPC hop is gone (the ext-proc collapsed into the in-process
router), so the benchmark measures the true hot path: model resolution +
weighted draw + filter-tree schedule over N engines' live metric snapshots.

    python benchmarks/bench_router.py --engines 200 --models 1000 --requests 100000
"""

from __future__ import annotations

import argparse
import json
import random
import statistics
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from llm_instance_gateway_amd.api.types import InferenceModel, TargetModel, random_weighted_draw
from llm_instance_gateway_amd.router.datastore import Datastore
from llm_instance_gateway_amd.router.scheduler import ResourceExhausted, Scheduler
from llm_instance_gateway_amd.router.types import Engine, EngineMetrics, EngineState, LLMRequest


class FakeProvider:
    """200 engines with randomized-but-fixed metrics (reference fakeMetrics,
    benchmark.go:92-104)."""

    def __init__(self, n_engines: int, adapters_per_engine: int, seed: int = 0):
        rng = random.Random(seed)
        self.states = []
        for i in range(n_engines):
            active = {
                f"adapter-{i}-{j}": 1 for j in range(adapters_per_engine)
            }
            self.states.append(
                EngineState(
                    engine=Engine(name=f"engine-{i}", gpu_id=i % 8),
                    metrics=EngineMetrics(
                        active_models=active,
                        max_active_models=adapters_per_engine + 2,
                        running_queue_size=rng.randrange(0, 8),
                        waiting_queue_size=rng.randrange(0, 10),
                        kv_cache_usage_percent=rng.random() * 0.7,
                        kv_cache_max_token_capacity=44448,
                    ),
                )
            )

    def all_engine_metrics(self):
        # reference semantics: snapshot by atomic swap, no per-call copy
        # (router/provider.py all_engine_metrics)
        return list(self.states)


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--engines", type=int, default=200)
    ap.add_argument("--adapters-per-engine", type=int, default=5)
    ap.add_argument("--models", type=int, default=1000,
                    help="registered InferenceModels round-robined by requests")
    ap.add_argument("--requests", type=int, default=100_000)
    ap.add_argument("--sheddable-frac", type=float, default=0.2)
    ap.add_argument("--native", action="store_true",
                    help="use the C++ filter tree (state synced once: the "
                    "fleet is static in this harness)")
    args = ap.parse_args()

    provider = FakeProvider(args.engines, args.adapters_per_engine)
    ds = Datastore()
    rng = random.Random(1)
    for m in range(args.models):
        target = f"adapter-{rng.randrange(args.engines)}-{rng.randrange(args.adapters_per_engine)}"
        ds.store_model(
            InferenceModel(
                model_name=f"model-{m}",
                target_models=[TargetModel(name=target, weight=100)],
            )
        )
    if args.native:
        from llm_instance_gateway_amd.router.native import NativeScheduler, native_available

        assert native_available(), "_router extension not built"
        sched = NativeScheduler(seed=1)
        sched.sync_states(provider.all_engine_metrics())
    else:
        sched = Scheduler(provider)

    lat = []
    shed = 0
    t0 = time.perf_counter()
    for i in range(args.requests):
        name = f"model-{i % args.models}"
        im = ds.fetch_model(name)
        resolved = random_weighted_draw(im)
        req = LLMRequest(
            model=name,
            resolved_target_model=resolved,
            critical=(i % 100) >= int(args.sheddable_frac * 100),
        )
        s = time.perf_counter()
        try:
            sched.schedule(req)
        except ResourceExhausted:
            shed += 1
        lat.append(time.perf_counter() - s)
    elapsed = time.perf_counter() - t0

    lat_us = sorted(x * 1e6 for x in lat)
    out = {
        "router": "native" if args.native else "python",
        "requests": args.requests,
        "engines": args.engines,
        "models": args.models,
        "elapsed_s": round(elapsed, 3),
        "throughput_rps": round(args.requests / elapsed, 1),
        "shed": shed,
        "schedule_us_mean": round(statistics.fmean(lat_us), 1),
        "schedule_us_p50": round(lat_us[len(lat_us) // 2], 1),
        "schedule_us_p99": round(lat_us[int(len(lat_us) * 0.99)], 1),
    }
    print(json.dumps(out))
    return 0


if __name__ == "__main__":
    sys.exit(main())
