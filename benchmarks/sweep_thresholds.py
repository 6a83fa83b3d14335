#!/usr/bin/env python3
"""Filter-tree threshold sweep over the simulated fleet (CPU only).

Answers SURVEY.md §7 hard-part #5: the reference's scheduler thresholds
(queueing_threshold_lora=50, queue_threshold_critical=5,
kv_cache_threshold=0.8 — scheduler.go:15-24) were tuned for metrics that
were up to 50 ms stale; this framework routes on ZERO-staleness
shared-memory gauges.  The sweep runs the PRODUCTION tree
(router/filters.py via sim/tree_router.py) against the discrete-event
fleet and reports goodput / SLO attainment / drops per setting.

    python benchmarks/sweep_thresholds.py --csv profiles/r02_threshold_sweep.csv
"""

from __future__ import annotations

import argparse
import csv
import json
import random
import sys
from pathlib import Path
from typing import Dict, List, Optional

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from llm_instance_gateway_amd.sim.core import SimEngine, SimEnv
from llm_instance_gateway_amd.sim.main import generate_workload, percentile
from llm_instance_gateway_amd.sim.tree_router import TreeSimRouter


def run_once(
    rate: float,
    lora_q: int,
    crit_q: int,
    kv_thr: float = 0.8,
    num_servers: int = 6,
    duration: float = 100.0,
    drain: float = 100.0,
    num_loras: int = 8,
    seed: int = 0,
) -> Dict:
    rng = random.Random(seed)
    env = SimEnv()
    engines = [SimEngine(env, i) for i in range(num_servers)]
    router = TreeSimRouter(
        env, engines,
        kv_cache_threshold=kv_thr,
        queue_threshold_critical=crit_q,
        queueing_threshold_lora=lora_q,
        rng=random.Random(seed + 1),
    )
    reqs = generate_workload(env, router, rate, rate, duration, rng, num_loras)
    env.run(until=duration + drain)
    done = [r for r in reqs if r.end_decode_time is not None]
    good = [r for r in done if r.within_slo]
    span = max((r.end_decode_time for r in done), default=duration)
    lats = [r.e2e_latency for r in done]
    return {
        "rate_per_tier": rate,
        "queueing_threshold_lora": lora_q,
        "queue_threshold_critical": crit_q,
        "kv_cache_threshold": kv_thr,
        "offered": len(reqs),
        "completed": len(done),
        "dropped": len(router.dropped),
        "goodput_rps": round(len(good) / span, 3) if span else 0.0,
        "pct_under_slo": round(100.0 * len(good) / len(done), 1) if done else 0.0,
        "latency_p99": round(percentile(lats, 99), 3) if lats else None,
        "recomputes": sum(r.recompute_count for r in reqs),
    }


def main(argv: Optional[List[str]] = None) -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--rates", type=float, nargs="+", default=[5, 10, 20])
    ap.add_argument("--lora-thresholds", type=int, nargs="+",
                    default=[2, 5, 10, 25, 50, 100])
    ap.add_argument("--crit-thresholds", type=int, nargs="+", default=[3, 5, 10])
    ap.add_argument("--num-servers", type=int, default=6)
    ap.add_argument("--duration", type=float, default=100.0)
    ap.add_argument("--num-loras", type=int, default=8)
    ap.add_argument("--seeds", type=int, default=3, help="runs averaged per point")
    ap.add_argument("--csv", default="")
    args = ap.parse_args(argv)

    def point(rate, lq, cq) -> Dict:
        runs = [
            run_once(rate, lq, cq, num_servers=args.num_servers,
                     duration=args.duration, num_loras=args.num_loras,
                     seed=s)
            for s in range(args.seeds)
        ]
        out = dict(runs[0])
        for k in ("offered", "completed", "dropped", "recomputes"):
            out[k] = sum(r[k] for r in runs)
        for k in ("goodput_rps", "pct_under_slo", "latency_p99"):
            vals = [r[k] for r in runs if r[k] is not None]
            out[k] = round(sum(vals) / len(vals), 3) if vals else None
        out["seeds"] = args.seeds
        return out

    rows: List[Dict] = []
    # axis 1: queueing_threshold_lora (crit fixed at reference 5)
    for rate in args.rates:
        for lq in args.lora_thresholds:
            rows.append(point(rate, lq, 5))
    # axis 2: queue_threshold_critical (lora fixed at reference 50)
    for rate in args.rates:
        for cq in args.crit_thresholds:
            if cq == 5:
                continue  # already covered above at lq=50
            rows.append(point(rate, 50, cq))

    for r in rows:
        print(json.dumps(r))
    if args.csv:
        with open(args.csv, "w", newline="") as f:
            w = csv.DictWriter(f, fieldnames=list(rows[0]))
            w.writeheader()
            w.writerows(rows)
        print(f"# wrote {args.csv}", file=sys.stderr)
    return 0


if __name__ == "__main__":
    sys.exit(main())
