"""Simulation CLI: routing-policy A/B + QPS sweep (BASELINE config #1).

Reference `simulations/llm_ig_simulation/src/main.py:16-41,207-247,332-355`:
two priority tiers with TPOT SLOs 0.025 s / 0.5 s, Poisson arrivals, input
202±20 / output 179±17 tokens, N simulated servers, CSV of latency / TTFT /
TPOT / throughput / recompute / %-under-SLO per policy and rate.

    python -m llm_instance_gateway_amd.sim.main --routing-type smart \
        --rates 5 10 20 --num-servers 6 --duration 100
"""

from __future__ import annotations

import argparse
import csv
import json
import math
import random
import sys
from typing import Dict, List, Optional

from .core import SimEngine, SimEnv, SimRequest
from .router import ROUTING_POLICIES, SimRouter

# Workload shape (reference main.py:17-37,62-63)
INPUT_MEAN, INPUT_STD = 202, 20
OUTPUT_MEAN, OUTPUT_STD = 179, 17
SLO_TPOT_LOW = 0.025   # latency-sensitive tier
SLO_TPOT_HIGH = 0.5    # latency-tolerant tier


def generate_workload(
    env: SimEnv,
    router: SimRouter,
    rate_low: float,
    rate_high: float,
    duration: float,
    rng: random.Random,
    num_loras: int = 0,
    max_messages: int = 2500,
) -> List[SimRequest]:
    """Poisson arrivals for both SLO tiers; returns all created requests."""
    reqs: List[SimRequest] = []
    next_id = [0]

    def spawn_arrivals(rate: float, slo: float) -> None:
        if rate <= 0:
            return
        t = 0.0
        n = 0
        while t < duration and n < max_messages:
            t += rng.expovariate(rate)
            if t >= duration:
                break
            ins = max(8, int(rng.gauss(INPUT_MEAN, INPUT_STD)))
            outs = max(4, int(rng.gauss(OUTPUT_MEAN, OUTPUT_STD)))
            lora = f"lora-{rng.randrange(num_loras)}" if num_loras else None
            req = SimRequest(
                id=next_id[0],
                arrival_time=t,
                input_size=ins,
                output_size=outs,
                target_latency_per_token=slo,
                lora=lora,
                critical=(slo == SLO_TPOT_LOW),
            )
            next_id[0] += 1
            reqs.append(req)
            env.schedule(t - env.now, lambda r=req: router.handle(r))
            n += 1

    spawn_arrivals(rate_low, SLO_TPOT_LOW)
    spawn_arrivals(rate_high, SLO_TPOT_HIGH)
    reqs.sort(key=lambda r: r.arrival_time)
    for i, r in enumerate(reqs):
        r.id = i  # ids in arrival order (recompute priority = oldest first)
    return reqs


def percentile(values: List[float], p: float) -> float:
    if not values:
        return float("nan")
    vs = sorted(values)
    k = min(len(vs) - 1, max(0, int(math.ceil(p / 100.0 * len(vs))) - 1))
    return vs[k]


def run_once(
    routing_type: str,
    rate_low: float,
    rate_high: float,
    num_servers: int = 6,
    duration: float = 100.0,
    drain: float = 100.0,
    num_loras: int = 0,
    seed: int = 0,
) -> Dict:
    rng = random.Random(seed)
    env = SimEnv()
    engines = [SimEngine(env, i) for i in range(num_servers)]
    router = SimRouter(env, engines, routing_type=routing_type, rng=random.Random(seed + 1))
    reqs = generate_workload(env, router, rate_low, rate_high, duration, rng, num_loras)
    env.run(until=duration + drain)

    done = [r for r in reqs if r.end_decode_time is not None]
    ttfts = [r.ttft for r in done]
    tpots = [r.tpot for r in done]
    lats = [r.e2e_latency for r in done]
    good = [r for r in done if r.within_slo]
    span = max((r.end_decode_time for r in done), default=duration)
    out = {
        "routing_type": routing_type,
        "rate_low": rate_low,
        "rate_high": rate_high,
        "num_servers": num_servers,
        "offered": len(reqs),
        "completed": len(done),
        "throughput_rps": len(done) / span if span else 0.0,
        "goodput_rps": len(good) / span if span else 0.0,
        "pct_under_slo": 100.0 * len(good) / len(done) if done else 0.0,
        "ttft_mean": sum(ttfts) / len(ttfts) if ttfts else float("nan"),
        "ttft_p99": percentile(ttfts, 99),
        "tpot_mean": sum(tpots) / len(tpots) if tpots else float("nan"),
        "tpot_p99": percentile(tpots, 99),
        "latency_mean": sum(lats) / len(lats) if lats else float("nan"),
        "latency_p99": percentile(lats, 99),
        "recomputes": sum(r.recompute_count for r in reqs),
        "tokens_generated": sum(r.output_size for r in done),
    }
    return out


def main(argv: Optional[List[str]] = None) -> int:
    ap = argparse.ArgumentParser(description="gateway fleet simulation")
    ap.add_argument("--routing-type", default="all",
                    help=f"one of {ROUTING_POLICIES} or 'all'")
    ap.add_argument("--rates", type=float, nargs="+",
                    default=[1, 5, 10, 20, 30, 40],
                    help="offered QPS per tier (sweep)")
    ap.add_argument("--num-servers", type=int, default=6)
    ap.add_argument("--duration", type=float, default=100.0)
    ap.add_argument("--num-loras", type=int, default=0)
    ap.add_argument("--seed", type=int, default=0)
    ap.add_argument("--csv", default="", help="write results CSV here")
    args = ap.parse_args(argv)

    policies = (
        list(ROUTING_POLICIES) if args.routing_type == "all" else [args.routing_type]
    )
    rows = []
    for pol in policies:
        for rate in args.rates:
            row = run_once(
                pol, rate, rate,
                num_servers=args.num_servers,
                duration=args.duration,
                num_loras=args.num_loras,
                seed=args.seed,
            )
            rows.append(row)
            print(json.dumps(row))
    if args.csv:
        with open(args.csv, "w", newline="") as f:
            w = csv.DictWriter(f, fieldnames=list(rows[0].keys()))
            w.writeheader()
            w.writerows(rows)
    return 0


if __name__ == "__main__":
    sys.exit(main())
