"""Simulated load balancer: the routing-policy A/B axis.

Reference `simulations/llm_ig_simulation/src/loadbalancer.py`: policies
``random`` / ``least`` (min expected KV after prefill) / ``leastPseudo``
(min pending) / ``leastlatency`` (min estimated request latency) /
``smart`` (max pending among pods estimated to meet the SLO), plus LoRA
affinity (`loadbalancer.py:130-139`), saturation-gated admission queues per
SLO tier with weighted dequeue (`loadbalancer.py:351-518`).

The latency estimator uses the sim's own analytic kernel model applied to
the candidate's current state (the reference fits per-token decode rates
from observation; in a sim both converge to the same signal).
"""

from __future__ import annotations

import random as _random
from typing import Dict, List, Optional

from .core import (
    MAX_KV_PERC_BEFORE_RECOMPUTE,
    SimEngine,
    SimEnv,
    SimRequest,
    decode_delay,
    prefill_delay,
)

ROUTING_POLICIES = ("random", "least", "leastPseudo", "leastlatency", "smart")


class SimRouter:
    def __init__(
        self,
        env: SimEnv,
        engines: List[SimEngine],
        routing_type: str = "smart",
        queueing_perc: float = 0.9,
        rng: Optional[_random.Random] = None,
    ) -> None:
        assert routing_type in ROUTING_POLICIES, routing_type
        self.env = env
        self.engines = engines
        self.routing_type = routing_type
        self.queueing_perc = queueing_perc
        self.rng = rng or _random.Random(0)
        # per-SLO-tier admission queues (reference: queue per target latency)
        self.queues: Dict[float, List[SimRequest]] = {}
        self.dropped: List[SimRequest] = []
        self.routed = 0

    # -- state signals -------------------------------------------------------
    def all_saturated(self, max_saturation: float = MAX_KV_PERC_BEFORE_RECOMPUTE) -> bool:
        return all(e.is_saturated(max_saturation) for e in self.engines)

    def queueing_signal(self) -> bool:
        """Hold new work at the router while the fleet is saturated
        (reference queueing_signal, loadbalancer.py:351-359)."""
        return self.all_saturated(self.queueing_perc)

    # -- latency estimation --------------------------------------------------
    def estimate_latency(self, e: SimEngine, req: SimRequest) -> float:
        """Rough completion-latency estimate if req lands on engine e: queue
        drain + own prefill + output_size decode steps at current batch."""
        kv = e.tokens_in_decode()
        batch = len(e.decode) + 1
        queue_cost = prefill_delay(e.pending_tokens()) if e.queue_len() else 0.0
        own = prefill_delay(req.input_size)
        dec = req.output_size * decode_delay(kv + req.input_size, batch)
        return queue_cost + own + dec

    # -- policies ------------------------------------------------------------
    def _pick(self, req: SimRequest) -> Optional[SimEngine]:
        rt = self.routing_type
        if rt == "random":
            return self.engines[self.rng.randrange(len(self.engines))]
        if rt == "least":
            best = min(e.expected_kv_after_prefill() for e in self.engines)
            cands = [e for e in self.engines if e.expected_kv_after_prefill() == best]
            return self.rng.choice(cands)
        if rt == "leastPseudo":
            best = min(e.pending_tokens() for e in self.engines)
            cands = [e for e in self.engines if e.pending_tokens() == best]
            return self.rng.choice(cands)
        if rt == "leastlatency":
            return min(self.engines, key=lambda e: self.estimate_latency(e, req))
        # smart: LoRA affinity first, then max-pending-under-SLO (pack the
        # busiest engine that still meets the target; reference
        # find_target_pod_based_on_max_pending, loadbalancer.py:320-348)
        cands = self.engines
        if req.lora is not None:
            affine = [e for e in self.engines if req.lora in e.lora_loaded]
            if affine:
                cands = affine
        # TPOT-SLO gate: would this engine's per-token decode latency (with
        # this request on board) still meet the request's target?  This is
        # the signal that actually segregates the tiers — tight-SLO traffic
        # stays off heavily-batched engines, loose-SLO traffic gets PACKED
        # onto the busiest qualifying engine, keeping lightly-loaded engines
        # free for the tight tier.
        def qualifying(pool):
            return [
                e
                for e in pool
                if not e.is_saturated()
                and decode_delay(
                    e.tokens_in_decode() + req.input_size + req.output_size,
                    len(e.decode) + 1,
                )
                <= req.target_latency_per_token
            ]

        under_slo = qualifying(cands)
        if not under_slo and cands is not self.engines:
            # adapter affinity must not trap a tight-SLO request on a packed
            # engine: loading the adapter elsewhere beats missing the SLO
            under_slo = qualifying(self.engines)
        if under_slo:
            # pack the latency-TOLERANT tier onto the busiest qualifying
            # engine (consolidation); SPREAD the tight tier onto the least
            # loaded one — max-pending packing of tight-SLO traffic
            # overshoots by the time the request reaches decode
            tight = req.target_latency_per_token <= 0.05
            if tight:
                return min(under_slo, key=lambda e: len(e.decode) + e.queue_len())
            return max(under_slo, key=lambda e: e.pending_tokens())
        unsat = [e for e in cands if not e.is_saturated()]
        if unsat:
            return min(unsat, key=lambda e: e.pending_tokens())
        unsat_all = [e for e in self.engines if not e.is_saturated()]
        if unsat_all:
            return min(unsat_all, key=lambda e: e.pending_tokens())
        return None

    # -- intake --------------------------------------------------------------
    def handle(self, req: SimRequest) -> None:
        """Route now, or park in the SLO-tier queue while saturated."""
        if self.queueing_signal():
            req.queued_time = self.env.now
            self.queues.setdefault(req.target_latency_per_token, []).append(req)
            self._schedule_dequeue()
            return
        self._route(req)

    def _route(self, req: SimRequest) -> None:
        target = self._pick(req)
        if target is None:
            req.queued_time = self.env.now
            self.queues.setdefault(req.target_latency_per_token, []).append(req)
            self._schedule_dequeue()
            return
        self.routed += 1
        target.submit(req)

    # -- admission queues (weighted dequeue, loadbalancer.py:431-454) --------
    _dequeue_pending = False

    def _schedule_dequeue(self) -> None:
        if not self._dequeue_pending:
            self._dequeue_pending = True
            self.env.schedule(0.01, self._dequeue_tick)

    def _dequeue_tick(self) -> None:
        self._dequeue_pending = False
        if not any(self.queues.values()):
            return
        if not self.queueing_signal():
            # tighter SLO tiers get weight inversely proportional to target
            tiers = sorted(
                (t for t, q in self.queues.items() if q),
                key=lambda t: t,
            )
            weights = [1.0 / t for t in tiers]
            total = sum(weights)
            r = self.rng.random() * total
            for t, w in zip(tiers, weights):
                if r < w:
                    tier = t
                    break
                r -= w
            else:  # pragma: no cover
                tier = tiers[-1]
            req = self.queues[tier].pop(0)
            self._route(req)
        self._schedule_dequeue()

    # -- results -------------------------------------------------------------
    def finished(self) -> List[SimRequest]:
        return [r for e in self.engines for r in e.decoded]
