"""Production filter tree driving the simulated fleet.

SURVEY.md §7 "hard parts" #5: the reference's affinity/queueing thresholds
(queueing_threshold_lora=50, queue_threshold_critical=5 — scheduler.go:15-24)
were tuned for 50 ms-stale scraped metrics.  This framework's router reads
shared-memory gauges with ZERO staleness, so the thresholds deserve a
re-examination — this adapter runs the REAL `router/filters.py` tree (the
code that serves production traffic) over the discrete-event fleet of
`sim/core.py`, so threshold sweeps measure end-to-end SLO attainment per
setting.  `benchmarks/sweep_thresholds.py` is the driver;
`profiles/r02_threshold_sweep.md` the findings.
"""

from __future__ import annotations

import random as _random
from typing import List, Optional

from ..router.filters import DropRequest, build_default_tree
from ..router.types import Engine, EngineMetrics, EngineState, LLMRequest
from .core import SimEngine, SimEnv, SimRequest
from .router import SimRouter


class TreeSimRouter(SimRouter):
    """SimRouter whose pick is the production filter tree.

    The SLO-tier admission queues / saturation gating of the base class are
    kept (they model the reference simulation's router intake); only the
    per-request engine choice is replaced.
    """

    def __init__(
        self,
        env: SimEnv,
        engines: List[SimEngine],
        kv_cache_threshold: float = 0.8,
        queue_threshold_critical: int = 5,
        queueing_threshold_lora: int = 50,
        queueing_perc: float = 0.9,
        rng: Optional[_random.Random] = None,
    ) -> None:
        super().__init__(
            env, engines, routing_type="smart",
            queueing_perc=queueing_perc, rng=rng,
        )
        self.tree = build_default_tree(
            kv_cache_threshold=kv_cache_threshold,
            queue_threshold_critical=queue_threshold_critical,
            queueing_threshold_lora=queueing_threshold_lora,
        )
        self.drops = 0

    def _state_of(self, e: SimEngine) -> EngineState:
        return EngineState(
            engine=Engine(name=f"sim-{e.id}", gpu_id=e.id),
            metrics=EngineMetrics(
                active_models={name: 1 for name in e.lora_loaded},
                max_active_models=max(len(e.lora_loaded), 8),
                running_queue_size=len(e.decode),
                waiting_queue_size=e.queue_len(),
                kv_cache_usage_percent=e.kv_usage(),
            ),
        )

    def _pick(self, req: SimRequest) -> Optional[SimEngine]:
        states = [self._state_of(e) for e in self.engines]
        lreq = LLMRequest(
            model="sim-model",
            resolved_target_model=req.lora or "sim-model",
            critical=req.critical,
            prompt_len=req.input_size,
            max_new_tokens=req.output_size,
        )
        try:
            survivors = self.tree.apply(lreq, states)
        except DropRequest:
            # sheddable overload: the serving layer would 429; the sim
            # counts it as a drop (request never completes)
            self.drops += 1
            self.dropped.append(req)
            req.dropped = True
            return None
        if not survivors:
            return None
        pick = self.rng.choice(survivors)
        idx = int(pick.engine.name.split("-")[1])
        return self.engines[idx]

    def _route(self, req: SimRequest) -> None:
        # unlike the base class, a DROPPED request must not be re-queued
        target = self._pick(req)
        if target is None:
            if getattr(req, "dropped", False):
                return
            req.queued_time = self.env.now
            self.queues.setdefault(req.target_latency_per_token, []).append(req)
            self._schedule_dequeue()
            return
        self.routed += 1
        target.submit(req)
