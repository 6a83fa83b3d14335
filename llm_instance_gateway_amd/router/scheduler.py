"""Request scheduler: filter chain + uniform pick over survivors.

Reference: `pkg/ext-proc/scheduling/scheduler.go:93-122`.  `Schedule` snapshots
all engine metrics from the provider, runs the decision tree, and picks
uniformly at random among the surviving engines.  A `DropRequest` from the
sheddable branch surfaces as `ResourceExhausted` (HTTP 429 at the front door,
reference handlers/server.go:95-113).
"""

from __future__ import annotations

import random
from typing import List, Optional, Protocol

from .filters import Filter, FilterError, DropRequest, build_default_tree
from .types import Engine, EngineState, LLMRequest


class ResourceExhausted(Exception):
    """Raised when a sheddable request is dropped for lack of capacity."""


class SchedulingError(Exception):
    """Filter chain yielded no engines (should never happen — reference
    scheduler.go:116-118)."""


class EngineMetricsProvider(Protocol):
    """Reference PodMetricsProvider (scheduler.go:106-110)."""

    def all_engine_metrics(self) -> List[EngineState]: ...


class Scheduler:
    def __init__(
        self,
        provider: EngineMetricsProvider,
        tree: Optional[Filter] = None,
        rng: Optional[random.Random] = None,
    ) -> None:
        self.provider = provider
        self.filter = tree if tree is not None else build_default_tree()
        self._rng = rng or random.Random()

    def set_thresholds(
        self,
        kv_cache_threshold: float,
        queue_threshold_critical: int,
        queueing_threshold_lora: int,
    ) -> None:
        """Rebuild the decision tree with new thresholds (runtime config
        reconcile — the reference carries these as compile-time consts with
        'make configurable' TODOs, scheduler.go:15-24).  Atomic swap: the
        tree is immutable once built, so in-flight schedule() calls keep
        their snapshot."""
        self.filter = build_default_tree(
            kv_cache_threshold=kv_cache_threshold,
            queue_threshold_critical=queue_threshold_critical,
            queueing_threshold_lora=queueing_threshold_lora,
        )

    def schedule(self, req: LLMRequest) -> Engine:
        """Reference Schedule (scheduler.go:113-122).

        Extension: if the request names a pool (mixed-pool serving — several
        sub-pools behind one front door, BASELINE config #4), only that
        pool's engines are candidates; untagged engines always qualify."""
        states = self.provider.all_engine_metrics()
        if req.pool_ref:
            states = [
                s for s in states if s.engine.pool in ("", req.pool_ref)
            ] or states
        try:
            survivors = self.filter.apply(req, states)
        except DropRequest as e:
            raise ResourceExhausted(str(e)) from e
        except FilterError as e:
            raise SchedulingError(f"failed to apply filter: {e}") from e
        if not survivors:
            raise SchedulingError("filter produced 0 engines, this should never happen")
        return survivors[self._rng.randrange(len(survivors))].engine
