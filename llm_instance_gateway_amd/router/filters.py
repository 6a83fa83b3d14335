"""Decision-tree filter chain — the heart of the model-aware router.

Semantics-compatible re-implementation of the reference scheduler's filter
tree (`pkg/ext-proc/scheduling/filter.go` + `scheduler.go:26-91`):

  * each node applies a filter function to the candidate engine set;
  * on success (non-empty result, no error) the *filtered* set flows to
    ``next_on_success`` (or ``next_on_success_or_failure``);
  * on failure the *original* set flows to ``next_on_failure`` (or
    ``next_on_success_or_failure``);
  * band filters (least-queuing, least-KV) keep every candidate within the
    first 1/N of the min..max range instead of the absolute minimum, to leave
    choices for downstream filters (reference filter.go:97-154, including the
    Go integer-division band for queue depth).

The tree shape and thresholds are the reference defaults
(`scheduler.go:15-24`): kv_cache_threshold=0.8, queue_threshold_critical=5,
queueing_threshold_lora=50 — made configurable here (the reference carries
"make configurable" TODOs for all three).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Callable, List, Optional

from .types import EngineState, LLMRequest

# Reference scheduler.go:15-24 defaults.
KV_CACHE_THRESHOLD = 0.8
QUEUE_THRESHOLD_CRITICAL = 5
QUEUEING_THRESHOLD_LORA = 50


class FilterError(Exception):
    """A filter produced no candidates (reference toFilterFunc 'no pods left')."""


FilterFunc = Callable[[LLMRequest, List[EngineState]], List[EngineState]]


@dataclass
class Filter:
    """One node of the decision tree (reference filter.go:20-41)."""

    name: str
    func: FilterFunc
    next_on_success: Optional["Filter"] = None
    next_on_failure: Optional["Filter"] = None
    next_on_success_or_failure: Optional["Filter"] = None

    def apply(self, req: LLMRequest, engines: List[EngineState]) -> List[EngineState]:
        """Reference filter.go:44-73 control flow."""
        err: Optional[Exception] = None
        filtered: List[EngineState] = []
        try:
            filtered = self.func(req, engines)
        except FilterError as e:
            err = e

        if err is None and len(filtered) > 0:
            nxt = self.next_on_success or self.next_on_success_or_failure
            if nxt is None:
                return filtered
            return nxt.apply(req, filtered)
        else:
            nxt = self.next_on_failure or self.next_on_success_or_failure
            if nxt is None:
                if err is not None:
                    raise err
                return filtered
            # On failure, pass the ORIGINAL set (reference filter.go:66-72).
            return nxt.apply(req, engines)


Predicate = Callable[[LLMRequest, EngineState], bool]


def to_filter_func(pred: Predicate) -> FilterFunc:
    """Reference toFilterFunc (filter.go:82-96): empty result is an error."""

    def f(req: LLMRequest, engines: List[EngineState]) -> List[EngineState]:
        out = [e for e in engines if pred(req, e)]
        if not out:
            raise FilterError("no engines left")
        return out

    return f


# -- band filters (reference filter.go:97-154) ------------------------------

def least_queuing_filter_func(
    req: LLMRequest, engines: List[EngineState]
) -> List[EngineState]:
    """Keep engines in the first 1/N band of the waiting-queue range.

    Exact reference semantics (filter.go:102-122) including Go integer
    division of (max-min)/len.
    """
    lo = min(e.metrics.waiting_queue_size for e in engines)
    hi = max(e.metrics.waiting_queue_size for e in engines)
    band = lo + (hi - lo) // len(engines)
    return [e for e in engines if lo <= e.metrics.waiting_queue_size <= band]


def least_kv_cache_filter_func(
    req: LLMRequest, engines: List[EngineState]
) -> List[EngineState]:
    """Keep engines in the first 1/N band of the KV-utilisation range
    (reference filter.go:134-154; float division)."""
    lo = min(e.metrics.kv_cache_usage_percent for e in engines)
    hi = max(e.metrics.kv_cache_usage_percent for e in engines)
    band = lo + (hi - lo) / float(len(engines))
    return [e for e in engines if lo <= e.metrics.kv_cache_usage_percent <= band]


# -- predicates (reference filter.go:124-126, 157-187) ----------------------

def low_queueing_predicate(req: LLMRequest, e: EngineState) -> bool:
    return e.metrics.waiting_queue_size < QUEUEING_THRESHOLD_LORA


def low_lora_cost_predicate(req: LLMRequest, e: EngineState) -> bool:
    """Adapter resident OR a free adapter slot (weak affinity that spreads an
    adapter across engines — reference filter.go:158-166)."""
    return (
        req.resolved_target_model in e.metrics.active_models
        or len(e.metrics.active_models) < e.metrics.max_active_models
    )


def lora_affinity_predicate(req: LLMRequest, e: EngineState) -> bool:
    return req.resolved_target_model in e.metrics.active_models


def can_accept_new_lora_predicate(req: LLMRequest, e: EngineState) -> bool:
    return len(e.metrics.active_models) < e.metrics.max_active_models


def critical_request_predicate(req: LLMRequest, e: EngineState) -> bool:
    return req.critical


def no_queue_and_less_than_kv_cache_threshold_predicate(
    queue_threshold: int, kv_cache_threshold: float
) -> Predicate:
    def pred(req: LLMRequest, e: EngineState) -> bool:
        return (
            e.metrics.waiting_queue_size <= queue_threshold
            and e.metrics.kv_cache_usage_percent <= kv_cache_threshold
        )

    return pred


class DropRequest(Exception):
    """Shed signal: maps to HTTP 429 at the front door (reference
    scheduler.go:83-89 ResourceExhausted → handlers/server.go:95-113)."""


def _drop_filter_func(req: LLMRequest, engines: List[EngineState]) -> List[EngineState]:
    raise DropRequest("dropping request due to limited backend resources")


def build_default_tree(
    kv_cache_threshold: float = KV_CACHE_THRESHOLD,
    queue_threshold_critical: int = QUEUE_THRESHOLD_CRITICAL,
    queueing_threshold_lora: int = QUEUEING_THRESHOLD_LORA,
) -> Filter:
    """Construct the reference default tree (scheduler.go:26-91).

    Shape:
      critical? ── yes → lowQueueing(<50)? ── yes → loRAAffinity? ─ yes → leastQueuing → leastKV
                │                        │                       └ no → canAcceptNewLora →(either)→ leastQueuing → leastKV
                │                        └ no → leastQueuing → lowLoRACost → leastKV
                └─ no (sheddable) → (queue≤5 ∧ KV≤0.8 on any engine)? ─ yes → leastQueuing → lowLoRACost → leastKV
                                                                      └ no → DROP (ResourceExhausted)
    """

    def low_q_pred(req: LLMRequest, e: EngineState) -> bool:
        return e.metrics.waiting_queue_size < queueing_threshold_lora

    # leastQueuing -> lowCostLoRA -> leastKV  (queueLoRAAndKVCacheFilter)
    queue_lora_kv = Filter(
        name="least queuing",
        func=least_queuing_filter_func,
        next_on_success_or_failure=Filter(
            name="low cost LoRA",
            func=to_filter_func(low_lora_cost_predicate),
            next_on_success_or_failure=Filter(
                name="least KV cache percent", func=least_kv_cache_filter_func
            ),
        ),
    )

    # leastQueuing -> leastKV  (queueAndKVCacheFilter)
    queue_kv = Filter(
        name="least queuing",
        func=least_queuing_filter_func,
        next_on_success_or_failure=Filter(
            name="least KV cache percent", func=least_kv_cache_filter_func
        ),
    )

    low_latency = Filter(
        name="low queueing filter",
        func=to_filter_func(low_q_pred),
        next_on_success=Filter(
            name="affinity LoRA",
            func=to_filter_func(lora_affinity_predicate),
            next_on_success=queue_kv,
            next_on_failure=Filter(
                name="can accept LoRA Adapter",
                func=to_filter_func(can_accept_new_lora_predicate),
                next_on_success_or_failure=queue_kv,
            ),
        ),
        next_on_failure=queue_lora_kv,
    )

    sheddable = Filter(
        name="has capacity for sheddable requests",
        func=to_filter_func(
            no_queue_and_less_than_kv_cache_threshold_predicate(
                queue_threshold_critical, kv_cache_threshold
            )
        ),
        next_on_success=queue_lora_kv,
        next_on_failure=Filter(name="drop request", func=_drop_filter_func),
    )

    return Filter(
        name="critical request",
        func=to_filter_func(critical_request_predicate),
        next_on_success=low_latency,
        next_on_failure=sheddable,
    )
