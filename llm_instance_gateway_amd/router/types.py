"""Router data model.

MI355X mapping of the reference backend types (`pkg/ext-proc/backend/types.go`):
a K8s pod running vLLM becomes a local per-GPU engine (or a remote engine
address in scrape mode).  The metric struct carries exactly the signals the
filter chain consumes (reference `types.go:20-53`):

  waiting/running queue depth, KV-cache utilisation %, the set of LoRA
  adapters resident on the GPU, and the adapter slot budget.
"""

from __future__ import annotations

import copy
from dataclasses import dataclass, field
from typing import Dict


@dataclass(frozen=True)
class Engine:
    """Identity of one serving engine (reference `backend.Pod`, types.go:8-14).

    ``name`` is the stable routing key (what the reference put in the
    `target-pod` header); ``gpu_id`` is the local HIP device index, or -1 for
    a remote engine addressed by ``address``.
    """

    name: str
    gpu_id: int = -1
    address: str = ""
    # Extensions over the reference (mixed-pool serving, BASELINE config #4):
    # the base model this engine serves and the pool it belongs to.  The
    # reference needs neither because one InferencePool == one homogeneous
    # fleet; here several sub-pools (8B TP=1 engines + 70B TP=4 engine) can
    # coexist behind one front door.
    model: str = ""
    pool: str = ""


@dataclass
class EngineMetrics:
    """Live signals from one engine (reference `backend.Metrics`, types.go:20-37).

    ``active_models`` maps adapter/model name -> resident flag (the reference
    uses a map for set semantics); ``max_active_models`` is the adapter slot
    budget (vLLM `--max-loras`, exported as `max_lora` in
    `vllm:lora_requests_info` — reference backend/vllm/metrics.go:20-31).
    """

    active_models: Dict[str, int] = field(default_factory=dict)
    max_active_models: int = 0
    running_queue_size: int = 0
    waiting_queue_size: int = 0
    kv_cache_usage_percent: float = 0.0
    kv_cache_max_token_capacity: int = 0
    # lifetime preemption count (vLLM exports vllm:num_preemptions_total;
    # the reference sim tracks the same as recompute_count)
    num_preemptions_total: int = 0


@dataclass
class EngineState:
    """An engine plus its latest metrics (reference `backend.PodMetrics`)."""

    engine: Engine
    metrics: EngineMetrics = field(default_factory=EngineMetrics)

    def clone(self) -> "EngineState":
        """Deep copy (reference `PodMetrics.Clone`, types.go:39-53).  Explicit
        field copy, not copy.deepcopy: clone runs per engine per schedule()
        call on the router hot path, and deepcopy's generic reflection is
        ~30x slower than rebuilding the one dict + five scalars."""
        m = self.metrics
        return EngineState(
            engine=self.engine,
            metrics=EngineMetrics(
                active_models=dict(m.active_models),
                max_active_models=m.max_active_models,
                running_queue_size=m.running_queue_size,
                waiting_queue_size=m.waiting_queue_size,
                kv_cache_usage_percent=m.kv_cache_usage_percent,
                kv_cache_max_token_capacity=m.kv_cache_max_token_capacity,
                num_preemptions_total=m.num_preemptions_total,
            ),
        )


@dataclass
class LLMRequest:
    """Parsed request fields the scheduler needs (reference scheduling/types.go:4-11)."""

    model: str
    target_models: Dict[str, int] = field(default_factory=dict)
    resolved_target_model: str = ""
    critical: bool = False
    # Extensions over the reference (used by admission/SLO accounting and
    # mixed-pool candidate selection, not by the parity filter chain):
    prompt_len: int = 0
    max_new_tokens: int = 0
    request_id: str = ""
    pool_ref: str = ""
