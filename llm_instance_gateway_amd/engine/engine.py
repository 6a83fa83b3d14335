"""LLMEngine: one GPU's serving engine.

The MI355X-native replacement for one vLLM pod of the reference's fleet
(SURVEY.md layer L5/L6): continuous batching over a paged KV cache, chunked
prefill, preemption-by-recompute, multi-LoRA slots, and — crucially for the
gateway — the live gauges the router's filter chain consumes
(waiting/running queue depth, KV utilisation %, resident adapter set;
reference `backend/vllm/metrics.go:19-32`), refreshed after every step with
zero staleness.
"""

from __future__ import annotations

import atexit
import logging
import os
import time
from typing import Callable, Dict, List, Optional

import torch

from ..router.types import EngineMetrics
from .config import EngineConfig
from .model_runner import ModelRunner
from .scheduler import EngineScheduler, SchedulerOutput
from .sequence import SamplingParams, Sequence, SequenceStatus

log = logging.getLogger(__name__)

# LIG_STEP_TIMING=1: accumulate per-step host/GPU phase times (schedule+lora
# vs execute-with-sync), dumped at exit — the host-gap diagnosis knob.
_STEP_TIMING = os.environ.get("LIG_STEP_TIMING", "0") == "1"
_STEP_STATS: Dict[str, list] = {}

if _STEP_TIMING:  # pragma: no cover - diagnostic only
    @atexit.register
    def _dump_step_stats() -> None:
        for kind, (n, sched_s, exec_s) in sorted(_STEP_STATS.items()):
            print(
                f"[step-timing] {kind}: n={n} schedule+lora={sched_s / max(1, n) * 1e3:.2f}ms "
                f"execute(sync)={exec_s / max(1, n) * 1e3:.2f}ms"
            )


class LLMEngine:
    def __init__(
        self,
        config: EngineConfig,
        device: str = "cpu",
        dtype: torch.dtype = torch.bfloat16,
        tp=None,
    ) -> None:
        self.config = config
        self.runner = ModelRunner(config, device=device, dtype=dtype, tp=tp)
        self.scheduler = EngineScheduler(config)
        # finished sequences ready for collection by the front door
        self._finished: List[Sequence] = []
        self._step_count = 0
        # adapter manager attaches itself here (lora.manager.AdapterManager)
        self.adapter_manager = None
        # callback(seq, token_id) fired on every generated token (streaming)
        self.on_token: Optional[Callable[[Sequence, int], None]] = None

    # -- request intake --------------------------------------------------------
    def add_request(
        self,
        prompt_token_ids: List[int],
        sampling: Optional[SamplingParams] = None,
        lora_name: str = "",
        request_id: str = "",
        critical: bool = False,
        arrival_time: Optional[float] = None,
    ) -> Sequence:
        if len(prompt_token_ids) > self.config.max_model_len:
            raise ValueError(
                f"prompt of {len(prompt_token_ids)} tokens exceeds max_model_len "
                f"{self.config.max_model_len}"
            )
        # vLLM semantics: generation never runs past the model's context
        # window — clamp max_new_tokens to the remaining budget (otherwise a
        # long request walks positions past the RoPE cache / KV capacity).
        if sampling is not None:
            room = self.config.max_model_len - len(prompt_token_ids)
            if sampling.max_new_tokens > room:
                sampling.max_new_tokens = max(1, room)
        seq = Sequence(
            prompt_token_ids,
            sampling=sampling,
            lora_name=lora_name,
            request_id=request_id,
            critical=critical,
            arrival_time=arrival_time,
        )
        if lora_name and self.adapter_manager is not None:
            self.adapter_manager.touch(lora_name)
        self.scheduler.add_seq(seq)
        return seq

    def abort_request(self, request_id: str) -> bool:
        return self.scheduler.abort_seq(request_id)

    # -- the serving loop body ---------------------------------------------------
    def step(self) -> List[Sequence]:
        """One continuous-batching iteration.  Returns sequences finished in
        this step."""
        timing = _STEP_TIMING
        t0 = time.perf_counter() if timing else 0.0
        out: SchedulerOutput = self.scheduler.schedule()
        if out.is_empty:
            return []
        if self.adapter_manager is not None:
            # make every scheduled adapter resident (async side-stream loads
            # fenced with events) before the step touches the slot weights
            self.adapter_manager.prepare(
                [ss.seq.lora_name for ss in out.scheduled if ss.seq.lora_name]
            )
            self.runner.adapter_manager = self.adapter_manager
        if timing:
            t1 = time.perf_counter()
        sampled = self.runner.execute(out)
        if timing:
            torch.cuda.synchronize() if self.runner.device.type == "cuda" else None
            t2 = time.perf_counter()
            kind = "prefill" if any(ss.num_tokens > 1 for ss in out.scheduled) else "decode"
            acc = _STEP_STATS.setdefault(kind, [0, 0.0, 0.0])
            acc[0] += 1
            acc[1] += t1 - t0
            acc[2] += t2 - t1
        for ss in out.scheduled:
            seq = ss.seq
            tok = sampled.get(seq.seq_id)
            if tok is not None:
                seq.append_token(tok)
                if self.on_token is not None:
                    self.on_token(seq, tok)
        finished = self.scheduler.finish_step(out)
        self._finished.extend(finished)
        self._step_count += 1
        return finished

    def has_work(self) -> bool:
        return self.scheduler.has_work()

    def collect_finished(self) -> List[Sequence]:
        out, self._finished = self._finished, []
        return out

    # -- gauges (the router's inputs; reference backend/vllm/metrics.go) ---------
    def get_metrics(self) -> EngineMetrics:
        active: Dict[str, int] = {}
        max_active = self.config.max_loras
        if self.adapter_manager is not None:
            active = {name: 1 for name in self.adapter_manager.resident_names()}
            max_active = self.adapter_manager.max_slots
        return EngineMetrics(
            active_models=active,
            max_active_models=max_active,
            running_queue_size=self.scheduler.num_running,
            waiting_queue_size=self.scheduler.num_waiting,
            kv_cache_usage_percent=self.scheduler.kv_usage,
            kv_cache_max_token_capacity=self.config.kv_tokens_capacity,
            num_preemptions_total=self.scheduler.total_preemptions,
        )

    # -- convenience: run to completion (tests / offline batch) -----------------
    def generate_all(self, max_steps: int = 100000) -> List[Sequence]:
        done: List[Sequence] = []
        for _ in range(max_steps):
            if not self.has_work():
                break
            self.step()
            done.extend(self.collect_finished())
        return done
