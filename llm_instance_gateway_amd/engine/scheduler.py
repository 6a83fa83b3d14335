"""Continuous-batching scheduler for one GPU engine.

The reference's SimPy model of vLLM's batching loop is the executable spec
(`simulations/llm_ig_simulation/src/continous_batching.py`):

  * admission caps: max sequences per batch and max batched (prefill) tokens
    per step (`continous_batching.py:10-43`, MAX_NUM_SEQ=256,
    MAX_NUM_BATCH_TOKENS per chunk);
  * chunked prefill: a long prompt is processed in token-budget-sized chunks
    across steps;
  * preemption-by-recompute: when KV utilisation crosses the threshold the
    NEWEST sequences are evicted back to the waiting queue and will re-prefill
    (`continous_batching.py:117-131`);
  * the waiting/running queue depths and KV utilisation exported after every
    step ARE the router's scheduling inputs (reference
    backend/vllm/metrics.go:24-31).

One `schedule()` call builds one mixed step: all running decodes (one token
each) plus as many prefill-chunk tokens as fit the budget.
"""

from __future__ import annotations

from collections import deque
from dataclasses import dataclass, field
from typing import Deque, List, Optional, Tuple

from .block_manager import BlockManager
from .config import EngineConfig
from .sequence import Sequence, SequenceStatus


@dataclass
class ScheduledSeq:
    seq: Sequence
    num_tokens: int  # 1 for decode; chunk size for prefill

    @property
    def is_prefill(self) -> bool:
        return self.seq.in_prefill


@dataclass
class SchedulerOutput:
    scheduled: List[ScheduledSeq] = field(default_factory=list)
    preempted: List[Sequence] = field(default_factory=list)

    @property
    def num_batched_tokens(self) -> int:
        return sum(s.num_tokens for s in self.scheduled)

    @property
    def is_empty(self) -> bool:
        return not self.scheduled


class EngineScheduler:
    def __init__(self, config: EngineConfig, block_manager: Optional[BlockManager] = None) -> None:
        self.config = config
        self.block_manager = block_manager or BlockManager(
            config.num_kv_blocks, config.block_size, config.watermark_blocks
        )
        self.waiting: Deque[Sequence] = deque()
        self.running: List[Sequence] = []
        # lifetime preemption count (vllm:num_preemptions_total parity)
        self.total_preemptions = 0

    # -- intake --------------------------------------------------------------
    def add_seq(self, seq: Sequence) -> None:
        """FIFO within a criticality tier; Critical requests are admitted
        ahead of waiting Default/Sheddable ones (the engine-level half of the
        reference's priority tiers — the router half sheds, scheduler.go:74-90;
        preempted sequences re-queue at the very front regardless, matching
        the sim's recompute priority)."""
        seq.status = SequenceStatus.WAITING
        if seq.critical:
            # insert after the last waiting critical (stable within tier),
            # before the first non-critical fresh arrival
            idx = 0
            for i, w in enumerate(self.waiting):
                if w.critical or w.num_computed_tokens > 0 or w.num_preemptions > 0:
                    idx = i + 1
                else:
                    break
            self.waiting.insert(idx, seq)
        else:
            self.waiting.append(seq)

    def abort_seq(self, request_id: str) -> bool:
        for q in (list(self.waiting), self.running):
            for seq in q:
                if seq.request_id == request_id:
                    seq.status = SequenceStatus.FINISHED_ABORTED
                    if seq in self.running:
                        self.running.remove(seq)
                        self.block_manager.free(seq)
                    else:
                        self.waiting.remove(seq)
                    return True
        return False

    # -- gauges (the router's inputs) -----------------------------------------
    @property
    def num_waiting(self) -> int:
        return len(self.waiting)

    @property
    def num_running(self) -> int:
        return len(self.running)

    @property
    def kv_usage(self) -> float:
        return self.block_manager.usage

    def has_work(self) -> bool:
        return bool(self.waiting or self.running)

    # -- preemption -----------------------------------------------------------
    def _preempt_one(
        self,
        out: SchedulerOutput,
        exclude: Optional[Sequence] = None,
        exclude_ids: Optional[set] = None,
    ) -> bool:
        """Evict the newest running sequence (reference
        continous_batching.py:117-131 preempts newest → recompute).  Sequences
        already scheduled in this step are never victims — evicting them would
        leave a scheduled seq without blocks."""
        exclude_ids = exclude_ids or set()
        victims = [
            s for s in self.running if s is not exclude and s.seq_id not in exclude_ids
        ]
        if not victims:
            return False
        # Prefer non-critical victims; among those, evict the newest arrival.
        noncrit = [s for s in victims if not s.critical]
        pool = noncrit if noncrit else victims
        victim = max(pool, key=lambda s: s.arrival_time)
        self.running.remove(victim)
        self.block_manager.free(victim)
        victim.preempt()
        self.total_preemptions += 1
        self.waiting.appendleft(victim)
        out.preempted.append(victim)
        return True

    # -- the step builder -------------------------------------------------------
    def schedule(self) -> SchedulerOutput:
        """Build one step.

        Default (``mixed_steps``): decodes are scheduled EVERY step (one
        token per running sequence) and prefill chunks fill the remaining
        token budget — vLLM's chunked-prefill semantics.  Decode-only steps
        still replay hipGraphs; mixed steps run the decode bucket's graph on
        a side stream overlapped with the eager prefill
        (engine/model_runner.py::_execute_overlapped), so a prefill chunk no
        longer stalls every in-flight decode for a full forward (~15 ms TPOT
        spikes at serving shapes).

        ``mixed_steps=False`` keeps the reference simulation's
        prefill_or_decode alternation (continous_batching.py:173-191):
        all-prefill steps prioritised, decodes only when no prefill is
        admittable."""
        if not self.config.mixed_steps:
            out = self._schedule_prefills()
            if not out.is_empty:
                return out
            return self._schedule_decodes(out)
        out = SchedulerOutput()
        self._schedule_decodes(out)
        budget = self.config.max_num_batched_tokens - len(out.scheduled)
        if budget > 0:
            self._schedule_prefills(out, budget)
        return out

    def _schedule_decodes(self, out: SchedulerOutput) -> SchedulerOutput:
        cfg = self.config
        bm = self.block_manager
        token_budget = cfg.max_num_batched_tokens

        # Decodes: every running seq past prefill needs one more slot
        #    (the token at position total_len-1 gets its KV cached this step).
        #    Oldest sequences are served first; when blocks run out, the
        #    newest not-yet-scheduled sequence is preempted.
        scheduled_ids: set = set()
        for seq in sorted(
            [s for s in self.running if not s.in_prefill], key=lambda s: s.arrival_time
        ):
            if seq not in self.running:
                continue  # preempted earlier in this loop
            target = seq.total_len
            while not bm.can_grow(seq, target):
                if not self._preempt_one(out, exclude=seq, exclude_ids=scheduled_ids):
                    # Nothing left to evict: preempt this seq itself.
                    self.running.remove(seq)
                    bm.free(seq)
                    seq.preempt()
                    self.total_preemptions += 1
                    self.waiting.appendleft(seq)
                    out.preempted.append(seq)
                    break
            else:
                bm.grow(seq, target)
                out.scheduled.append(ScheduledSeq(seq, 1))
                scheduled_ids.add(seq.seq_id)
                token_budget -= 1
                continue
        return out

    def _schedule_prefills(
        self, out: Optional[SchedulerOutput] = None, budget: Optional[int] = None
    ) -> SchedulerOutput:
        if out is None:
            out = SchedulerOutput()
        cfg = self.config
        bm = self.block_manager
        token_budget = budget if budget is not None else cfg.max_num_batched_tokens
        # sequences preempted earlier in THIS step must not be re-admitted in
        # the same step (their blocks were freed to make room)
        preempted_now = {s.seq_id for s in out.preempted}

        # 1) Continue chunked prefills already running.
        for seq in list(self.running):
            if not seq.in_prefill or token_budget <= 0:
                continue
            chunk = seq.next_chunk_tokens(token_budget)
            if chunk <= 0:
                continue
            target = seq.num_computed_tokens + chunk
            if not bm.can_grow(seq, target):
                continue  # stay running; retry next step after frees
            bm.grow(seq, target)
            out.scheduled.append(ScheduledSeq(seq, chunk))
            token_budget -= chunk

        # 2) Admit waiting sequences (FIFO) under caps.  A batch may not need
        # more DISTINCT adapters than the engine has slots (vLLM's curr_loras
        # rule): adapter-blocked sequences are SKIPPED (not head-of-line
        # blocking — a later request with a resident adapter may proceed).
        adapters_in_use = {
            s.lora_name for s in self.running if s.lora_name
        }
        skip = 0
        while (
            skip < len(self.waiting)
            and token_budget > 0
            and len(self.running) < cfg.max_num_seqs
            and bm.usage < cfg.preempt_kv_threshold
        ):
            seq = self.waiting[skip]
            if seq.seq_id in preempted_now:
                skip += 1
                continue
            if (
                seq.lora_name
                and cfg.max_loras > 0
                and seq.lora_name not in adapters_in_use
                and len(adapters_in_use) >= cfg.max_loras
            ):
                skip += 1  # adapter slots exhausted for this batch
                continue
            # Recompute path included: a preempted seq re-prefills prompt +
            # prior output (next_chunk_tokens covers total_len).
            chunk = seq.next_chunk_tokens(token_budget)
            if chunk <= 0:
                break
            target = seq.num_computed_tokens + chunk
            if not bm.can_grow(seq, target, respect_watermark=True):
                break  # head-of-line blocking, same as vLLM FIFO admission
            del self.waiting[skip]
            if seq.lora_name:
                adapters_in_use.add(seq.lora_name)
            bm.grow(seq, target)
            seq.status = SequenceStatus.RUNNING
            if seq.first_scheduled_time is None:
                import time

                seq.first_scheduled_time = time.monotonic()
            self.running.append(seq)
            out.scheduled.append(ScheduledSeq(seq, chunk))
            token_budget -= chunk

        return out

    # -- post-step bookkeeping ---------------------------------------------------
    def finish_step(self, out: SchedulerOutput) -> List[Sequence]:
        """Called after the model step appended tokens; retires finished seqs
        and returns them."""
        finished: List[Sequence] = []
        for ss in out.scheduled:
            seq = ss.seq
            if seq.is_finished():
                continue
            if not seq.in_prefill and seq.check_stop():
                import time

                seq.status = SequenceStatus.FINISHED
                seq.finish_time = time.monotonic()
                if seq in self.running:
                    self.running.remove(seq)
                self.block_manager.free(seq)
                finished.append(seq)
        return finished
