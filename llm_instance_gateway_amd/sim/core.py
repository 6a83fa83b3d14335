"""Discrete-event simulation of the gateway + engine fleet.

Reference: `simulations/llm_ig_simulation/src/` (SimPy).  Re-built on a plain
heapq event loop (no SimPy dependency in this image) with the reference's
calibrated behavior:

  * latency model (`constants.py:1-9`):
      prefill = max(0.04, 6.769e-5·tok + 0.01969) s
      decode  = 5.353e-7·KV_tokens + 0.014 + 1.026e-4·batch s
  * KV pool 2810 blocks × 16 tokens − 512 (`constants.py:13-15`)
  * admission caps: 512 prefill-batch tokens, 256 sequences, preempt to
    recompute above 90 % KV (`constants.py:11,16-18`)
  * LoRA adapters consume KV budget on first use
    (`continous_batching.py:94-97`, `LORA_DICT constants.py:21`)
  * the actor loop prefers prefill, else evict-then-decode
    (`continous_batching.py:173-191`)

This is the executable spec for the real engine's continuous-batching
scheduler and the plumbing harness for BASELINE config #1.
"""

from __future__ import annotations

import heapq
import itertools
from dataclasses import dataclass
from typing import Callable, Dict, List, Optional, Set

# Latency model constants (reference constants.py:1-9)
PREFILL_LATENCY_CONST_1 = 0.00006769375513
PREFILL_LATENCY_CONST_0 = 0.01969
PREFILL_LATENCY_CONST_MIN = 0.04
DECODE_LATENCY_CONST_BATCH = 0.0001026494433
DECODE_LATENCY_CONST_1 = 0.0000005353485087
DECODE_LATENCY_CONST_0 = 0.014

# Batching caps (reference constants.py:11-18)
MAX_NUM_BATCH_TOKENS = 512
TOTAL_NUM_GPU_BLOCKS = 2810
TOKENS_PER_BLOCK = 16
MAX_NUM_TOKENS_ALLOWED = TOTAL_NUM_GPU_BLOCKS * TOKENS_PER_BLOCK - MAX_NUM_BATCH_TOKENS
MAX_KV_PERC_BEFORE_RECOMPUTE = 0.9
MAX_NUM_SEQ = 256

# KV cost of named adapters (reference constants.py:21)
DEFAULT_LORA_COST = 1600


def prefill_delay(tokens: int) -> float:
    return max(
        PREFILL_LATENCY_CONST_MIN,
        tokens * PREFILL_LATENCY_CONST_1 + PREFILL_LATENCY_CONST_0,
    )


def decode_delay(kv_tokens: int, batch: int) -> float:
    return (
        kv_tokens * DECODE_LATENCY_CONST_1
        + DECODE_LATENCY_CONST_0
        + DECODE_LATENCY_CONST_BATCH * batch
    )


class SimEnv:
    """Minimal event loop: schedule(delay, fn) + run(until)."""

    def __init__(self) -> None:
        self.now = 0.0
        self._heap: List = []
        self._counter = itertools.count()

    def schedule(self, delay: float, fn: Callable[[], None]) -> None:
        heapq.heappush(self._heap, (self.now + delay, next(self._counter), fn))

    def run(self, until: float) -> None:
        while self._heap and self._heap[0][0] <= until:
            t, _, fn = heapq.heappop(self._heap)
            self.now = t
            fn()
        self.now = until


@dataclass
class SimRequest:
    """Per-request record (reference request.py:3-27)."""

    id: int
    arrival_time: float
    input_size: int
    output_size: int
    target_latency_per_token: float  # TPOT SLO tier
    lora: Optional[str] = None
    critical: bool = False
    # runtime state
    output_size_remaining: int = 0
    recompute_count: int = 0
    start_prefill_time: Optional[float] = None
    end_prefill_time: Optional[float] = None   # == TTFT instant
    end_decode_time: Optional[float] = None
    queued_time: float = 0.0

    def __post_init__(self) -> None:
        self.output_size_remaining = self.output_size

    # -- derived metrics -----------------------------------------------------
    @property
    def ttft(self) -> Optional[float]:
        if self.end_prefill_time is None:
            return None
        return self.end_prefill_time - self.arrival_time

    @property
    def e2e_latency(self) -> Optional[float]:
        if self.end_decode_time is None:
            return None
        return self.end_decode_time - self.arrival_time

    @property
    def tpot(self) -> Optional[float]:
        if self.end_decode_time is None or self.end_prefill_time is None:
            return None
        return (self.end_decode_time - self.end_prefill_time) / max(
            1, self.output_size - 1
        )

    @property
    def within_slo(self) -> bool:
        t = self.tpot
        return t is not None and t <= self.target_latency_per_token


class SimEngine:
    """One simulated per-GPU engine (reference LLMActor, llmactor.py:8-76).

    Queues: prefill (FIFO), recompute (priority: oldest seq id first),
    decode (batch), decoded (finished).  The actor loop runs one batch step
    at a time, sleeping the modeled kernel latency.
    """

    def __init__(self, env: SimEnv, engine_id: int,
                 lora_cost: Optional[Dict[str, int]] = None) -> None:
        self.env = env
        self.id = engine_id
        self.prefill_q: List[SimRequest] = []
        self.recompute_q: List[SimRequest] = []   # kept sorted oldest-first
        self.decode: List[SimRequest] = []
        self.decoded: List[SimRequest] = []
        self.lora_loaded: Set[str] = set()
        self.lora_cost = dict(lora_cost or {})
        self.max_num_tokens_allowed = MAX_NUM_TOKENS_ALLOWED
        self._running = False
        self.busy_until = 0.0

    # -- gauges (the router's scheduler inputs) ------------------------------
    def tokens_in_decode(self) -> int:
        return sum(r.input_size + r.output_size - r.output_size_remaining
                   for r in self.decode)

    def pending_tokens(self) -> int:
        """Total work queued but not yet resident (prefill + recompute)."""
        return sum(r.input_size + r.output_size - r.output_size_remaining
                   for r in self.prefill_q + self.recompute_q)

    def kv_usage(self) -> float:
        return self.tokens_in_decode() / self.max_num_tokens_allowed

    def expected_kv_after_prefill(self) -> float:
        """Reference get_min_expected_num_tokens_in_kvcache_after_prefill
        (the 'least' policy's signal)."""
        nxt = 0
        pool = self.recompute_q + self.prefill_q
        if pool:
            r = pool[0]
            nxt = r.input_size + r.output_size - r.output_size_remaining
        return (self.tokens_in_decode() + nxt) / self.max_num_tokens_allowed

    def queue_len(self) -> int:
        return len(self.prefill_q) + len(self.recompute_q)

    def is_saturated(self, max_saturation: float = MAX_KV_PERC_BEFORE_RECOMPUTE) -> bool:
        return self.expected_kv_after_prefill() >= max_saturation

    # -- intake --------------------------------------------------------------
    def submit(self, req: SimRequest) -> None:
        self.prefill_q.append(req)
        self._wake()

    def _wake(self) -> None:
        if not self._running:
            self._running = True
            self.env.schedule(0.0, self._step)

    # -- the actor loop (reference prefill_or_decode) ------------------------
    def _can_prefill(self) -> bool:
        pool = self.recompute_q or self.prefill_q
        if not pool:
            return False
        r = pool[0]
        if len(self.decode) + 1 > MAX_NUM_SEQ:
            return False
        if r.input_size > MAX_NUM_BATCH_TOKENS:
            return False
        if self.tokens_in_decode() / self.max_num_tokens_allowed >= MAX_KV_PERC_BEFORE_RECOMPUTE:
            return False
        return True

    def _fetch_prefill_batch(self) -> List[SimRequest]:
        """Admission caps per reference fetch_prefill_items
        (continous_batching.py:45-85): recompute first, then fresh prefills;
        batch bounded by MAX_NUM_SEQ / MAX_NUM_BATCH_TOKENS / KV headroom."""
        items: List[SimRequest] = []
        batch_tokens = 0
        for q in (self.recompute_q, self.prefill_q):
            while q:
                r = q[0]
                r_len = r.input_size + r.output_size - r.output_size_remaining
                if (
                    len(self.decode) + len(items) + 1 > MAX_NUM_SEQ
                    or batch_tokens + r.input_size > MAX_NUM_BATCH_TOKENS
                    or (batch_tokens + len(items) + self.tokens_in_decode())
                    / self.max_num_tokens_allowed
                    >= MAX_KV_PERC_BEFORE_RECOMPUTE
                ):
                    break
                batch_tokens += r_len
                items.append(q.pop(0))
        return items

    def _evict_for_recompute(self) -> None:
        """Preempt newest decode items while projected KV exceeds the
        threshold (reference remove_from_decode_store + should_recompute)."""
        while (
            self.decode
            and (self.tokens_in_decode() + len(self.decode))
            / self.max_num_tokens_allowed
            > MAX_KV_PERC_BEFORE_RECOMPUTE
        ):
            victim = self.decode.pop()  # newest
            victim.recompute_count += 1
            self.recompute_q.append(victim)
            self.recompute_q.sort(key=lambda r: r.id)  # oldest-first priority

    def _step(self) -> None:
        if not (self.prefill_q or self.recompute_q or self.decode):
            self._running = False
            return
        if self._can_prefill():
            items = self._fetch_prefill_batch()
            tokens = sum(r.input_size + r.output_size - r.output_size_remaining
                         for r in items)
            delay = prefill_delay(tokens)
            for r in items:
                if r.lora is not None and r.lora not in self.lora_loaded:
                    # adapter residency shrinks the KV budget
                    self.lora_loaded.add(r.lora)
                    self.max_num_tokens_allowed -= self.lora_cost.get(
                        r.lora, DEFAULT_LORA_COST
                    )
                if r.start_prefill_time is None:
                    r.start_prefill_time = self.env.now
                    r.end_prefill_time = self.env.now + delay  # first token
                r.output_size_remaining -= 1
                if r.output_size_remaining == 0:
                    r.end_decode_time = self.env.now + delay
                    self.decoded.append(r)
                else:
                    self.decode.append(r)
        else:
            self._evict_for_recompute()
            if not self.decode:
                self.env.schedule(0.001, self._step)
                return
            kv = self.tokens_in_decode()
            delay = decode_delay(kv, len(self.decode))
            still: List[SimRequest] = []
            for r in self.decode:
                r.output_size_remaining -= 1
                if r.output_size_remaining == 0:
                    r.end_decode_time = self.env.now + delay
                    self.decoded.append(r)
                else:
                    still.append(r)
            self.decode = still
        self.busy_until = self.env.now + delay
        self.env.schedule(delay, self._step)
