"""Sequence state for the continuous-batching engine.

A sequence is one in-flight request on one GPU engine: prompt tokens, emitted
tokens, paged-KV block table, status, and the latency bookkeeping (arrival,
first-token time) that feeds the TTFT/TPOT histograms and the goodput metric
(the reference tracks the same per-request record in its simulation —
`simulations/llm_ig_simulation/src/request.py:3-27`).
"""

from __future__ import annotations

import enum
import itertools
import time
from dataclasses import dataclass, field
from typing import List, Optional

_seq_counter = itertools.count()


class SequenceStatus(enum.Enum):
    WAITING = "waiting"
    RUNNING = "running"
    PREEMPTED = "preempted"  # reference sim: recompute_store
    FINISHED = "finished"
    FINISHED_ABORTED = "aborted"


@dataclass
class SamplingParams:
    max_new_tokens: int = 128
    temperature: float = 0.0  # 0 => greedy
    top_p: float = 1.0
    top_k: int = 0
    ignore_eos: bool = True  # synthetic workloads have no meaningful EOS
    eos_token_id: int = -1
    seed: int = 0
    # OpenAI `stop`: token-id sequences; generation halts when the output
    # tail matches one, and the matched tail is trimmed from the output.
    # Token-id matching equals string matching under the front door's
    # byte-level codec (utils/tokenizer.py: one byte == one id, context-free).
    stop_sequences: List[List[int]] = field(default_factory=list)
    # String-level stops (vLLM semantics): with a subword tokenizer a stop
    # string can tokenize differently in generation context, so the engine
    # ALSO matches stops on the decoded output tail (utils.tokenizer.decode
    # of a bounded suffix).  The front door trims the stop text from the
    # decoded response; token counts are not trimmed on this path.
    stop_texts: List[str] = field(default_factory=list)
    # OpenAI `logprobs`: when > 0, record the chosen token's logprob and the
    # top-N alternatives per step (engine/sampling.py fills
    # Sequence.output_logprobs).  0 = off (no cost on the sampling path).
    logprobs: int = 0
    # vLLM-style penalties, applied to the logits row before sampling;
    # neutral values are skipped entirely.
    presence_penalty: float = 0.0   # subtract once per already-present token
    frequency_penalty: float = 0.0  # subtract per occurrence
    repetition_penalty: float = 1.0  # divide positive / multiply negative


class Sequence:
    def __init__(
        self,
        prompt_token_ids: List[int],
        sampling: Optional[SamplingParams] = None,
        lora_name: str = "",
        request_id: str = "",
        critical: bool = False,
        arrival_time: Optional[float] = None,
    ) -> None:
        self.seq_id: int = next(_seq_counter)
        self.request_id = request_id or f"req-{self.seq_id}"
        self.prompt_token_ids = list(prompt_token_ids)
        self.output_token_ids: List[int] = []
        # per-step logprob records (only when sampling.logprobs > 0):
        # [(chosen_logprob, {top_token_id: logprob, ...}), ...]
        self.output_logprobs: List = []
        self.sampling = sampling or SamplingParams()
        self.lora_name = lora_name
        self.critical = critical
        self.status = SequenceStatus.WAITING
        # Number of tokens (prompt + prior output) whose KV is computed and
        # cached.  A step processes tokens [num_computed, num_computed+n);
        # when the processed range reaches total_len, the step samples a new
        # token from the last logit.  Decode invariant between steps:
        # num_computed == total_len - 1.
        self.num_computed_tokens = 0
        # Paged KV block table (block ids owned by this sequence).
        self.block_table: List[int] = []
        # Latency bookkeeping.
        self.arrival_time = arrival_time if arrival_time is not None else time.monotonic()
        self.first_scheduled_time: Optional[float] = None
        self.first_token_time: Optional[float] = None
        self.finish_time: Optional[float] = None
        self.num_preemptions = 0
        self.finish_reason: str = "length"

    # -- lengths ------------------------------------------------------------
    @property
    def prompt_len(self) -> int:
        return len(self.prompt_token_ids)

    @property
    def num_output_tokens(self) -> int:
        return len(self.output_token_ids)

    @property
    def total_len(self) -> int:
        return self.prompt_len + self.num_output_tokens

    @property
    def in_prefill(self) -> bool:
        """More than one token still unprocessed ⇒ the next step for this
        sequence is a (chunked) prefill, not a single-token decode.  Covers
        both fresh prompts and recompute-after-preemption (where prior output
        tokens also need their KV rebuilt)."""
        return self.total_len - self.num_computed_tokens > 1

    @property
    def all_token_ids(self) -> List[int]:
        return self.prompt_token_ids + self.output_token_ids

    def next_chunk_tokens(self, budget: int) -> int:
        """How many tokens the next prefill chunk would process (the final
        chunk includes the last known token, whose logit is sampled)."""
        return min(budget, self.total_len - self.num_computed_tokens)

    # -- transitions ---------------------------------------------------------
    def append_token(self, token_id: int) -> None:
        self.output_token_ids.append(token_id)
        if self.first_token_time is None:
            self.first_token_time = time.monotonic()

    def is_finished(self) -> bool:
        return self.status in (SequenceStatus.FINISHED, SequenceStatus.FINISHED_ABORTED)

    def check_stop(self) -> bool:
        s = self.sampling
        if self.num_output_tokens >= s.max_new_tokens:
            self.finish_reason = "length"
            return True
        if (
            not s.ignore_eos
            and s.eos_token_id >= 0
            and self.output_token_ids
            and self.output_token_ids[-1] == s.eos_token_id
        ):
            self.finish_reason = "stop"
            return True
        for stop in s.stop_sequences:
            n = len(stop)
            if n and self.num_output_tokens >= n and self.output_token_ids[-n:] == stop:
                # OpenAI semantics: the stop text itself is not returned
                del self.output_token_ids[-n:]
                self.finish_reason = "stop"
                return True
        if s.stop_texts and self.output_token_ids:
            from ..utils import tokenizer as _tok

            # A stop of L chars spans at most L tokens (every token decodes
            # to >= 1 char); +8 slack for multi-char boundary tokens.
            longest = max(len(t) for t in s.stop_texts)
            ntail = min(self.num_output_tokens, longest + 8)
            tail = _tok.decode(self.output_token_ids[-ntail:])
            for t in s.stop_texts:
                if t and t in tail:
                    self.finish_reason = "stop"
                    return True
        return False

    def preempt(self) -> None:
        """Preemption-by-recompute (reference continous_batching.py:117-131):
        drop all cached KV; the sequence re-prefills prompt+outputs later."""
        self.status = SequenceStatus.PREEMPTED
        self.num_computed_tokens = 0
        self.block_table = []
        self.num_preemptions += 1

    # -- SLO accounting -------------------------------------------------------
    def ttft(self) -> Optional[float]:
        if self.first_token_time is None:
            return None
        return self.first_token_time - self.arrival_time

    def tpot(self) -> Optional[float]:
        if self.finish_time is None or self.first_token_time is None:
            return None
        n = max(1, self.num_output_tokens - 1)
        return (self.finish_time - self.first_token_time) / n

    def __repr__(self) -> str:  # pragma: no cover
        return (
            f"Sequence(id={self.seq_id}, status={self.status.value}, "
            f"prompt={self.prompt_len}, out={self.num_output_tokens}, "
            f"computed={self.num_computed_tokens}, lora={self.lora_name!r})"
        )
