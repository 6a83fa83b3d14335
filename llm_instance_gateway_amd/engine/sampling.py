"""Token sampling.

Greedy / temperature / top-k / top-p over the logits of the sampled positions,
with vLLM-style presence/frequency/repetition penalties and OpenAI
``logprobs`` recording.  Per-sequence params ride `Sequence.sampling`.
(The reference delegates sampling to vLLM; this owns it natively.)

The common serving fast path (all-greedy, no penalties, no logprobs — the
benchmark workload) stays one batched argmax; every extension below is
skipped unless some sequence in the batch asks for it.
"""

from __future__ import annotations

from typing import List

import torch

from .sequence import Sequence


def _wants_extras(sp) -> bool:
    return (
        sp.logprobs > 0
        or sp.presence_penalty != 0.0
        or sp.frequency_penalty != 0.0
        or sp.repetition_penalty != 1.0
    )


def _apply_penalties(row: torch.Tensor, seq: Sequence) -> torch.Tensor:
    """vLLM semantics: repetition divides positive logits / multiplies
    negative ones for seen tokens; presence subtracts once per distinct
    output token; frequency subtracts per occurrence."""
    sp = seq.sampling
    if not seq.output_token_ids:
        return row
    seen = torch.tensor(
        sorted(set(seq.output_token_ids)), dtype=torch.int64, device=row.device
    )
    if sp.repetition_penalty != 1.0:
        vals = row[seen]
        row = row.clone()
        row[seen] = torch.where(
            vals > 0, vals / sp.repetition_penalty, vals * sp.repetition_penalty
        )
    if sp.presence_penalty != 0.0 or sp.frequency_penalty != 0.0:
        if sp.repetition_penalty == 1.0:
            row = row.clone()
        if sp.presence_penalty != 0.0:
            row[seen] -= sp.presence_penalty
        if sp.frequency_penalty != 0.0:
            counts = torch.bincount(
                torch.tensor(
                    seq.output_token_ids, dtype=torch.int64, device=row.device
                ),
                minlength=row.numel(),
            )
            row -= sp.frequency_penalty * counts.to(row.dtype)
    return row


def _record_logprobs(row: torch.Tensor, chosen: int, seq: Sequence) -> None:
    """Record the chosen token's logprob + the top-N alternatives, computed
    on the (penalised, pre-temperature) logits row in fp32."""
    n = seq.sampling.logprobs
    lp = torch.log_softmax(row.float(), dim=-1)
    top_v, top_i = torch.topk(lp, min(n, lp.numel()))
    seq.output_logprobs.append(
        (
            float(lp[chosen]),
            {int(i): float(v) for i, v in zip(top_i.tolist(), top_v.tolist())},
        )
    )


@torch.no_grad()
def sample_tokens(logits: torch.Tensor, seqs: List[Sequence]) -> List[int]:
    """logits: [N, vocab] fp32 — one row per sampling sequence."""
    assert logits.shape[0] == len(seqs)
    if all(
        s.sampling.temperature == 0.0 and not _wants_extras(s.sampling)
        for s in seqs
    ):
        return logits.argmax(dim=-1).tolist()
    out: List[int] = []
    for i, seq in enumerate(seqs):
        sp = seq.sampling
        row = logits[i]
        if _wants_extras(sp):
            row = _apply_penalties(row, seq)
        base_row = row  # pre-temperature row for logprob reporting
        if sp.temperature == 0.0:
            tok = int(row.argmax())
            if sp.logprobs > 0:
                _record_logprobs(base_row, tok, seq)
            out.append(tok)
            continue
        row = row / max(1e-5, sp.temperature)
        if sp.top_k > 0 and sp.top_k < row.numel():
            kth = torch.topk(row, sp.top_k).values[-1]
            row = row.masked_fill(row < kth, float("-inf"))
        probs = torch.softmax(row, dim=-1)
        if 0.0 < sp.top_p < 1.0:
            sorted_p, idx = probs.sort(descending=True)
            cum = sorted_p.cumsum(-1)
            keep = cum - sorted_p <= sp.top_p
            keep[0] = True
            mask = torch.zeros_like(probs, dtype=torch.bool)
            mask[idx[keep]] = True
            probs = probs.masked_fill(~mask, 0.0)
            probs = probs / probs.sum()
        gen = torch.Generator(device=probs.device)
        if sp.seed:
            # OpenAI `seed` contract: the same request + seed reproduces the
            # same tokens, independent of engine-local sequence numbering
            gen.manual_seed(sp.seed * 1_000_003 + seq.num_output_tokens)
        else:
            # unseeded: decorrelate concurrent sequences via the seq id
            gen.manual_seed(seq.seq_id * 7919 + seq.num_output_tokens)
        tok = int(torch.multinomial(probs, 1, generator=gen))
        if sp.logprobs > 0:
            _record_logprobs(base_row, tok, seq)
        out.append(tok)
    return out
