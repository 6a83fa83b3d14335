"""Token sampling.

Greedy / temperature / top-k / top-p over the logits of the sampled positions.
Per-sequence params ride `Sequence.sampling`.  (The reference delegates
sampling to vLLM; this owns it natively.)
"""

from __future__ import annotations

from typing import List

import torch

from .sequence import Sequence


@torch.no_grad()
def sample_tokens(logits: torch.Tensor, seqs: List[Sequence]) -> List[int]:
    """logits: [N, vocab] fp32 — one row per sampling sequence."""
    assert logits.shape[0] == len(seqs)
    if all(s.sampling.temperature == 0.0 for s in seqs):
        return logits.argmax(dim=-1).tolist()
    out: List[int] = []
    for i, seq in enumerate(seqs):
        sp = seq.sampling
        row = logits[i]
        if sp.temperature == 0.0:
            out.append(int(row.argmax()))
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
        out.append(int(torch.multinomial(probs, 1, generator=gen)))
    return out
