"""Sampling extensions: logprobs recording + presence/frequency/repetition
penalties (vLLM semantics) + the OpenAI response shape through the front
door."""

import math

import torch

from llm_instance_gateway_amd.engine.sampling import sample_tokens
from llm_instance_gateway_amd.engine.sequence import SamplingParams, Sequence


def _seq(**kw) -> Sequence:
    return Sequence([1, 2], SamplingParams(**kw))


def test_greedy_fast_path_unchanged():
    logits = torch.tensor([[0.0, 5.0, 1.0], [9.0, 0.0, 0.0]])
    seqs = [_seq(), _seq()]
    assert sample_tokens(logits, seqs) == [1, 0]
    assert seqs[0].output_logprobs == []


def test_logprobs_recorded_greedy():
    logits = torch.tensor([[0.0, 5.0, 1.0]])
    s = _seq(logprobs=2)
    toks = sample_tokens(logits, [s])
    assert toks == [1]
    (chosen_lp, top), = s.output_logprobs
    ref = torch.log_softmax(logits[0], dim=-1)
    assert math.isclose(chosen_lp, float(ref[1]), rel_tol=1e-5)
    assert set(top) == {1, 2}  # top-2 alternatives
    assert math.isclose(top[1], float(ref[1]), rel_tol=1e-5)


def test_repetition_penalty_discourages_seen():
    # token 1 dominates but was already emitted; a large penalty flips the pick
    logits = torch.tensor([[0.0, 2.0, 1.9]])
    s = _seq(repetition_penalty=4.0)
    s.output_token_ids.append(1)
    assert sample_tokens(logits.clone(), [s] )[0] == 2
    # neutral penalty keeps the argmax
    s2 = _seq()
    s2.output_token_ids.append(1)
    assert sample_tokens(logits.clone(), [s2])[0] == 1


def test_frequency_penalty_scales_with_count():
    logits = torch.tensor([[0.0, 1.0, 0.7]])
    s = _seq(frequency_penalty=0.2)
    s.output_token_ids.extend([1, 1])  # 2 occurrences -> -0.4
    assert sample_tokens(logits.clone(), [s])[0] == 2
    s2 = _seq(frequency_penalty=0.2)
    s2.output_token_ids.extend([1])  # -0.2 only: 0.8 > 0.7
    assert sample_tokens(logits.clone(), [s2])[0] == 1


def test_presence_penalty_once_per_distinct_token():
    logits = torch.tensor([[0.0, 1.0, 0.9]])
    s = _seq(presence_penalty=0.15)
    s.output_token_ids.extend([1, 1, 1])  # subtracted ONCE: 0.85 < 0.9
    assert sample_tokens(logits.clone(), [s])[0] == 2


def test_penalties_do_not_mutate_input_logits():
    logits = torch.tensor([[0.0, 1.0, 0.5]])
    keep = logits.clone()
    s = _seq(presence_penalty=1.0, frequency_penalty=1.0, repetition_penalty=2.0)
    s.output_token_ids.extend([1, 2])
    sample_tokens(logits, [s])
    assert torch.equal(logits, keep)


