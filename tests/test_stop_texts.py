"""Decoded-text stop matching (vLLM string-level semantics).

With a subword tokenizer a stop string can tokenize differently in
generation context than standalone, so token-id tail matching misses it;
`SamplingParams.stop_texts` matches on the decoded output tail instead
(engine/sequence.py), and the front door trims the stop from the decoded
response text (frontdoor._trim_stop_text).
"""

import pytest

from llm_instance_gateway_amd.engine.sequence import SamplingParams, Sequence
from llm_instance_gateway_amd.serving.frontdoor import _trim_stop_text
from llm_instance_gateway_amd.utils import tokenizer as tok


class _Enc:
    def __init__(self, ids):
        self.ids = ids


class FakeSubwordTok:
    """Backend where 'STOP' encodes standalone to ONE token (42) but the
    model emits it as two tokens 7+8 ('ST' + 'OP')."""

    VOCAB = {7: "ST", 8: "OP", 42: "STOP", 5: "hello ", 6: "world"}

    def encode(self, text):
        if text == "STOP":
            return _Enc([42])
        return _Enc([5])

    def decode(self, ids):
        return "".join(self.VOCAB.get(i, "?") for i in ids)


@pytest.fixture
def fake_backend():
    tok.set_backend(FakeSubwordTok())
    try:
        yield
    finally:
        tok.set_backend(None)


def _step(seq, token):
    seq.output_token_ids.append(token)
    return seq.check_stop()


def test_stop_text_matches_when_token_ids_miss(fake_backend):
    sp = SamplingParams(
        max_new_tokens=10, stop_sequences=[[42]], stop_texts=["STOP"]
    )
    seq = Sequence([1, 2, 3], sp)
    assert not _step(seq, 5)   # "hello "
    assert not _step(seq, 7)   # "ST" — no match yet
    assert _step(seq, 8)       # "ST"+"OP" = "STOP" in decoded tail
    assert seq.finish_reason == "stop"
    # token ids are NOT trimmed on the text path (front door trims the text)
    assert seq.output_token_ids == [5, 7, 8]


def test_stop_token_ids_still_trim_first(fake_backend):
    sp = SamplingParams(
        max_new_tokens=10, stop_sequences=[[42]], stop_texts=["STOP"]
    )
    seq = Sequence([1], sp)
    assert not _step(seq, 5)
    assert _step(seq, 42)      # exact token match wins and trims
    assert seq.finish_reason == "stop"
    assert seq.output_token_ids == [5]


def test_stop_texts_byte_codec_equivalent():
    """Under the byte codec (no backend) text matching agrees with the
    token-id path — the ids ARE the bytes."""
    stop_ids = tok.encode("END", add_bos=False)
    sp = SamplingParams(max_new_tokens=20, stop_texts=["END"])
    seq = Sequence([1], sp)
    for t in tok.encode("abc", add_bos=False):
        assert not _step(seq, t)
    *head, last = stop_ids
    for t in head:
        assert not _step(seq, t)
    assert _step(seq, last)
    assert seq.finish_reason == "stop"


def test_trim_stop_text():
    assert _trim_stop_text("hello STOP tail", "STOP") == "hello "
    assert _trim_stop_text("hello", "STOP") == "hello"
    assert _trim_stop_text("a END b STOP", ["STOP", "END"]) == "a "
    assert _trim_stop_text("text", None) == "text"
    assert _trim_stop_text("", "x") == ""


def test_no_stop_texts_zero_cost_path():
    sp = SamplingParams(max_new_tokens=4)
    seq = Sequence([1], sp)
    assert not _step(seq, 9)
    assert not _step(seq, 9)
    assert not _step(seq, 9)
    assert _step(seq, 9)  # length
    assert seq.finish_reason == "length"
