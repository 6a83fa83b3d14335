"""Utils coverage: byte tokenizer round-trip, span log, tracing toggles."""

from llm_instance_gateway_amd.utils import tokenizer
from llm_instance_gateway_amd.utils.tracing import SpanLog, gpu_range, set_gpu_tracing


def test_tokenizer_roundtrip():
    for text in ("hello world", "", "unicode: ü€漢字", "a" * 1000):
        ids = tokenizer.encode(text)
        assert ids[0] == tokenizer.BOS_ID
        assert tokenizer.decode(ids) == text


def test_tokenizer_accepts_token_ids():
    assert tokenizer.to_token_ids([5, 6, 7]) == [5, 6, 7]
    assert tokenizer.to_token_ids("hi") == tokenizer.encode("hi")


def test_tokenizer_ids_bounded():
    ids = tokenizer.encode("ÿ\x00")
    assert all(0 <= i < 512 for i in ids)  # fits the tiny model's vocab


def test_span_log_bounded_and_filterable():
    log = SpanLog(capacity=4)
    for i in range(10):
        log.event(f"req-{i % 2}", "arrived", n=i)
    evs = log.tail()
    assert len(evs) == 4  # ring bound
    only0 = log.tail(request_id="req-0")
    assert all(e["request_id"] == "req-0" for e in only0)
    assert only0[-1]["n"] == 8


def test_gpu_range_noop_on_cpu():
    set_gpu_tracing(True)
    try:
        with gpu_range("x"):
            pass  # must not raise without a GPU
    finally:
        set_gpu_tracing(False)
