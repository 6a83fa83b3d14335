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


def test_tokenizer_real_backend_roundtrip(tmp_path):
    """A real HF vocabulary plugs into the front-door codec (the vLLM API
    surface the reference pins); built in-test since no assets ship offline."""
    from tokenizers import Tokenizer, models, pre_tokenizers, trainers

    from llm_instance_gateway_amd.utils import tokenizer as tok

    t = Tokenizer(models.BPE(unk_token="<unk>"))
    t.pre_tokenizer = pre_tokenizers.Whitespace()
    trainer = trainers.BpeTrainer(special_tokens=["<unk>"], vocab_size=200)
    t.train_from_iterator(
        ["hello world", "hello gateway", "routing requests to engines"] * 8,
        trainer,
    )
    path = tmp_path / "tokenizer.json"
    t.save(str(path))
    try:
        assert tok.load_backend(str(path))
        assert tok.backend_name() == "Tokenizer"
        ids = tok.encode("hello world")
        assert ids and ids != [tok.BOS_ID] + [b + tok.OFFSET for b in b"hello world"]
        assert "hello" in tok.decode(ids)
    finally:
        tok.set_backend(None)  # restore byte codec for other tests
    # byte codec restored
    assert tok.decode(tok.encode("abc", add_bos=False)) == "abc"


def test_chat_template_shape():
    from llm_instance_gateway_amd.utils import tokenizer as tok

    out = tok.apply_chat_template(
        [{"role": "user", "content": "hi"}], add_generation_prompt=True
    )
    assert "<|start_header_id|>user<|end_header_id|>" in out
    assert out.endswith("assistant<|end_header_id|>\n\n")
