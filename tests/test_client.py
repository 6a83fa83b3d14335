"""Client SDK against the real app (TestClient as transport)."""

import pytest
from fastapi.testclient import TestClient

from llm_instance_gateway_amd.api.client import GatewayClient, GatewayError
from llm_instance_gateway_amd.api.types import InferenceModel
from llm_instance_gateway_amd.serving import EnginePool, build_app


@pytest.fixture(scope="module")
def gw():
    pool = EnginePool(
        "tiny",
        devices=["cpu"],
        engine_kwargs=dict(
            num_kv_blocks=64, max_num_seqs=4,
            max_num_batched_tokens=128, max_model_len=128, max_loras=2,
        ),
        lora_rank=8,
    )
    pool.start(watchdog=False)
    pool.datastore.store_model(InferenceModel(model_name="m"))
    client = GatewayClient("http://testserver", session=TestClient(build_app(pool)))
    yield client
    pool.stop()


def test_client_completion_and_usage(gw):
    out = gw.completion("m", "hello", max_tokens=3)
    assert out["usage"]["completion_tokens"] == 3
    assert len(out["choices"][0]["token_ids"]) == 3


def test_client_sampling_surface(gw):
    """seed reproduces, logprobs shape present, penalties accepted."""
    a = gw.completion("m", "abc", max_tokens=4, temperature=0.7, seed=9,
                      logprobs=1, presence_penalty=0.1)
    b = gw.completion("m", "abc", max_tokens=4, temperature=0.7, seed=9,
                      logprobs=1, presence_penalty=0.1)
    assert a["choices"][0]["token_ids"] == b["choices"][0]["token_ids"]
    lp = a["choices"][0]["logprobs"]
    assert len(lp["token_logprobs"]) == 4 and len(lp["top_logprobs"]) == 4


def test_client_admin_roundtrip(gw):
    gw.register_model("extra", criticality="Sheddable")
    names = [m["model_name"] for m in gw.list_inference_models()]
    assert "extra" in names
    gw.delete_model("extra")
    assert "extra" not in [m["model_name"] for m in gw.list_inference_models()]


def test_client_adapters_engines_metrics(gw):
    gw.load_adapter("sdk-adapter")
    assert any(m["id"] == "sdk-adapter" for m in gw.models())
    gw.unload_adapter("sdk-adapter")
    assert gw.health()
    engines = gw.engines()
    assert engines and engines[0]["alive"]
    assert "vllm:num_requests_running" in gw.metrics_text()
    assert isinstance(gw.trace(n=5), list)


def test_client_error_surface(gw):
    with pytest.raises(GatewayError) as ei:
        gw.completion("no-such-model", "x")
    assert ei.value.status == 404


def test_client_streaming(gw):
    chunks = list(gw.completion_stream("m", "go", max_tokens=3))
    token_chunks = [c for c in chunks if c["choices"][0].get("token") is not None]
    assert len(token_chunks) == 3
    assert chunks[-1]["usage"]["completion_tokens"] == 3
