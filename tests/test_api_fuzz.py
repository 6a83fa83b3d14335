"""Randomized full-surface fuzz of /v1/completions through the hermetic
stack: any combination of the documented parameters must produce a valid
response (200 with consistent fields, or a documented 4xx) — never a 500
and never a hung ticket."""

import random

import pytest
from fastapi.testclient import TestClient

from llm_instance_gateway_amd.api.types import InferenceModel
from llm_instance_gateway_amd.serving import EnginePool, build_app


@pytest.fixture(scope="module")
def fuzz_client():
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
    pool.datastore.store_model(InferenceModel(model_name="fuzz"))
    client = TestClient(build_app(pool))
    client.post("/v1/load_lora_adapter", json={"lora_name": "fz-a"})
    yield client
    pool.stop()


@pytest.mark.timeout(240)
def test_completions_parameter_fuzz(fuzz_client):
    rng = random.Random(0xF022)
    for i in range(60):
        body = {"model": "fuzz", "max_tokens": rng.randint(1, 20)}
        body["prompt"] = (
            "fz " * rng.randint(1, 30)
            if rng.random() < 0.7
            else [rng.randint(3, 250) for _ in range(rng.randint(1, 40))]
        )
        if rng.random() < 0.5:
            body["temperature"] = rng.choice([0.0, 0.3, 0.8, 1.5])
        if rng.random() < 0.3:
            body["top_p"] = rng.choice([0.5, 0.9, 1.0])
        if rng.random() < 0.3:
            body["seed"] = rng.randint(0, 10_000)
        if rng.random() < 0.3:
            body["logprobs"] = rng.randint(1, 3)
        if rng.random() < 0.3:
            body["stop"] = rng.choice(["z", ["q", "xx"], "\n"])
        if rng.random() < 0.2:
            body["n"] = rng.randint(1, 3)
        if rng.random() < 0.2:
            body["echo"] = True
        if rng.random() < 0.2:
            body["presence_penalty"] = rng.choice([0.1, 0.5])
        if rng.random() < 0.2:
            body["repetition_penalty"] = rng.choice([1.1, 1.5])
        if rng.random() < 0.3:
            body["lora"] = "fz-a"
        r = fuzz_client.post("/v1/completions", json=body, timeout=60)
        assert r.status_code == 200, (i, body, r.status_code, r.text[:200])
        out = r.json()
        n = body.get("n", 1)
        assert len(out["choices"]) == n, (i, body)
        total = 0
        for ch in out["choices"]:
            assert ch["finish_reason"] in ("length", "stop")
            toks = ch["token_ids"]
            assert 0 < len(toks) <= body["max_tokens"]
            total += len(toks)
            if body.get("logprobs"):
                lp = ch["logprobs"]
                assert len(lp["token_logprobs"]) == len(toks)
                assert all(len(t) <= body["logprobs"] for t in lp["top_logprobs"])
            if body.get("echo") and isinstance(body["prompt"], str):
                assert ch["text"].startswith(body["prompt"])
        assert out["usage"]["completion_tokens"] == total


@pytest.mark.timeout(120)
def test_streaming_fuzz(fuzz_client):
    rng = random.Random(0x5EED)
    for i in range(10):
        body = {
            "model": "fuzz", "max_tokens": rng.randint(1, 16), "stream": True,
            "prompt": "s " * rng.randint(1, 20),
        }
        if rng.random() < 0.5:
            body["stop"] = rng.choice(["zz", ["a", "bb"]])
        with fuzz_client.stream("POST", "/v1/completions", json=body) as r:
            assert r.status_code == 200
            events = [ln for ln in r.iter_lines() if ln.startswith("data: ")]
        assert events[-1] == "data: [DONE]"
        import json as _json

        final = _json.loads(events[-2][len("data: "):])
        assert final["choices"][0]["finish_reason"] in ("length", "stop")
        assert final["usage"]["completion_tokens"] >= 1
