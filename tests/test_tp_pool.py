"""Mixed-pool serving test (BASELINE config #4 shape, CPU scale-down):
one TP=2 engine + one TP=1 engine of a different "size" behind one front
door, pool_ref-scoped routing, completions on both."""

import time

import pytest
from fastapi.testclient import TestClient

from llm_instance_gateway_amd.api.types import Criticality, InferenceModel
from llm_instance_gateway_amd.router.types import LLMRequest
from llm_instance_gateway_amd.serving import EnginePool, EngineSpec, build_app

EKW = dict(
    num_kv_blocks=64,
    max_num_seqs=4,
    max_num_batched_tokens=128,
    max_model_len=128,
    max_loras=2,
)


@pytest.fixture(scope="module")
def mixed_pool():
    specs = [
        EngineSpec(
            name="small-0", model="tiny", devices=["cpu"],
            engine_kwargs=EKW, lora_rank=8, pool_ref="small-pool",
        ),
        EngineSpec(
            name="big-0", model="tiny", devices=["cpu", "cpu"],  # TP=2
            engine_kwargs=EKW, lora_rank=8, pool_ref="big-pool",
        ),
    ]
    p = EnginePool(specs=specs)
    p.start()
    p.datastore.store_model(
        InferenceModel(model_name="small-model", pool_ref="small-pool",
                       criticality=Criticality.CRITICAL)
    )
    p.datastore.store_model(
        InferenceModel(model_name="big-model", pool_ref="big-pool",
                       criticality=Criticality.CRITICAL)
    )
    yield p
    p.stop()


def test_tp2_engine_serves(mixed_pool):
    client = TestClient(build_app(mixed_pool))
    r = client.post(
        "/v1/completions",
        json={"model": "big-model", "prompt": "tensor parallel", "max_tokens": 4},
    )
    assert r.status_code == 200, r.text
    assert r.headers["x-target-engine"] == "big-0"
    assert len(r.json()["choices"][0]["token_ids"]) == 4


def test_pool_ref_scopes_routing(mixed_pool):
    """Requests whose InferenceModel names a pool only land on that pool's
    engines (mixed-pool extension; reference has one pool per deployment)."""
    for _ in range(6):
        eng = mixed_pool.route(
            LLMRequest(model="small-model", critical=True, pool_ref="small-pool")
        )
        assert eng.name == "small-0"
    for _ in range(6):
        eng = mixed_pool.route(
            LLMRequest(model="big-model", critical=True, pool_ref="big-pool")
        )
        assert eng.name == "big-0"


def test_tp2_and_tp1_coexist_under_load(mixed_pool):
    client = TestClient(build_app(mixed_pool))
    results = []
    for i in range(6):
        model = "big-model" if i % 2 else "small-model"
        r = client.post(
            "/v1/completions",
            json={"model": model, "prompt": f"req {i}", "max_tokens": 3},
        )
        results.append((model, r))
    for model, r in results:
        assert r.status_code == 200, (model, r.text)
        expect = "big-0" if model == "big-model" else "small-0"
        assert r.headers["x-target-engine"] == expect


def test_tp_engine_adapter_load(mixed_pool):
    h = mixed_pool.handles["big-0"]
    assert h.load_adapter("tp-adapter") is None
    deadline = time.monotonic() + 5
    while time.monotonic() < deadline and "tp-adapter" not in h.active_adapters:
        time.sleep(0.02)
    assert "tp-adapter" in h.active_adapters


def test_tp_engine_follower_death_detected_and_restarted():
    """A dead TP FOLLOWER rank must take the whole engine out of routing
    (the front rank's broadcast would hang); restart re-spawns the group."""
    p = EnginePool(
        specs=[
            EngineSpec(
                name="tp-x", model="tiny", devices=["cpu", "cpu"],
                engine_kwargs=EKW, lora_rank=8,
            )
        ]
    )
    p.start()  # watchdog on
    try:
        h = p.handles["tp-x"]
        assert h.alive() and len(h.procs) == 2
        h.procs[1].terminate()  # kill the follower
        deadline = time.monotonic() + 15
        while time.monotonic() < deadline:
            if not p.all_engine_metrics():
                break
            time.sleep(0.05)
        assert not p.all_engine_metrics(), "dead TP engine still routable"
        p.restart_engine("tp-x")
        deadline = time.monotonic() + 30
        while time.monotonic() < deadline:
            if p.handles["tp-x"].alive():
                break
            time.sleep(0.05)
        assert p.handles["tp-x"].alive()
        from llm_instance_gateway_amd.serving import PoolRequest

        t = p.submit("tp-x", PoolRequest(request_id="tpr-1",
                                         prompt_token_ids=[1, 2, 3],
                                         max_new_tokens=2))
        assert len(t.wait(60).output_token_ids) == 2
    finally:
        p.stop()
