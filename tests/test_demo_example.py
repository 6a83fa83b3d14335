"""examples/demo_client.py stays runnable: the same API walk, hermetic."""

import os
import sys

import pytest
from fastapi.testclient import TestClient

from llm_instance_gateway_amd.api.client import GatewayClient
from llm_instance_gateway_amd.api.types import InferenceModel
from llm_instance_gateway_amd.serving import EnginePool, build_app

sys.path.insert(
    0, os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "examples")
)


@pytest.mark.timeout(120)
def test_demo_walk(capsys):
    import demo_client

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
    try:
        pool.datastore.store_model(InferenceModel(model_name="demo"))
        gw = GatewayClient(
            "http://testserver", session=TestClient(build_app(pool))
        )
        demo_client.run(gw)
        out = capsys.readouterr().out
        assert "greedy:" in out
        assert "seeded sampling" in out
        assert "finish:" in out
        assert "adapter completion tokens: 8" in out
        assert "metric:" in out
    finally:
        pool.stop()
