"""Remote-router mode: the reference's deployment shape — standalone router
over remote model servers (scrape + schedule + forward).  Two stdlib HTTP
servers play the engine fleet; assertions mirror the hermetic e2e's contract
(target header, body model rewrite, 429 shed)."""

import json
import threading
from http.server import BaseHTTPRequestHandler, HTTPServer

import pytest
from fastapi.testclient import TestClient

from llm_instance_gateway_amd.api.types import Criticality, InferenceModel, TargetModel
from llm_instance_gateway_amd.serving.remote_router import build_remote_router_app


def make_engine_server(name, waiting, kv, adapters):
    """A canned vLLM-ish server: /metrics + /v1/completions echo."""

    class H(BaseHTTPRequestHandler):
        requests_seen = []

        def log_message(self, *a):
            pass

        def do_GET(self):
            if self.path == "/metrics":
                body = (
                    f"vllm:num_requests_running 1\n"
                    f"vllm:num_requests_waiting {waiting}\n"
                    f"vllm:gpu_cache_usage_perc {kv}\n"
                    f'vllm:lora_requests_info{{running_lora_adapters="{adapters}",max_lora="4"}} 1\n'
                ).encode()
                self.send_response(200)
            else:
                body, _ = b"{}", self.send_response(200)
            self.send_header("Content-Length", str(len(body)))
            self.end_headers()
            self.wfile.write(body)

        def do_POST(self):
            n = int(self.headers.get("Content-Length", 0))
            payload = json.loads(self.rfile.read(n) or b"{}")
            H.requests_seen.append((self.path, payload))
            body = json.dumps(
                {
                    "id": f"cmpl-{name}",
                    "model": payload.get("model"),
                    "choices": [{"index": 0, "text": f"from-{name}"}],
                    "usage": {"prompt_tokens": 3, "completion_tokens": 2,
                              "total_tokens": 5},
                }
            ).encode()
            self.send_response(200)
            self.send_header("Content-Type", "application/json")
            self.send_header("Content-Length", str(len(body)))
            self.end_headers()
            self.wfile.write(body)

    srv = HTTPServer(("127.0.0.1", 0), H)
    threading.Thread(target=srv.serve_forever, daemon=True).start()
    return srv, H


@pytest.fixture(scope="module")
def stack():
    # engine-0: idle, has lora-a; engine-1: deeply queued
    s0, h0 = make_engine_server("e0", waiting=0, kv=0.1, adapters="lora-a")
    s1, h1 = make_engine_server("e1", waiting=90, kv=0.9, adapters="")
    addrs = [f"127.0.0.1:{s.server_address[1]}" for s in (s0, s1)]
    models = [
        InferenceModel(model_name="my-model", criticality=Criticality.CRITICAL),
        InferenceModel(
            model_name="alias-model",
            criticality=Criticality.CRITICAL,
            target_models=[TargetModel(name="lora-a", weight=100)],
        ),
        InferenceModel(model_name="shed-model", criticality=Criticality.SHEDDABLE),
    ]
    app = build_remote_router_app(addrs, models, start_refresh=False)
    app.state.provider.refresh_metrics_once()
    client = TestClient(app)
    yield client, h0, h1
    s0.shutdown()
    s1.shutdown()


def test_routes_to_low_queue_engine(stack):
    client, h0, h1 = stack
    r = client.post(
        "/v1/completions",
        json={"model": "my-model", "prompt": "hello", "max_tokens": 4},
    )
    assert r.status_code == 200, r.text
    assert r.headers["x-target-engine"] == "engine-0"  # low queue wins
    assert r.json()["choices"][0]["text"] == "from-e0"
    assert r.json()["usage"]["total_tokens"] == 5


def test_body_model_rewritten_to_adapter(stack):
    """Reference request.go:62-70: the forwarded body carries the resolved
    target model (adapter), not the client alias."""
    client, h0, h1 = stack
    r = client.post(
        "/v1/completions",
        json={"model": "alias-model", "prompt": "x", "max_tokens": 2},
    )
    assert r.status_code == 200
    assert h0.requests_seen[-1][1]["model"] == "lora-a"


def test_unknown_model_404(stack):
    client, _, _ = stack
    r = client.post("/v1/completions", json={"model": "nope", "prompt": "x"})
    assert r.status_code == 404


def test_sheddable_not_dropped_with_capacity(stack):
    """engine-0 has capacity (queue 0 < 5, kv 0.1 < 0.8) -> sheddable served."""
    client, _, _ = stack
    r = client.post(
        "/v1/completions", json={"model": "shed-model", "prompt": "x"}
    )
    assert r.status_code == 200


def test_metrics_page_aggregates_fleet(stack):
    client, _, _ = stack
    text = client.get("/metrics").text
    assert 'vllm:num_requests_waiting{engine="engine-1"} 90' in text
    assert "gateway_requests_total" in text
    assert client.get("/health").status_code == 200


def test_remote_router_admin_models(stack):
    client, _, _ = stack
    client.post("/admin/models", json={"model_name": "late-model",
                                       "criticality": "Critical"})
    names = [m["model_name"] for m in client.get("/admin/models").json()]
    assert "late-model" in names
    r = client.post("/v1/completions", json={"model": "late-model", "prompt": "x"})
    assert r.status_code == 200
    client.delete("/admin/models/late-model")
    names = [m["model_name"] for m in client.get("/admin/models").json()]
    assert "late-model" not in names


def test_remote_router_v1_models(stack):
    client, _, _ = stack
    ids = [m["id"] for m in client.get("/v1/models").json()["data"]]
    assert "my-model" in ids
