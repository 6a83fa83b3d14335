"""Remote-mode tests: HTTP metric scraping + adapter actuation against a
local stdlib HTTP server standing in for a remote engine (the reference's
deployment shape: router scrapes pods over HTTP, provider.go:134-179)."""

import json
import threading
from http.server import BaseHTTPRequestHandler, HTTPServer

import pytest

from llm_instance_gateway_amd.router.datastore import Datastore
from llm_instance_gateway_amd.router.provider import Provider
from llm_instance_gateway_amd.router.remote import HTTPMetricsSource, RemoteAdapterActuator
from llm_instance_gateway_amd.router.types import Engine, EngineMetrics

METRICS_PAGE = """\
vllm:num_requests_running 3
vllm:num_requests_waiting 7
vllm:gpu_cache_usage_perc 0.42
vllm:lora_requests_info{running_lora_adapters="a1,a2",max_lora="4"} 100
"""


class _Handler(BaseHTTPRequestHandler):
    calls = []

    def log_message(self, *a):  # quiet
        pass

    def do_GET(self):
        if self.path == "/metrics":
            body = METRICS_PAGE.encode()
            self.send_response(200)
        elif self.path == "/health":
            body = b"{}"
            self.send_response(200)
        elif self.path == "/v1/models":
            body = json.dumps(
                {"data": [{"id": "base"}, {"id": "a1"}, {"id": "a2"}]}
            ).encode()
            self.send_response(200)
        else:
            body = b"nope"
            self.send_response(404)
        self.send_header("Content-Length", str(len(body)))
        self.end_headers()
        self.wfile.write(body)

    def do_POST(self):
        n = int(self.headers.get("Content-Length", 0))
        payload = json.loads(self.rfile.read(n) or b"{}")
        _Handler.calls.append((self.path, payload))
        body = b"ok"
        self.send_response(200)
        self.send_header("Content-Length", str(len(body)))
        self.end_headers()
        self.wfile.write(body)


@pytest.fixture(scope="module")
def fake_server():
    srv = HTTPServer(("127.0.0.1", 0), _Handler)
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    yield f"127.0.0.1:{srv.server_address[1]}"
    srv.shutdown()


def test_http_metrics_source(fake_server):
    src = HTTPMetricsSource()
    m = src.fetch_metrics(Engine("e0", address=fake_server), EngineMetrics())
    assert m.running_queue_size == 3
    assert m.waiting_queue_size == 7
    assert m.kv_cache_usage_percent == pytest.approx(0.42)
    assert sorted(m.active_models) == ["a1", "a2"]
    assert m.max_active_models == 4


def test_provider_refresh_mode_with_http_source(fake_server):
    """The full reference loop: datastore engines -> fan-out scrape ->
    snapshot (provider.go:60-179), minus the daemon threads."""
    ds = Datastore(engines=[Engine("e0", address=fake_server)])
    p = Provider(HTTPMetricsSource(), ds)
    p.init(start_loops=False)
    states = p.all_engine_metrics()
    assert len(states) == 1
    assert states[0].metrics.waiting_queue_size == 7


def test_scrape_error_keeps_last_known(fake_server):
    ds = Datastore(
        engines=[Engine("e0", address=fake_server), Engine("bad", address="127.0.0.1:1")]
    )
    p = Provider(HTTPMetricsSource(timeout_s=0.3), ds)
    p.refresh_engines_once()
    errs = p.refresh_metrics_once(timeout_s=3.0)
    assert len(errs) == 1  # the dead engine
    by_name = {s.engine.name: s for s in p.all_engine_metrics()}
    assert by_name["e0"].metrics.waiting_queue_size == 7
    assert by_name["bad"].metrics.waiting_queue_size == 0  # zero-value retained


def test_remote_adapter_actuator(fake_server):
    act = RemoteAdapterActuator(f"http://{fake_server}")
    assert act.wait_healthy(timeout_s=5, interval_s=0.1)
    assert act.list_models() == ["base", "a1", "a2"]
    assert act.load("new-adapter", "/weights/x.pt") is None
    assert act.unload("a2") is None
    assert ("/v1/load_lora_adapter", {
        "lora_name": "new-adapter", "lora_path": "/weights/x.pt",
        "base_model_name": "",
    }) in _Handler.calls
    assert ("/v1/unload_lora_adapter", {"lora_name": "a2"}) in _Handler.calls


def test_remote_config_reconciler(fake_server, tmp_path):
    """Sidecar mode against a remote server: diff vs GET /v1/models, actuate
    over HTTP, models into a local datastore."""
    from llm_instance_gateway_amd.api.config import ConfigReconciler
    from llm_instance_gateway_amd.router.remote import RemoteAdapterActuator

    path = tmp_path / "remote.yaml"
    path.write_text(
        "pool: {name: remote-pool, engine_addresses: ['%s']}\n"
        "models: [{model_name: remote-model}]\n"
        "ensureExist: {models: [{id: a1}, {id: fresh-adapter}]}\n"
        "ensureNotExist: {models: [{id: a2}]}\n" % fake_server
    )
    ds = Datastore()
    before = len(_Handler.calls)
    rec = ConfigReconciler(
        str(path),
        datastore=ds,
        remote_actuators=[RemoteAdapterActuator(f"http://{fake_server}")],
        health_timeout_s=5,
        health_interval_s=0.1,
    )
    assert rec.reconcile_once()
    assert ds.fetch_model("remote-model") is not None
    calls = _Handler.calls[before:]
    # a1 already listed -> skipped; fresh-adapter loaded; a2 unloaded
    paths = [(p, d.get("lora_name")) for p, d in calls]
    assert ("/v1/load_lora_adapter", "fresh-adapter") in paths
    assert ("/v1/unload_lora_adapter", "a2") in paths
    assert ("/v1/load_lora_adapter", "a1") not in paths
