"""Remote-engine adapters: HTTP metric scraping + adapter actuation.

Co-resident engines publish gauges through shared memory (serving/pool.py),
but the reference's deployment mode — router on one box, model servers
elsewhere — is also supported for capability parity:

  * `HTTPMetricsSource` — the reference's vLLM scraper
    (`backend/vllm/metrics.go:38-68`): GET http://<addr>/metrics, parse the
    Prometheus text page into EngineMetrics.  Plug into
    `router.provider.Provider` refresh mode to reproduce the 50 ms scrape
    loop (`provider.go:60-101`).
  * `RemoteAdapterActuator` — the dynamic-LoRA sidecar's actuation calls
    (`sidecar.py:177-213`): POST /v1/load_lora_adapter /
    /v1/unload_lora_adapter against a remote server, with the sidecar's
    health gate (`sidecar.py:158-175`).
"""

from __future__ import annotations

import logging
import time
from typing import Optional

import requests

from ..metrics.prometheus import parse_engine_metrics
from .types import Engine, EngineMetrics

log = logging.getLogger(__name__)

DEFAULT_TIMEOUT_S = 5.0  # reference fetchMetricsTimeout (provider.go:14)


class HTTPMetricsSource:
    """MetricsSource over HTTP (reference PodMetricsClientImpl)."""

    def __init__(self, timeout_s: float = DEFAULT_TIMEOUT_S, scheme: str = "http"):
        self.timeout_s = timeout_s
        self.scheme = scheme
        self.session = requests.Session()

    def fetch_metrics(self, engine: Engine, existing: EngineMetrics) -> EngineMetrics:
        if not engine.address:
            raise ValueError(f"engine {engine.name} has no address to scrape")
        url = f"{self.scheme}://{engine.address}/metrics"
        resp = self.session.get(url, timeout=self.timeout_s)
        if resp.status_code != 200:
            # non-200 keeps last-known metrics upstream (provider.go:150-159)
            raise RuntimeError(f"scrape {url}: HTTP {resp.status_code}")
        return parse_engine_metrics(resp.text, existing)


class RemoteAdapterActuator:
    """Drive a remote server's dynamic-LoRA API (sidecar parity)."""

    def __init__(self, base_url: str, timeout_s: float = 30.0):
        self.base_url = base_url.rstrip("/")
        self.timeout_s = timeout_s
        self.session = requests.Session()

    def wait_healthy(
        self, timeout_s: float = 300.0, interval_s: float = 15.0
    ) -> bool:
        """Poll GET /health until 200 (sidecar.py:158-175)."""
        deadline = time.monotonic() + timeout_s
        while True:
            try:
                if (
                    self.session.get(
                        f"{self.base_url}/health", timeout=self.timeout_s
                    ).status_code
                    == 200
                ):
                    return True
            except requests.RequestException as e:
                log.debug("health poll failed: %s", e)
            if time.monotonic() >= deadline:
                return False
            time.sleep(interval_s)

    def list_models(self) -> list:
        """GET /v1/models — what the sidecar diffs against (sidecar.py:143-156)."""
        r = self.session.get(f"{self.base_url}/v1/models", timeout=self.timeout_s)
        r.raise_for_status()
        return [m["id"] for m in r.json().get("data", [])]

    def load(
        self, lora_name: str, lora_path: str = "", base_model_name: str = ""
    ) -> Optional[str]:
        """POST /v1/load_lora_adapter (sidecar.py:177-195); None on success."""
        r = self.session.post(
            f"{self.base_url}/v1/load_lora_adapter",
            json={
                "lora_name": lora_name,
                "lora_path": lora_path,
                "base_model_name": base_model_name,
            },
            timeout=self.timeout_s,
        )
        return None if r.status_code == 200 else f"HTTP {r.status_code}: {r.text}"

    def unload(self, lora_name: str) -> Optional[str]:
        """POST /v1/unload_lora_adapter (sidecar.py:197-213)."""
        r = self.session.post(
            f"{self.base_url}/v1/unload_lora_adapter",
            json={"lora_name": lora_name},
            timeout=self.timeout_s,
        )
        return None if r.status_code == 200 else f"HTTP {r.status_code}: {r.text}"
