"""Typed gateway client SDK.

The analogue of the reference's generated clientset (`client-go/`, 2.8k LoC
of kubebuilder-generated typed clients/informers/listers): one hand-written
client covering the whole HTTP surface — OpenAI inference calls, the
dynamic-LoRA API, the admin (CRD-analogue) plane, metrics and traces.

    gw = GatewayClient("http://localhost:8081")
    gw.completion("chat-prod", "hello", max_tokens=32)
    gw.register_model("canary", criticality="Sheddable",
                      target_models=[{"name": "lora-x", "weight": 100}])
    gw.load_adapter("lora-x")

``session`` is injectable (anything with requests-style get/post/delete) so
tests can pass a starlette TestClient.
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional, Sequence, Union


class GatewayError(RuntimeError):
    def __init__(self, status: int, body: str):
        super().__init__(f"HTTP {status}: {body}")
        self.status = status
        self.body = body


class GatewayClient:
    def __init__(self, base_url: str, session=None, timeout_s: float = 600.0):
        self.base_url = base_url.rstrip("/")
        if session is None:
            import requests

            session = requests.Session()
        self.session = session
        self.timeout_s = timeout_s

    # -- internals -----------------------------------------------------------
    def _check(self, resp):
        if resp.status_code >= 400:
            raise GatewayError(resp.status_code, resp.text)
        return resp

    def _get(self, path: str, **kw):
        return self._check(self.session.get(self.base_url + path, **kw))

    def _post(self, path: str, json: Optional[dict] = None, **kw):
        return self._check(self.session.post(self.base_url + path, json=json, **kw))

    # -- inference -----------------------------------------------------------
    def completion(
        self,
        model: str,
        prompt: Union[str, Sequence[int]],
        max_tokens: int = 16,
        temperature: float = 0.0,
        top_p: float = 1.0,
        lora: Optional[str] = None,
        stop: Optional[Union[str, Sequence[str]]] = None,
        seed: Optional[int] = None,
        logprobs: Optional[int] = None,
        n: int = 1,
        echo: bool = False,
        presence_penalty: float = 0.0,
        frequency_penalty: float = 0.0,
        repetition_penalty: float = 1.0,
    ) -> Dict[str, Any]:
        body: Dict[str, Any] = {
            "model": model,
            "prompt": list(prompt) if not isinstance(prompt, str) else prompt,
            "max_tokens": max_tokens,
            "temperature": temperature,
            "top_p": top_p,
            "n": n,
            "echo": echo,
            "presence_penalty": presence_penalty,
            "frequency_penalty": frequency_penalty,
            "repetition_penalty": repetition_penalty,
        }
        if lora is not None:
            body["lora"] = lora
        if stop is not None:
            body["stop"] = stop if isinstance(stop, str) else list(stop)
        if seed is not None:
            body["seed"] = seed
        if logprobs is not None:
            body["logprobs"] = logprobs
        return self._post("/v1/completions", json=body).json()

    def completion_stream(
        self,
        model: str,
        prompt: Union[str, Sequence[int]],
        max_tokens: int = 16,
        temperature: float = 0.0,
    ):
        """Yield streamed completion chunks (parsed SSE `data:` events);
        the final yielded dict carries the usage object."""
        body = {
            "model": model,
            "prompt": list(prompt) if not isinstance(prompt, str) else prompt,
            "max_tokens": max_tokens,
            "temperature": temperature,
            "stream": True,
        }
        import json as _json

        url = self.base_url + "/v1/completions"
        if hasattr(self.session, "stream"):  # httpx / starlette TestClient
            ctx = self.session.stream("POST", url, json=body)
            with ctx as resp:
                self._check(resp)
                for line in resp.iter_lines():
                    line = line.decode() if isinstance(line, bytes) else line
                    if not line.startswith("data: "):
                        continue
                    payload = line[len("data: "):]
                    if payload == "[DONE]":
                        return
                    yield _json.loads(payload)
            return
        # requests-style session
        resp = self._check(self.session.post(url, json=body, stream=True))
        for line in resp.iter_lines():
            line = line.decode() if isinstance(line, bytes) else line
            if not line or not line.startswith("data: "):
                continue
            payload = line[len("data: "):]
            if payload == "[DONE]":
                return
            yield _json.loads(payload)

    def chat(
        self, model: str, messages: List[Dict[str, str]], max_tokens: int = 16
    ) -> Dict[str, Any]:
        return self._post(
            "/v1/chat/completions",
            json={"model": model, "messages": messages, "max_tokens": max_tokens},
        ).json()

    def models(self) -> List[Dict[str, Any]]:
        return self._get("/v1/models").json()["data"]

    def health(self) -> bool:
        try:
            return self._get("/health").json().get("status") == "ok"
        except GatewayError:
            return False

    # -- dynamic LoRA (sidecar API) -------------------------------------------
    def load_adapter(
        self, lora_name: str, lora_path: str = "", engine: Optional[str] = None
    ) -> None:
        self._post(
            "/v1/load_lora_adapter",
            json={"lora_name": lora_name, "lora_path": lora_path, "engine": engine},
        )

    def unload_adapter(self, lora_name: str, engine: Optional[str] = None) -> None:
        self._post(
            "/v1/unload_lora_adapter",
            json={"lora_name": lora_name, "engine": engine},
        )

    # -- admin plane (CRD analogue) -------------------------------------------
    def list_inference_models(self) -> List[Dict[str, Any]]:
        return self._get("/admin/models").json()

    def register_model(
        self,
        model_name: str,
        criticality: str = "Default",
        target_models: Optional[List[Dict[str, Any]]] = None,
        pool_ref: str = "",
    ) -> None:
        self._post(
            "/admin/models",
            json={
                "model_name": model_name,
                "criticality": criticality,
                "target_models": target_models or [],
                "pool_ref": pool_ref,
            },
        )

    def delete_model(self, model_name: str) -> None:
        self._check(self.session.delete(f"{self.base_url}/admin/models/{model_name}"))

    def engines(self) -> List[Dict[str, Any]]:
        return self._get("/admin/engines").json()

    def drain(self, engine_name: str, on: bool = True) -> None:
        self._post(f"/admin/drain/{engine_name}?on={'true' if on else 'false'}")

    def restart_engine(self, engine_name: str) -> None:
        self._post(f"/admin/restart/{engine_name}")

    def metrics_text(self) -> str:
        return self._get("/metrics").text

    def trace(self, request_id: Optional[str] = None, n: int = 100) -> List[Dict]:
        q = f"?n={n}" + (f"&request_id={request_id}" if request_id else "")
        return self._get(f"/admin/trace{q}").json()
