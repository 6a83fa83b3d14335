"""Remote-router mode: the reference's own deployment topology.

The co-resident mode (`serving/pool.py`) replaces the reference's fleet with
in-process engines; THIS mode keeps the reference's shape — a standalone L7
router in front of **remote** OpenAI-compatible model servers (vLLM, or this
framework's own front door on other nodes):

    client ── POST /v1/completions ──► router ──► chosen engine (HTTP)
                   ▲                        │
           filter-tree scheduler ◄── 50 ms Prometheus scrapes

Reference mapping: Envoy + ext-proc + ORIGINAL_DST (`patch_policy.yaml`)
collapse into direct request forwarding; the scrape loop is
`router/provider.py` refresh mode over `router/remote.py::HTTPMetricsSource`;
model resolution / traffic split / criticality shedding identical to the
co-resident front door.
"""

from __future__ import annotations

import logging
import time
from typing import List, Optional

import httpx
from fastapi import FastAPI
from fastapi.responses import JSONResponse, PlainTextResponse, StreamingResponse

from ..api.types import InferenceModel, is_critical, random_weighted_draw
from ..metrics.prometheus import GatewayMetrics, render_engine_metrics
from ..router.datastore import Datastore
from ..router.provider import Provider
from ..router.remote import HTTPMetricsSource
from ..router.scheduler import ResourceExhausted, Scheduler, SchedulingError
from ..router.types import Engine, LLMRequest
from .frontdoor import RESP_DEBUG_HEADER, TARGET_ENGINE_HEADER, CompletionRequest

log = logging.getLogger(__name__)


def build_remote_router_app(
    engine_addresses: List[str],
    models: Optional[List[InferenceModel]] = None,
    refresh_metrics_interval_s: float = 0.050,
    start_refresh: bool = True,
    request_timeout_s: float = 600.0,
) -> FastAPI:
    """Router-only gateway over remote OpenAI-compatible engines."""
    ds = Datastore(
        engines=[
            Engine(name=f"engine-{i}", address=addr)
            for i, addr in enumerate(engine_addresses)
        ]
    )
    for m in models or []:
        ds.store_model(m)
    provider = Provider(HTTPMetricsSource(), ds)
    provider.init(
        refresh_metrics_interval_s=refresh_metrics_interval_s,
        start_loops=start_refresh,
    )
    from ..router.native import NativeScheduler, native_available

    if native_available():
        scheduler = NativeScheduler.for_provider(provider)
        log.info("remote-router using the native (C++) filter tree")
    else:  # pragma: no cover - extension always built in-tree
        scheduler = Scheduler(provider)
    gw_metrics = GatewayMetrics()
    client = httpx.AsyncClient(timeout=request_timeout_s)

    app = FastAPI(title="llm-instance-gateway-amd (remote-router mode)")
    app.state.datastore = ds
    app.state.provider = provider

    @app.on_event("shutdown")
    async def _shutdown():
        provider.stop()
        await client.aclose()

    @app.post("/v1/completions")
    async def completions(req: CompletionRequest):
        t0 = time.monotonic()
        im = ds.fetch_model(req.model)
        if im is None:
            return JSONResponse(
                status_code=404,
                content={"error": f"error getting target model name for model {req.model}"},
            )
        target = random_weighted_draw(im)
        lreq = LLMRequest(
            model=req.model,
            target_models={t.name: t.weight for t in im.targets},
            resolved_target_model=target,
            critical=is_critical(im),
            pool_ref=im.pool_ref,
        )
        try:
            engine = scheduler.schedule(lreq)
        except ResourceExhausted as e:
            gw_metrics.shed_total.inc(model=req.model)
            gw_metrics.requests_total.inc(model=req.model, outcome="shed")
            return JSONResponse(
                status_code=429,
                content={"error": f"dropped sheddable request: {e}"},
                headers={RESP_DEBUG_HEADER: "true"},
            )
        except SchedulingError as e:
            return JSONResponse(status_code=503, content={"error": str(e)})
        gw_metrics.route_latency.observe(time.monotonic() - t0)

        # body rewrite: resolved adapter replaces the client model name
        # (reference request.go:62-70)
        body = req.model_dump(exclude_none=True)
        body.pop("request_id", None)
        body.pop("lora", None)
        if target and target != im.model_name:
            body["model"] = target
        url = f"http://{engine.address}/v1/completions"

        if req.stream:
            async def relay():
                async with client.stream("POST", url, json=body) as resp:
                    async for chunk in resp.aiter_bytes():
                        yield chunk

            return StreamingResponse(
                relay(),
                media_type="text/event-stream",
                headers={TARGET_ENGINE_HEADER: engine.name, RESP_DEBUG_HEADER: "true"},
            )

        try:
            resp = await client.post(url, json=body)
        except httpx.HTTPError as e:
            gw_metrics.requests_total.inc(model=req.model, outcome="error")
            return JSONResponse(status_code=502, content={"error": str(e)})
        gw_metrics.requests_total.inc(model=req.model, outcome="ok")
        is_json = resp.headers.get("content-type", "").startswith("application/json")
        gw_metrics.e2e.observe(time.monotonic() - t0)
        fwd_headers = {TARGET_ENGINE_HEADER: engine.name, RESP_DEBUG_HEADER: "true"}
        if not is_json:
            # pass non-JSON upstream bodies through verbatim
            return PlainTextResponse(
                resp.text, status_code=resp.status_code, headers=fwd_headers
            )
        payload = resp.json()
        usage = payload.get("usage") if isinstance(payload, dict) else None
        if usage:
            gw_metrics.prompt_tokens_total.inc(usage.get("prompt_tokens", 0))
            gw_metrics.completion_tokens_total.inc(usage.get("completion_tokens", 0))
        return JSONResponse(
            status_code=resp.status_code, content=payload, headers=fwd_headers
        )

    @app.get("/v1/models")
    async def list_models():
        return {
            "object": "list",
            "data": [
                {"id": m.model_name, "object": "model",
                 "criticality": m.criticality.value}
                for m in ds.all_models()
            ],
        }

    @app.get("/admin/models")
    async def admin_models():
        return [
            {
                "model_name": m.model_name,
                "criticality": m.criticality.value,
                "target_models": [
                    {"name": t.name, "weight": t.weight} for t in m.targets
                ],
                "pool_ref": m.pool_ref,
            }
            for m in ds.all_models()
        ]

    @app.post("/admin/models")
    async def register_model(body: dict):
        from ..api.types import Criticality, TargetModel

        ds.store_model(
            InferenceModel(
                model_name=str(body["model_name"]),
                criticality=Criticality(body.get("criticality", "Default")),
                target_models=[
                    TargetModel(name=t["name"], weight=int(t.get("weight", 1)))
                    for t in body.get("target_models", [])
                ],
                pool_ref=str(body.get("pool_ref", "")),
            )
        )
        return {"status": "ok"}

    @app.delete("/admin/models/{model_name}")
    async def delete_model(model_name: str):
        ds.delete_model(model_name)
        return {"status": "ok"}

    @app.get("/health")
    async def health():
        ok = bool(provider.all_engine_metrics())
        return JSONResponse(
            status_code=200 if ok else 503,
            content={"status": "ok" if ok else "no engines"},
        )

    @app.get("/metrics")
    async def metrics():
        parts = [
            render_engine_metrics(s.engine.name, s.metrics)
            for s in provider.all_engine_metrics()
        ]
        parts.append(gw_metrics.render())
        return PlainTextResponse("".join(parts))

    return app
