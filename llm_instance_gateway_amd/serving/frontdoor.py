"""OpenAI-compatible HTTP front door.

Collapses the reference's Envoy-gateway + ext-proc pipeline
into one process: body parsing + model resolution + traffic split
(`handlers/request.go:19-120`), filter-tree routing (`scheduling/`), direct
engine dispatch (replacing the `target-pod` header + ORIGINAL_DST hop,
`patch_policy.yaml:24-36`), usage accounting on the response
(`handlers/response.go:64-93`), and 429 for shed sheddable traffic
(`handlers/server.go:95-113`).

Model-server API parity (the contract the reference pins on vLLM,
`backend/vllm/metrics.go`, `sidecar.py:143-213`):

    POST /v1/completions           POST /v1/chat/completions
    GET  /v1/models                GET  /health
    POST /v1/load_lora_adapter     POST /v1/unload_lora_adapter
    GET  /metrics                  (Prometheus text)

Admin plane (replaces the InferenceModel/InferencePool CRDs):

    GET/POST /admin/models         POST /admin/drain/{engine}
    GET /admin/engines
"""

from __future__ import annotations

import asyncio
import json
import time
import uuid
from typing import Any, AsyncIterator, Dict, List, Optional, Union

from fastapi import FastAPI, Request
from fastapi.responses import JSONResponse, PlainTextResponse, Response, StreamingResponse
from pydantic import BaseModel, Field

from ..api.types import Criticality, InferenceModel, TargetModel, is_critical, random_weighted_draw
from ..metrics.prometheus import GatewayMetrics, render_engine_metrics
from ..router.scheduler import ResourceExhausted, SchedulingError
from ..router.types import LLMRequest
from ..utils import tokenizer
from ..utils.tracing import SpanLog
from .pool import EnginePool, PoolRequest, RequestTicket

# Debug header the reference sets on every response (handlers/response.go:13-38).
RESP_DEBUG_HEADER = "x-went-into-resp-headers"
TARGET_ENGINE_HEADER = "x-target-engine"  # the reference's `target-pod` header


class CompletionRequest(BaseModel):
    model: str
    prompt: Union[str, List[int]] = ""
    max_tokens: int = 16
    temperature: float = 0.0
    top_p: float = 1.0
    stream: bool = False
    stop: Optional[Union[str, List[str]]] = None
    n: int = 1            # number of choices (non-streaming only)
    echo: bool = False    # prepend the prompt text to each choice
    seed: Optional[int] = None  # deterministic sampling (OpenAI seed field)
    logprobs: Optional[int] = None      # top-N logprobs per generated token
    presence_penalty: float = 0.0
    frequency_penalty: float = 0.0
    repetition_penalty: float = 1.0     # vLLM extension
    # non-standard extensions (the bench/tests use these)
    lora: Optional[str] = None  # explicit adapter override
    request_id: Optional[str] = None


class ChatMessage(BaseModel):
    role: str
    content: str


class ChatCompletionRequest(BaseModel):
    model: str
    messages: List[ChatMessage] = Field(default_factory=list)
    max_tokens: int = 16
    temperature: float = 0.0
    top_p: float = 1.0
    stream: bool = False
    stop: Optional[Union[str, List[str]]] = None
    seed: Optional[int] = None
    presence_penalty: float = 0.0
    frequency_penalty: float = 0.0
    lora: Optional[str] = None


class LoadAdapterRequest(BaseModel):
    """vLLM dynamic-LoRA API shape (`sidecar.py:184-195`)."""

    lora_name: str
    lora_path: str = ""
    base_model_name: str = ""
    engine: Optional[str] = None  # None: load on every engine of the pool


class UnloadAdapterRequest(BaseModel):
    lora_name: str
    engine: Optional[str] = None


class RegisterModelRequest(BaseModel):
    """InferenceModel CRD analogue (`inferencemodel_types.go:40-135`)."""

    model_name: str
    criticality: str = "Default"
    target_models: List[Dict[str, Any]] = Field(default_factory=list)
    pool_ref: str = ""


class Gateway:
    """The server's state: pool + registry + instruments."""

    def __init__(self, pool: EnginePool):
        self.pool = pool
        self.metrics = GatewayMetrics()
        # per-request lifecycle spans (reference has none — SURVEY.md §5)
        self.spans = SpanLog()

    # -- request resolution (reference HandleRequestBody, request.go:19-70) --
    def resolve(self, model_name: str) -> tuple:
        im = self.pool.datastore.fetch_model(model_name)
        if im is None:
            raise KeyError(f"error getting target model name for model {model_name}")
        target = random_weighted_draw(im)
        return im, target

    async def complete(
        self, req: CompletionRequest, critical_override: Optional[bool] = None,
        choice_idx: int = 0,
    ) -> tuple:
        """Route + dispatch + await; returns (ticket|None already raised)."""
        t0 = time.monotonic()
        im, target = self.resolve(req.model)
        critical = is_critical(im) if critical_override is None else critical_override
        prompt_ids = tokenizer.to_token_ids(req.prompt)
        # adapter: explicit override, else a drawn target that differs from the
        # client model name is an adapter version (reference rewrites the body
        # "model" field the same way, request.go:62-70)
        lora = req.lora if req.lora is not None else (target if target != im.model_name else "")
        lreq = LLMRequest(
            model=req.model,
            target_models={t.name: t.weight for t in im.targets},
            # affinity routing keys on the adapter actually being used: an
            # explicit `lora` override supersedes the weighted draw
            resolved_target_model=lora or target,
            critical=critical,
            prompt_len=len(prompt_ids),
            max_new_tokens=req.max_tokens,
            pool_ref=im.pool_ref,
        )
        stops = req.stop if req.stop is not None else []
        if isinstance(stops, str):
            stops = [stops]
        base_id = req.request_id or f"cmpl-{uuid.uuid4().hex[:24]}"
        preq = PoolRequest(
            request_id=base_id if choice_idx == 0 else f"{base_id}-c{choice_idx}",
            prompt_token_ids=prompt_ids,
            max_new_tokens=req.max_tokens,
            temperature=req.temperature,
            top_p=req.top_p,
            seed=(req.seed if req.seed is not None else 0) + choice_idx,
            logprobs=req.logprobs or 0,
            presence_penalty=req.presence_penalty,
            frequency_penalty=req.frequency_penalty,
            repetition_penalty=req.repetition_penalty,
            lora_name=lora,
            critical=critical,
            stream=req.stream,
            stop_sequences=[tokenizer.encode(t, add_bos=False) for t in stops if t],
            # string-level matching too (vLLM semantics): with a subword
            # tokenizer the stop can tokenize differently in generation
            # context and the token-id match above would miss it.  Under the
            # byte codec token matching IS string matching, so skip the
            # per-token decoded-tail checks there.
            stop_texts=[t for t in stops if t] if tokenizer.has_backend() else [],
        )
        self.spans.event(preq.request_id, "arrived", model=req.model,
                         target=target, critical=critical)
        ticket = self.pool.route_and_submit(lreq, preq)
        self.metrics.route_latency.observe(time.monotonic() - t0)
        self.spans.event(preq.request_id, "routed", engine=ticket.engine_name)
        return ticket, target

    def record_finished(self, ticket: RequestTicket, model: str) -> None:
        fin = ticket.result
        if fin is None:
            return
        self.metrics.requests_total.inc(model=model, outcome="ok")
        self.metrics.prompt_tokens_total.inc(fin.prompt_len)
        self.metrics.completion_tokens_total.inc(len(fin.output_token_ids))
        if fin.ttft is not None:
            self.metrics.ttft.observe(fin.ttft)
        if fin.tpot is not None:
            self.metrics.tpot.observe(fin.tpot)
        self.metrics.e2e.observe(fin.e2e)
        self.spans.event(fin.request_id, "finished", engine=fin.engine_name,
                         ttft=fin.ttft, e2e=fin.e2e,
                         output_tokens=len(fin.output_token_ids))


def _trim_stop_text(text: str, req_stop) -> str:
    """OpenAI semantics: the matched stop text (and anything after it) is
    not returned.  Applied to DECODED text when the engine finished on a
    stop — covers the decoded-text match path, where output token ids are
    not trimmed (engine/sequence.py stop_texts)."""
    stops = [req_stop] if isinstance(req_stop, str) else (req_stop or [])
    cut = len(text)
    for s in stops:
        if s:
            i = text.find(s)
            if i >= 0:
                cut = min(cut, i)
    return text[:cut]


def _usage(prompt_len: int, completion_len: int) -> Dict[str, int]:
    """OpenAI usage object (reference parses exactly this, response.go:85-93)."""
    return {
        "prompt_tokens": prompt_len,
        "completion_tokens": completion_len,
        "total_tokens": prompt_len + completion_len,
    }


def build_app(pool: EnginePool, request_timeout_s: float = 600.0) -> FastAPI:
    gw = Gateway(pool)
    app = FastAPI(title="llm-instance-gateway-amd")
    app.state.gateway = gw

    async def _await_ticket(ticket: RequestTicket):
        loop = asyncio.get_running_loop()
        await loop.run_in_executor(None, ticket.wait, request_timeout_s)

    @app.post("/v1/completions")
    async def completions(req: CompletionRequest):
        try:
            ticket, target = await gw.complete(req)
        except KeyError as e:
            return JSONResponse(status_code=404, content={"error": str(e)})
        except ResourceExhausted as e:
            # reference: ImmediateResponse 429 (server.go:95-113)
            gw.metrics.shed_total.inc(model=req.model)
            gw.metrics.requests_total.inc(model=req.model, outcome="shed")
            return JSONResponse(
                status_code=429,
                content={"error": f"dropped sheddable request: {e}"},
                headers={RESP_DEBUG_HEADER: "true"},
            )
        except SchedulingError as e:
            return JSONResponse(status_code=503, content={"error": str(e)})

        created = int(time.time())
        rid = ticket.request_id

        if req.stream:
            # OpenAI semantics: a matched stop sequence is never streamed.
            # Hold back a tail of len(longest stop) tokens until the stop
            # check clears them; at stream end flush from the engine's
            # trimmed output so matched stop tokens are dropped.
            stops_s = [s for s in ([req.stop] if isinstance(req.stop, str)
                                   else (req.stop or [])) if s]
            # a stop of L chars spans at most L generated tokens, which can
            # exceed its standalone encoding length under a subword tokenizer
            holdback = max(
                [len(tokenizer.encode(s, add_bos=False)) for s in stops_s]
                + [len(s) for s in stops_s],
                default=0,
            )

            async def sse() -> AsyncIterator[str]:
                loop = asyncio.get_running_loop()
                sent = 0
                pending: List[int] = []

                def chunk_for(tok: int) -> str:
                    chunk = {
                        "id": rid,
                        "object": "text_completion",
                        "created": created,
                        "model": req.model,
                        "choices": [
                            {"index": 0, "text": tokenizer.decode([tok]),
                             "token": tok, "finish_reason": None}
                        ],
                    }
                    return f"data: {json.dumps(chunk)}\n\n"

                while True:
                    try:
                        tok = await loop.run_in_executor(None, ticket.tokens.get)
                    except asyncio.CancelledError:
                        # client disconnected mid-stream: stop decoding for a
                        # dead connection (engine abort + block free)
                        pool.abort(rid)
                        raise
                    if tok is None:
                        break
                    pending.append(tok)
                    while len(pending) > holdback:
                        sent += 1
                        yield chunk_for(pending.pop(0))
                await _await_ticket(ticket)
                gw.record_finished(ticket, req.model)
                fin = ticket.result
                if fin is not None:
                    # flush survivors of the stop trim (held-back tail) as one
                    # text chunk, with any decoded-text-matched stop removed
                    tail_text = tokenizer.decode(fin.output_token_ids[sent:])
                    if fin.finish_reason == "stop":
                        tail_text = _trim_stop_text(tail_text, req.stop)
                    if tail_text:
                        yield (
                            "data: "
                            + json.dumps({
                                "id": rid, "object": "text_completion",
                                "created": created, "model": req.model,
                                "choices": [{"index": 0, "text": tail_text,
                                             "finish_reason": None}],
                            })
                            + "\n\n"
                        )
                final = {
                    "id": rid,
                    "object": "text_completion",
                    "created": created,
                    "model": req.model,
                    "choices": [{"index": 0, "text": "",
                                 "finish_reason": fin.finish_reason if fin else "length"}],
                    "usage": _usage(fin.prompt_len, len(fin.output_token_ids)) if fin else None,
                }
                yield f"data: {json.dumps(final)}\n\n"
                yield "data: [DONE]\n\n"

            return StreamingResponse(
                sse(),
                media_type="text/event-stream",
                headers={
                    TARGET_ENGINE_HEADER: ticket.engine_name,
                    RESP_DEBUG_HEADER: "true",
                },
            )

        # n > 1: route + dispatch the remaining choices concurrently
        extra_tickets = []
        for ci in range(1, max(1, min(req.n, 16))):
            try:
                t_extra, _ = await gw.complete(req, choice_idx=ci)
                extra_tickets.append(t_extra)
            except (ResourceExhausted, SchedulingError):
                break  # partial degradation: return the choices we can serve
        tickets = [ticket] + extra_tickets
        try:
            for t in tickets:
                await _await_ticket(t)
        except TimeoutError:
            for t in tickets:
                pool.abort(t.request_id)
            return JSONResponse(status_code=504, content={"error": "request timed out"})
        except RuntimeError as e:
            gw.metrics.requests_total.inc(model=req.model, outcome="error")
            msg = str(e)
            if "max_model_len" in msg:
                status = 400
            elif "unknown LoRA adapter" in msg:
                status = 404
            else:
                status = 500
            return JSONResponse(status_code=status, content={"error": msg})
        echo_text = (
            req.prompt if req.echo and isinstance(req.prompt, str) else ""
        )
        choices = []
        total_completion = 0
        for i, t in enumerate(tickets):
            gw.record_finished(t, req.model)
            fin_i = t.result
            total_completion += len(fin_i.output_token_ids)
            out_text = tokenizer.decode(fin_i.output_token_ids)
            if fin_i.finish_reason == "stop":
                out_text = _trim_stop_text(out_text, req.stop)
            choice = {
                "index": i,
                "text": echo_text + out_text,
                "token_ids": fin_i.output_token_ids,
                "finish_reason": fin_i.finish_reason,
            }
            if req.logprobs and fin_i.output_logprobs:
                choice["logprobs"] = {
                    "tokens": [
                        tokenizer.decode([t]) for t in fin_i.output_token_ids
                    ],
                    "token_logprobs": [
                        lp for lp, _ in fin_i.output_logprobs
                    ],
                    "top_logprobs": [
                        {tokenizer.decode([t]): v for t, v in top.items()}
                        for _, top in fin_i.output_logprobs
                    ],
                }
            choices.append(choice)
        fin = ticket.result
        body = {
            "id": rid,
            "object": "text_completion",
            "created": created,
            "model": req.model,
            "choices": choices,
            "usage": _usage(fin.prompt_len, total_completion),
        }
        return JSONResponse(
            content=body,
            headers={
                TARGET_ENGINE_HEADER: fin.engine_name,
                RESP_DEBUG_HEADER: "true",
            },
        )

    @app.post("/v1/chat/completions")
    async def chat_completions(req: ChatCompletionRequest):
        prompt = tokenizer.apply_chat_template(req.messages)
        creq = CompletionRequest(
            model=req.model,
            prompt=prompt,
            max_tokens=req.max_tokens,
            temperature=req.temperature,
            top_p=req.top_p,
            stream=False,
            stop=req.stop,
            seed=req.seed,
            presence_penalty=req.presence_penalty,
            frequency_penalty=req.frequency_penalty,
            lora=req.lora,
        )
        resp = await completions(creq)
        if isinstance(resp, JSONResponse) and resp.status_code == 200:
            payload = json.loads(bytes(resp.body))
            payload["object"] = "chat.completion"
            payload["choices"] = [
                {
                    "index": 0,
                    "message": {"role": "assistant",
                                "content": payload["choices"][0]["text"]},
                    "finish_reason": payload["choices"][0]["finish_reason"],
                }
            ]
            return JSONResponse(content=payload, headers=dict(resp.headers))
        return resp

    @app.get("/v1/models")
    async def list_models():
        """Registered client models + resident adapters (vLLM GET /v1/models is
        what the sidecar diffs against, `sidecar.py:143-156`)."""
        data = []
        for im in pool.datastore.all_models():
            data.append({"id": im.model_name, "object": "model",
                         "criticality": im.criticality.value})
        seen = set()
        for h in pool.handles.values():
            for name in h.active_adapters:
                if name not in seen:
                    seen.add(name)
                    data.append({"id": name, "object": "model", "adapter": True})
        return {"object": "list", "data": data}

    @app.post("/v1/load_lora_adapter")
    async def load_lora_adapter(req: LoadAdapterRequest):
        targets = (
            [pool.handles[req.engine]] if req.engine else list(pool.handles.values())
        )
        loop = asyncio.get_running_loop()
        errs = []
        for h in targets:
            err = await loop.run_in_executor(
                None, h.load_adapter, req.lora_name, req.lora_path or None
            )
            if err:
                errs.append(f"{h.engine.name}: {err}")
        if errs:
            return JSONResponse(status_code=400, content={"error": "; ".join(errs)})
        return PlainTextResponse(f"Success: LoRA adapter '{req.lora_name}' added")

    @app.post("/v1/unload_lora_adapter")
    async def unload_lora_adapter(req: UnloadAdapterRequest):
        targets = (
            [pool.handles[req.engine]] if req.engine else list(pool.handles.values())
        )
        loop = asyncio.get_running_loop()
        for h in targets:
            await loop.run_in_executor(None, h.unload_adapter, req.lora_name)
        return PlainTextResponse(f"Success: LoRA adapter '{req.lora_name}' removed")

    @app.get("/health")
    async def health():
        """gRPC health-check parity (reference main.go:45-48): serving iff at
        least one engine is routable."""
        ok = any(h.alive() for h in pool.handles.values())
        return JSONResponse(
            status_code=200 if ok else 503,
            content={"status": "ok" if ok else "no healthy engines"},
        )

    @app.get("/metrics")
    async def metrics():
        parts = [
            render_engine_metrics(h.engine.name, h.metrics())
            for h in pool.handles.values()
            if h.alive()
        ]
        parts.append(gw.metrics.render())
        return PlainTextResponse("".join(parts))

    # -- admin plane (CRD analogue) ------------------------------------------
    @app.get("/admin/models")
    async def admin_models():
        return [
            {
                "model_name": im.model_name,
                "criticality": im.criticality.value,
                "target_models": [{"name": t.name, "weight": t.weight} for t in im.targets],
                "pool_ref": im.pool_ref,
            }
            for im in pool.datastore.all_models()
        ]

    @app.post("/admin/models")
    async def register_model(req: RegisterModelRequest):
        im = InferenceModel(
            model_name=req.model_name,
            criticality=Criticality(req.criticality),
            target_models=[
                TargetModel(name=t["name"], weight=int(t.get("weight", 1)))
                for t in req.target_models
            ],
            pool_ref=req.pool_ref,
        )
        pool.datastore.store_model(im)
        return {"status": "ok"}

    @app.delete("/admin/models/{model_name}")
    async def delete_model(model_name: str):
        pool.datastore.delete_model(model_name)
        return {"status": "ok"}

    @app.get("/admin/engines")
    async def admin_engines():
        return [
            {
                "name": h.engine.name,
                "gpu_id": h.engine.gpu_id,
                "alive": h.alive(),
                "draining": h.is_draining(),
                "heartbeat_age_s": round(h.heartbeat_age_s(), 3),
                "restarts": pool.restart_counts.get(h.engine.name, 0),
                "metrics": h.metrics().__dict__,
            }
            for h in pool.handles.values()
        ]

    @app.get("/admin/trace")
    async def trace_tail(n: int = 100, request_id: Optional[str] = None):
        """Tail of the per-request span log (arrived/routed/finished)."""
        return gw.spans.tail(n=n, request_id=request_id)

    @app.post("/admin/drain/{engine_name}")
    async def drain_engine(engine_name: str, on: bool = True):
        if engine_name not in pool.handles:
            return JSONResponse(status_code=404, content={"error": "no such engine"})
        pool.handles[engine_name].drain(on)
        return {"status": "ok", "draining": on}

    @app.post("/admin/restart/{engine_name}")
    async def restart_engine(engine_name: str):
        if engine_name not in pool.handles:
            return JSONResponse(status_code=404, content={"error": "no such engine"})
        loop = asyncio.get_running_loop()
        await loop.run_in_executor(None, pool.restart_engine, engine_name)
        return {"status": "ok"}

    return app
