"""Engine pool: N per-GPU engine worker processes behind one router.

MI355X-native replacement for the reference's pod fleet + Envoy dispatch
(SURVEY.md §7): instead of K8s pods scraped over HTTP every 50 ms
(`pkg/ext-proc/backend/provider.go:134-179`) and routed via a `target-pod`
header (`pkg/manifests/patch_policy.yaml:24-36`), the engines are worker
processes on the same node — one per GPU, required because each engine's
host-side continuous-batching loop must not share a GIL — publishing their
gauges into a **shared-memory array the router reads with zero staleness**,
and receiving requests as direct IPC messages.

Per-engine shared gauge slots (doubles, written lock-free by the worker after
every scheduler iteration, read on every route):

    [heartbeat_ts, step_count, waiting, running, kv_usage, kv_capacity,
     max_loras, draining]

The variable-size resident-adapter set travels as an event message only when
it changes (rare: adapter load/evict), cached parent-side.

Failure semantics (reference parity): a worker that stops
heartbeating is marked unhealthy and drops out of the routing set (the
reference's EndpointSlice `Ready` condition, `endpointslice_reconciler.go:
107-111`); `restart_engine` re-spawns it and replays the adapter registry
(the reference's pod restart + sidecar reconcile).
"""

from __future__ import annotations

import logging
import multiprocessing as mp
import os
import queue
import threading
import time
from dataclasses import dataclass, field
from typing import Any, Callable, Dict, List, Optional, Sequence

from ..router.datastore import Datastore
from ..router.scheduler import ResourceExhausted, Scheduler, SchedulingError
from ..router.types import Engine, EngineMetrics, EngineState, LLMRequest

log = logging.getLogger(__name__)

# shared gauge slot layout
_SLOT_HEARTBEAT = 0
_SLOT_STEP = 1
_SLOT_WAITING = 2
_SLOT_RUNNING = 3
_SLOT_KV_USAGE = 4
_SLOT_KV_CAP = 5
_SLOT_MAX_LORAS = 6
_SLOT_DRAINING = 7
_SLOT_PREEMPTIONS = 8
SLOTS_PER_ENGINE = 9

HEARTBEAT_STALE_S = 30.0


@dataclass
class PoolRequest:
    """One request as handed to an engine worker."""

    request_id: str
    prompt_token_ids: List[int]
    max_new_tokens: int = 128
    temperature: float = 0.0
    top_p: float = 1.0
    seed: int = 0
    logprobs: int = 0
    presence_penalty: float = 0.0
    frequency_penalty: float = 0.0
    repetition_penalty: float = 1.0
    lora_name: str = ""
    critical: bool = False
    stream: bool = False
    stop_sequences: List[List[int]] = field(default_factory=list)
    stop_texts: List[str] = field(default_factory=list)
    arrival_time: float = field(default_factory=time.monotonic)


@dataclass
class FinishedRequest:
    request_id: str
    engine_name: str
    prompt_len: int
    output_token_ids: List[int]
    ttft: Optional[float]
    tpot: Optional[float]
    e2e: float
    finish_reason: str = "length"
    # only populated when the request asked for logprobs:
    # [(chosen_logprob, {top_token_id: logprob}), ...]
    output_logprobs: List = field(default_factory=list)


@dataclass
class EngineSpec:
    """One engine of a (possibly mixed) pool: base model, device set
    (len == TP degree), per-engine knobs."""

    name: str
    model: str
    devices: List[str]
    engine_kwargs: Dict[str, Any] = field(default_factory=dict)
    lora_rank: int = 16
    pool_ref: str = ""


class RequestTicket:
    """Parent-side handle for one in-flight request (thread-safe; the async
    front door bridges it onto the event loop)."""

    def __init__(self, request_id: str, engine_name: str, stream: bool):
        self.request_id = request_id
        self.engine_name = engine_name
        self.stream = stream
        self.tokens: "queue.Queue[Optional[int]]" = queue.Queue()
        self.done = threading.Event()
        self.result: Optional[FinishedRequest] = None
        self.error: Optional[str] = None

    def wait(self, timeout: Optional[float] = None) -> FinishedRequest:
        if not self.done.wait(timeout):
            raise TimeoutError(f"request {self.request_id} timed out")
        if self.error:
            raise RuntimeError(self.error)
        assert self.result is not None
        return self.result


def _worker_main(
    conn,
    shared,
    slot_base: int,
    engine_name: str,
    model_name: str,
    device: str,
    engine_kwargs: Dict[str, Any],
    lora_rank: int,
    tp_rank: int = 0,
    tp_size: int = 1,
    master_port: int = 0,
) -> None:
    """Engine worker process body: one LLMEngine, one GPU, one event loop.

    TP engines (tp_size > 1, BASELINE config #4's 70B TP=4) run this body on
    every rank with its own process group (gloo on CPU, RCCL over xGMI on
    GPU): rank 0 fronts the request pipe and broadcasts each iteration's
    command batch; all ranks then run the same deterministic scheduler +
    model step, synchronised by the per-layer collectives."""
    import torch

    from ..engine import EngineConfig, LLMEngine, SamplingParams
    from ..lora.manager import AdapterManager
    from ..models.config import get_model_config

    if device.startswith("cuda"):
        torch.cuda.set_device(device)
        from ..utils.tunableop import maybe_enable_tunableop

        maybe_enable_tunableop()
    # Worker processes need the same tokenizer backend as the front door for
    # decoded-text stop matching (server.py exports LIG_TOKENIZER; spawn
    # children inherit the env).
    tok_path = os.environ.get("LIG_TOKENIZER", "")
    if tok_path:
        from ..utils import tokenizer as _worker_tok

        _worker_tok.load_backend(tok_path)
    tp = None
    if tp_size > 1:
        from ..parallel import TPGroup, init_distributed

        init_distributed(
            backend="nccl" if device.startswith("cuda") else "gloo",
            rank=tp_rank,
            world_size=tp_size,
            master_port=master_port,
        )
        tp = TPGroup(device=torch.device(device))
    is_front = tp_rank == 0
    mc = get_model_config(model_name)
    cfg = EngineConfig(model=mc, engine_id=engine_name, **engine_kwargs)
    torch.manual_seed(1234)  # same base weights on every engine of the pool
    eng = LLMEngine(cfg, device=device, tp=tp)
    mgr: Optional[AdapterManager] = None
    if cfg.max_loras > 0:
        mgr = AdapterManager(eng, max_slots=cfg.max_loras, rank=lora_rank)

    seq_by_id: Dict[str, Any] = {}
    stream_ids: set = set()
    token_buf: List = []
    draining = False

    def publish_gauges() -> None:
        m = eng.get_metrics()
        b = slot_base
        shared[b + _SLOT_HEARTBEAT] = time.time()
        shared[b + _SLOT_STEP] = float(eng._step_count)
        shared[b + _SLOT_WAITING] = float(m.waiting_queue_size)
        shared[b + _SLOT_RUNNING] = float(m.running_queue_size)
        shared[b + _SLOT_KV_USAGE] = m.kv_cache_usage_percent
        shared[b + _SLOT_KV_CAP] = float(m.kv_cache_max_token_capacity)
        shared[b + _SLOT_MAX_LORAS] = float(m.max_active_models)
        shared[b + _SLOT_DRAINING] = 1.0 if draining else 0.0
        shared[b + _SLOT_PREEMPTIONS] = float(m.num_preemptions_total)

    def send(msg) -> None:
        if is_front:
            conn.send(msg)

    def publish_adapters() -> None:
        if mgr is not None:
            send(("adapters", mgr.resident_names(), mgr.max_slots))

    def on_token(seq, tok):
        if seq.request_id in stream_ids:
            token_buf.append((seq.request_id, tok))

    eng.on_token = on_token

    # TP: commands are broadcast from the front rank over a dedicated gloo
    # group (host-side metadata must not ride the RCCL compute stream).
    tp_cmd = None
    if tp is not None:
        import torch.distributed as dist

        from ..parallel import TPGroup

        tp_cmd = TPGroup(group=dist.new_group(backend="gloo"))

    nonlocal_draining = [draining]

    def apply_cmd(msg) -> bool:
        """Apply one command on this rank; False means stop."""
        op = msg[0]
        if op == "add":
            r: PoolRequest = msg[1]
            if r.stream and is_front:
                stream_ids.add(r.request_id)
            try:
                seq = eng.add_request(
                    r.prompt_token_ids,
                    SamplingParams(
                        max_new_tokens=r.max_new_tokens,
                        temperature=r.temperature,
                        top_p=r.top_p,
                        seed=r.seed,
                        logprobs=r.logprobs,
                        presence_penalty=r.presence_penalty,
                        frequency_penalty=r.frequency_penalty,
                        repetition_penalty=r.repetition_penalty,
                        stop_sequences=r.stop_sequences,
                        stop_texts=r.stop_texts,
                    ),
                    lora_name=r.lora_name,
                    request_id=r.request_id,
                    critical=r.critical,
                    arrival_time=r.arrival_time,
                )
                seq_by_id[r.request_id] = seq
            except Exception as e:  # e.g. prompt > max_model_len
                send(("error", r.request_id, str(e)))
        elif op == "abort":
            eng.abort_request(msg[1])
        elif op == "load_adapter":
            _, name, source, rank = msg
            try:
                if mgr is None:
                    raise RuntimeError("engine has no adapter slots")
                if isinstance(source, dict):
                    mgr.register_weights_full(name, source, rank=rank)
                elif isinstance(source, str) and os.path.exists(source):
                    weights = torch.load(source, weights_only=True)
                    # full (unsharded) checkpoints: TP engines shard in-place
                    mgr.register_weights_full(name, weights, rank=rank)
                else:
                    mgr.register_random(
                        name, seed=abs(hash(name)) % (2**31), rank=rank
                    )
                mgr.prepare([name])
                send(("adapter_loaded", name, None))
            except Exception as e:
                send(("adapter_loaded", name, str(e)))
            publish_adapters()
        elif op == "unload_adapter":
            name = msg[1]
            ok = mgr.unregister(name) if mgr is not None else False
            send(("adapter_unloaded", name, ok))
            publish_adapters()
        elif op == "drain":
            nonlocal_draining[0] = bool(msg[1])
        elif op == "ping":
            send(("pong", msg[1]))
        elif op == "stop":
            return False
        return True

    if is_front:
        publish_gauges()
        conn.send(("ready",))
    prev_resident: List[str] = []

    while True:
        # front rank drains its pipe into a command batch ...
        cmds: List = []
        if is_front:
            timeout = 0.0 if eng.has_work() else 0.002
            try:
                while conn.poll(timeout):
                    cmds.append(conn.recv())
                    timeout = 0.0
            except (EOFError, OSError):
                cmds.append(("stop",))
        # ... and every TP rank applies the identical batch in order
        if tp_cmd is not None:
            cmds = tp_cmd.broadcast_obj(cmds)
        for msg in cmds:
            if not apply_cmd(msg):
                return
        draining = nonlocal_draining[0]

        if eng.has_work():
            try:
                eng.step()
            except Exception as e:  # OOM / kernel error: report, then die so
                # the watchdog fails in-flight tickets and drops the engine
                import traceback

                send(("error", None, f"engine step failed: {e}\n{traceback.format_exc()}"))
                raise
            finished = eng.collect_finished()
            if is_front and token_buf:
                conn.send(("tokens", token_buf))
                token_buf = []
            if finished and is_front:
                now = time.monotonic()
                out = []
                for seq in finished:
                    seq_by_id.pop(seq.request_id, None)
                    stream_ids.discard(seq.request_id)
                    out.append(
                        FinishedRequest(
                            request_id=seq.request_id,
                            engine_name=engine_name,
                            prompt_len=seq.prompt_len,
                            output_token_ids=list(seq.output_token_ids),
                            ttft=seq.ttft(),
                            tpot=seq.tpot(),
                            e2e=(seq.finish_time or now) - seq.arrival_time,
                            finish_reason=seq.finish_reason,
                            output_logprobs=list(seq.output_logprobs),
                        )
                    )
                conn.send(("finished", out))
            if mgr is not None and is_front:
                resident = mgr.resident_names()
                if resident != prev_resident:
                    prev_resident = resident
                    publish_adapters()
        if is_front:
            publish_gauges()


class EngineHandle:
    """Parent-side wrapper of one engine worker process."""

    def __init__(
        self,
        pool: "EnginePool",
        engine: Engine,
        slot_base: int,
        model_name: str,
        devices: List[str],
        engine_kwargs: Dict[str, Any],
        lora_rank: int,
    ):
        self.pool = pool
        self.engine = engine
        self.slot_base = slot_base
        self.model_name = model_name
        # len(devices) == tensor-parallel degree; devices[0] hosts the front
        # rank that owns the request pipe (BASELINE config #4's TP=4 engine)
        self.devices = list(devices)
        self.engine_kwargs = engine_kwargs
        self.lora_rank = lora_rank
        self.active_adapters: List[str] = []
        self.max_slots = int(engine_kwargs.get("max_loras", 8))
        self.healthy = False
        self.procs: List[mp.process.BaseProcess] = []
        self.conn = None
        self._reader: Optional[threading.Thread] = None
        # adapter registry for replay-on-restart: name -> (source, rank)
        self.adapter_registry: Dict[str, Any] = {}
        self._adapter_events: Dict[str, threading.Event] = {}
        self._adapter_errors: Dict[str, Optional[str]] = {}
        # Connection.send splits payloads >16 KiB into multiple OS writes;
        # submit/abort run on the event loop while adapter ops run in executor
        # threads, so every send must hold this lock or the command pipe can
        # interleave and corrupt (kills the worker under adapter rollout +
        # live traffic).
        self._send_lock = threading.Lock()
        # True once the pool has decided this engine should STOP (scale-in /
        # shutdown): the watchdog must never resurrect a retired engine.
        self.retired = False

    def _send(self, msg) -> None:
        with self._send_lock:
            self.conn.send(msg)

    @property
    def tp_size(self) -> int:
        return len(self.devices)

    @property
    def proc(self) -> Optional[mp.process.BaseProcess]:
        return self.procs[0] if self.procs else None

    def start(self, ready_timeout_s: float = 300.0) -> None:
        ctx = mp.get_context("spawn")
        parent_conn, child_conn = ctx.Pipe()
        self.conn = parent_conn
        master_port = 0
        if self.tp_size > 1:
            # each TP engine is its own process-group world on 127.0.0.1
            import socket

            with socket.socket() as s:
                s.bind(("127.0.0.1", 0))
                master_port = s.getsockname()[1]
        self.procs = []
        for r, dev in enumerate(self.devices):
            p = ctx.Process(
                target=_worker_main,
                args=(
                    child_conn if r == 0 else None,
                    self.pool.shared,
                    self.slot_base,
                    self.engine.name,
                    self.model_name,
                    dev,
                    self.engine_kwargs,
                    self.lora_rank,
                    r,
                    self.tp_size,
                    master_port,
                ),
                daemon=True,
                name=f"engine-worker-{self.engine.name}-tp{r}",
            )
            p.start()
            self.procs.append(p)
        child_conn.close()
        if not parent_conn.poll(ready_timeout_s):
            raise TimeoutError(f"engine {self.engine.name} did not become ready")
        msg = parent_conn.recv()
        assert msg[0] == "ready", msg
        self.retired = False  # (re)started deliberately
        self.healthy = True
        self._reader = threading.Thread(
            target=self._read_loop, daemon=True, name=f"reader-{self.engine.name}"
        )
        self._reader.start()

    def _read_loop(self) -> None:
        # Bind the connection ONCE: this thread serves exactly the pipe it
        # was started for.  Re-reading self.conn each iteration raced with
        # restart_engine — the superseded reader could consume messages from
        # the NEW worker's pipe, and its late EOFError flipped healthy=False
        # on the freshly restarted engine (rare flake caught by soak runs).
        conn = self.conn
        while True:
            try:
                msg = conn.recv()
            except (EOFError, OSError):
                if self.conn is conn:
                    # the live connection died -> the engine really is gone
                    self.healthy = False
                # else: a restart superseded this reader; exit silently
                return
            op = msg[0]
            if op == "finished":
                for fin in msg[1]:
                    self.pool._on_finished(fin)
            elif op == "tokens":
                self.pool._on_tokens(msg[1])
            elif op == "adapters":
                self.active_adapters = msg[1]
                self.max_slots = msg[2]
            elif op == "adapter_loaded":
                _, name, err = msg
                self._adapter_errors[name] = err
                ev = self._adapter_events.get(name)
                if ev:
                    ev.set()
            elif op == "adapter_unloaded":
                _, name, _ok = msg
                ev = self._adapter_events.get(name)
                if ev:
                    ev.set()
            elif op == "error":
                self.pool._on_error(msg[1], msg[2])

    # -- command API --------------------------------------------------------
    def submit(self, req: PoolRequest) -> None:
        self._send(("add", req))

    def abort(self, request_id: str) -> None:
        self._send(("abort", request_id))

    def load_adapter(
        self, name: str, source: Any = None, rank: Optional[int] = None,
        wait: bool = True, timeout_s: float = 120.0,
    ) -> Optional[str]:
        """Hot-load one adapter (sidecar POST /v1/load_lora_adapter parity,
        `sidecar.py:177-195`).  Returns an error string or None."""
        if rank is None:
            rank = self.lora_rank
        self.adapter_registry[name] = (source, rank)
        ev = threading.Event()
        self._adapter_events[name] = ev
        self._send(("load_adapter", name, source, rank))
        if not wait:
            return None
        if not ev.wait(timeout_s):
            return f"timeout loading adapter {name}"
        return self._adapter_errors.get(name)

    def unload_adapter(self, name: str, wait: bool = True, timeout_s: float = 60.0) -> None:
        """Sidecar POST /v1/unload_lora_adapter parity (`sidecar.py:197-213`)."""
        self.adapter_registry.pop(name, None)
        ev = threading.Event()
        self._adapter_events[name] = ev
        self._send(("unload_adapter", name))
        if wait:
            ev.wait(timeout_s)

    def drain(self, on: bool = True) -> None:
        self._send(("drain", on))

    def stop(self) -> None:
        self.retired = True
        if self.conn is not None:
            try:
                self._send(("stop",))
            except (BrokenPipeError, OSError):
                pass
        for p in self.procs:
            p.join(timeout=10)
            if p.is_alive():
                p.terminate()
        self.healthy = False

    def kill(self) -> None:
        for p in self.procs:
            p.terminate()
            p.join(timeout=10)
        self.healthy = False

    # -- state reads --------------------------------------------------------
    def metrics(self) -> EngineMetrics:
        b = self.slot_base
        s = self.pool.shared
        return EngineMetrics(
            active_models={n: 1 for n in self.active_adapters},
            max_active_models=int(s[b + _SLOT_MAX_LORAS]),
            running_queue_size=int(s[b + _SLOT_RUNNING]),
            waiting_queue_size=int(s[b + _SLOT_WAITING]),
            kv_cache_usage_percent=s[b + _SLOT_KV_USAGE],
            kv_cache_max_token_capacity=int(s[b + _SLOT_KV_CAP]),
            num_preemptions_total=int(s[b + _SLOT_PREEMPTIONS]),
        )

    def heartbeat_age_s(self) -> float:
        return time.time() - self.pool.shared[self.slot_base + _SLOT_HEARTBEAT]

    def is_draining(self) -> bool:
        return self.pool.shared[self.slot_base + _SLOT_DRAINING] > 0.5

    def alive(self) -> bool:
        return (
            bool(self.procs)
            and all(p.is_alive() for p in self.procs)
            and self.healthy
        )


class EnginePool:
    """N engine workers + shared gauges + routing entry point.

    Satisfies the router's `EngineMetricsProvider` protocol directly from
    shared memory — the 50 ms scrape staleness window of the reference
    collapses to a memory read (SURVEY.md §5 'Distributed communication').
    """

    def __init__(
        self,
        model_name: str = "",
        devices: Sequence[str] = (),
        engine_kwargs: Optional[Dict[str, Any]] = None,
        lora_rank: int = 16,
        datastore: Optional[Datastore] = None,
        scheduler: Optional[Scheduler] = None,
        specs: Optional[Sequence["EngineSpec"]] = None,
        max_engines: int = 0,
    ):
        """Homogeneous pool: ``EnginePool(model, devices=[...])`` — one TP=1
        engine per device (the reference's replica fleet).  Mixed pool:
        ``EnginePool(specs=[EngineSpec(...), ...])`` — arbitrary per-engine
        model / device-set / TP degree (BASELINE config #4).

        ``max_engines`` bounds LIFETIME membership capacity (shared gauge
        slots are pre-sized so the pool can grow at runtime — the reference's
        EndpointSlice add/prune, `endpointslice_reconciler.go:50-79`)."""
        if specs is None:
            specs = [
                EngineSpec(
                    name=f"engine-{i}",
                    model=model_name,
                    devices=[dev],
                    engine_kwargs=dict(engine_kwargs or {}),
                    lora_rank=lora_rank,
                )
                for i, dev in enumerate(devices)
            ]
        self.model_name = model_name or (specs[0].model if specs else "")
        self.engine_kwargs = dict(engine_kwargs or {})
        self.lora_rank = lora_rank
        self.max_engines = max(max_engines, 16, 2 * len(specs))
        self.shared = mp.get_context("spawn").Array(
            "d", SLOTS_PER_ENGINE * self.max_engines, lock=False
        )
        self._free_slots: List[int] = [
            SLOTS_PER_ENGINE * i for i in range(self.max_engines - 1, -1, -1)
        ]
        self.datastore = datastore or Datastore()
        self.handles: Dict[str, EngineHandle] = {}
        for spec in specs:
            self._make_handle(spec)
        self.scheduler = scheduler or Scheduler(self)
        self._tickets: Dict[str, RequestTicket] = {}
        self._tickets_lock = threading.Lock()
        # lifetime restarts per engine (observability: /admin/engines)
        self.restart_counts: Dict[str, int] = {}
        self.on_finished: Optional[Callable[[FinishedRequest], None]] = None
        self._watchdog: Optional[threading.Thread] = None
        self._stop = threading.Event()

    def _make_handle(self, spec: "EngineSpec") -> EngineHandle:
        """Allocate a gauge slot + handle + datastore entry (not started)."""
        if spec.name in self.handles:
            raise ValueError(f"engine {spec.name} already in the pool")
        if not self._free_slots:
            raise RuntimeError(f"pool at max_engines={self.max_engines}")
        slot_base = self._free_slots.pop()
        dev0 = spec.devices[0]
        gpu_id = int(dev0.split(":")[1]) if dev0.startswith("cuda:") else (
            0 if dev0 == "cuda" else -1
        )
        eng = Engine(
            name=spec.name,
            gpu_id=gpu_id,
            model=spec.model,
            pool=spec.pool_ref,
        )
        h = EngineHandle(
            self,
            eng,
            slot_base=slot_base,
            model_name=spec.model,
            devices=spec.devices,
            engine_kwargs=dict(spec.engine_kwargs),
            lora_rank=spec.lora_rank,
        )
        self.handles[eng.name] = h
        self.datastore.add_engine(eng)
        return h

    # -- lifecycle ----------------------------------------------------------
    def start(self, watchdog: bool = True, auto_restart: bool = False) -> None:
        """``auto_restart``: the watchdog re-spawns a dead engine (with
        adapter-registry replay) instead of only dropping it — the kubelet
        restart semantics the reference's pod fleet gets for free.  Capped
        at 3 consecutive failures per engine (crash-loop guard); a
        successful heartbeat resets the count."""
        self._auto_restart = auto_restart
        self._restart_failures: Dict[str, int] = {}
        threads = []
        for h in self.handles.values():
            t = threading.Thread(target=h.start, daemon=True)
            t.start()
            threads.append(t)
        for t in threads:
            t.join()
        for h in self.handles.values():
            if not h.alive():
                raise RuntimeError(f"engine {h.engine.name} failed to start")
        if watchdog:
            self._watchdog = threading.Thread(
                target=self._watchdog_loop, daemon=True, name="pool-watchdog"
            )
            self._watchdog.start()

    def stop(self, drain_timeout_s: float = 0.0) -> None:
        """Stop all engines.  ``drain_timeout_s`` > 0: graceful — engines
        stop ADMITTING (drain on), in-flight requests run to completion (or
        the deadline), then workers exit.  The reference's SIGTERM grace
        period (main.go graceful gRPC shutdown)."""
        self._stop.set()
        if drain_timeout_s > 0:
            for h in list(self.handles.values()):
                try:
                    h.drain(True)
                except (BrokenPipeError, OSError):  # already dead
                    pass
            deadline = time.monotonic() + drain_timeout_s
            while time.monotonic() < deadline:
                with self._tickets_lock:
                    if not self._tickets:
                        break
                time.sleep(0.05)
        for h in list(self.handles.values()):
            h.stop()

    def _watchdog_loop(self, interval_s: float = 1.0) -> None:
        """Engine health: stale heartbeat (hung kernel / dead worker) drops
        the engine from the routing set — reference Ready-condition semantics
        (`endpointslice_reconciler.go:107-111`)."""
        while not self._stop.wait(interval_s):
            for h in list(self.handles.values()):  # membership may change live
                stale = h.heartbeat_age_s() > HEARTBEAT_STALE_S
                dead = bool(h.procs) and any(not p.is_alive() for p in h.procs)
                name = h.engine.name
                if h.healthy and not (stale or dead):
                    self._restart_failures.pop(name, None)  # recovered
                if (
                    not h.healthy and not stale and not dead and h.procs
                    and not h.retired
                ):
                    # Heartbeats resumed and the workers are alive: the
                    # engine was only STALLED (host CPU starvation, long
                    # kernel), not dead — return it to the routing set (the
                    # reference's EndpointSlice Ready condition flipping
                    # back to Ready).  Crash recovery still goes through
                    # restart_engine (procs dead ⇒ `dead` stays true).
                    log.warning("engine %s heartbeat recovered; re-adding", name)
                    h.healthy = True
                    self.datastore.add_engine(h.engine)
                    continue
                if (stale or dead) and h.healthy:
                    log.error(
                        "engine %s unhealthy (stale=%s dead=%s); dropping from pool",
                        name, stale, dead,
                    )
                    h.healthy = False
                    self.datastore.remove_engine(name)
                    self._fail_inflight(name, "engine died")
                # Auto-restart is gated on the OBSERVED DOWN STATE, not on
                # the healthy->unhealthy transition above: the pipe reader
                # thread usually notices a dead worker first (EOF within ms)
                # and flips healthy itself, so a transition-gated restart
                # would ~never fire.
                if (
                    getattr(self, "_auto_restart", False)
                    and not h.healthy
                    and not h.retired
                    and (stale or dead)
                ):
                        fails = self._restart_failures.get(name, 0)
                        if fails >= 3:
                            log.error(
                                "engine %s crash-looping (%d restarts); "
                                "leaving down", name, fails,
                            )
                            continue
                        self._restart_failures[name] = fails + 1
                        log.warning("auto-restarting engine %s (attempt %d)",
                                    name, fails + 1)
                        try:
                            self.restart_engine(name)
                        except Exception as e:
                            log.error("auto-restart of %s failed: %s", name, e)

    def _fail_inflight(self, engine_name: str, reason: str) -> None:
        with self._tickets_lock:
            doomed = [t for t in self._tickets.values() if t.engine_name == engine_name]
            for t in doomed:
                del self._tickets[t.request_id]
        for t in doomed:
            t.error = reason
            if t.stream:
                t.tokens.put(None)
            t.done.set()

    def restart_engine(self, name: str) -> None:
        """Re-spawn a dead engine and replay its adapter registry (reference:
        pod restart + sidecar reconcile, SURVEY.md §5 failure detection)."""
        self.restart_counts[name] = self.restart_counts.get(name, 0) + 1
        h = self.handles[name]
        h.kill()
        registry = dict(h.adapter_registry)
        h.start()
        for aname, (source, rank) in registry.items():
            h.load_adapter(aname, source, rank, wait=False)
        self.datastore.add_engine(h.engine)

    # -- runtime membership (reference EndpointSlice add/prune,
    #    endpointslice_reconciler.go:50-79) ----------------------------------
    def add_engine_live(self, spec: "EngineSpec") -> EngineHandle:
        """Scale-out: spawn a new engine worker into the running pool; it
        joins the routing set once its first heartbeat lands."""
        h = self._make_handle(spec)
        try:
            h.start()
        except Exception:
            self.datastore.remove_engine(spec.name)
            del self.handles[spec.name]
            self._free_slots.append(h.slot_base)
            raise
        log.info("engine %s added to pool (device %s)", spec.name, spec.devices)
        return h

    def remove_engine_live(
        self, name: str, drain: bool = True, drain_timeout_s: float = 60.0
    ) -> None:
        """Scale-in: drop the engine from the routing set immediately, drain
        in-flight work (bounded), then stop the worker and free its slot."""
        h = self.handles.get(name)
        if h is None:
            return
        h.retired = True  # watchdog must not re-add during drain/stop
        self.datastore.remove_engine(name)
        if drain and h.alive():
            h.drain(True)
            deadline = time.monotonic() + drain_timeout_s
            while time.monotonic() < deadline:
                with self._tickets_lock:
                    inflight = any(
                        t.engine_name == name for t in self._tickets.values()
                    )
                if not inflight:
                    break
                time.sleep(0.02)
        h.stop()
        self._fail_inflight(name, "engine removed from pool")
        del self.handles[name]
        self._free_slots.append(h.slot_base)
        log.info("engine %s removed from pool", name)

    def reconcile_membership(
        self, specs: Sequence["EngineSpec"], drain_timeout_s: float = 60.0
    ) -> tuple:
        """Make the live engine set match ``specs`` by name: start engines
        that are missing, drain-and-remove engines no longer desired.
        Returns (added_names, removed_names)."""
        desired = {s.name: s for s in specs}
        current = set(self.handles)
        added, removed = [], []
        for name in sorted(current - set(desired)):
            self.remove_engine_live(name, drain=True, drain_timeout_s=drain_timeout_s)
            removed.append(name)
        for name, spec in desired.items():
            if name not in self.handles:
                self.add_engine_live(spec)
                added.append(name)
        return added, removed

    # -- EngineMetricsProvider protocol -------------------------------------
    def all_engine_metrics(self) -> List[EngineState]:
        out = []
        for e in self.datastore.get_engines():
            h = self.handles.get(e.name)
            if h is None or not h.healthy or h.is_draining():
                continue
            out.append(EngineState(engine=e, metrics=h.metrics()))
        return out

    # -- request path -------------------------------------------------------
    def route(self, req: LLMRequest) -> Engine:
        """Router decision only (reference Schedule, scheduler.go:113-122)."""
        return self.scheduler.schedule(req)

    def submit(self, engine_name: str, req: PoolRequest) -> RequestTicket:
        h = self.handles.get(engine_name)
        if h is None:
            # engine removed between route and dispatch (scale-in race):
            # surface as retryable, same as the died-mid-dispatch path below
            raise SchedulingError(f"engine {engine_name} left the pool")
        ticket = RequestTicket(req.request_id, engine_name, req.stream)
        with self._tickets_lock:
            self._tickets[req.request_id] = ticket
        try:
            h.submit(req)
        except (BrokenPipeError, OSError) as e:
            # engine died between routing and dispatch: drop it from the
            # routing set now (don't wait for the watchdog tick) and surface
            # a retryable scheduling error
            h.healthy = False
            self.datastore.remove_engine(engine_name)
            with self._tickets_lock:
                self._tickets.pop(req.request_id, None)
            raise SchedulingError(f"engine {engine_name} unavailable: {e}") from e
        return ticket

    def route_and_submit(self, lreq: LLMRequest, preq: PoolRequest) -> RequestTicket:
        """The full hot path: filter-tree routing + direct dispatch (replaces
        the reference's header-mutation + Envoy ORIGINAL_DST hop).  Raises
        ResourceExhausted for shed sheddable traffic (HTTP 429 upstream)."""
        engine = self.route(lreq)
        return self.submit(engine.name, preq)

    def abort(self, request_id: str) -> None:
        with self._tickets_lock:
            ticket = self._tickets.pop(request_id, None)
        if ticket is not None:
            self.handles[ticket.engine_name].abort(request_id)
            ticket.error = "aborted"
            if ticket.stream:
                ticket.tokens.put(None)
            ticket.done.set()

    # -- worker event sinks --------------------------------------------------
    def _on_finished(self, fin: FinishedRequest) -> None:
        with self._tickets_lock:
            ticket = self._tickets.pop(fin.request_id, None)
        if ticket is not None:
            ticket.result = fin
            if ticket.stream:
                ticket.tokens.put(None)  # sentinel: stream end
            ticket.done.set()
        if self.on_finished is not None:
            self.on_finished(fin)

    def _on_tokens(self, toks: List) -> None:
        with self._tickets_lock:
            for rid, tok in toks:
                t = self._tickets.get(rid)
                if t is not None:
                    t.tokens.put(tok)

    def _on_error(self, request_id: Optional[str], msg: str) -> None:
        if request_id is None:
            log.error("engine error: %s", msg)
            return
        with self._tickets_lock:
            ticket = self._tickets.pop(request_id, None)
        if ticket is not None:
            ticket.error = msg
            if ticket.stream:
                ticket.tokens.put(None)
            ticket.done.set()
