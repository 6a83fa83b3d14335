"""YAML config plane: the file-driven analogue of the reference's CRDs +
dynamic-LoRA ConfigMap.

One watched YAML file replaces three reference inputs:

  * ``pool:`` — InferencePool CRD (`api/v1alpha1/inferencepool_types.go:26-46`):
    which GPUs serve, base model, TP degree.
  * ``models:`` — InferenceModel CRDs (`inferencemodel_types.go:40-135`):
    client model name -> criticality + weighted traffic split.
  * ``ensureExist/ensureNotExist:`` — the dynamic-LoRA sidecar's
    ``vLLMLoRAConfig`` schema (`tools/dynamic-lora-sidecar/sidecar/
    validation.yaml`, `sidecar.py:215-239`): declarative adapter rollout.
  * ``thresholds:`` / ``refresh:`` — the scheduler consts the reference marks
    "TODO: make configurable" (`scheduler.go:15-24`) and the provider
    intervals (`main.go:38-39`).

`ConfigReconciler` polls the file for changes (the sidecar uses a watchdog
PollingObserver, `sidecar.py:33-44,247-252`) and reconciles datastore +
adapters; `reconcile_adapters` implements the sidecar diff semantics:
load ensureExist − ensureNotExist, unload ensureNotExist − ensureExist, and
**skip adapters present in both lists with a warning**
(`sidecar.py:223-239`), gated on engine health with timeout
(`sidecar.py:158-175`).
"""

from __future__ import annotations

import logging
import threading
import time
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional, Set, Tuple

import yaml

from .types import Criticality, InferenceModel, InferencePool, TargetModel

log = logging.getLogger(__name__)

# Sidecar health-gate defaults (sidecar.py:158-175).
DEFAULT_HEALTH_TIMEOUT_S = 300.0
DEFAULT_HEALTH_INTERVAL_S = 15.0


class ConfigError(ValueError):
    """Schema violation (the sidecar jsonschema-validates, sidecar.py:73-80)."""


@dataclass
class AdapterSpec:
    """One adapter entry (sidecar LoraAdapter: id + source + base)."""

    id: str
    source: str = ""  # path to weights; empty -> deterministic random init
    base_model: str = ""
    rank: int = 16

    @classmethod
    def parse(cls, raw: Any) -> "AdapterSpec":
        if isinstance(raw, str):
            return cls(id=raw)
        if not isinstance(raw, dict) or "id" not in raw:
            raise ConfigError(f"adapter entry needs an 'id': {raw!r}")
        return cls(
            id=str(raw["id"]),
            source=str(raw.get("source", "")),
            base_model=str(raw.get("base-model", raw.get("base_model", ""))),
            rank=int(raw.get("rank", 16)),
        )


@dataclass
class GatewayConfig:
    pool: InferencePool = field(default_factory=lambda: InferencePool(name="default"))
    models: List[InferenceModel] = field(default_factory=list)
    ensure_exist: List[AdapterSpec] = field(default_factory=list)
    ensure_not_exist: List[AdapterSpec] = field(default_factory=list)
    # scheduler thresholds (reference scheduler.go:15-24)
    kv_cache_threshold: float = 0.8
    queue_threshold_critical: int = 5
    queueing_threshold_lora: int = 50
    # provider refresh intervals (reference main.go:38-39)
    refresh_metrics_interval_s: float = 0.050
    refresh_engines_interval_s: float = 10.0
    # ops: watchdog re-spawns dead engines (kubelet restart semantics)
    auto_restart_engines: bool = False
    # front door
    port: int = 8081
    health_check_timeout_s: float = DEFAULT_HEALTH_TIMEOUT_S
    health_check_interval_s: float = DEFAULT_HEALTH_INTERVAL_S


def parse_config(raw: Dict[str, Any]) -> GatewayConfig:
    """Validate + parse one YAML document into a GatewayConfig."""
    if not isinstance(raw, dict):
        raise ConfigError("config root must be a mapping")
    cfg = GatewayConfig()

    p = raw.get("pool", {})
    if p:
        if not isinstance(p, dict) or "name" not in p:
            raise ConfigError("pool needs a 'name'")
        cfg.pool = InferencePool(
            name=str(p["name"]),
            gpu_ids=[int(g) for g in p.get("gpu_ids", [])],
            engine_addresses=[str(a) for a in p.get("engine_addresses", [])],
            base_model=str(p.get("base_model", "")),
            tensor_parallel=int(p.get("tensor_parallel", 1)),
            devices=[str(d) for d in p.get("devices", [])],
            quant=str(p.get("quant", "")),
        )
        if cfg.pool.quant not in ("", "fp8"):
            raise ConfigError(f"pool.quant must be '' or 'fp8', got {cfg.pool.quant!r}")

    for m in raw.get("models", []) or []:
        if not isinstance(m, dict) or "model_name" not in m:
            raise ConfigError(f"model entry needs 'model_name': {m!r}")
        crit = m.get("criticality", "Default")
        try:
            criticality = Criticality(crit)
        except ValueError as e:
            raise ConfigError(f"bad criticality {crit!r}") from e
        targets = [
            TargetModel(name=str(t["name"]), weight=int(t.get("weight", 1)))
            for t in m.get("target_models", []) or []
        ]
        cfg.models.append(
            InferenceModel(
                model_name=str(m["model_name"]),
                criticality=criticality,
                target_models=targets,
                pool_ref=str(m.get("pool_ref", cfg.pool.name)),
            )
        )

    cfg.ensure_exist = [AdapterSpec.parse(a) for a in raw.get("ensureExist", {}).get("models", []) or []] if isinstance(raw.get("ensureExist"), dict) else [AdapterSpec.parse(a) for a in raw.get("ensureExist", []) or []]
    ene = raw.get("ensureNotExist")
    if isinstance(ene, dict):
        cfg.ensure_not_exist = [AdapterSpec.parse(a) for a in ene.get("models", []) or []]
    else:
        cfg.ensure_not_exist = [AdapterSpec.parse(a) for a in ene or []]

    cfg.auto_restart_engines = bool(raw.get("auto_restart_engines", False))
    th = raw.get("thresholds", {}) or {}
    cfg.kv_cache_threshold = float(th.get("kv_cache_threshold", cfg.kv_cache_threshold))
    cfg.queue_threshold_critical = int(
        th.get("queue_threshold_critical", cfg.queue_threshold_critical)
    )
    cfg.queueing_threshold_lora = int(
        th.get("queueing_threshold_lora", cfg.queueing_threshold_lora)
    )
    if not (0.0 <= cfg.kv_cache_threshold <= 1.0):
        raise ConfigError("kv_cache_threshold must be in [0,1]")

    rf = raw.get("refresh", {}) or {}
    cfg.refresh_metrics_interval_s = float(
        rf.get("metrics_interval_s", cfg.refresh_metrics_interval_s)
    )
    cfg.refresh_engines_interval_s = float(
        rf.get("engines_interval_s", cfg.refresh_engines_interval_s)
    )

    srv = raw.get("server", {}) or {}
    cfg.port = int(srv.get("port", cfg.port))
    cfg.health_check_timeout_s = float(
        srv.get("health_check_timeout_s", cfg.health_check_timeout_s)
    )
    cfg.health_check_interval_s = float(
        srv.get("health_check_interval_s", cfg.health_check_interval_s)
    )
    return cfg


def load_config(path: str) -> GatewayConfig:
    with open(path) as f:
        raw = yaml.safe_load(f) or {}
    return parse_config(raw)


def reconcile_models(cfg: GatewayConfig, datastore) -> None:
    """Config models -> datastore (reference InferenceModelReconciler
    store/delete semantics, `inferencemodel_reconciler.go:45-55`: only models
    whose pool_ref matches the serving pool are stored; vanished ones are
    deleted)."""
    datastore.set_pool(cfg.pool)
    wanted = {
        m.model_name: m for m in cfg.models if m.pool_ref in ("", cfg.pool.name)
    }
    for name, m in wanted.items():
        datastore.store_model(m)
    for existing in datastore.all_models():
        if existing.model_name not in wanted:
            datastore.delete_model(existing.model_name)


def desired_engine_specs(
    cfg: GatewayConfig,
    default_model: str = "",
    engine_kwargs: Optional[Dict[str, Any]] = None,
    lora_rank: int = 16,
) -> Optional[List[Any]]:
    """Derive the desired engine set from the pool config (the reference's
    EndpointSlice -> Pod-set mapping, `endpointslice_reconciler.go:50-79`):
    ``pool.devices`` (explicit device names) or ``pool.gpu_ids`` (cuda:<id>),
    chunked into TP groups of ``tensor_parallel`` devices.  Returns None when
    the config does not pin membership (empty devices AND gpu_ids) — then the
    reconciler leaves the running set alone."""
    from ..serving.pool import EngineSpec

    p = cfg.pool
    devs = list(p.devices) if p.devices else [f"cuda:{g}" for g in p.gpu_ids]
    if not devs:
        return None
    tp = max(1, p.tensor_parallel)
    if len(devs) % tp != 0:
        raise ConfigError(
            f"pool lists {len(devs)} devices, not a multiple of tensor_parallel={tp}"
        )
    model = p.base_model or default_model
    specs = []
    for i in range(0, len(devs), tp):
        specs.append(
            EngineSpec(
                name=f"engine-{i // tp}",
                model=model,
                devices=devs[i : i + tp],
                engine_kwargs=dict(engine_kwargs or {}),
                lora_rank=lora_rank,
                pool_ref=p.name,
            )
        )
    return specs


def reconcile_pool_membership(cfg: GatewayConfig, pool) -> Tuple[List[str], List[str]]:
    """Grow/shrink the live engine set to match the config (scale-out spawns
    workers, scale-in drains then stops).  Returns (added, removed)."""
    specs = desired_engine_specs(
        cfg, pool.model_name, pool.engine_kwargs, pool.lora_rank
    )
    if specs is None:
        return [], []
    return pool.reconcile_membership(specs)


def reconcile_adapters(cfg: GatewayConfig, pool) -> Tuple[Set[str], Set[str], Set[str]]:
    """Sidecar reconcile (`sidecar.py:215-239`): returns (loaded, unloaded,
    skipped) adapter-id sets.  Adapters named in BOTH lists are skipped with a
    warning (`sidecar.py:223-229`)."""
    exist = {a.id: a for a in cfg.ensure_exist}
    not_exist = {a.id for a in cfg.ensure_not_exist}
    dupes = set(exist) & not_exist
    for d in sorted(dupes):
        log.warning(
            "skipping adapter %r: present in both ensureExist and ensureNotExist", d
        )
    to_load = {i: a for i, a in exist.items() if i not in dupes}
    to_unload = not_exist - dupes

    loaded: Set[str] = set()
    unloaded: Set[str] = set()
    for h in pool.handles.values():
        if not h.alive():
            continue
        registered = set(h.adapter_registry)
        for aid, spec in to_load.items():
            if aid not in registered:
                err = h.load_adapter(aid, spec.source or None, rank=spec.rank)
                if err:
                    log.error("load adapter %s on %s failed: %s", aid, h.engine.name, err)
                else:
                    loaded.add(aid)
        for aid in to_unload & registered:
            h.unload_adapter(aid)
            unloaded.add(aid)
    return loaded, unloaded, dupes


def reconcile_adapters_remote(cfg: GatewayConfig, actuator) -> Tuple[Set[str], Set[str], Set[str]]:
    """Sidecar reconcile against a REMOTE server (`sidecar.py:215-239`): diff
    ensureExist − ensureNotExist against the server's GET /v1/models listing
    (`sidecar.py:143-156`), actuate over its dynamic-LoRA HTTP API.
    ``actuator`` is a `router.remote.RemoteAdapterActuator`."""
    exist = {a.id: a for a in cfg.ensure_exist}
    not_exist = {a.id for a in cfg.ensure_not_exist}
    dupes = set(exist) & not_exist
    for d in sorted(dupes):
        log.warning(
            "skipping adapter %r: present in both ensureExist and ensureNotExist", d
        )
    listed = set(actuator.list_models())
    loaded: Set[str] = set()
    unloaded: Set[str] = set()
    for aid, spec in exist.items():
        if aid in dupes or aid in listed:
            continue
        err = actuator.load(aid, spec.source, spec.base_model)
        if err:
            log.error("remote load %s failed: %s", aid, err)
        else:
            loaded.add(aid)
    for aid in (not_exist - dupes) & listed:
        err = actuator.unload(aid)
        if err:
            log.error("remote unload %s failed: %s", aid, err)
        else:
            unloaded.add(aid)
    return loaded, unloaded, dupes


def wait_healthy(pool, timeout_s: float, interval_s: float) -> bool:
    """Sidecar health gate (`sidecar.py:158-175`): poll until an engine is
    serving or timeout."""
    deadline = time.monotonic() + timeout_s
    while True:
        if any(h.alive() for h in pool.handles.values()):
            return True
        if time.monotonic() >= deadline:
            return False
        time.sleep(interval_s)


class ConfigReconciler:
    """Polling file watcher + reconcile loop (sidecar PollingObserver
    `sidecar.py:247-252` + the reference's three K8s reconcilers collapsed
    into one).  A malformed config is logged and skipped; the last good state
    stays active (CRD validation analogue)."""

    def __init__(
        self,
        path: str,
        pool=None,
        poll_interval_s: float = 1.0,
        health_timeout_s: float = DEFAULT_HEALTH_TIMEOUT_S,
        health_interval_s: float = DEFAULT_HEALTH_INTERVAL_S,
        datastore=None,
        remote_actuators=None,
    ) -> None:
        """In-process mode: pass ``pool``.  Remote (sidecar) mode: pass
        ``datastore`` + ``remote_actuators`` (one per model server)."""
        self.path = path
        self.pool = pool
        self.datastore = datastore if datastore is not None else (
            pool.datastore if pool is not None else None
        )
        self.remote_actuators = list(remote_actuators or [])
        self.poll_interval_s = poll_interval_s
        self.health_timeout_s = health_timeout_s
        self.health_interval_s = health_interval_s
        self._mtime: Optional[float] = None
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self.last_config: Optional[GatewayConfig] = None
        self.reconcile_count = 0

    def reconcile_once(self) -> bool:
        """Load + apply the file; returns True if applied."""
        try:
            cfg = load_config(self.path)
        except (OSError, ConfigError, yaml.YAMLError) as e:
            log.error("config %s rejected: %s", self.path, e)
            return False
        if self.pool is not None:
            if not wait_healthy(self.pool, self.health_timeout_s, self.health_interval_s):
                log.error("no healthy engine within %.0fs; skipping reconcile", self.health_timeout_s)
                return False
        else:
            for act in self.remote_actuators:
                if not act.wait_healthy(self.health_timeout_s, self.health_interval_s):
                    log.error("remote engine %s unhealthy; skipping reconcile", act.base_url)
                    return False
        if self.datastore is not None:
            reconcile_models(cfg, self.datastore)
        if self.pool is not None:
            # membership first: engines added here pick up the adapter set in
            # the same pass (the reference's EndpointSlice reconcile runs
            # independently of the sidecar; collapsing them keeps ordering)
            try:
                added, removed = reconcile_pool_membership(cfg, self.pool)
                if added or removed:
                    log.info("pool membership: +%s -%s", added, removed)
            except (ConfigError, RuntimeError) as e:
                log.error("pool membership reconcile failed: %s", e)
            reconcile_adapters(cfg, self.pool)
            # scheduler thresholds: rebuild the tree when a live edit changes
            # them (the reference's compile-time consts, made runtime here)
            last = self.last_config
            if last is None or (
                last.kv_cache_threshold,
                last.queue_threshold_critical,
                last.queueing_threshold_lora,
            ) != (
                cfg.kv_cache_threshold,
                cfg.queue_threshold_critical,
                cfg.queueing_threshold_lora,
            ):
                sched = getattr(self.pool, "scheduler", None)
                if sched is not None and hasattr(sched, "set_thresholds"):
                    sched.set_thresholds(
                        cfg.kv_cache_threshold,
                        cfg.queue_threshold_critical,
                        cfg.queueing_threshold_lora,
                    )
                    if last is not None:
                        log.info(
                            "scheduler thresholds updated: kv=%.2f qc=%d ql=%d",
                            cfg.kv_cache_threshold,
                            cfg.queue_threshold_critical,
                            cfg.queueing_threshold_lora,
                        )
        for act in self.remote_actuators:
            reconcile_adapters_remote(cfg, act)
        self.last_config = cfg
        self.reconcile_count += 1
        return True

    def start(self) -> None:
        import os

        try:
            self._mtime = os.stat(self.path).st_mtime
        except OSError:
            self._mtime = None
        self.reconcile_once()
        self._thread = threading.Thread(target=self._loop, daemon=True, name="config-reconciler")
        self._thread.start()

    def stop(self) -> None:
        self._stop.set()

    def _loop(self) -> None:
        import os

        while not self._stop.wait(self.poll_interval_s):
            try:
                mtime = os.stat(self.path).st_mtime
            except OSError:
                continue
            if mtime != self._mtime:
                self._mtime = mtime
                self.reconcile_once()
