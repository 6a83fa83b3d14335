"""Engine-metrics provider: the router's view of live engine state.

Reference: `pkg/ext-proc/backend/provider.go`.  The reference keeps a
sync.Map[Pod]*PodMetrics and refreshes it with two daemon goroutines — pod set
every 10 s, metrics every 50 ms via concurrent HTTP scrapes with a 5 s
timeout; failed scrapes keep last-known metrics.

The MI355X build has two modes:

  * **direct mode** (default, co-resident engines): each per-GPU engine
    publishes its gauges into a shared-memory snapshot after every scheduler
    step; `all_engine_metrics` reads them with zero staleness — the 50 ms
    scrape window of the reference collapses to a memory read.  See
    `llm_instance_gateway_amd.metrics.shm`.
  * **refresh mode** (parity / remote engines): background threads poll a
    `MetricsSource` per engine at the reference's intervals and retain stale
    metrics on error (reference provider.go:134-179).
"""

from __future__ import annotations

import logging
import threading
import time
from typing import Callable, Dict, List, Optional, Protocol

from .datastore import Datastore
from .types import Engine, EngineMetrics, EngineState

log = logging.getLogger(__name__)

# Reference defaults: main.go:38-39, provider.go:14.
DEFAULT_REFRESH_ENGINES_INTERVAL_S = 10.0
DEFAULT_REFRESH_METRICS_INTERVAL_S = 0.050
DEFAULT_FETCH_TIMEOUT_S = 5.0


class MetricsSource(Protocol):
    """Reference PodMetricsClient (provider.go:34-36)."""

    def fetch_metrics(
        self, engine: Engine, existing: EngineMetrics
    ) -> EngineMetrics: ...


class Provider:
    """Reference Provider (provider.go:17-101)."""

    def __init__(self, source: Optional[MetricsSource], datastore: Datastore) -> None:
        self._source = source
        self._datastore = datastore
        self._lock = threading.Lock()
        self._states: Dict[str, EngineState] = {}
        # bumped on every state change: cheap staleness check for consumers
        # that mirror the snapshot (router/native.py syncs only on change)
        self.version = 0
        self._stop = threading.Event()
        self._threads: List[threading.Thread] = []

    # -- snapshot API (provider.go:38-58) -----------------------------------
    def all_engine_metrics(self) -> List[EngineState]:
        """Snapshot of every engine's state.

        The reference deep-copies each PodMetrics here (provider.go:38-46)
        because its scrape goroutines mutate entries concurrently.  This
        provider only ever *replaces* whole EngineState objects under the
        lock (update_engine_metrics / refresh_metrics_once), so handing out
        references is race-free as long as consumers treat them read-only —
        which the filter chain does.  Skipping the per-call copy is ~70% of
        schedule() latency at 200 engines (benchmarks/bench_router.py)."""
        with self._lock:
            return list(self._states.values())

    def get_engine_metrics(self, name: str) -> Optional[EngineState]:
        with self._lock:
            return self._states.get(name)

    def update_engine_metrics(self, engine: Engine, metrics: EngineMetrics) -> None:
        """Atomic swap of one engine's metrics (provider.go:52-58).  In direct
        mode the engine worker calls this after every scheduler iteration."""
        with self._lock:
            self._states[engine.name] = EngineState(engine=engine, metrics=metrics)
            self.version += 1

    # -- lifecycle (provider.go:60-101) -------------------------------------
    def init(
        self,
        refresh_engines_interval_s: float = DEFAULT_REFRESH_ENGINES_INTERVAL_S,
        refresh_metrics_interval_s: float = DEFAULT_REFRESH_METRICS_INTERVAL_S,
        start_loops: bool = True,
    ) -> None:
        """Populate once, then (optionally) start the two refresh loops."""
        self.refresh_engines_once()
        if self._source is not None:
            self.refresh_metrics_once()
        if not start_loops:
            return
        t1 = threading.Thread(
            target=self._loop,
            args=(self.refresh_engines_once, refresh_engines_interval_s),
            daemon=True,
            name="provider-refresh-engines",
        )
        t1.start()
        self._threads.append(t1)
        if self._source is not None:
            t2 = threading.Thread(
                target=self._loop,
                args=(self.refresh_metrics_once, refresh_metrics_interval_s),
                daemon=True,
                name="provider-refresh-metrics",
            )
            t2.start()
            self._threads.append(t2)

    def stop(self) -> None:
        self._stop.set()

    def _loop(self, fn: Callable[[], None], interval_s: float) -> None:
        while not self._stop.wait(interval_s):
            try:
                fn()
            except Exception:  # pragma: no cover - defensive, mirrors multierr log
                log.exception("provider refresh failed")

    def refresh_engines_once(self) -> None:
        """Sync the tracked engine set with the datastore (provider.go:105-132):
        add new engines with empty metrics, drop vanished ones."""
        engines = {e.name: e for e in self._datastore.get_engines()}
        with self._lock:
            changed = False
            for name, eng in engines.items():
                if name not in self._states:
                    self._states[name] = EngineState(engine=eng)
                    changed = True
            for name in list(self._states):
                if name not in engines:
                    del self._states[name]
                    changed = True
            if changed:
                self.version += 1

    def refresh_metrics_once(self, timeout_s: float = DEFAULT_FETCH_TIMEOUT_S) -> List[Exception]:
        """Fan-out fetch for every engine (provider.go:134-179).  Errors are
        collected; the engine keeps its last-known metrics."""
        assert self._source is not None
        with self._lock:
            snapshot = [(s.engine, s.metrics) for s in self._states.values()]
        errs: List[Exception] = []
        errs_lock = threading.Lock()
        results: Dict[str, EngineMetrics] = {}

        def fetch(engine: Engine, existing: EngineMetrics) -> None:
            try:
                results[engine.name] = self._source.fetch_metrics(engine, existing)
            except Exception as e:
                with errs_lock:
                    errs.append(e)

        threads = [
            threading.Thread(target=fetch, args=(e, m), daemon=True)
            for e, m in snapshot
        ]
        for t in threads:
            t.start()
        deadline = time.monotonic() + timeout_s
        for t in threads:
            t.join(max(0.0, deadline - time.monotonic()))
        with self._lock:
            for name, metrics in results.items():
                if name in self._states:
                    self._states[name] = EngineState(
                        engine=self._states[name].engine, metrics=metrics
                    )
            if results:
                self.version += 1
        if errs:
            log.debug("refresh_metrics_once: %d fetch errors", len(errs))
        return errs
