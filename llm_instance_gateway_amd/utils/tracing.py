"""Tracing: rocprof-friendly GPU markers + per-request span log.

The reference has no tracing at all (the only timing is a
debug log with a "TODO: add a metric instead" — `provider.go:138-142`).
Here:

  * `gpu_range(name)` emits roctx ranges (torch.cuda.nvtx IS roctx on ROCm)
    around engine phases, so `rocprofv3 --marker-trace` attributes kernel
    time to build_batch / forward / sample / graph_replay per step.
    Enabled with LIG_TRACE=1 (markers cost ~1 us each — off for benches).
  * `SpanLog` records per-request lifecycle events (arrived, routed, queued,
    first_token, finished) in a bounded ring; the front door exposes the
    tail at /admin/trace.
"""

from __future__ import annotations

import collections
import contextlib
import os
import threading
import time
from typing import Deque, Dict, List, Optional

import torch

_enabled = os.environ.get("LIG_TRACE", "0") not in ("", "0", "false")


def set_gpu_tracing(on: bool) -> None:
    global _enabled
    _enabled = on


def gpu_tracing_enabled() -> bool:
    return _enabled


@contextlib.contextmanager
def gpu_range(name: str):
    """roctx range around a GPU-dispatching region (no-op unless enabled)."""
    if _enabled and torch.cuda.is_available():
        torch.cuda.nvtx.range_push(name)
        try:
            yield
        finally:
            torch.cuda.nvtx.range_pop()
    else:
        yield


class SpanLog:
    """Bounded per-request event log (thread-safe)."""

    def __init__(self, capacity: int = 4096):
        self._lock = threading.Lock()
        self._events: Deque = collections.deque(maxlen=capacity)

    def event(self, request_id: str, name: str, **attrs) -> None:
        with self._lock:
            self._events.append(
                {"ts": time.time(), "request_id": request_id, "event": name, **attrs}
            )

    def tail(self, n: int = 100, request_id: Optional[str] = None) -> List[Dict]:
        with self._lock:
            evs = list(self._events)
        if request_id is not None:
            evs = [e for e in evs if e["request_id"] == request_id]
        return evs[-n:]
