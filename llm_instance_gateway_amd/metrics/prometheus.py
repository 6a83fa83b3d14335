"""Prometheus text-format metrics: exposition + parsing.

Two parity roles:

  * **Exposition** — each engine exports the exact gauge family the reference
    router consumes from vLLM (`pkg/ext-proc/backend/vllm/metrics.go:19-32`):
    ``vllm:num_requests_running``, ``vllm:num_requests_waiting``,
    ``vllm:gpu_cache_usage_perc`` and ``vllm:lora_requests_info`` (whose
    *labels* carry the comma-separated resident adapter set and the
    ``max_lora`` slot budget, and whose value is a timestamp).  On top, the
    gateway exports what the reference left as a TODO
    (`provider.go:138-142`, `response.go:64-83` parses usage but never
    exports): request counters, shed (429) counters, token counters and
    TTFT/TPOT/E2E histograms.

  * **Parsing** — `parse_engine_metrics` maps a Prometheus text page back to
    `EngineMetrics` with the reference's semantics
    (`promToPodMetrics`, metrics.go:73-129): missing families keep the
    existing values; among multiple ``vllm:lora_requests_info`` series pick
    the one with the greatest value (gauge-value-as-timestamp,
    `getLatestLoraMetric` metrics.go:135-150).  This is the scrape-mode
    adapter for remote engines.
"""

from __future__ import annotations

import bisect
import re
import threading
import time
from typing import Dict, Iterable, Optional, Tuple

from ..router.types import EngineMetrics

# Metric names the reference hard-codes (backend/vllm/metrics.go:19-32).
LORA_INFO_METRIC = "vllm:lora_requests_info"
LORA_INFO_RUNNING_LABEL = "running_lora_adapters"
LORA_INFO_MAX_LABEL = "max_lora"
RUNNING_QUEUE_METRIC = "vllm:num_requests_running"
WAITING_QUEUE_METRIC = "vllm:num_requests_waiting"
KV_CACHE_USAGE_METRIC = "vllm:gpu_cache_usage_perc"
PREEMPTIONS_METRIC = "vllm:num_preemptions_total"


def _escape_label(v: str) -> str:
    return v.replace("\\", "\\\\").replace('"', '\\"').replace("\n", "\\n")


def render_engine_metrics(
    engine_name: str, m: EngineMetrics, now: Optional[float] = None
) -> str:
    """Render one engine's gauges in Prometheus text format.

    The ``vllm:lora_requests_info`` encoding matches what the reference's
    parser expects (metrics.go:105-126): adapter set and slot budget ride in
    labels, the sample value is a unix timestamp used for latest-wins.
    """
    ts = now if now is not None else time.time()
    lab = f'{{engine="{_escape_label(engine_name)}"}}'
    lines = [
        f"# TYPE {RUNNING_QUEUE_METRIC} gauge",
        f"{RUNNING_QUEUE_METRIC}{lab} {m.running_queue_size}",
        f"# TYPE {WAITING_QUEUE_METRIC} gauge",
        f"{WAITING_QUEUE_METRIC}{lab} {m.waiting_queue_size}",
        f"# TYPE {KV_CACHE_USAGE_METRIC} gauge",
        f"{KV_CACHE_USAGE_METRIC}{lab} {m.kv_cache_usage_percent:.6f}",
        f"# TYPE {PREEMPTIONS_METRIC} counter",
        f"{PREEMPTIONS_METRIC}{lab} {m.num_preemptions_total}",
        f"# TYPE {LORA_INFO_METRIC} gauge",
        "{}{{engine=\"{}\",{}=\"{}\",{}=\"{}\"}} {:.3f}".format(
            LORA_INFO_METRIC,
            _escape_label(engine_name),
            LORA_INFO_RUNNING_LABEL,
            _escape_label(",".join(sorted(m.active_models))),
            LORA_INFO_MAX_LABEL,
            m.max_active_models,
            ts,
        ),
    ]
    return "\n".join(lines) + "\n"


_LABEL_RE = re.compile(r'([A-Za-z_][\w.:-]*)="((?:[^"\\]|\\.)*)"')
_UNESCAPE_RE = re.compile(r"\\(.)")


def _parse_line(line: str) -> Optional[Tuple[str, Dict[str, str], float]]:
    """Parse one Prometheus sample line -> (name, labels, value)."""
    line = line.strip()
    if not line or line.startswith("#"):
        return None
    labels: Dict[str, str] = {}
    if "{" in line:
        name, rest = line.split("{", 1)
        labpart, _, valpart = rest.rpartition("}")
        for key, raw in _LABEL_RE.findall(labpart):
            labels[key] = _UNESCAPE_RE.sub(
                lambda m: {"n": "\n"}.get(m.group(1), m.group(1)), raw
            )
        value_str = valpart.strip()
    else:
        parts = line.split()
        if len(parts) < 2:
            return None
        name, value_str = parts[0], parts[1]
    try:
        value = float(value_str.split()[0])
    except (ValueError, IndexError):
        return None
    return name.strip(), labels, value


def parse_engine_metrics(text: str, existing: EngineMetrics) -> EngineMetrics:
    """Prometheus page -> EngineMetrics (reference promToPodMetrics,
    metrics.go:73-129).  Starts from a clone of ``existing`` so missing
    families keep last-known values (reference provider.go:150-159)."""
    out = EngineMetrics(
        active_models=dict(existing.active_models),
        max_active_models=existing.max_active_models,
        running_queue_size=existing.running_queue_size,
        waiting_queue_size=existing.waiting_queue_size,
        kv_cache_usage_percent=existing.kv_cache_usage_percent,
        kv_cache_max_token_capacity=existing.kv_cache_max_token_capacity,
        num_preemptions_total=existing.num_preemptions_total,
    )
    latest_lora: Optional[Tuple[float, Dict[str, str]]] = None
    for raw in text.splitlines():
        parsed = _parse_line(raw)
        if parsed is None:
            continue
        name, labels, value = parsed
        if name == RUNNING_QUEUE_METRIC:
            out.running_queue_size = int(value)
        elif name == WAITING_QUEUE_METRIC:
            out.waiting_queue_size = int(value)
        elif name == KV_CACHE_USAGE_METRIC:
            out.kv_cache_usage_percent = value
        elif name == PREEMPTIONS_METRIC:
            out.num_preemptions_total = int(value)
        elif name == LORA_INFO_METRIC:
            # latest-wins by gauge value as timestamp (metrics.go:135-150)
            if latest_lora is None or value >= latest_lora[0]:
                latest_lora = (value, labels)
    if latest_lora is not None:
        labels = latest_lora[1]
        running = labels.get(LORA_INFO_RUNNING_LABEL, "")
        out.active_models = {a: 1 for a in running.split(",") if a}
        try:
            out.max_active_models = int(labels.get(LORA_INFO_MAX_LABEL, "0"))
        except ValueError:
            out.max_active_models = 0
    return out


# ---------------------------------------------------------------------------
# Gateway-side instruments (the reference's unexported TODO, provider.go:138).
# ---------------------------------------------------------------------------

# TTFT/TPOT/E2E bucket boundaries in seconds.
DEFAULT_TIME_BUCKETS = (
    0.001, 0.005, 0.01, 0.025, 0.05, 0.1, 0.25, 0.5,
    1.0, 2.5, 5.0, 10.0, 30.0, 60.0,
)


class Histogram:
    """Minimal thread-safe Prometheus histogram."""

    def __init__(self, name: str, help_: str, buckets: Iterable[float] = DEFAULT_TIME_BUCKETS):
        self.name = name
        self.help = help_
        self.buckets = sorted(buckets)
        self._counts = [0] * (len(self.buckets) + 1)
        self._sum = 0.0
        self._n = 0
        self._lock = threading.Lock()

    def observe(self, v: float) -> None:
        idx = bisect.bisect_left(self.buckets, v)
        with self._lock:
            self._counts[idx] += 1
            self._sum += v
            self._n += 1

    def render(self) -> str:
        with self._lock:
            counts, total, s = list(self._counts), self._n, self._sum
        lines = [f"# HELP {self.name} {self.help}", f"# TYPE {self.name} histogram"]
        cum = 0
        for b, c in zip(self.buckets, counts):
            cum += c
            lines.append(f'{self.name}_bucket{{le="{b}"}} {cum}')
        lines.append(f'{self.name}_bucket{{le="+Inf"}} {total}')
        lines.append(f"{self.name}_sum {s:.6f}")
        lines.append(f"{self.name}_count {total}")
        return "\n".join(lines) + "\n"


class Counter:
    """Minimal thread-safe Prometheus counter with optional label sets."""

    def __init__(self, name: str, help_: str):
        self.name = name
        self.help = help_
        self._vals: Dict[Tuple[Tuple[str, str], ...], float] = {}
        self._lock = threading.Lock()

    def inc(self, amount: float = 1.0, **labels: str) -> None:
        key = tuple(sorted(labels.items()))
        with self._lock:
            self._vals[key] = self._vals.get(key, 0.0) + amount

    def value(self, **labels: str) -> float:
        key = tuple(sorted(labels.items()))
        with self._lock:
            return self._vals.get(key, 0.0)

    def render(self) -> str:
        lines = [f"# HELP {self.name} {self.help}", f"# TYPE {self.name} counter"]
        with self._lock:
            items = list(self._vals.items())
        for key, v in sorted(items):
            if key:
                lab = "{" + ",".join(f'{k}="{_escape_label(val)}"' for k, val in key) + "}"
            else:
                lab = ""
            lines.append(f"{self.name}{lab} {v:g}")
        return "\n".join(lines) + "\n"


class GatewayMetrics:
    """All gateway-level instruments, rendered on GET /metrics.

    Exports the signals the reference parsed but never exported
    (usage tokens, response.go:64-93) plus the SLO histograms the goodput
    metric needs (TTFT/TPOT — the simulation's per-request record,
    request.py:3-27).
    """

    def __init__(self) -> None:
        self.requests_total = Counter(
            "gateway_requests_total", "Requests by client model and outcome"
        )
        self.shed_total = Counter(
            "gateway_requests_shed_total",
            "Sheddable requests dropped with 429 (reference scheduler.go:74-90)",
        )
        self.prompt_tokens_total = Counter(
            "gateway_prompt_tokens_total", "Prompt tokens processed"
        )
        self.completion_tokens_total = Counter(
            "gateway_completion_tokens_total", "Completion tokens generated"
        )
        self.ttft = Histogram("gateway_ttft_seconds", "Time to first token")
        self.tpot = Histogram(
            "gateway_tpot_seconds", "Time per output token (per request mean)"
        )
        self.e2e = Histogram("gateway_e2e_latency_seconds", "Request end-to-end latency")
        self.route_latency = Histogram(
            "gateway_route_seconds",
            "Router schedule() latency",
            buckets=(1e-6, 5e-6, 1e-5, 5e-5, 1e-4, 5e-4, 1e-3, 5e-3, 1e-2),
        )

    def render(self) -> str:
        return "".join(
            x.render()
            for x in (
                self.requests_total,
                self.shed_total,
                self.prompt_tokens_total,
                self.completion_tokens_total,
                self.ttft,
                self.tpot,
                self.e2e,
                self.route_latency,
            )
        )
