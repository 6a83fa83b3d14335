from .prometheus import (
    Counter,
    GatewayMetrics,
    Histogram,
    parse_engine_metrics,
    render_engine_metrics,
)

__all__ = [
    "Counter",
    "GatewayMetrics",
    "Histogram",
    "parse_engine_metrics",
    "render_engine_metrics",
]
