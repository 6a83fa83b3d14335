"""Server wiring unit tests (no uvicorn): build_pool honors config
thresholds and registers the base model (reference main.go wiring)."""

import argparse

from llm_instance_gateway_amd.api.config import parse_config
from llm_instance_gateway_amd.server import build_pool


def _args(**kw):
    ns = argparse.Namespace(model="", devices="cpu", config="")
    for k, v in kw.items():
        setattr(ns, k, v)
    return ns


def test_build_pool_registers_base_model_and_thresholds():
    cfg = parse_config(
        {
            "pool": {"name": "p", "base_model": "tiny"},
            "thresholds": {"queue_threshold_critical": 2, "kv_cache_threshold": 0.5},
        }
    )
    pool = build_pool(_args(model="tiny", devices="cpu"), cfg)
    try:
        assert pool.datastore.fetch_model("tiny") is not None
        # the scheduler got a tree built from the config thresholds: a
        # sheddable request against a mildly-loaded engine must now drop
        from llm_instance_gateway_amd.router.filters import DropRequest
        from llm_instance_gateway_amd.router.types import (
            EngineMetrics,
            EngineState,
            Engine,
            LLMRequest,
        )

        states = [
            EngineState(
                engine=Engine("e0"),
                metrics=EngineMetrics(waiting_queue_size=3, kv_cache_usage_percent=0.1),
            )
        ]
        try:
            pool.scheduler.filter.apply(LLMRequest(model="m", critical=False), states)
            raise AssertionError("expected DropRequest under tightened thresholds")
        except DropRequest:
            pass
    finally:
        pool.stop()


def test_build_pool_defaults_without_config():
    pool = build_pool(_args(model="tiny", devices="cpu"), None)
    try:
        assert list(pool.handles) == ["engine-0"]
        assert pool.datastore.fetch_model("tiny") is not None
    finally:
        pool.stop()


def test_port_precedence():
    """--port wins; else the config's `port:`; else 8081 (a truthy argparse
    default used to shadow the config value)."""
    from llm_instance_gateway_amd.api.config import GatewayConfig
    from llm_instance_gateway_amd.server import resolve_port

    cfg = GatewayConfig()
    cfg.port = 9099
    assert resolve_port(0, cfg) == 9099
    assert resolve_port(7000, cfg) == 7000
    assert resolve_port(0, None) == 8081
