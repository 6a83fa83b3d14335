"""Gateway server entrypoint.

The reference ext-proc `main()` (`pkg/ext-proc/main.go:59-160`): flag parsing,
datastore + reconcilers + provider + scheduler + gRPC server, graceful
shutdown.  Here the same wiring, MI355X-native: build the per-GPU engine pool,
attach the filter-tree router over shared-memory gauges, start the config
reconciler (replaces the three K8s controllers), serve the OpenAI-compatible
HTTP front door, drain on SIGTERM/SIGINT.

Flags mirror `main.go:31-41` where they still make sense in-process
(port, pool name, refresh intervals collapse to config), plus the
config-file path that replaces the K8s API.

    python -m llm_instance_gateway_amd.server --config gateway.yaml
    python -m llm_instance_gateway_amd.server --model tiny --devices cpu,cpu
"""

from __future__ import annotations

import argparse
import os
import logging
import signal
import sys
from typing import List, Optional

log = logging.getLogger(__name__)


def resolve_port(args_port: int, cfg) -> int:
    """--port wins when given; else the config's `port:`; else 8081."""
    if args_port:
        return args_port
    if cfg and cfg.port:
        return cfg.port
    return 8081


def build_pool(args, cfg) -> "EnginePool":
    import torch

    from .api.types import InferenceModel
    from .router.filters import build_default_tree
    from .router.scheduler import Scheduler
    from .serving.pool import EnginePool

    model = args.model or (cfg.pool.base_model if cfg else "") or "llama3-8b"
    if args.devices:
        devices: List[str] = args.devices.split(",")
    elif cfg and (cfg.pool.devices or cfg.pool.gpu_ids):
        devices = cfg.pool.devices or [f"cuda:{g}" for g in cfg.pool.gpu_ids]
    elif torch.cuda.is_available():
        devices = [f"cuda:{i}" for i in range(torch.cuda.device_count())]
    else:
        devices = ["cpu"]

    engine_kwargs = {}
    if not any(d.startswith("cuda") for d in devices):
        engine_kwargs = dict(
            num_kv_blocks=256, max_num_seqs=16,
            max_num_batched_tokens=512, max_model_len=512,
        )
    if cfg and cfg.pool.quant:
        engine_kwargs["quant"] = cfg.pool.quant
    tp = cfg.pool.tensor_parallel if cfg else 1
    if cfg and tp > 1 and not args.devices:
        # TP pools: chunk devices into TP groups, one engine per group —
        # the same shape the config reconciler derives at runtime
        from .api.config import desired_engine_specs

        specs = desired_engine_specs(cfg, model, engine_kwargs)
        pool = EnginePool(model, engine_kwargs=engine_kwargs, specs=specs)
    else:
        pool = EnginePool(model, devices=devices, engine_kwargs=engine_kwargs)
    if cfg:
        tree = build_default_tree(
            kv_cache_threshold=cfg.kv_cache_threshold,
            queue_threshold_critical=cfg.queue_threshold_critical,
            queueing_threshold_lora=cfg.queueing_threshold_lora,
        )
        pool.scheduler = Scheduler(pool, tree=tree)
    # the base model is always addressable (engines exist even with no config)
    pool.datastore.store_model(InferenceModel(model_name=model))
    return pool


def main(argv: Optional[List[str]] = None) -> int:
    ap = argparse.ArgumentParser(description=__doc__.splitlines()[0])
    ap.add_argument("--config", default="", help="gateway YAML (pool/models/adapters/thresholds)")
    ap.add_argument(
        "--port", type=int, default=0,
        help="HTTP front-door port (0 = the config's `port:`, else 8081; "
        "reference gateway.yaml:13-15)",
    )
    ap.add_argument("--model", default="", help="base model preset override")
    ap.add_argument("--devices", default="", help="comma list, e.g. cuda:0,cuda:1 or cpu")
    ap.add_argument(
        "--remote-engines", default="",
        help="comma list of host:port model servers — remote-router mode "
        "(the reference's topology: scrape + schedule + forward, no local GPUs)",
    )
    ap.add_argument(
        "--native-router", action="store_true",
        help="route with the C++ filter tree reading shared-memory gauges "
        "(default: Python tree; both are semantically identical)",
    )
    ap.add_argument(
        "--tokenizer", default="",
        help="HuggingFace tokenizer.json for the front door (default: the "
        "asset-free byte codec)",
    )
    ap.add_argument("--v", type=int, default=1, help="log verbosity (klog analogue)")
    args = ap.parse_args(argv)

    logging.basicConfig(
        level=logging.DEBUG if args.v >= 3 else logging.INFO,
        format="%(asctime)s %(name)s %(levelname)s %(message)s",
    )

    from .api.config import ConfigReconciler, load_config
    from .serving.frontdoor import build_app
    from .utils import tokenizer as _tok

    if args.tokenizer:
        if _tok.load_backend(args.tokenizer):
            log.info("front-door tokenizer: %s", _tok.backend_name())
            # engine workers (spawned later) load the same backend for
            # decoded-text stop matching
            os.environ["LIG_TOKENIZER"] = args.tokenizer
        else:
            log.warning("could not load tokenizer %s; using byte codec", args.tokenizer)

    cfg = load_config(args.config) if args.config else None

    if args.remote_engines or (cfg and cfg.pool.engine_addresses and not cfg.pool.gpu_ids):
        from .serving.remote_router import build_remote_router_app

        addrs = (
            args.remote_engines.split(",")
            if args.remote_engines
            else cfg.pool.engine_addresses
        )
        app = build_remote_router_app(
            addrs,
            models=cfg.models if cfg else None,
            refresh_metrics_interval_s=(
                cfg.refresh_metrics_interval_s if cfg else 0.050
            ),
        )
        import uvicorn

        port = resolve_port(args.port, cfg)
        log.info("remote-router mode over %d engines", len(addrs))
        uvicorn.run(app, host="0.0.0.0", port=port, log_level="warning")
        return 0
    pool = build_pool(args, cfg)
    log.info("starting %d engine(s)...", len(pool.handles))
    pool.start(auto_restart=bool(cfg and cfg.auto_restart_engines))

    if args.native_router:
        from .router.native import NativeScheduler, native_available

        if native_available():
            kw = {}
            if cfg:
                kw = dict(
                    kv_cache_threshold=cfg.kv_cache_threshold,
                    queue_threshold_critical=cfg.queue_threshold_critical,
                    queueing_threshold_lora=cfg.queueing_threshold_lora,
                )
            pool.scheduler = NativeScheduler.for_pool(pool, **kw)
            log.info("routing with the native (C++) filter tree")
        else:  # pragma: no cover
            log.warning("--native-router requested but _router is not built")

    reconciler = None
    if args.config:
        reconciler = ConfigReconciler(
            args.config,
            pool,
            health_timeout_s=(
                cfg.health_check_timeout_s if cfg else 300.0
            ),
            health_interval_s=(
                cfg.health_check_interval_s if cfg else 15.0
            ),
        )
        reconciler.start()

    app = build_app(pool)

    def shutdown(signum, frame):
        log.info("signal %s: draining and stopping", signum)
        if reconciler:
            reconciler.stop()
        pool.stop(drain_timeout_s=30.0)  # in-flight requests finish
        sys.exit(0)

    signal.signal(signal.SIGTERM, shutdown)
    signal.signal(signal.SIGINT, shutdown)

    import uvicorn

    port = resolve_port(args.port, cfg)
    uvicorn.run(app, host="0.0.0.0", port=port, log_level="warning")
    pool.stop()
    return 0


if __name__ == "__main__":
    sys.exit(main())
