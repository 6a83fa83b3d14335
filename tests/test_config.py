"""Config plane tests.

Mirrors the sidecar's reconcile tests (`tools/dynamic-lora-sidecar/sidecar/
test_sidecar.py`: load ensureExist−ensureNotExist, skip both-lists dupes) and
the reconciler unit tests (`backend/inferencemodel_reconciler_test.go`
store/delete semantics), against a real CPU engine pool.
"""

import time

import pytest
import yaml

from llm_instance_gateway_amd.api.config import (
    AdapterSpec,
    ConfigError,
    ConfigReconciler,
    GatewayConfig,
    load_config,
    parse_config,
    reconcile_adapters,
    reconcile_models,
)
from llm_instance_gateway_amd.api.types import Criticality
from llm_instance_gateway_amd.router.datastore import Datastore
from llm_instance_gateway_amd.serving import EnginePool

CFG_YAML = """
pool:
  name: mi355x-pool
  base_model: tiny
  gpu_ids: [0, 1]
  quant: fp8
models:
  - model_name: chat
    criticality: Critical
    target_models:
      - {name: chat, weight: 60}
      - {name: lora-a, weight: 40}
  - model_name: batch
    criticality: Sheddable
  - model_name: other-pool-model
    criticality: Default
    pool_ref: some-other-pool
ensureExist:
  models:
    - id: lora-a
      rank: 8
    - id: lora-dupe
ensureNotExist:
  models:
    - id: lora-gone
    - id: lora-dupe
thresholds:
  kv_cache_threshold: 0.7
  queue_threshold_critical: 3
  queueing_threshold_lora: 25
refresh:
  metrics_interval_s: 0.01
server:
  port: 9099
"""


def test_parse_config_full():
    cfg = parse_config(yaml.safe_load(CFG_YAML))
    assert cfg.pool.name == "mi355x-pool"
    assert cfg.pool.gpu_ids == [0, 1]
    assert cfg.pool.quant == "fp8"
    assert [m.model_name for m in cfg.models] == ["chat", "batch", "other-pool-model"]
    assert cfg.models[0].criticality == Criticality.CRITICAL
    assert cfg.models[0].target_models[1].weight == 40
    assert [a.id for a in cfg.ensure_exist] == ["lora-a", "lora-dupe"]
    assert cfg.ensure_exist[0].rank == 8
    assert [a.id for a in cfg.ensure_not_exist] == ["lora-gone", "lora-dupe"]
    assert cfg.kv_cache_threshold == 0.7
    assert cfg.queue_threshold_critical == 3
    assert cfg.queueing_threshold_lora == 25
    assert cfg.refresh_metrics_interval_s == 0.01
    assert cfg.port == 9099


def test_parse_config_rejects_bad_quant():
    with pytest.raises(ConfigError, match="quant"):
        parse_config(yaml.safe_load("pool: {name: p, quant: int4}"))


@pytest.mark.parametrize(
    "raw",
    [
        {"pool": {"gpu_ids": [0]}},  # pool without name
        {"models": [{"criticality": "Critical"}]},  # model without name
        {"models": [{"model_name": "x", "criticality": "Bogus"}]},
        {"thresholds": {"kv_cache_threshold": 1.5}},
        {"ensureExist": {"models": [{"source": "p"}]}},  # adapter without id
    ],
)
def test_parse_config_rejects(raw):
    with pytest.raises(ConfigError):
        parse_config(raw)


def test_reconcile_models_store_and_delete():
    """InferenceModelReconciler semantics: store models whose pool_ref matches,
    delete vanished ones (`inferencemodel_reconciler.go:45-55`)."""
    cfg = parse_config(yaml.safe_load(CFG_YAML))
    ds = Datastore()
    reconcile_models(cfg, ds)
    assert ds.fetch_model("chat") is not None
    assert ds.fetch_model("batch") is not None
    # pool_ref mismatch filtered out (reference PoolRef gate)
    assert ds.fetch_model("other-pool-model") is None
    # now remove 'batch' from config -> deleted from datastore
    cfg2 = GatewayConfig(pool=cfg.pool, models=[cfg.models[0]])
    reconcile_models(cfg2, ds)
    assert ds.fetch_model("batch") is None
    assert ds.fetch_model("chat") is not None


@pytest.fixture(scope="module")
def pool():
    p = EnginePool(
        "tiny",
        devices=["cpu"],
        engine_kwargs=dict(
            num_kv_blocks=64, max_num_seqs=4,
            max_num_batched_tokens=128, max_model_len=128, max_loras=4,
        ),
        lora_rank=8,
    )
    p.start(watchdog=False)
    yield p
    p.stop()


def test_reconcile_adapters_diff_and_dupe_skip(pool):
    cfg = GatewayConfig(
        ensure_exist=[AdapterSpec(id="lora-a", rank=8), AdapterSpec(id="lora-dupe", rank=8)],
        ensure_not_exist=[AdapterSpec(id="lora-gone"), AdapterSpec(id="lora-dupe")],
    )
    loaded, unloaded, skipped = reconcile_adapters(cfg, pool)
    assert loaded == {"lora-a"}
    assert skipped == {"lora-dupe"}
    h = pool.handles["engine-0"]
    assert "lora-a" in h.adapter_registry
    assert "lora-dupe" not in h.adapter_registry
    # idempotent: second run loads nothing new
    loaded2, _, _ = reconcile_adapters(cfg, pool)
    assert loaded2 == set()
    # flip lora-a to ensureNotExist -> unloaded
    cfg2 = GatewayConfig(ensure_not_exist=[AdapterSpec(id="lora-a")])
    _, unloaded2, _ = reconcile_adapters(cfg2, pool)
    assert unloaded2 == {"lora-a"}
    assert "lora-a" not in h.adapter_registry


def test_config_reconciler_watches_file(tmp_path, pool):
    path = tmp_path / "gw.yaml"
    path.write_text(
        "pool: {name: p, base_model: tiny}\n"
        "models: [{model_name: watched-model}]\n"
    )
    rec = ConfigReconciler(
        str(path), pool, poll_interval_s=0.05,
        health_timeout_s=5, health_interval_s=0.05,
    )
    rec.start()
    try:
        assert rec.reconcile_count == 1
        assert pool.datastore.fetch_model("watched-model") is not None
        # mutate the file -> picked up by the polling observer
        time.sleep(0.05)
        path.write_text(
            "pool: {name: p, base_model: tiny}\n"
            "models: [{model_name: watched-model-v2}]\n"
        )
        deadline = time.monotonic() + 5
        while time.monotonic() < deadline:
            if pool.datastore.fetch_model("watched-model-v2") is not None:
                break
            time.sleep(0.05)
        assert pool.datastore.fetch_model("watched-model-v2") is not None
        assert pool.datastore.fetch_model("watched-model") is None  # deleted
        # malformed update is rejected; last good state stays
        count = rec.reconcile_count
        time.sleep(0.05)
        path.write_text("models: [{criticality: Critical}]\n")
        deadline = time.monotonic() + 3
        while time.monotonic() < deadline and rec.reconcile_count == count:
            time.sleep(0.05)
        assert pool.datastore.fetch_model("watched-model-v2") is not None
    finally:
        rec.stop()


def test_load_config_file(tmp_path):
    p = tmp_path / "c.yaml"
    p.write_text(CFG_YAML)
    cfg = load_config(str(p))
    assert cfg.pool.base_model == "tiny"


def test_reconcile_flips_under_live_traffic(tmp_path, pool):
    """Adapter rollout flips (ensureExist <-> ensureNotExist) while requests
    stream through the engine: no crashes, traffic keeps completing, final
    registry state matches the last config."""
    import threading

    from llm_instance_gateway_amd.serving import PoolRequest

    path = tmp_path / "flip.yaml"

    def write(exist, not_exist):
        path.write_text(
            "pool: {name: p, base_model: tiny}\n"
            "models: [{model_name: flip-model}]\n"
            f"ensureExist: {{models: [{', '.join('{id: %s, rank: 8}' % a for a in exist)}]}}\n"
            f"ensureNotExist: {{models: [{', '.join('{id: %s}' % a for a in not_exist)}]}}\n"
        )

    write(["flip-a", "flip-b"], [])
    rec = ConfigReconciler(
        str(path), pool, poll_interval_s=0.05,
        health_timeout_s=5, health_interval_s=0.05,
    )
    rec.start()
    stop = threading.Event()
    errors = []
    done_count = [0]

    def traffic():
        i = 0
        while not stop.is_set():
            i += 1
            h = pool.handles["engine-0"]
            if "flip-a" not in h.adapter_registry:
                time.sleep(0.01)
                continue
            try:
                t = pool.submit(
                    "engine-0",
                    PoolRequest(request_id=f"flip-{i}",
                                prompt_token_ids=[1, 2, 3, 4],
                                max_new_tokens=2, lora_name="flip-a"),
                )
                t.wait(30)
                done_count[0] += 1
            except Exception as e:  # pragma: no cover
                errors.append(e)
                return

    th = threading.Thread(target=traffic, daemon=True)
    th.start()
    try:
        for round_ in range(3):
            time.sleep(0.15)
            write(["flip-a"], ["flip-b"])
            time.sleep(0.15)
            write(["flip-a", "flip-b"], [])
        deadline = time.monotonic() + 10
        while time.monotonic() < deadline and done_count[0] < 3:
            time.sleep(0.05)
        # wait for the reconciler to apply the FINAL config before stopping
        h = pool.handles["engine-0"]
        deadline = time.monotonic() + 10
        while time.monotonic() < deadline and "flip-b" not in h.adapter_registry:
            time.sleep(0.05)
    finally:
        stop.set()
        th.join(timeout=30)
        rec.stop()
    assert not errors, errors[:1]
    assert done_count[0] >= 3
    assert {"flip-a", "flip-b"} <= set(h.adapter_registry)


def test_runtime_threshold_reconcile(tmp_path, pool):
    """Live edits to `thresholds:` rebuild the scheduler tree (the
    reference's compile-time consts, runtime here)."""
    path = tmp_path / "thr.yaml"
    path.write_text(
        "pool: {name: p, base_model: tiny}\n"
        "thresholds: {queue_threshold_critical: 5}\n"
    )
    rec = ConfigReconciler(
        str(path), pool, poll_interval_s=0.05,
        health_timeout_s=5, health_interval_s=0.05,
    )
    assert rec.reconcile_once()
    tree_v1 = pool.scheduler.filter
    # same thresholds -> tree untouched
    assert rec.reconcile_once()
    assert pool.scheduler.filter is tree_v1
    # changed thresholds -> atomic tree swap
    path.write_text(
        "pool: {name: p, base_model: tiny}\n"
        "thresholds: {queue_threshold_critical: 2, kv_cache_threshold: 0.5}\n"
    )
    assert rec.reconcile_once()
    assert pool.scheduler.filter is not tree_v1


def test_scheduler_set_thresholds_changes_shed_behavior():
    """Functional check: tightening queue_threshold_critical flips a
    sheddable request from routed to dropped."""
    import pytest as _pytest

    from llm_instance_gateway_amd.router.scheduler import (
        ResourceExhausted,
        Scheduler,
    )
    from llm_instance_gateway_amd.router.types import (
        Engine,
        EngineMetrics,
        EngineState,
        LLMRequest,
    )

    class OneBusyEngine:
        def all_engine_metrics(self):
            return [
                EngineState(
                    engine=Engine(name="e0", gpu_id=0),
                    metrics=EngineMetrics(
                        waiting_queue_size=3, kv_cache_usage_percent=0.5
                    ),
                )
            ]

    sched = Scheduler(OneBusyEngine())
    shed_req = LLMRequest(model="m", critical=False)
    assert sched.schedule(shed_req).name == "e0"  # 3 <= default 5
    sched.set_thresholds(0.8, 2, 50)  # queue gate now 2 < waiting 3
    with _pytest.raises(ResourceExhausted):
        sched.schedule(shed_req)
