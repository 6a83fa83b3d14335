"""Runtime pool-membership reconciliation (reference EndpointSlice
add/prune, `endpointslice_reconciler.go:50-79`): a config flip grows or
shrinks the live engine set while requests stream, with zero failures and
traffic migrating to the surviving/new engines."""

import threading
import time

import pytest

from llm_instance_gateway_amd.api.config import (
    ConfigReconciler,
    GatewayConfig,
    desired_engine_specs,
    parse_config,
    reconcile_pool_membership,
)
from llm_instance_gateway_amd.api.types import InferenceModel
from llm_instance_gateway_amd.router.scheduler import SchedulingError
from llm_instance_gateway_amd.router.types import LLMRequest
from llm_instance_gateway_amd.serving import EnginePool, PoolRequest
from llm_instance_gateway_amd.serving.pool import EngineSpec

ENGINE_KWARGS = dict(
    num_kv_blocks=128,
    max_num_seqs=4,
    max_num_batched_tokens=256,
    max_model_len=256,
    max_loras=2,
)


def _spec(i: int) -> EngineSpec:
    return EngineSpec(
        name=f"engine-{i}", model="tiny", devices=["cpu"],
        engine_kwargs=dict(ENGINE_KWARGS), lora_rank=8,
    )


def test_desired_engine_specs_derivation():
    cfg = parse_config(
        {"pool": {"name": "p", "base_model": "m70", "gpu_ids": [0, 1, 2, 3],
                  "tensor_parallel": 2}}
    )
    specs = desired_engine_specs(cfg)
    assert [s.name for s in specs] == ["engine-0", "engine-1"]
    assert specs[0].devices == ["cuda:0", "cuda:1"]
    assert specs[1].devices == ["cuda:2", "cuda:3"]
    assert specs[0].model == "m70"
    # explicit devices override gpu_ids
    cfg2 = parse_config(
        {"pool": {"name": "p", "base_model": "tiny", "devices": ["cpu", "cpu"]}}
    )
    specs2 = desired_engine_specs(cfg2)
    assert [s.devices for s in specs2] == [["cpu"], ["cpu"]]
    # no membership pinned -> None (reconciler leaves the live set alone)
    cfg3 = parse_config({"pool": {"name": "p", "base_model": "tiny"}})
    assert desired_engine_specs(cfg3) is None


def test_scale_out_and_in_under_live_traffic():
    """Grow 1 -> 2 engines, then shrink back, while a traffic thread streams
    requests: zero failed requests; routed set follows membership."""
    pool = EnginePool(specs=[_spec(0)])
    pool.start()
    pool.datastore.store_model(InferenceModel(model_name="m"))
    stop = threading.Event()
    failures = []
    served_by = {}
    lock = threading.Lock()

    def traffic(tid: int = 0):
        i = 0
        while not stop.is_set():
            i += 1
            rid = f"mem-{tid}-{i}"
            try:
                t = pool.route_and_submit(
                    LLMRequest(model="m", critical=True, request_id=rid),
                    PoolRequest(request_id=rid,
                                prompt_token_ids=[1, 2, 3, 4, 5],
                                max_new_tokens=3),
                )
                fin = t.wait(30)
                with lock:
                    served_by[fin.engine_name] = served_by.get(fin.engine_name, 0) + 1
            except SchedulingError:
                time.sleep(0.01)  # transient: retryable by contract
            except Exception as e:  # noqa: BLE001
                failures.append((rid, repr(e)))
                return

    threads = [
        threading.Thread(target=traffic, args=(k,), daemon=True) for k in range(3)
    ]
    try:
        for t in threads:
            t.start()
        time.sleep(0.5)
        # scale OUT: engine-1 joins live
        added, removed = pool.reconcile_membership([_spec(0), _spec(1)])
        assert added == ["engine-1"] and removed == []
        assert pool.handles["engine-1"].alive()
        deadline = time.monotonic() + 20
        while time.monotonic() < deadline:
            with lock:
                if served_by.get("engine-1", 0) >= 3:
                    break
            time.sleep(0.05)
        with lock:
            assert served_by.get("engine-1", 0) >= 3, f"no traffic migrated: {served_by}"
        # scale IN: engine-1 drains and leaves; traffic keeps flowing on engine-0
        with lock:
            base0 = served_by.get("engine-0", 0)
        added, removed = pool.reconcile_membership([_spec(0)])
        assert removed == ["engine-1"] and added == []
        assert "engine-1" not in pool.handles
        assert [e.name for e in pool.datastore.get_engines()] == ["engine-0"]
        deadline = time.monotonic() + 20
        while time.monotonic() < deadline:
            with lock:
                if served_by.get("engine-0", 0) > base0 + 3:
                    break
            time.sleep(0.05)
        with lock:
            assert served_by.get("engine-0", 0) > base0 + 3
        assert not failures, failures
    finally:
        stop.set()
        for t in threads:
            t.join(timeout=30)
        pool.stop()
    assert not failures, failures


def test_config_reconciler_flips_membership(tmp_path):
    """The watched-file path end-to-end: editing pool.devices in the YAML
    grows then shrinks the engine set (VERDICT #6 done-criterion)."""
    pool = EnginePool(specs=[_spec(0)])
    pool.start()
    path = tmp_path / "gw.yaml"

    def write(n_devices: int):
        devs = ", ".join(["cpu"] * n_devices)
        path.write_text(
            f"pool: {{name: p, base_model: tiny, devices: [{devs}]}}\n"
            "models: [{model_name: m}]\n"
        )

    write(1)
    rec = ConfigReconciler(
        str(path), pool, poll_interval_s=0.05,
        health_timeout_s=5, health_interval_s=0.05,
    )
    # engine_kwargs for new engines come from the pool's defaults; set them
    # so the dynamically-added tiny engine stays small
    pool.engine_kwargs.update(ENGINE_KWARGS)
    rec.start()
    try:
        assert rec.reconcile_count == 1
        assert set(pool.handles) == {"engine-0"}
        time.sleep(0.05)
        write(2)
        deadline = time.monotonic() + 30
        while time.monotonic() < deadline and not (
            "engine-1" in pool.handles and pool.handles["engine-1"].alive()
        ):
            time.sleep(0.05)
        assert "engine-1" in pool.handles and pool.handles["engine-1"].alive()
        # routable
        names = {s.engine.name for s in pool.all_engine_metrics()}
        assert names == {"engine-0", "engine-1"}
        time.sleep(0.05)
        write(1)
        deadline = time.monotonic() + 30
        while time.monotonic() < deadline and "engine-1" in pool.handles:
            time.sleep(0.05)
        assert set(pool.handles) == {"engine-0"}
    finally:
        rec.stop()
        pool.stop()


def test_membership_capacity_bound():
    pool = EnginePool(specs=[_spec(0)], max_engines=16)
    assert len(pool._free_slots) == 15
    # slot round-trips through remove/add
    pool.datastore.store_model(InferenceModel(model_name="m"))
    h = pool._make_handle(_spec(1))
    base = h.slot_base
    pool.remove_engine_live("engine-1", drain=False)
    assert base in pool._free_slots
    h2 = pool._make_handle(_spec(2))
    assert h2.slot_base == base


def test_graceful_stop_drains_inflight():
    """pool.stop(drain_timeout_s>0): engines stop admitting but in-flight
    requests run to completion before workers exit (SIGTERM grace)."""
    from llm_instance_gateway_amd.router.types import LLMRequest
    from llm_instance_gateway_amd.serving.pool import EnginePool, PoolRequest

    pool = EnginePool(
        "tiny",
        devices=["cpu"],
        engine_kwargs=dict(
            num_kv_blocks=64, max_num_seqs=4,
            max_num_batched_tokens=128, max_model_len=128, max_loras=0,
        ),
    )
    pool.start(watchdog=False)
    try:
        tickets = [
            pool.route_and_submit(
                LLMRequest(model="tiny", prompt_len=8, max_new_tokens=24),
                PoolRequest(
                    request_id=f"drain-{i}",
                    prompt_token_ids=list(range(4, 12)),
                    max_new_tokens=24,
                ),
            )
            for i in range(3)
        ]
    except Exception:
        pool.stop()
        raise
    pool.stop(drain_timeout_s=60.0)
    for t in tickets:
        fin = t.wait(timeout=5)  # already done: drained before workers exited
        assert len(fin.output_token_ids) == 24


def test_watchdog_auto_restart_replays_adapters():
    """auto_restart=True: the watchdog re-spawns a dead engine (kubelet
    semantics) and the adapter registry is replayed."""
    import os
    import time as _time

    from llm_instance_gateway_amd.serving.pool import EnginePool

    pool = EnginePool(
        "tiny",
        devices=["cpu"],
        engine_kwargs=dict(
            num_kv_blocks=64, max_num_seqs=4,
            max_num_batched_tokens=128, max_model_len=128, max_loras=2,
        ),
        lora_rank=8,
    )
    pool.start(watchdog=True, auto_restart=True)
    try:
        h = pool.handles["engine-0"]
        assert h.load_adapter("keep-me", "", 8) is None
        # murder the worker process out-of-band and WAIT for the death to
        # be observed (multiprocessing is_alive() lags the signal, so a
        # naive healthy-check here passes vacuously against the old proc)
        victim = h.procs[0]
        victim.terminate()
        victim.join(timeout=60)
        assert not victim.is_alive()
        # generous deadlines: a fresh spawn re-imports torch, which can take
        # minutes on a heavily loaded machine (e.g. 4-way pytest-xdist with
        # several engine pools spawning concurrently)
        deadline = _time.monotonic() + 300
        while _time.monotonic() < deadline:
            if (
                pool.restart_counts.get("engine-0", 0) >= 1
                and h.healthy
                and h.alive()
            ):
                break
            _time.sleep(0.2)
        assert pool.restart_counts.get("engine-0", 0) >= 1, "no restart fired"
        assert h.healthy and h.alive(), (
            f"restart incomplete: healthy={h.healthy} alive={h.alive()}"
        )
        assert h.procs[0] is not victim  # genuinely re-spawned
        assert "keep-me" in h.adapter_registry
        deadline = _time.monotonic() + 120
        while _time.monotonic() < deadline:
            if "keep-me" in h.active_adapters:
                break
            _time.sleep(0.2)
        assert "keep-me" in h.active_adapters  # replayed onto the new worker
    finally:
        pool.stop()


def test_watchdog_recovers_stalled_engine(monkeypatch):
    """A STALLED engine (heartbeats paused, workers alive — host CPU
    starvation / long kernel) is dropped from the routing set and then
    RE-ADDED when heartbeats resume (EndpointSlice Ready flipping back),
    without a restart."""
    import os
    import signal
    import time as _time

    import llm_instance_gateway_amd.serving.pool as pool_mod
    from llm_instance_gateway_amd.serving.pool import EnginePool

    monkeypatch.setattr(pool_mod, "HEARTBEAT_STALE_S", 1.0)
    pool = EnginePool(
        "tiny",
        devices=["cpu"],
        engine_kwargs=dict(
            num_kv_blocks=64, max_num_seqs=4,
            max_num_batched_tokens=128, max_model_len=128, max_loras=0,
        ),
    )
    pool.start(watchdog=True)
    try:
        h = pool.handles["engine-0"]
        pid = h.procs[0].pid
        os.kill(pid, signal.SIGSTOP)  # freeze: heartbeats stop, proc alive
        try:
            deadline = _time.monotonic() + 30
            while _time.monotonic() < deadline and h.healthy:
                _time.sleep(0.1)
            assert not h.healthy, "stalled engine was not dropped"
            assert pool.datastore.get_engine("engine-0") is None
        finally:
            os.kill(pid, signal.SIGCONT)
        deadline = _time.monotonic() + 30
        while _time.monotonic() < deadline and not h.healthy:
            _time.sleep(0.1)
        assert h.healthy, "recovered engine was not re-added"
        assert pool.datastore.get_engine("engine-0") is not None
        assert h.procs[0].is_alive()  # same worker, no restart
        assert pool.restart_counts.get("engine-0", 0) == 0
    finally:
        pool.stop()
