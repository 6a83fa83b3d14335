"""Native (C++) router equivalence + integration.

The C++ tree must be SEMANTICALLY IDENTICAL to the Python reference tree
(router/filters.py, itself mirroring the reference scheduler.go): hypothesis
drives random fleets through both and compares survivor sets and drop
decisions exactly."""

import hypothesis.strategies as st
import pytest
from hypothesis import given, settings

from llm_instance_gateway_amd.router.filters import DropRequest, build_default_tree
from llm_instance_gateway_amd.router.native import NativeScheduler, native_available
from llm_instance_gateway_amd.router.scheduler import ResourceExhausted
from llm_instance_gateway_amd.router.types import Engine, EngineMetrics, EngineState, LLMRequest

pytestmark = pytest.mark.skipif(
    not native_available(), reason="_router extension not built"
)

metrics_st = st.builds(
    EngineMetrics,
    active_models=st.dictionaries(
        st.sampled_from(["a1", "a2", "a3"]), st.just(1), max_size=3
    ),
    max_active_models=st.integers(0, 4),
    running_queue_size=st.integers(0, 200),
    waiting_queue_size=st.integers(0, 200),
    kv_cache_usage_percent=st.floats(0.0, 1.0, allow_nan=False),
)

fleet_st = st.lists(metrics_st, min_size=1, max_size=10).map(
    lambda ms: [
        EngineState(engine=Engine(name=f"e{i}"), metrics=m)
        for i, m in enumerate(ms)
    ]
)

req_st = st.builds(
    LLMRequest,
    model=st.just("m"),
    resolved_target_model=st.sampled_from(["a1", "a2", "zz"]),
    critical=st.booleans(),
)


@settings(max_examples=300, deadline=None)
@given(states=fleet_st, req=req_st)
def test_native_tree_equals_python_tree(states, req):
    tree = build_default_tree()
    try:
        py_out = {s.engine.name for s in tree.apply(req, states)}
        py_drop = False
    except DropRequest:
        py_out, py_drop = set(), True

    ns = NativeScheduler(seed=1)
    ns.sync_states(states)
    nat_out = set(ns.survivors(req))
    if py_drop:
        assert nat_out == set()
        with pytest.raises(ResourceExhausted):
            ns.schedule(req)
    else:
        assert nat_out == py_out
        assert ns.schedule(req).name in py_out


def test_native_pool_ref_scoping():
    ns = NativeScheduler(seed=0)
    states = [
        EngineState(engine=Engine("p1", pool="pool-1"), metrics=EngineMetrics()),
        EngineState(engine=Engine("p2", pool="pool-2"), metrics=EngineMetrics()),
    ]
    ns.sync_states(states)
    for _ in range(5):
        assert ns.schedule(
            LLMRequest(model="m", critical=True, pool_ref="pool-2")
        ).name == "p2"
    # unmatched pool falls back to the full set
    assert ns.schedule(
        LLMRequest(model="m", critical=True, pool_ref="ghost")
    ).name in {"p1", "p2"}


def test_native_engine_removal():
    ns = NativeScheduler(seed=0)
    ns.sync_states(
        [EngineState(engine=Engine("e0"), metrics=EngineMetrics())]
    )
    assert ns.schedule(LLMRequest(model="m", critical=True)).name == "e0"
    ns.sync_states([])  # fleet emptied
    from llm_instance_gateway_amd.router.scheduler import SchedulingError

    with pytest.raises(SchedulingError):
        ns.schedule(LLMRequest(model="m", critical=True))


def test_native_reads_pool_shared_memory():
    """Pool mode: the C++ tree reads gauges straight from the pool's shared
    array — mutate the array, the routing decision follows instantly."""
    from llm_instance_gateway_amd.serving import EnginePool
    from llm_instance_gateway_amd.serving import pool as pool_mod

    p = EnginePool.__new__(EnginePool)  # skeleton: no workers needed
    import multiprocessing as mp

    from llm_instance_gateway_amd.router.datastore import Datastore

    p.shared = mp.get_context("spawn").Array("d", pool_mod.SLOTS_PER_ENGINE * 2, lock=False)
    p.datastore = Datastore()
    p.handles = {}

    class H:  # minimal handle stub
        def __init__(self, name, base):
            self.healthy = True
            self.active_adapters = []
            self.max_slots = 4
            self.slot_base = base
            self.engine = Engine(name=name)

    for i, name in enumerate(["engine-0", "engine-1"]):
        p.handles[name] = H(name, pool_mod.SLOTS_PER_ENGINE * i)
        p.datastore.add_engine(Engine(name=name))

    ns = NativeScheduler.for_pool(p, seed=2)
    # engine-1 heavily queued -> engine-0 wins
    p.shared[pool_mod.SLOTS_PER_ENGINE * 1 + pool_mod._SLOT_WAITING] = 80.0
    for _ in range(5):
        assert ns.schedule(LLMRequest(model="m", critical=True)).name == "engine-0"
    # flip the load in shared memory: decision follows with NO resync
    p.shared[pool_mod.SLOTS_PER_ENGINE * 1 + pool_mod._SLOT_WAITING] = 0.0
    p.shared[pool_mod.SLOTS_PER_ENGINE * 0 + pool_mod._SLOT_WAITING] = 80.0
    for _ in range(5):
        assert ns.schedule(LLMRequest(model="m", critical=True)).name == "engine-1"
    # draining flag in shm removes an engine from consideration
    p.shared[pool_mod.SLOTS_PER_ENGINE * 1 + pool_mod._SLOT_DRAINING] = 1.0
    assert ns.schedule(LLMRequest(model="m", critical=True)).name == "engine-0"


def test_native_for_pool_end_to_end():
    """Real engine pool + native router: route, dispatch, adapter-affinity
    sync on adapter load, drain visibility through shared memory."""
    from llm_instance_gateway_amd.api.types import InferenceModel
    from llm_instance_gateway_amd.serving import EnginePool, PoolRequest

    p = EnginePool(
        "tiny",
        devices=["cpu", "cpu"],
        engine_kwargs=dict(
            num_kv_blocks=64, max_num_seqs=4,
            max_num_batched_tokens=128, max_model_len=128, max_loras=2,
        ),
        lora_rank=8,
    )
    p.start(watchdog=False)
    try:
        p.datastore.store_model(InferenceModel(model_name="m"))
        ns = NativeScheduler.for_pool(p, seed=4)
        p.scheduler = ns  # route through the native tree
        eng = p.route(LLMRequest(model="m", critical=True))
        assert eng.name in p.handles
        t = p.route_and_submit(
            LLMRequest(model="m", critical=True),
            PoolRequest(request_id="nr-1", prompt_token_ids=[1, 2, 3],
                        max_new_tokens=2),
        )
        assert len(t.wait(60).output_token_ids) == 2
        # adapter residency flows into native affinity
        assert p.handles["engine-0"].load_adapter("nat-ad") is None
        import time as _t

        deadline = _t.monotonic() + 5
        while _t.monotonic() < deadline:
            if "nat-ad" in p.handles["engine-0"].active_adapters:
                break
            _t.sleep(0.02)
        for _ in range(6):
            eng = ns.schedule(
                LLMRequest(model="m", resolved_target_model="nat-ad", critical=True)
            )
            assert eng.name == "engine-0"  # affinity wins
    finally:
        p.stop()
