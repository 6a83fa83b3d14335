"""HBM allocation planning for TP engines (BASELINE config #4 readiness):
the 70B TP=4 engine must fit 288 GB/GPU with generous KV headroom BEFORE
the first 8-GPU run."""

from llm_instance_gateway_amd.models.config import get_model_config
from llm_instance_gateway_amd.utils.memplan import (
    MI355X_HBM_BYTES,
    lora_stack_bytes,
    params_per_gpu,
    plan_tp_engine,
)

GiB = 1024**3


def test_70b_tp4_fits_with_headroom():
    mc = get_model_config("llama3-70b")
    plan = plan_tp_engine(mc, tp=4, max_loras=8, lora_rank=16)
    # sharded weights ~19B params (~38 GiB) per rank incl. replicated
    # embeddings + lm_head
    assert 30 * GiB < plan.weights_bytes < 45 * GiB
    # KV budget after weights+adapters leaves > 180 GiB
    assert plan.kv_budget_bytes > 180 * GiB
    # >= 2M KV tokens per rank at block_size 16
    assert plan.max_kv_blocks * 16 > 2_000_000
    assert plan.fits(100_000)


def test_8b_tp1_plan_matches_runtime_sizer():
    """The plan's block cost equals the model runner's bytes_per_block."""
    mc = get_model_config("llama3-8b")
    plan = plan_tp_engine(mc, tp=1)
    expect = mc.num_layers * 2 * mc.num_kv_heads * 16 * mc.head_dim * 2
    assert plan.kv_bytes_per_block == expect
    # 8B on one 288 GB GPU: weights ~16 GiB, KV budget > 200 GiB
    assert 14 * GiB < plan.weights_bytes < 20 * GiB
    assert plan.kv_budget_bytes > 200 * GiB


def test_params_per_gpu_shards_correctly():
    mc = get_model_config("llama3-70b")
    full = params_per_gpu(mc, 1)
    tp4 = params_per_gpu(mc, 4)
    # replicated part (embeddings + lm_head + norms) stays; sharded part /4
    emb = mc.vocab_size * mc.hidden_size * 2
    sharded_full = full - emb - mc.hidden_size - mc.num_layers * 2 * mc.hidden_size
    assert abs((tp4 - emb - mc.hidden_size - mc.num_layers * 2 * mc.hidden_size)
               - sharded_full // 4) < mc.num_layers  # integer-division slack


def test_lora_stack_bytes_scale():
    mc = get_model_config("llama3-8b")
    b1 = lora_stack_bytes(mc, tp=1, max_slots=32, rank=16)
    b4 = lora_stack_bytes(mc, tp=4, max_slots=32, rank=16)
    assert b4 < b1  # sharded modules shrink
    assert b1 < 4 * GiB  # 32 slots of rank-16 adapters are cheap vs 288 GB
