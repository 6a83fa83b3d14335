"""HBM allocation planning for TP engines (BASELINE config #4 readiness).

The reference sizes its fleet implicitly through vLLM's
``--gpu-memory-utilization`` (`tools/dynamic-lora-sidecar/deployment.yaml:40`);
here the plan is explicit so a 70B TP=4 engine is known to fit BEFORE the
first 8-GPU run: per-GPU bytes for sharded weights, the paged KV pool, and
the stacked LoRA slot tensors, against the MI355X's 288 GB HBM3E.

Sharding model (models/llama.py, lora/manager.py):
  * qkv/gate_up column-sharded, o/down row-sharded  -> / tp
  * embeddings, lm_head, norms replicated
  * LoRA stacks follow the module they attach to    -> / tp
"""

from __future__ import annotations

from dataclasses import dataclass

from ..models.config import ModelConfig

MI355X_HBM_BYTES = 288 * 1024**3
BF16 = 2


@dataclass
class MemPlan:
    weights_bytes: int
    kv_bytes_per_block: int
    lora_bytes: int
    hbm_bytes: int
    gpu_memory_utilization: float

    @property
    def budget_bytes(self) -> int:
        return int(self.hbm_bytes * self.gpu_memory_utilization)

    @property
    def kv_budget_bytes(self) -> int:
        return self.budget_bytes - self.weights_bytes - self.lora_bytes

    @property
    def max_kv_blocks(self) -> int:
        return max(0, self.kv_budget_bytes // self.kv_bytes_per_block)

    def fits(self, num_kv_blocks: int) -> bool:
        return num_kv_blocks <= self.max_kv_blocks


def params_per_gpu(mc: ModelConfig, tp: int) -> int:
    """Sharded parameter count on one rank of a TP group."""
    h, i, v = mc.hidden_size, mc.intermediate_size, mc.vocab_size
    kv = mc.num_kv_heads * mc.head_dim
    q = mc.num_heads * mc.head_dim
    attn = h * q + 2 * h * kv + q * h  # qkv + o, all sharded
    if mc.is_moe:
        mlp = 3 * h * i * mc.num_experts + h * mc.num_experts
    else:
        mlp = 3 * h * i  # gate_up + down, sharded
    per_layer = (attn + mlp) // tp + 2 * h  # norms replicated
    emb = v * h * (1 if mc.tie_word_embeddings else 2)  # replicated
    return mc.num_layers * per_layer + emb + h


def lora_stack_bytes(mc: ModelConfig, tp: int, max_slots: int, rank: int) -> int:
    """Bytes of the stacked per-(layer, module) A/B slot tensors on one rank
    (lora/manager.py pre-allocates all slots at manager construction)."""
    h = mc.hidden_size
    q = mc.num_heads * mc.head_dim // tp
    kv = mc.num_kv_heads * mc.head_dim // tp
    dims = {"qkv": (h, q + 2 * kv), "o": (q, h)}
    if not mc.is_moe:
        i = mc.intermediate_size // tp
        dims["gate_up"] = (h, 2 * i)
        dims["down"] = (i, h)
    per_layer = sum(rank * fin + fout * rank for fin, fout in dims.values())
    return mc.num_layers * max_slots * per_layer * BF16


def plan_tp_engine(
    mc: ModelConfig,
    tp: int,
    block_size: int = 16,
    max_loras: int = 0,
    lora_rank: int = 16,
    hbm_bytes: int = MI355X_HBM_BYTES,
    gpu_memory_utilization: float = 0.90,
) -> MemPlan:
    kv_bytes_per_block = (
        mc.num_layers * 2 * (mc.num_kv_heads // tp) * block_size * mc.head_dim * BF16
    )
    return MemPlan(
        weights_bytes=params_per_gpu(mc, tp) * BF16,
        kv_bytes_per_block=kv_bytes_per_block,
        lora_bytes=lora_stack_bytes(mc, tp, max_loras, lora_rank),
        hbm_bytes=hbm_bytes,
        gpu_memory_utilization=gpu_memory_utilization,
    )
