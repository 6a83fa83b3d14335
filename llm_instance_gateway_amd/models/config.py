"""Model architecture configs.

The reference delegates all model execution to vLLM pods; the
model families its configs pin are Llama-class dense transformers
(`examples/poc/manifests/vllm/vllm-lora-deployment.yaml`: Llama-2-7b) and —
per BASELINE.json — Llama-3-8B, Llama-3-70B (TP=4) and Mixtral 8x7B.  These
presets are random-init shapes (no network ⇒ no checkpoints); weights are
initialised randomly and the benchmark is synthetic-data.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Optional


@dataclass
class ModelConfig:
    name: str
    vocab_size: int = 128256
    hidden_size: int = 4096
    num_layers: int = 32
    num_heads: int = 32
    num_kv_heads: int = 8
    intermediate_size: int = 14336
    head_dim: int = 128
    rms_norm_eps: float = 1e-5
    rope_theta: float = 500000.0
    max_position_embeddings: int = 8192
    tie_word_embeddings: bool = False
    # MoE (Mixtral-style); num_experts == 0 means dense.
    num_experts: int = 0
    num_experts_per_tok: int = 2
    # Multi-GPU MoE mode: False = TP-shard every expert's intermediate dim;
    # True = expert parallelism (each rank owns num_experts/world full
    # experts; models/mixtral.py MoEBlock docstring).
    expert_parallel: bool = False
    dtype: str = "bfloat16"

    @property
    def is_moe(self) -> bool:
        return self.num_experts > 0

    @property
    def kv_bytes_per_token_per_layer(self) -> int:
        # K + V, bf16.
        return 2 * self.num_kv_heads * self.head_dim * 2

    def num_params(self) -> int:
        """Approximate parameter count (dense path)."""
        h, i, v = self.hidden_size, self.intermediate_size, self.vocab_size
        kv = self.num_kv_heads * self.head_dim
        q = self.num_heads * self.head_dim
        attn = h * q + 2 * h * kv + q * h
        if self.is_moe:
            mlp = 3 * h * i * self.num_experts + h * self.num_experts
        else:
            mlp = 3 * h * i
        per_layer = attn + mlp + 2 * h
        emb = v * h * (1 if self.tie_word_embeddings else 2)
        return self.num_layers * per_layer + emb + h


LLAMA3_8B = ModelConfig(name="llama3-8b")

LLAMA3_70B = ModelConfig(
    name="llama3-70b",
    hidden_size=8192,
    num_layers=80,
    num_heads=64,
    num_kv_heads=8,
    intermediate_size=28672,
)

MIXTRAL_8X7B = ModelConfig(
    name="mixtral-8x7b",
    vocab_size=32000,
    hidden_size=4096,
    num_layers=32,
    num_heads=32,
    num_kv_heads=8,
    intermediate_size=14336,
    rope_theta=1e6,
    num_experts=8,
    num_experts_per_tok=2,
)

# 70B-shaped tiny: keeps every ratio TP=4 sharding depends on (64 q heads /
# 8 kv heads GQA grouping, 3.5x intermediate, separate lm_head) at CPU-test
# dims — the 4-rank gloo token-identity test for BASELINE config #4 runs on
# this shape.
TINY_70B = ModelConfig(
    name="tiny-70b",
    vocab_size=512,
    hidden_size=128,
    num_layers=2,
    num_heads=64,
    num_kv_heads=8,
    intermediate_size=448,
    head_dim=16,
    max_position_embeddings=512,
)

# Tiny shapes for CPU tests and smoke runs.
TINY = ModelConfig(
    name="tiny",
    vocab_size=512,
    hidden_size=64,
    num_layers=2,
    num_heads=4,
    num_kv_heads=2,
    intermediate_size=128,
    head_dim=16,
    max_position_embeddings=512,
)

TINY_MOE = ModelConfig(
    name="tiny-moe",
    vocab_size=512,
    hidden_size=64,
    num_layers=2,
    num_heads=4,
    num_kv_heads=2,
    intermediate_size=128,
    head_dim=16,
    max_position_embeddings=512,
    num_experts=4,
    num_experts_per_tok=2,
)

PRESETS = {
    "llama3-8b": LLAMA3_8B,
    "llama3-70b": LLAMA3_70B,
    "tiny-70b": TINY_70B,
    "mixtral-8x7b": MIXTRAL_8X7B,
    "tiny": TINY,
    "tiny-moe": TINY_MOE,
}


def get_model_config(name: str) -> ModelConfig:
    if name not in PRESETS:
        raise KeyError(f"unknown model preset {name!r}; have {sorted(PRESETS)}")
    return PRESETS[name]
