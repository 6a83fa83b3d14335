"""Per-GPU engine configuration.

The knobs reproduce the serving contract the reference pins on its vLLM fleet
(SURVEY.md §2.3): block size 16 (`simulations/.../constants.py:13-15`),
max_num_seqs 256 and chunked-prefill token cap (`constants.py:11,16`),
preemption-by-recompute at 90% KV (`constants.py:17-18`,
`continous_batching.py:117-131`), adapter slot budget `--max-loras`
(`vllm-lora-deployment.yaml:44`), gpu memory utilisation 0.95
(`tools/dynamic-lora-sidecar/deployment.yaml:40`).
"""

from __future__ import annotations

from dataclasses import dataclass

from ..models.config import ModelConfig


@dataclass
class EngineConfig:
    model: ModelConfig
    # Paged KV cache (reference sim: BLOCK_SIZE=16, TOTAL_BLOCKS=2810).
    block_size: int = 16
    num_kv_blocks: int = 2810  # overridden by profile_memory on GPU
    gpu_memory_utilization: float = 0.90
    # Continuous-batching admission caps (reference sim constants.py:11,16).
    max_num_seqs: int = 256
    max_num_batched_tokens: int = 4096
    # Preempt (recompute) newest sequences above this KV utilisation
    # (reference continous_batching.py:117-131).
    preempt_kv_threshold: float = 0.90
    # Mixed steps (default): decodes run EVERY step, prefill chunks fill the
    # remaining token budget (vLLM chunked-prefill semantics; lets the runner
    # overlap the graphed decode bucket with the eager prefill).  False keeps
    # the reference sim's prefill_or_decode alternation.
    mixed_steps: bool = True
    # Opt-in quantization: "" (bf16, default) or "fp8" (e4m3 weights +
    # dynamic per-token activations on the projection GEMMs — models/quant.py;
    # vLLM --quantization fp8 analogue).  Dense models only.
    quant: str = ""
    # Multi-LoRA slots (vLLM --max-loras analogue).
    max_loras: int = 8
    max_lora_rank: int = 16
    # Watermark of blocks kept free when admitting prefills.
    watermark_blocks: int = 8
    max_model_len: int = 4096
    # Engine identity.
    engine_id: str = "engine-0"
    gpu_id: int = 0
    enforce_eager: bool = False  # False: decode-only steps replay hipGraphs
    # Pad mixed (eager) batches to a multiple of this many tokens so the
    # prefill GEMMs see a handful of stable, tile-friendly M shapes instead
    # of a new M every step (hipBLASLt heuristic picks at variable M ran at
    # ~17% of bf16 peak — profiles/r01_steady_state_breakdown.md).  Pad rows
    # write K/V to the scratch block and are excluded from attention,
    # LoRA, and sampling.  0 disables; GPU only; dense models only.
    pad_batch_multiple: int = 256

    def __post_init__(self) -> None:
        if self.block_size <= 0:
            raise ValueError("block_size must be positive")
        if self.max_model_len % self.block_size != 0:
            # round up to a whole number of blocks
            self.max_model_len = (
                (self.max_model_len + self.block_size - 1) // self.block_size
            ) * self.block_size

    @property
    def kv_tokens_capacity(self) -> int:
        return self.num_kv_blocks * self.block_size
