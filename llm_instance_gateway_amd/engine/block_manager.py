"""Paged KV-cache block manager.

The MI355X-native ownership of what the reference delegates to vLLM and
encodes as ground truth in its simulation: a pool of fixed-size KV blocks
(16 tokens/block, `simulations/.../constants.py:13-15`), per-sequence block
tables, allocation at admission, preemption when the pool runs dry.  The
``usage`` gauge here is exactly the `vllm:gpu_cache_usage_perc` signal the
router's least-KV filter consumes (reference backend/vllm/metrics.go:24-31).

Pure host-side logic (no GPU): the physical cache tensors live in the model
runner; this class only tracks block ids.
"""

from __future__ import annotations

from typing import List

from .sequence import Sequence


class BlockManager:
    def __init__(self, num_blocks: int, block_size: int, watermark_blocks: int = 0) -> None:
        self.num_blocks = num_blocks
        self.block_size = block_size
        self.watermark_blocks = watermark_blocks
        # LIFO free list: recently-freed blocks are re-used first (their cache
        # lines may still be resident in L2).
        self._free: List[int] = list(range(num_blocks - 1, -1, -1))

    # -- accounting ----------------------------------------------------------
    @property
    def num_free_blocks(self) -> int:
        return len(self._free)

    @property
    def num_used_blocks(self) -> int:
        return self.num_blocks - len(self._free)

    @property
    def usage(self) -> float:
        """KV utilisation in [0,1] — the router's KV% gauge."""
        if self.num_blocks == 0:
            return 0.0
        return self.num_used_blocks / self.num_blocks

    def blocks_needed(self, seq: Sequence, target_len: int) -> int:
        total = (target_len + self.block_size - 1) // self.block_size
        return max(0, total - len(seq.block_table))

    def can_grow(self, seq: Sequence, target_len: int, respect_watermark: bool = False) -> bool:
        need = self.blocks_needed(seq, target_len)
        reserve = self.watermark_blocks if respect_watermark else 0
        return need <= len(self._free) - reserve

    # -- mutation ------------------------------------------------------------
    def grow(self, seq: Sequence, target_len: int) -> None:
        """Extend the sequence's block table to cover target_len tokens."""
        need = self.blocks_needed(seq, target_len)
        if need > len(self._free):
            raise RuntimeError(
                f"out of KV blocks: need {need}, free {len(self._free)}"
            )
        for _ in range(need):
            seq.block_table.append(self._free.pop())

    def free(self, seq: Sequence) -> None:
        self._free.extend(reversed(seq.block_table))
        seq.block_table = []
