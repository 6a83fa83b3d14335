"""Flat-batch metadata handed from the model runner to the attention ops.

One engine step runs ONE model forward over a flat token batch laid out as
[all prefill-chunk tokens..., all decode tokens...] with no padding.  The
attention layer consumes this metadata to (a) scatter new K/V into the paged
cache and (b) run the prefill and decode attention kernels on their
partitions.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch


@dataclass
class BatchMeta:
    # Tokens: prefill partition first, then decodes.
    num_prefill_tokens: int
    num_decode_tokens: int
    slot_mapping: torch.Tensor  # [T] int64 — cache slot per token

    # Prefill partition (S_p sequences, chunk sizes vary).
    prefill_query_start_locs: Optional[torch.Tensor] = None  # [S_p+1] int32
    prefill_context_lens: Optional[torch.Tensor] = None  # [S_p] int32 (AFTER chunk cached)
    prefill_block_tables: Optional[torch.Tensor] = None  # [S_p, max_blocks] int32

    # Decode partition (S_d == num_decode_tokens sequences).
    decode_block_tables: Optional[torch.Tensor] = None  # [S_d, max_blocks] int32
    decode_context_lens: Optional[torch.Tensor] = None  # [S_d] int32

    # Per-TOKEN prefill expansion (computed ONCE per step by the model
    # runner; the per-layer attention calls reuse it — computing it per layer
    # cost 32x redundant arange/repeat kernels, see profiles/r01_*.md):
    prefill_tok_block_tables: Optional[torch.Tensor] = None  # [Tp, max_blocks]
    prefill_tok_context_lens: Optional[torch.Tensor] = None  # [Tp] int32
    # 64-row query tiles for the MFMA flash-prefill kernel (GPU path).
    prefill_tile_seq: Optional[torch.Tensor] = None  # [ntiles] int32
    prefill_tile_qstart: Optional[torch.Tensor] = None  # [ntiles] int32

    # Multi-LoRA (SGMV): per-token adapter slot index, -1 = base model only.
    lora_slot_per_token: Optional[torch.Tensor] = None  # [T] int32
    # host-side max decode context this step (drives the flash-decode split
    # heuristic without a device sync)
    max_decode_ctx: int = 0
    # pinned context-split factor for the decode kernel (0 = let the ops
    # heuristic decide from max_decode_ctx; the hipGraph path captures a
    # bucket-static value here since replay cannot branch on context)
    decode_nsplits: int = 0
    # Prefill LoRA segment tiles (64-token runs sharing one adapter slot) for
    # the MFMA SGMV kernels: token start, length (<=64), adapter slot.
    lora_tile_t0: Optional[torch.Tensor] = None    # [ntiles] int32
    lora_tile_len: Optional[torch.Tensor] = None   # [ntiles] int32
    lora_tile_slot: Optional[torch.Tensor] = None  # [ntiles] int32
    # Slot-grouped SGMV units (r02 default path): tokens sorted by adapter
    # slot (`lora_order`), chunked into <=32-token same-slot units so the
    # kernels read each adapter's weights once per unit.
    lora_order: Optional[torch.Tensor] = None      # [T] int32
    lora_unit_t0: Optional[torch.Tensor] = None    # [U] int32 (into order)
    lora_unit_len: Optional[torch.Tensor] = None   # [U] int32
    lora_unit_slot: Optional[torch.Tensor] = None  # [U] int32
    # Private LoRA shrink workspace for this batch (the decode graph owns its
    # own so graph replay and a concurrent eager prefill never share u).
    lora_ws: Optional[torch.Tensor] = None

    @property
    def num_tokens(self) -> int:
        return self.num_prefill_tokens + self.num_decode_tokens
