"""hipGraph capture of decode-only steps.

MI355X-first use of HIP graphs instead of a tracing compiler (build brief):
a decode step of Llama-3-8B issues ~400 kernels (32 layers x {norm, rope,
cache-scatter, attention, 4 GEMMs, swiglu, 2x SGMV x 4 modules}); dispatched
eagerly from Python the GPU sits ~50-60% idle (profiles/r01_*.md).  Decode
batches have a fixed shape per batch-size bucket, so the whole step is
captured once per bucket into a hipGraph and replayed with one launch.

Static-buffer contract: per padded batch size B the graph reads
  ids[B], pos[B], slot_mapping[B], block_tables[B, MAXB], context_lens[B],
  lora_slots[B]
and writes logits[B, vocab] (bf16).  Padding rows decode a scratch KV block
(one extra block appended to the cache pool) with context_len=1 so they
compute garbage without touching live state.

Prefill / mixed steps stay eager (their shapes vary too much to bucket).
"""

from __future__ import annotations

import logging
from typing import Dict, List, Optional

import torch

from .metadata import BatchMeta

log = logging.getLogger(__name__)

CAPTURE_SIZES = [1, 2, 4, 8, 16, 24, 32, 48, 64, 96, 128, 192, 256, 320, 384, 448, 512]


class DecodeGraphRunner:
    def __init__(self, runner) -> None:
        self.runner = runner
        self.device = runner.device
        cfg = runner.config
        mc = cfg.model
        self.max_bs = cfg.max_num_seqs
        self.max_blocks = cfg.max_model_len // cfg.block_size
        self.block_size = cfg.block_size
        self.sizes = [s for s in CAPTURE_SIZES if s <= self.max_bs]
        if self.sizes[-1] != self.max_bs:
            self.sizes.append(self.max_bs)
        dev = self.device
        B = self.max_bs
        self.ids = torch.zeros(B, dtype=torch.int64, device=dev)
        self.pos = torch.zeros(B, dtype=torch.int64, device=dev)
        self.slot_mapping = torch.zeros(B, dtype=torch.int64, device=dev)
        self.block_tables = torch.zeros(
            B, self.max_blocks, dtype=torch.int32, device=dev
        )
        self.context_lens = torch.ones(B, dtype=torch.int32, device=dev)
        self.lora_slots = torch.full((B,), -1, dtype=torch.int32, device=dev)
        # slot-grouped SGMV unit buffers (U == B, empty units padded len=0)
        self.lora_order = torch.zeros(B, dtype=torch.int32, device=dev)
        self.lora_ut0 = torch.zeros(B, dtype=torch.int32, device=dev)
        self.lora_ulen = torch.zeros(B, dtype=torch.int32, device=dev)
        self.lora_uslot = torch.full((B,), -1, dtype=torch.int32, device=dev)
        # graph-private LoRA shrink workspace (see metadata.lora_ws): replay
        # must not share u with a concurrently-running eager prefill
        self.lora_ws: Optional[torch.Tensor] = None
        # pinned host staging (one copy per tensor per step)
        self.h_ids = torch.zeros(B, dtype=torch.int64, pin_memory=True)
        self.h_pos = torch.zeros(B, dtype=torch.int64, pin_memory=True)
        self.h_slot = torch.zeros(B, dtype=torch.int64, pin_memory=True)
        self.h_bt = torch.zeros(B, self.max_blocks, dtype=torch.int32, pin_memory=True)
        self.h_ctx = torch.ones(B, dtype=torch.int32, pin_memory=True)
        self.h_lora = torch.full((B,), -1, dtype=torch.int32, pin_memory=True)
        self.h_order = torch.zeros(B, dtype=torch.int32, pin_memory=True)
        self.h_ut0 = torch.zeros(B, dtype=torch.int32, pin_memory=True)
        self.h_ulen = torch.zeros(B, dtype=torch.int32, pin_memory=True)
        self.h_uslot = torch.full((B,), -1, dtype=torch.int32, pin_memory=True)
        # numpy views of the pinned staging (fast per-row fills from lists)
        self.n_ids = self.h_ids.numpy()
        self.n_pos = self.h_pos.numpy()
        self.n_slot = self.h_slot.numpy()
        self.n_bt = self.h_bt.numpy()
        self.n_ctx = self.h_ctx.numpy()
        self.n_lora = self.h_lora.numpy()
        self.n_order = self.h_order.numpy()
        self.n_ut0 = self.h_ut0.numpy()
        self.n_ulen = self.h_ulen.numpy()
        self.n_uslot = self.h_uslot.numpy()
        # scratch KV block for padding rows = last block of the (extended) pool
        self.scratch_block = runner.config.num_kv_blocks  # extra block id
        self.graphs: Dict[int, torch.cuda.CUDAGraph] = {}
        self.logits: Dict[int, torch.Tensor] = {}
        self._pool = None
        self.with_lora = False

    def _graph_nsplits(self, B: int) -> int:
        """Bucket-static flash-decode split factor: replay can't branch on
        live context lengths, so pick from the bucket size + the engine's max
        context (empty split ranges merge as no-ops — bit-exact either way,
        profiles/r01_flash_decode_split.txt)."""
        from .. import ops

        attn = self.runner.model.layers[0].self_attn
        return ops.decode_nsplits(
            B, attn.num_kv_heads, self.max_blocks * self.block_size
        )

    def _meta(self, B: int) -> BatchMeta:
        return BatchMeta(
            num_prefill_tokens=0,
            num_decode_tokens=B,
            slot_mapping=self.slot_mapping[:B],
            decode_block_tables=self.block_tables[:B],
            decode_context_lens=self.context_lens[:B],
            lora_slot_per_token=self.lora_slots[:B] if self.with_lora else None,
            lora_order=self.lora_order[:B] if self.with_lora else None,
            lora_unit_t0=self.lora_ut0[:B] if self.with_lora else None,
            lora_unit_len=self.lora_ulen[:B] if self.with_lora else None,
            lora_unit_slot=self.lora_uslot[:B] if self.with_lora else None,
            lora_ws=self.lora_ws,
            decode_nsplits=self._graph_nsplits(B),
        )

    def _forward(self, B: int) -> torch.Tensor:
        model = self.runner.model
        hidden = model(self.ids[:B], self.pos[:B], self.runner.kv_caches, self._meta(B))
        idx = torch.arange(B, device=self.device)
        return model.compute_logits(hidden, idx).to(torch.bfloat16)

    def capture(self, with_lora: bool) -> None:
        self.with_lora = with_lora
        if with_lora and self.lora_ws is None:
            mgr = self.runner.adapter_manager
            self.lora_ws = torch.zeros(
                self.max_bs, mgr.rank, dtype=mgr.dtype, device=self.device
            )
        # Preallocate the flash-decode split scratch at the max any bucket
        # needs so every captured graph bakes one stable pointer (the ops
        # cache grows monotonically; growth during capture would leave
        # earlier graphs pointing at freed memory).
        from .. import ops

        attn = self.runner.model.layers[0].self_attn
        hq, d = attn.num_heads, attn.head_dim
        max_scratch = max(
            (B * hq * self._graph_nsplits(B) * (d + 2) for B in self.sizes),
            default=0,
        )
        if max_scratch > 0 and self.device.type == "cuda":
            ops.split_scratch(self.device, max_scratch)
        # seed static buffers with safe values (padding semantics)
        self.block_tables.fill_(0)
        self.block_tables[:, 0] = self.scratch_block
        self.context_lens.fill_(1)
        self.slot_mapping.fill_(self.scratch_block * self.block_size)
        self.lora_slots.fill_(-1)
        torch.cuda.synchronize(self.device)
        # Warmup on a side stream (torch.cuda.graph requirement) — at EVERY
        # capture batch size: hipBLASLt picks its algorithm + allocates its
        # workspace on the FIRST call of each GEMM shape, and a first call
        # that happens inside graph capture is restricted to slow no-workspace
        # fallbacks which then get baked into the graph (measured: in-graph
        # decode GEMMs 2-8x slower than the same shapes eager —
        # profiles/r01_c256_breakdown.md).
        s = torch.cuda.Stream(device=self.device)
        s.wait_stream(torch.cuda.current_stream(self.device))
        with torch.cuda.stream(s):
            for B in self.sizes:
                self._forward(B)
        torch.cuda.current_stream(self.device).wait_stream(s)
        torch.cuda.synchronize(self.device)
        for B in self.sizes:
            g = torch.cuda.CUDAGraph()
            if self._pool is None:
                with torch.cuda.graph(g):
                    self.logits[B] = self._forward(B)
                self._pool = g.pool()
            else:
                with torch.cuda.graph(g, pool=self._pool):
                    self.logits[B] = self._forward(B)
            self.graphs[B] = g
        torch.cuda.synchronize(self.device)
        log.info("captured %d decode graphs (lora=%s)", len(self.graphs), with_lora)

    def pick_size(self, n: int) -> Optional[int]:
        for s in self.sizes:
            if s >= n:
                return s
        return None

    def run(
        self,
        decodes,  # List[ScheduledSeq]
        slot_of,  # callable(name) -> int, or None
    ) -> torch.Tensor:
        """Fill static buffers, replay the right graph, return logits[:n]."""
        n = len(decodes)
        B = self.pick_size(n)
        assert B is not None
        bs = self.block_size
        n_ids, n_pos, n_slot = self.n_ids, self.n_pos, self.n_slot
        n_bt, n_ctx, n_lora = self.n_bt, self.n_ctx, self.n_lora
        for i, ss in enumerate(decodes):
            seq = ss.seq
            p = seq.num_computed_tokens
            n_ids[i] = seq.all_token_ids[p]
            n_pos[i] = p
            n_slot[i] = seq.block_table[p // bs] * bs + p % bs
            nb = len(seq.block_table)
            n_bt[i, :nb] = seq.block_table
            n_ctx[i] = p + 1
            if self.with_lora:
                n_lora[i] = slot_of(seq.lora_name) if seq.lora_name else -1
        # padding rows
        if B > n:
            n_ids[n:B] = 0
            n_pos[n:B] = 0
            n_slot[n:B] = self.scratch_block * bs
            n_bt[n:B, :] = 0
            n_bt[n:B, 0] = self.scratch_block
            n_ctx[n:B] = 1
            n_lora[n:B] = -1
        self.ids[:B].copy_(self.h_ids[:B], non_blocking=True)
        self.pos[:B].copy_(self.h_pos[:B], non_blocking=True)
        self.slot_mapping[:B].copy_(self.h_slot[:B], non_blocking=True)
        self.block_tables[:B].copy_(self.h_bt[:B], non_blocking=True)
        self.context_lens[:B].copy_(self.h_ctx[:B], non_blocking=True)
        if self.with_lora:
            from ..lora.manager import build_lora_units

            order, ut0, ulen, uslot = build_lora_units(self.n_lora[:B])
            u = len(ut0)
            self.n_order[:B] = order
            self.n_ut0[:u] = ut0
            self.n_ulen[:u] = ulen
            self.n_ulen[u:B] = 0  # padding units: no-op in the kernels
            self.n_uslot[:u] = uslot
            self.n_uslot[u:B] = -1
            self.lora_slots[:B].copy_(self.h_lora[:B], non_blocking=True)
            self.lora_order[:B].copy_(self.h_order[:B], non_blocking=True)
            self.lora_ut0[:B].copy_(self.h_ut0[:B], non_blocking=True)
            self.lora_ulen[:B].copy_(self.h_ulen[:B], non_blocking=True)
            self.lora_uslot[:B].copy_(self.h_uslot[:B], non_blocking=True)
        self.graphs[B].replay()
        return self.logits[B][:n]
