"""Adapter lifecycle manager: multi-LoRA slots + hot load/evict.

MI355X-native ownership of two reference capabilities:

  * the per-GPU adapter slot table vLLM exposes as `--max-loras` /
    `vllm:lora_requests_info` (reference backend/vllm/metrics.go:20-31) — the
    resident set feeds the router's LoRA-affinity / can-accept filters;
  * the dynamic-LoRA sidecar's load/unload actuation
    (tools/dynamic-lora-sidecar/sidecar/sidecar.py:177-213) — here hot
    load is a pinned-host → HBM `hipMemcpyAsync` on a dedicated side stream
    (torch.cuda.Stream), overlapped with serving, with an event fence before
    the adapter's first use.

Weight organisation: per (layer, module) stacked device tensors
  A[max_slots, rank, in_features], B[max_slots, out_features, rank]
so the SGMV kernels index adapters by slot id with zero per-step rewiring.
LoRA scaling (alpha/rank) is folded into B at load time so a mixed batch
needs no per-token scaling factor.
"""

from __future__ import annotations

import logging
import os
import threading
import time
from dataclasses import dataclass
from typing import Dict, List, Optional, Tuple

import torch

log = logging.getLogger(__name__)

# MFMA segment-tiled prefill-LoRA path: wins its microbench (2.5x at
# T=2048/O=28672, benchmarks/bench_sgmv.py) but LOSES end-to-end in the
# engine (same-box A/B: 26.9 vs 31.0 req/s goodput) — real prefill batches
# have many partial 64-token tiles and the scalar y read-modify-write
# epilogue costs more than the adapter-read amortization saves.  Opt-in for
# further work; the wide per-token kernels are the default.
_TILED_SGMV = os.environ.get("LIG_TILED_SGMV", "0") == "1"

# Slot-grouped unit kernels (r02): default ON; LIG_SGMV_UNITS=0 falls back to
# the per-token/wide kernels for A/B.
_UNITS_SGMV = os.environ.get("LIG_SGMV_UNITS", "1") == "1"

def build_lora_units(slots_np, cap: int = 32):
    """Sort a per-token adapter-slot array into slot-grouped units.

    Returns int32 numpy arrays (order[T], unit_t0[U], unit_len[U],
    unit_slot[U]): tokens sorted by slot (stable — prefill runs stay
    contiguous), slot<0 tokens excluded, runs chunked to <= ``cap`` tokens
    (bounds per-thread serialization in the unit kernels while keeping the
    adapter-weight read amortized)."""
    import numpy as np

    arr = np.asarray(slots_np, dtype=np.int32)
    order = np.argsort(arr, kind="stable").astype(np.int32)
    sorted_slots = arr[order]
    t0s, lens, uslots = [], [], []
    i = int(np.searchsorted(sorted_slots, 0))  # skip slot -1 tokens
    n = len(arr)
    while i < n:
        s = int(sorted_slots[i])
        k = int(np.searchsorted(sorted_slots, s, side="right"))
        while i < k:
            ln = min(cap, k - i)
            t0s.append(i)
            lens.append(ln)
            uslots.append(s)
            i += ln
    return (
        order,
        np.asarray(t0s, dtype=np.int32),
        np.asarray(lens, dtype=np.int32),
        np.asarray(uslots, dtype=np.int32),
    )


# module name -> (in_features, out_features) resolver
def _module_dims(mc, tp_size: int) -> Dict[str, Tuple[int, int]]:
    h = mc.hidden_size
    d = mc.head_dim
    q = mc.num_heads * d // tp_size
    kv = mc.num_kv_heads * d // tp_size
    dims = {
        "qkv": (h, q + 2 * kv),
        "o": (q, h),
    }
    if not mc.is_moe:
        i = mc.intermediate_size // tp_size
        dims["gate_up"] = (h, 2 * i)
        dims["down"] = (i, h)
    return dims


@dataclass
class _Adapter:
    name: str
    rank: int
    scaling: float
    seed: Optional[int] = None  # random-init adapters
    host_weights: Optional[Dict] = None  # {(layer, module): (A, B)} pinned
    slot: int = -1  # -1 = not resident
    last_used: float = 0.0
    load_event: Optional[torch.cuda.Event] = None


class AdapterManager:
    def __init__(
        self,
        engine,
        max_slots: int = 8,
        rank: int = 16,
        alpha: float = 32.0,
        modules: Optional[List[str]] = None,
    ) -> None:
        self.engine = engine
        runner = engine.runner
        self.device = runner.device
        self.dtype = runner.dtype
        mc = runner.config.model
        tp_size = runner.tp.world_size if runner.tp is not None else 1
        self.num_layers = mc.num_layers
        self.rank = rank
        self.alpha = alpha
        self.max_slots = max_slots
        self.dims = _module_dims(mc, tp_size)
        if modules:
            self.dims = {k: v for k, v in self.dims.items() if k in modules}
        # stacked slot tensors per (layer, module)
        self.a_stacks: Dict[Tuple[int, str], torch.Tensor] = {}
        self.b_stacks: Dict[Tuple[int, str], torch.Tensor] = {}
        for layer in range(self.num_layers):
            for m, (fin, fout) in self.dims.items():
                self.a_stacks[(layer, m)] = torch.zeros(
                    max_slots, rank, fin, dtype=self.dtype, device=self.device
                )
                self.b_stacks[(layer, m)] = torch.zeros(
                    max_slots, fout, rank, dtype=self.dtype, device=self.device
                )
        self._adapters: Dict[str, _Adapter] = {}
        self._slots: List[Optional[str]] = [None] * max_slots
        # Shrink-output workspace u[T, R]: allocated ONCE at the max flat
        # batch (token budget rounded up to the GEMM pad multiple).  Decode
        # hipGraphs bake this pointer, so it must never be reallocated.
        cfg = runner.config
        padm = max(getattr(cfg, "pad_batch_multiple", 0), 1)
        max_t = (cfg.max_num_batched_tokens + padm - 1) // padm * padm
        self._u_ws = torch.zeros(
            max(max_t, 256), rank, dtype=self.dtype, device=self.device
        )
        self._lock = threading.RLock()
        self._side_stream = (
            torch.cuda.Stream(device=self.device)
            if self.device.type == "cuda"
            else None
        )
        # attach to the model's layers
        self._attach(runner.model)
        engine.adapter_manager = self

    # -- model hookup ---------------------------------------------------------
    def _attach(self, model) -> None:
        for idx, layer in enumerate(model.layers):
            attn = layer.self_attn
            attn._lora = self
            attn._layer_idx = idx
            mlp = getattr(layer, "mlp", None)
            if mlp is not None and "gate_up" in self.dims:
                mlp._lora = self
                mlp._layer_idx = idx

    # -- registry (the sidecar's load/unload surface) --------------------------
    def register_random(self, name: str, seed: int, rank: Optional[int] = None) -> None:
        """Register an adapter with deterministic random weights (synthetic
        workloads; no checkpoint files in this environment)."""
        with self._lock:
            if name in self._adapters:
                return
            r = rank or self.rank
            self._adapters[name] = _Adapter(
                name=name, rank=r, scaling=self.alpha / r, seed=seed
            )

    def register_weights(self, name: str, weights: Dict, rank: Optional[int] = None) -> None:
        """weights: {(layer_idx, module): (A [r,in], B [out,r]) cpu tensors}."""
        with self._lock:
            r = rank or self.rank
            self._adapters[name] = _Adapter(
                name=name, rank=r, scaling=self.alpha / r, host_weights=weights
            )

    def register_weights_full(self, name: str, weights: Dict, rank: Optional[int] = None) -> None:
        """Register an adapter from FULL (unsharded) weights on a TP engine:
        shards each module's (A, B) to this rank's slice using the base
        model's layout (column-parallel qkv/gate_up: A replicated, B rows
        sharded; row-parallel o/down: A columns sharded with the input, B
        replicated — the SGMV delta lands before the block all-reduce, so
        the partial sums add up exactly; tests/test_tp.py validates
        token-identity vs TP=1).  On TP=1 engines this is register_weights."""
        runner = self.engine.runner
        tp = runner.tp
        if tp is None or tp.world_size == 1:
            self.register_weights(name, weights, rank=rank)
            return
        mc = runner.config.model
        w = tp.world_size
        r_ = tp.rank
        hd, nh, nkv = mc.head_dim, mc.num_heads, mc.num_kv_heads
        qs, kvs = nh * hd // w, nkv * hd // w
        il = getattr(mc, "intermediate_size", 0)
        ils = il // w if il else 0
        sharded: Dict = {}
        for (layer, mod), (A, B) in weights.items():
            if mod == "qkv":
                bq, bk, bv = B.split([nh * hd, nkv * hd, nkv * hd], dim=0)
                Bs = torch.cat(
                    [
                        bq[r_ * qs:(r_ + 1) * qs],
                        bk[r_ * kvs:(r_ + 1) * kvs],
                        bv[r_ * kvs:(r_ + 1) * kvs],
                    ],
                    dim=0,
                )
                As = A
            elif mod == "o":
                As = A[:, r_ * qs:(r_ + 1) * qs]
                Bs = B
            elif mod == "gate_up":
                bg, bu = B.chunk(2, dim=0)
                Bs = torch.cat(
                    [bg[r_ * ils:(r_ + 1) * ils], bu[r_ * ils:(r_ + 1) * ils]],
                    dim=0,
                )
                As = A
            elif mod == "down":
                As = A[:, r_ * ils:(r_ + 1) * ils]
                Bs = B
            else:
                raise KeyError(f"unknown LoRA module {mod!r}")
            sharded[(layer, mod)] = (As.contiguous(), Bs.contiguous())
        self.register_weights(name, sharded, rank=rank)

    def unregister(self, name: str) -> bool:
        """Sidecar 'unload': evict from GPU and drop from the registry."""
        with self._lock:
            ad = self._adapters.pop(name, None)
            if ad is None:
                return False
            if ad.slot >= 0:
                self._slots[ad.slot] = None
            return True

    def registered_names(self) -> List[str]:
        with self._lock:
            return list(self._adapters)

    def resident_names(self) -> List[str]:
        """The `running_lora_adapters` gauge (reference metrics.go:105-126)."""
        with self._lock:
            return [n for n in self._slots if n is not None]

    def slot_of(self, name: str) -> int:
        if not name:
            return -1
        with self._lock:
            ad = self._adapters.get(name)
            return ad.slot if ad is not None else -1

    # -- residency -------------------------------------------------------------
    def touch(self, name: str) -> None:
        """Mark intent-to-use (called at request admission); starts an async
        load if the adapter is registered but not resident."""
        with self._lock:
            ad = self._adapters.get(name)
            if ad is None:
                raise KeyError(f"unknown LoRA adapter {name!r}")
            ad.last_used = time.monotonic()
            if ad.slot < 0:
                self._load(ad)

    def prepare(self, names: List[str]) -> None:
        """Ensure every adapter in `names` is resident and its load copy has
        completed before the compute stream uses the slot weights."""
        pending: List[_Adapter] = []
        with self._lock:
            needed = set(n for n in names if n)
            for n in needed:
                ad = self._adapters.get(n)
                if ad is None:
                    raise KeyError(f"unknown LoRA adapter {n!r}")
                ad.last_used = time.monotonic()
                if ad.slot < 0:
                    self._load(ad, in_use=needed)
                if ad.load_event is not None:
                    pending.append(ad)
        for ad in pending:
            if ad.load_event is not None:
                torch.cuda.current_stream(self.device).wait_event(ad.load_event)
                ad.load_event = None

    def _free_slot(self, in_use: Optional[set] = None) -> int:
        for i, n in enumerate(self._slots):
            if n is None:
                return i
        # LRU-evict a resident adapter not needed by the current batch
        candidates = [
            self._adapters[n]
            for n in self._slots
            if n is not None and (in_use is None or n not in in_use)
        ]
        if not candidates:
            raise RuntimeError(
                f"all {self.max_slots} LoRA slots are in use by the current batch"
            )
        victim = min(candidates, key=lambda a: a.last_used)
        log.info("evicting LoRA %s from slot %d", victim.name, victim.slot)
        slot = victim.slot
        victim.slot = -1
        self._slots[slot] = None
        return slot

    def _materialize_host(self, ad: _Adapter) -> Dict:
        """Generate/cached pinned host weights (scaling folded into B)."""
        if ad.host_weights is not None:
            return ad.host_weights
        g = torch.Generator().manual_seed(ad.seed or 0)
        weights = {}
        pin = self.device.type == "cuda"
        for layer in range(self.num_layers):
            for m, (fin, fout) in self.dims.items():
                a = torch.randn(ad.rank, fin, generator=g, dtype=torch.float32)
                b = torch.randn(fout, ad.rank, generator=g, dtype=torch.float32)
                a = (a * (1.0 / fin**0.5)).to(self.dtype)
                b = (b * 0.01 * ad.scaling).to(self.dtype)
                if pin:
                    a = a.pin_memory()
                    b = b.pin_memory()
                weights[(layer, m)] = (a, b)
        ad.host_weights = weights
        return weights

    def _load(self, ad: _Adapter, in_use: Optional[set] = None) -> None:
        slot = self._free_slot(in_use)
        weights = self._materialize_host(ad)
        r = ad.rank
        if self._side_stream is not None:
            stream_ctx = torch.cuda.stream(self._side_stream)
        else:
            import contextlib

            stream_ctx = contextlib.nullcontext()
        with stream_ctx:
            for layer in range(self.num_layers):
                for m in self.dims:
                    a, b = weights[(layer, m)]
                    self.a_stacks[(layer, m)][slot, :r].copy_(a, non_blocking=True)
                    self.b_stacks[(layer, m)][slot, :, :r].copy_(b, non_blocking=True)
            if r < self.rank:
                for layer in range(self.num_layers):
                    for m in self.dims:
                        self.a_stacks[(layer, m)][slot, r:].zero_()
                        self.b_stacks[(layer, m)][slot, :, r:].zero_()
        if self._side_stream is not None:
            ev = torch.cuda.Event()
            ev.record(self._side_stream)
            ad.load_event = ev
        ad.slot = slot
        self._slots[slot] = ad.name

    # -- the compute hook (called from model layers) ----------------------------
    def apply(self, layer_idx: int, module: str, x: torch.Tensor, y: torch.Tensor, meta) -> None:
        slots = getattr(meta, "lora_slot_per_token", None)
        if slots is None:
            return
        from .. import ops

        key = (layer_idx, module)
        if ops.native_available() and x.is_cuda:
            T = x.shape[0]
            ws = getattr(meta, "lora_ws", None)
            if ws is None:
                ws = self._u_ws
            assert ws.shape[0] >= T, "LoRA workspace smaller than flat batch"
            # slot-grouped units (r02 default): shrink on the occupancy-first
            # v2 kernels, expand via the LDS-staged unit kernel (B read once
            # per <=32-token unit; profiles/r02_kernel_stats.md)
            order = getattr(meta, "lora_order", None)
            if order is not None and _UNITS_SGMV and y.shape[1] % 256 == 0:
                wide = bool(
                    getattr(meta, "num_prefill_tokens", 0) > 0 and T >= 256
                )
                ops._EXT.sgmv_apply_units(
                    y, x, self.a_stacks[key], self.b_stacks[key], ws,
                    slots, order, meta.lora_unit_t0, meta.lora_unit_len,
                    meta.lora_unit_slot, 1.0, wide, ops._stream(),
                )
                return
            tile_t0 = getattr(meta, "lora_tile_t0", None)
            if (
                tile_t0 is not None
                and y.shape[1] % 64 == 0
                and x.shape[1] % 32 == 0
                and _TILED_SGMV
            ):
                # prefill: MFMA over 64-token same-adapter segment tiles
                ops._EXT.sgmv_apply_tiled(
                    y, x, self.a_stacks[key], self.b_stacks[key], ws,
                    tile_t0, meta.lora_tile_len, meta.lora_tile_slot, 1.0,
                    ops._stream(),
                )
                # tiles cover only the prefill partition; a mixed batch's
                # decode suffix still needs the per-token path or those
                # tokens silently lose their LoRA delta
                np_ = int(getattr(meta, "num_prefill_tokens", 0))
                if np_ < T:
                    ops._EXT.sgmv_apply(
                        y[np_:], x[np_:], self.a_stacks[key],
                        self.b_stacks[key], ws, slots[np_:],
                        1.0, False, ops._stream(),
                    )
            else:
                # decode (or odd dims): per-token kernels; wide variants only
                # help when consecutive tokens share an adapter
                wide = bool(
                    getattr(meta, "num_prefill_tokens", 0) > 0 and x.shape[0] >= 256
                )
                ops._EXT.sgmv_apply(
                    y, x, self.a_stacks[key], self.b_stacks[key], ws, slots,
                    1.0, wide, ops._stream(),
                )
        else:
            u = ops.sgmv_shrink(x, self.a_stacks[key], slots)
            ops.sgmv_expand(u, self.b_stacks[key], slots, y, 1.0)
