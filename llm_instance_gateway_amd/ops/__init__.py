"""Engine ops: dispatch between the CDNA4 HIP extension and torch references.

Policy:
  * CUDA (= ROCm/HIP) tensors → the in-tree native extension
    ``llm_instance_gateway_amd._C`` (hand-written gfx950 kernels).  If the
    extension is missing on a GPU machine the op RAISES — no silent eager
    fallback (the driver records which .so files GPU processes load).
  * CPU tensors → the fp32 torch reference (`torch_ref`), which is also the
    ground truth the kernels are tested against.
  * ``LIG_FORCE_TORCH_OPS=1`` forces the torch path on GPU — for A/B numerics
    and perf comparisons only.
"""

from __future__ import annotations

import os
from typing import Optional, Tuple

import torch

from . import torch_ref

_EXT = None
_EXT_ERR: Optional[Exception] = None


def _load_ext():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    try:
        from llm_instance_gateway_amd import _C  # built in-tree by setup.py

        _EXT = _C
    except ImportError as e:
        _EXT_ERR = e
    return _EXT


def native_available() -> bool:
    return _load_ext() is not None


def _stream() -> int:
    return torch.cuda.current_stream().cuda_stream


def _use_native(t: torch.Tensor) -> bool:
    if not t.is_cuda:
        return False
    if os.environ.get("LIG_FORCE_TORCH_OPS") == "1":
        return False
    ext = _load_ext()
    if ext is None:
        raise RuntimeError(
            "llm_instance_gateway_amd._C native extension is not built but a "
            "GPU tensor was passed; build it with `python setup.py "
            f"build_ext --inplace` (import error: {_EXT_ERR})"
        )
    return True


# -- public op surface -------------------------------------------------------

def rms_norm(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    if _use_native(x):
        out = torch.empty_like(x)
        _EXT.rms_norm(out, x, weight, eps, _stream())
        return out
    return torch_ref.rms_norm(x, weight, eps)


def fused_add_rms_norm(
    x: torch.Tensor, residual: torch.Tensor, weight: torch.Tensor, eps: float
) -> Tuple[torch.Tensor, torch.Tensor]:
    if _use_native(x):
        # In-place: residual += x; x = rmsnorm(residual) written to out.
        out = torch.empty_like(x)
        _EXT.fused_add_rms_norm(out, x, residual, weight, eps, _stream())
        return out, residual
    return torch_ref.fused_add_rms_norm(x, residual, weight, eps)


def rotary_embedding(
    positions: torch.Tensor,
    q: torch.Tensor,
    k: torch.Tensor,
    cos_sin_cache: torch.Tensor,
    head_dim: int,
) -> Tuple[torch.Tensor, torch.Tensor]:
    if _use_native(q):
        _EXT.rotary_embedding(positions, q, k, cos_sin_cache, head_dim, _stream())
        return q, k
    return torch_ref.rotary_embedding(positions, q, k, cos_sin_cache, head_dim)


def reshape_and_cache(
    k: torch.Tensor,
    v: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    slot_mapping: torch.Tensor,
) -> None:
    if _use_native(k):
        _EXT.reshape_and_cache(k, v, k_cache, v_cache, slot_mapping, _stream())
        return
    torch_ref.reshape_and_cache(k, v, k_cache, v_cache, slot_mapping)


# Flash-decoding context split (ROADMAP r01 #1, measured in
# profiles/r01_flash_decode_split.txt: 13.7x at N=4/ctx=16k, neutral at
# chip-filling batch).  The single-pass kernel launches N*Hkv workgroups;
# when that is far below the chip's 256 CUs and the context is long, split
# the context across `ns` workgroups + a combine pass (bit-exact merge).
_NUM_CUS = 256
_SPLIT_MIN_CTX = 1024      # short contexts: single-pass latency already fine
_SPLIT_MIN_PER_SPLIT = 256  # keep each split's K/V walk worth a workgroup
_SPLIT_MAX = 32
_SPLIT_SCRATCH: dict = {}   # device -> fp32 scratch tensor
# Superseded scratches stay referenced: a captured hipGraph bakes the pointer
# it saw, so freeing an old scratch after a grow would let the allocator hand
# that memory to live tensors the graph then scribbles over.
_SPLIT_SCRATCH_RETIRED: list = []


def decode_nsplits(n_seqs: int, n_kv_heads: int, max_ctx: int) -> int:
    """Context-split factor for the paged decode kernel (1 = single-pass)."""
    wg = n_seqs * n_kv_heads
    target = 2 * _NUM_CUS
    if wg == 0 or wg >= target or max_ctx < _SPLIT_MIN_CTX:
        return 1
    ns = min(_SPLIT_MAX, (target + wg - 1) // wg)
    ns = min(ns, max(1, max_ctx // _SPLIT_MIN_PER_SPLIT))
    return max(1, ns)


def split_scratch(device, numel: int) -> torch.Tensor:
    """fp32 partials scratch [N, Hq, ns, D+2] (flat).  Cached per device and
    grown monotonically; graph capture preallocates its max so every captured
    graph keeps one stable pointer."""
    t = _SPLIT_SCRATCH.get(device)
    if t is None or t.numel() < numel:
        if t is not None:
            _SPLIT_SCRATCH_RETIRED.append(t)
        t = torch.empty(numel, dtype=torch.float32, device=device)
        _SPLIT_SCRATCH[device] = t
    return t


def paged_attention_decode(
    q: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,
    context_lens: torch.Tensor,
    scale: float,
    max_ctx: int = 0,
    nsplits: int = 0,
) -> torch.Tensor:
    """`max_ctx` is a host-side hint (no device sync) driving the split
    heuristic; `nsplits` > 0 pins the factor (the hipGraph path captures a
    bucket-static value)."""
    if _use_native(q):
        out = torch.empty_like(q)
        N, Hq, D = q.shape
        Hkv = k_cache.shape[1]
        ns = nsplits if nsplits > 0 else decode_nsplits(N, Hkv, max_ctx)
        if ns > 1:
            scratch = split_scratch(q.device, N * Hq * ns * (D + 2))
            _EXT.paged_attention_decode_split(
                out, q, k_cache, v_cache, block_tables, context_lens,
                scratch, scale, ns, _stream()
            )
        else:
            _EXT.paged_attention_decode(
                out, q, k_cache, v_cache, block_tables, context_lens, scale,
                _stream()
            )
        return out
    return torch_ref.paged_attention_decode(
        q, k_cache, v_cache, block_tables, context_lens, scale
    )


def paged_attention_prefill(
    q: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,
    query_start_locs: torch.Tensor,
    context_lens: torch.Tensor,
    scale: float,
    tok_block_tables: "Optional[torch.Tensor]" = None,
    tok_context_lens: "Optional[torch.Tensor]" = None,
    tile_seq: "Optional[torch.Tensor]" = None,
    tile_qstart: "Optional[torch.Tensor]" = None,
) -> torch.Tensor:
    if _use_native(q):
        head_dim = q.shape[-1]
        if head_dim in (64, 128):
            # MFMA flash-prefill kernel over 64-row query tiles.
            if tile_seq is None or tile_qstart is None:
                qsl = query_start_locs.cpu().tolist()
                ts, tq = [], []
                for s in range(len(qsl) - 1):
                    for qb in range(0, qsl[s + 1] - qsl[s], 64):
                        ts.append(s)
                        tq.append(qb)
                dev = q.device
                tile_seq = torch.tensor(ts, dtype=torch.int32, device=dev)
                tile_qstart = torch.tensor(tq, dtype=torch.int32, device=dev)
            out = torch.empty_like(q)
            _EXT.paged_attention_prefill(
                out, q, k_cache, v_cache, tile_seq, tile_qstart,
                query_start_locs, context_lens, block_tables, scale, _stream()
            )
            return out
        if tok_block_tables is not None and tok_context_lens is not None:
            # per-token expansion precomputed once per step by the model runner
            return paged_attention_decode(
                q, k_cache, v_cache, tok_block_tables, tok_context_lens, scale
            )
        # v1 native path: run each chunk query row as a "decode" against its
        # causal context prefix (context = L - n + j + 1).  The GROUP waves of
        # a row share K/V lines via L1/L2, and consecutive rows of a chunk
        # walk overlapping context, so the re-read is largely cache-absorbed.
        # A fused MFMA flash-prefill kernel replaces this (see csrc/).
        n_per = (query_start_locs[1:] - query_start_locs[:-1]).long()
        ctx_end = context_lens.long()
        # per-token causal context length
        tok_in_chunk = torch.cat(
            [torch.arange(int(n), device=q.device) for n in n_per]
        )
        ctx_per_tok = (
            torch.repeat_interleave(ctx_end - n_per, n_per) + tok_in_chunk + 1
        ).int()
        bt_per_tok = torch.repeat_interleave(block_tables, n_per, dim=0).contiguous()
        return paged_attention_decode(
            q, k_cache, v_cache, bt_per_tok, ctx_per_tok, scale
        )
    return torch_ref.paged_attention_prefill(
        q, k_cache, v_cache, block_tables, query_start_locs, context_lens, scale
    )


def quant_fp8_rowwise(x: torch.Tensor):
    """Fused dynamic per-row e4m3 quantization: (x [T,K] bf16) ->
    (xq [T,K] float8_e4m3fn, xs [T,1] fp32) with x ≈ xq.float() * xs.
    One kernel launch vs ~4 for the torch elementwise path (models/quant.py)."""
    if _use_native(x):
        xq = torch.empty_like(x, dtype=torch.float8_e4m3fn)
        xs = torch.empty(x.shape[0], dtype=torch.float32, device=x.device)
        _EXT.quant_fp8_rowwise(xq, xs, x, _stream())
        return xq, xs.unsqueeze(1)
    # CPU reference of the KERNEL's exact semantics (amax clamp 1e-6,
    # matching models/quant.Fp8Linear._quant_input's activation path)
    amax = x.abs().amax(dim=-1, keepdim=True).float().clamp_(min=1e-6)
    xs = amax / 448.0
    xq = (x / xs).clamp_(-448.0, 448.0).to(torch.float8_e4m3fn)
    return xq, xs


def silu_and_mul(x: torch.Tensor) -> torch.Tensor:
    if _use_native(x):
        d = x.shape[-1] // 2
        out = torch.empty(*x.shape[:-1], d, dtype=x.dtype, device=x.device)
        _EXT.silu_and_mul(out, x, _stream())
        return out
    return torch_ref.silu_and_mul(x)


build_cos_sin_cache = torch_ref.build_cos_sin_cache


def sgmv_shrink(x, a_stack, slots):
    if _use_native(x):
        T = x.shape[0]
        R = a_stack.shape[1]
        u = torch.zeros(T, R, dtype=x.dtype, device=x.device)
        _EXT.sgmv_shrink(u, x, a_stack, slots, _stream())
        return u
    return torch_ref.sgmv_shrink(x, a_stack, slots)


def sgmv_expand(u, b_stack, slots, y, scaling=1.0):
    if _use_native(u):
        _EXT.sgmv_expand(y, u, b_stack, slots, float(scaling), _stream())
        return
    torch_ref.sgmv_expand(u, b_stack, slots, y, scaling)


# MEASURED DEAD END (benchmarks/bench_gemm.py on MI355X): the hand-written
# skinny-M MFMA GEMM (ops/csrc/skinny_gemm.hip) reaches only ~0.5 TB/s of W
# streaming vs hipBLASLt's 2.4 TB/s at M=128 — and the CDNA guide's own
# best-effort decode-projection recipe tops out at 1.3-1.6 TB/s ("per-CU
# load-path/latency-bound ... nowhere near the HBM roofline").  hipBLASLt
# wins this regime; dispatch is disabled (LIG_SKINNY_GEMM_MAX_M to
# re-enable for experiments), kernel + microbench kept as evidence.
SKINNY_GEMM_MAX_M = int(os.environ.get("LIG_SKINNY_GEMM_MAX_M", "0"))


def linear(x: torch.Tensor, weight: torch.Tensor) -> torch.Tensor:
    """y = x @ weight.T (nn.Linear, no bias).  Dispatch: custom MFMA kernel
    for decode-sized M on GPU; library GEMM otherwise."""
    if (
        x.is_cuda
        and x.dtype == torch.bfloat16
        and 0 < x.shape[0] <= SKINNY_GEMM_MAX_M
        and weight.shape[1] % 32 == 0
        and weight.shape[0] % 64 == 0
        and _use_native(x)
    ):
        out = torch.empty(
            x.shape[0], weight.shape[0], dtype=x.dtype, device=x.device
        )
        _EXT.skinny_gemm(out, x.contiguous(), weight, _stream())
        return out
    return torch.nn.functional.linear(x, weight)
