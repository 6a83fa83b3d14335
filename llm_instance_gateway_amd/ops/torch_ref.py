"""Pure-PyTorch reference implementations of the engine ops.

These are (a) the CPU execution path for tests/simulation, and (b) the fp32
numerics references the HIP/CDNA4 kernels are validated against (SURVEY.md §4:
"GPU kernel unit tests vs PyTorch-ROCm reference outputs").  All math is done
in fp32 and cast back to the input dtype.

Shapes convention (flat token batch, no padding):
  x:            [T, hidden]
  q:            [T, num_heads * head_dim]
  k, v:         [T, num_kv_heads * head_dim]
  k_cache/v_cache: [num_blocks, num_kv_heads, block_size, head_dim]
  slot_mapping: [T] int64, slot = block_id * block_size + offset
  block_tables: [num_seqs, max_blocks] int32
"""

from __future__ import annotations

import math
from typing import Optional, Tuple

import torch


def rms_norm(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    xf = x.float()
    var = xf.pow(2).mean(dim=-1, keepdim=True)
    y = xf * torch.rsqrt(var + eps) * weight.float()
    return y.to(x.dtype)


def fused_add_rms_norm(
    x: torch.Tensor, residual: torch.Tensor, weight: torch.Tensor, eps: float
) -> Tuple[torch.Tensor, torch.Tensor]:
    """y = rmsnorm(x + residual); returns (y, x + residual)."""
    s = (x.float() + residual.float())
    var = s.pow(2).mean(dim=-1, keepdim=True)
    y = s * torch.rsqrt(var + eps) * weight.float()
    return y.to(x.dtype), s.to(x.dtype)


def build_cos_sin_cache(
    head_dim: int, max_positions: int, theta: float, device, dtype=torch.float32
) -> torch.Tensor:
    """[max_positions, head_dim] — first half cos, second half sin, matching
    GPT-NeoX-style (Llama) half-rotation RoPE.  Precomputed on host: on-device
    sinf/cosf per element turns a memory-bound op VALU-bound (guide App. B)."""
    inv_freq = 1.0 / (
        theta ** (torch.arange(0, head_dim, 2, device=device, dtype=torch.float64) / head_dim)
    )
    t = torch.arange(max_positions, device=device, dtype=torch.float64)
    freqs = torch.outer(t, inv_freq)  # [P, head_dim/2]
    cache = torch.cat([freqs.cos(), freqs.sin()], dim=-1).to(dtype)
    return cache  # [P, head_dim]


def rotary_embedding(
    positions: torch.Tensor,  # [T] int64
    q: torch.Tensor,  # [T, num_heads * head_dim]
    k: torch.Tensor,  # [T, num_kv_heads * head_dim]
    cos_sin_cache: torch.Tensor,  # [P, head_dim]
    head_dim: int,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Llama (NeoX-style half-rotation) RoPE applied to q and k."""

    def rot(x: torch.Tensor) -> torch.Tensor:
        T = x.shape[0]
        xh = x.float().view(T, -1, head_dim)
        half = head_dim // 2
        cs = cos_sin_cache[positions].float()  # [T, head_dim]
        cos = cs[:, :half].unsqueeze(1)  # [T, 1, half]
        sin = cs[:, half:].unsqueeze(1)
        x1, x2 = xh[..., :half], xh[..., half:]
        o1 = x1 * cos - x2 * sin
        o2 = x2 * cos + x1 * sin
        return torch.cat([o1, o2], dim=-1).view(T, -1).to(x.dtype)

    return rot(q), rot(k)


def reshape_and_cache(
    k: torch.Tensor,
    v: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    slot_mapping: torch.Tensor,
) -> None:
    """Scatter new K/V rows into the paged cache (in-place)."""
    num_blocks, kvh, bs, hd = k_cache.shape
    T = k.shape[0]
    kf = k.view(T, kvh, hd)
    vf = v.view(T, kvh, hd)
    blk = torch.div(slot_mapping, bs, rounding_mode="floor")
    off = slot_mapping % bs
    k_cache[blk, :, off, :] = kf.to(k_cache.dtype)
    v_cache[blk, :, off, :] = vf.to(v_cache.dtype)


def _gather_kv(
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    block_table: torch.Tensor,  # [max_blocks] int32
    context_len: int,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Rebuild contiguous [L, kv_heads, head_dim] K/V for one sequence."""
    nb, kvh, bs, hd = k_cache.shape
    nblk = (context_len + bs - 1) // bs
    ids = block_table[:nblk].long()
    k = k_cache[ids].permute(0, 2, 1, 3).reshape(nblk * bs, kvh, hd)[:context_len]
    v = v_cache[ids].permute(0, 2, 1, 3).reshape(nblk * bs, kvh, hd)[:context_len]
    return k, v


def _sdpa(
    q: torch.Tensor,  # [Tq, H, D]
    k: torch.Tensor,  # [L, KVH, D]
    v: torch.Tensor,
    scale: float,
    causal_offset: Optional[int] = None,
) -> torch.Tensor:
    """fp32 attention; if causal_offset is given, query row i attends to
    keys [0, causal_offset + i]."""
    Tq, H, D = q.shape
    L, KVH, _ = k.shape
    group = H // KVH
    qf = q.float().permute(1, 0, 2)  # [H, Tq, D]
    kf = k.float().repeat_interleave(group, dim=1).permute(1, 0, 2)  # [H, L, D]
    vf = v.float().repeat_interleave(group, dim=1).permute(1, 0, 2)
    scores = torch.bmm(qf, kf.transpose(1, 2)) * scale  # [H, Tq, L]
    if causal_offset is not None:
        pos = torch.arange(Tq, device=q.device).unsqueeze(1) + causal_offset
        keyp = torch.arange(L, device=q.device).unsqueeze(0)
        mask = keyp > pos  # [Tq, L]
        scores.masked_fill_(mask.unsqueeze(0), float("-inf"))
    p = torch.softmax(scores, dim=-1)
    out = torch.bmm(p, vf)  # [H, Tq, D]
    return out.permute(1, 0, 2).to(q.dtype)


def paged_attention_decode(
    q: torch.Tensor,  # [N, num_heads, head_dim]
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,  # [N, max_blocks]
    context_lens: torch.Tensor,  # [N]
    scale: float,
) -> torch.Tensor:
    N, H, D = q.shape
    out = torch.empty_like(q)
    for i in range(N):
        L = int(context_lens[i])
        k, v = _gather_kv(k_cache, v_cache, block_tables[i], L)
        out[i] = _sdpa(q[i : i + 1], k, v, scale)[0]
    return out


def paged_attention_prefill(
    q: torch.Tensor,  # [T, num_heads, head_dim] — all prefill-chunk tokens, flat
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,  # [S, max_blocks]
    query_start_locs: torch.Tensor,  # [S+1] cu lens of chunks
    context_lens: torch.Tensor,  # [S] ctx len AFTER this chunk's KV was cached
    scale: float,
) -> torch.Tensor:
    """Chunked-prefill attention: the chunk's own K/V are already in the cache
    (reshape_and_cache runs first), so query row j of sequence s (global
    position ctx - n + j) attends to cache positions [0, ctx - n + j]."""
    out = torch.empty_like(q)
    S = block_tables.shape[0]
    for s in range(S):
        lo, hi = int(query_start_locs[s]), int(query_start_locs[s + 1])
        n = hi - lo
        L = int(context_lens[s])
        k, v = _gather_kv(k_cache, v_cache, block_tables[s], L)
        out[lo:hi] = _sdpa(q[lo:hi], k, v, scale, causal_offset=L - n)
    return out


def silu_and_mul(x: torch.Tensor) -> torch.Tensor:
    d = x.shape[-1] // 2
    a = x[..., :d].float()
    b = x[..., d:].float()
    return (a * torch.sigmoid(a) * b).to(x.dtype)


def sgmv_shrink(
    x: torch.Tensor,  # [T, H]
    a_stack: torch.Tensor,  # [S, R, H]
    slots: torch.Tensor,  # [T] int32, -1 = no adapter
) -> torch.Tensor:
    """u[t] = A[slot_t] @ x[t]; zero rows for slot -1."""
    T = x.shape[0]
    R = a_stack.shape[1]
    u = torch.zeros(T, R, dtype=torch.float32, device=x.device)
    for t in range(T):
        s = int(slots[t])
        if s >= 0:
            u[t] = a_stack[s].float() @ x[t].float()
    return u.to(x.dtype)


def sgmv_expand(
    u: torch.Tensor,  # [T, R]
    b_stack: torch.Tensor,  # [S, O, R]
    slots: torch.Tensor,  # [T]
    y: torch.Tensor,  # [T, O] accumulated in-place
    scaling: float = 1.0,
) -> None:
    for t in range(u.shape[0]):
        s = int(slots[t])
        if s >= 0:
            y[t] += (b_stack[s].float() @ u[t].float() * scaling).to(y.dtype)
