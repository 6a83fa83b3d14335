// SGMV (segmented grouped matrix-vector) kernels for batched multi-LoRA,
// bf16, gfx950.
//
// The batched multi-LoRA capability the reference routes around but delegates
// to vLLM's punica kernels (SURVEY.md §2.3 "Batched multi-LoRA serving"):
// every token in a mixed batch may use a different adapter; slot_per_token
// maps token -> adapter slot (-1 = base model only).
//
//   shrink:  u[t, :R]  = A[slot_t] (R x H)  · x[t]      (R = lora rank, <=64)
//   expand:  y[t, :O] += s * B[slot_t] (O x R) · u[t]
//
// Adapter working sets are small (32 rank-16 adapters' A for one module =
// 4 MiB) so both kernels run out of L2 — the binding constraint is LATENCY,
// not bandwidth.  v1 (one block per token / one y-vector per thread with
// serialized loads) sat at ~6% of effective bandwidth: with T=128 decode
// batches it put only 512 waves on a 256-CU chip with ~2 loads in flight
// each.  v2 maximizes occupancy x memory-level parallelism:
//
// shrink: one WAVE per (token, rank-row) -> T*R waves (2048 at T=128);
//   lanes stride the H dim in u16x8 vectors, 4x unrolled so 8 independent
//   16 B loads are in flight per wave; wave-reduce, lane 0 writes.
// expand: one thread per 8 consecutive outputs of one token; ALL B-row
//   vectors (8 * R/8), u[t] and y are loaded before any FMA so ~20
//   independent loads are outstanding per thread.
#include "common.h"

__global__ void sgmv_shrink_kernel(
    uint16_t* __restrict__ u,        // [T, R] bf16
    const uint16_t* __restrict__ x,  // [T, H]
    const uint16_t* __restrict__ a,  // [S, R, H]
    const int* __restrict__ slots,   // [T]
    const int T, const int R, const int H) {
  // global wave id -> (token, rank row)
  const int wid = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  const int t = wid / R;
  if (t >= T) return;
  const int r = wid - t * R;
  const int s = slots[t];
  if (s < 0) return;
  const int lane = threadIdx.x & 63;
  const uint16_t* xrow = x + (size_t)t * H;
  const uint16_t* arow = a + ((size_t)s * R + r) * H;
  const int chunks = H / 8;  // 8 bf16 per lane-step

  float acc = 0.0f;
#pragma unroll 4
  for (int c = lane; c < chunks; c += 64) {
    f32x8 xv = bf16x8_to_f32x8(*reinterpret_cast<const u16x8*>(xrow + c * 8));
    f32x8 av = bf16x8_to_f32x8(*reinterpret_cast<const u16x8*>(arow + c * 8));
#pragma unroll
    for (int e = 0; e < 8; ++e) acc += xv[e] * av[e];
  }
  acc = wave_reduce_sum(acc);
  if (lane == 0) u[(size_t)t * R + r] = f32_to_bf16(acc);
}

template <int R>
__global__ void sgmv_expand_kernel(
    uint16_t* __restrict__ y,        // [T, O] accumulated in place
    const uint16_t* __restrict__ u,  // [T, R]
    const uint16_t* __restrict__ b,  // [S, O, R]
    const int* __restrict__ slots,   // [T]
    const float scaling,
    const int T, const int O) {
  // thread handles 8 consecutive outputs of one token
  const long total = (long)T * (O / 8);
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < total;
       idx += gridDim.x * (long)blockDim.x) {
    const int oc = idx % (O / 8);
    const long t = idx / (O / 8);
    const int s = slots[t];
    if (s < 0) continue;
    const uint16_t* brow = b + ((size_t)s * O + oc * 8) * R;
    // issue EVERY load before any arithmetic: 8*(R/8) B vectors + u + y
    u16x8 breg[8 * (R / 8)];
#pragma unroll
    for (int e = 0; e < 8; ++e)
#pragma unroll
      for (int rc = 0; rc < R / 8; ++rc)
        breg[e * (R / 8) + rc] = *reinterpret_cast<const u16x8*>(
            brow + (size_t)e * R + rc * 8);
    u16x8 ureg[R / 8];
#pragma unroll
    for (int rc = 0; rc < R / 8; ++rc)
      ureg[rc] =
          *reinterpret_cast<const u16x8*>(u + (size_t)t * R + rc * 8);
    u16x8 yv = *reinterpret_cast<u16x8*>(y + (size_t)t * O + oc * 8);

    float uv[R];
#pragma unroll
    for (int rc = 0; rc < R / 8; ++rc) {
      f32x8 v = bf16x8_to_f32x8(ureg[rc]);
#pragma unroll
      for (int e = 0; e < 8; ++e) uv[rc * 8 + e] = v[e];
    }
    f32x8 acc;
#pragma unroll
    for (int e = 0; e < 8; ++e) acc[e] = bf16_to_f32(yv[e]);
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      float a = 0.0f;
#pragma unroll
      for (int rc = 0; rc < R / 8; ++rc) {
        f32x8 bv = bf16x8_to_f32x8(breg[e * (R / 8) + rc]);
#pragma unroll
        for (int q = 0; q < 8; ++q) a += bv[q] * uv[rc * 8 + q];
      }
      acc[e] += a * scaling;
    }
    *reinterpret_cast<u16x8*>(y + (size_t)t * O + oc * 8) = f32x8_to_bf16x8(acc);
  }
}

// ---- large-T (prefill) variants -------------------------------------------
//
// At prefill shapes (T up to 4096) the v2 kernels become BANDWIDTH-bound on
// redundant adapter-weight reads: shrink re-reads A per (token, row) — T*R*H
// traffic; expand re-reads B per token — T*O*R.  Prefill batches are
// sequence-contiguous, so consecutive tokens share an adapter: amortize the
// weight reads across a token group.
//
// shrink_wide: one wave per (8 consecutive tokens, rank row).  Lane = 8
//   token-subgroups x 8 H-stripes; the 8 subgroups read the SAME A chunk in
//   the same cycle (one L1 fetch) -> A traffic / 8.
// expand_wide: one thread per (4 consecutive tokens, 8 outputs).  B rows are
//   loaded once per thread and reused for 4 tokens (per-token reload only on
//   a slot boundary) -> B traffic / 4; y RMW traffic is the floor.

__global__ void sgmv_shrink_wide_kernel(
    uint16_t* __restrict__ u, const uint16_t* __restrict__ x,
    const uint16_t* __restrict__ a, const int* __restrict__ slots,
    const int T, const int R, const int H) {
  // wave handles tokens [t8*8, t8*8+8) x rank row r
  const int wid = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  const int nt8 = (T + 7) / 8;
  const int t8 = wid / R;
  if (t8 >= nt8) return;
  const int r = wid - t8 * R;
  const int lane = threadIdx.x & 63;
  const int tg = lane >> 3;   // token subgroup 0..7
  const int li8 = lane & 7;   // H stripe 0..7
  const int t = t8 * 8 + tg;
  const int tc = t < T ? t : T - 1;  // clamp reads; store is guarded
  const int s = slots[tc];
  const int chunks = H / 8;

  float acc = 0.0f;
  if (s >= 0) {
    const uint16_t* xrow = x + (size_t)tc * H;
    const uint16_t* arow = a + ((size_t)s * R + r) * H;
#pragma unroll 4
    for (int c = li8; c < chunks; c += 8) {
      f32x8 xv = bf16x8_to_f32x8(*reinterpret_cast<const u16x8*>(xrow + c * 8));
      f32x8 av = bf16x8_to_f32x8(*reinterpret_cast<const u16x8*>(arow + c * 8));
#pragma unroll
      for (int e = 0; e < 8; ++e) acc += xv[e] * av[e];
    }
  }
  // reduce across the 8 H-stripes of this token subgroup
#pragma unroll
  for (int st = 4; st >= 1; st >>= 1)
    acc += __shfl_xor(acc, st, 64);
  if (li8 == 0 && t < T && s >= 0) u[(size_t)t * R + r] = f32_to_bf16(acc);
}

template <int R>
__global__ void sgmv_expand_wide_kernel(
    uint16_t* __restrict__ y, const uint16_t* __restrict__ u,
    const uint16_t* __restrict__ b, const int* __restrict__ slots,
    const float scaling, const int T, const int O) {
  const int nt4 = (T + 3) / 4;
  const long total = (long)nt4 * (O / 8);
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < total;
       idx += gridDim.x * (long)blockDim.x) {
    const int oc = idx % (O / 8);
    const long t4 = idx / (O / 8);
    const long t0 = t4 * 4;
    int cur_slot = -2;  // force first load
    u16x8 breg[8 * (R / 8)];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const long t = t0 + j;
      if (t >= T) break;
      const int s = slots[t];
      if (s < 0) continue;
      if (s != cur_slot) {
        cur_slot = s;
        const uint16_t* brow = b + ((size_t)s * O + oc * 8) * R;
#pragma unroll
        for (int e = 0; e < 8; ++e)
#pragma unroll
          for (int rc = 0; rc < R / 8; ++rc)
            breg[e * (R / 8) + rc] = *reinterpret_cast<const u16x8*>(
                brow + (size_t)e * R + rc * 8);
      }
      float uv[R];
#pragma unroll
      for (int rc = 0; rc < R / 8; ++rc) {
        f32x8 v = bf16x8_to_f32x8(
            *reinterpret_cast<const u16x8*>(u + (size_t)t * R + rc * 8));
#pragma unroll
        for (int e = 0; e < 8; ++e) uv[rc * 8 + e] = v[e];
      }
      u16x8 yv = *reinterpret_cast<u16x8*>(y + (size_t)t * O + oc * 8);
      f32x8 acc;
#pragma unroll
      for (int e = 0; e < 8; ++e) acc[e] = bf16_to_f32(yv[e]);
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        float aa = 0.0f;
#pragma unroll
        for (int rc = 0; rc < R / 8; ++rc) {
          f32x8 bv = bf16x8_to_f32x8(breg[e * (R / 8) + rc]);
#pragma unroll
          for (int q = 0; q < 8; ++q) aa += bv[q] * uv[rc * 8 + q];
        }
        acc[e] += aa * scaling;
      }
      *reinterpret_cast<u16x8*>(y + (size_t)t * O + oc * 8) =
          f32x8_to_bf16x8(acc);
    }
  }
}

// ---- MFMA segment-tiled kernels (prefill) ---------------------------------
//
// Prefill batches carry 64-token segment tiles (host metadata: same adapter
// slot per tile — engine/model_runner.py).  With the segment known per tile,
// both halves become small MFMA GEMMs that read the adapter matrices ONCE
// per tile instead of per token:
//   shrink: per tile, u[64,R] = x[64,H] @ A_s[R,H]^T — K-loop over H.
//   expand: per (tile, 64-out chunk), y[64,64] += u[64,R] @ B_s[64,R]^T —
//           K = R zero-padded to the mfma's K=32.
// Fragment layout as verified by csrc/mfma_probe.hip (see skinny_gemm.hip):
// mfma_f32_16x16x32_bf16(a,b,c): C[i][j] += sum_k A[i,k]B[j,k]; a: lane li=i;
// b: lane li=j; both k = lg*8..+7; C: row i = lg*4+r, col j = li.

typedef __bf16 sgmv_bf16x8 __attribute__((ext_vector_type(8)));
typedef float sgmv_f32x4 __attribute__((ext_vector_type(4)));

template <int R>
__global__ __launch_bounds__(256) void sgmv_shrink_tiled_kernel(
    uint16_t* __restrict__ u,        // [T, R]
    const uint16_t* __restrict__ x,  // [T, H]
    const uint16_t* __restrict__ a,  // [S, R, H]
    const int* __restrict__ tile_t0, const int* __restrict__ tile_len,
    const int* __restrict__ tile_slot, const int T, const int H) {
  constexpr int NR = (R + 15) / 16;  // 16-rank chunks
  const int tile = blockIdx.x;
  const int t0 = tile_t0[tile];
  const int len = tile_len[tile];
  const int s = tile_slot[tile];
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int li = lane & 15;
  const int lg = lane >> 4;

  const int trow = wave * 16 + li;  // token within tile (a-fragment row)
  const uint16_t* xrow = x + (size_t)(t0 + min(trow, len - 1)) * H;

  sgmv_f32x4 acc[NR];
#pragma unroll
  for (int nr = 0; nr < NR; ++nr) acc[nr] = sgmv_f32x4{0.f, 0.f, 0.f, 0.f};

  const uint16_t* arow[NR];
#pragma unroll
  for (int nr = 0; nr < NR; ++nr) {
    const int r = min(nr * 16 + li, R - 1);  // clamp for R < 16
    arow[nr] = a + ((size_t)s * R + r) * H;
  }

#pragma unroll 4
  for (int k0 = 0; k0 < H; k0 += 32) {
    u16x8 araw = *reinterpret_cast<const u16x8*>(xrow + k0 + lg * 8);
    sgmv_bf16x8 a_frag = *reinterpret_cast<sgmv_bf16x8*>(&araw);
#pragma unroll
    for (int nr = 0; nr < NR; ++nr) {
      u16x8 braw = *reinterpret_cast<const u16x8*>(arow[nr] + k0 + lg * 8);
      sgmv_bf16x8 b_frag = *reinterpret_cast<sgmv_bf16x8*>(&braw);
      acc[nr] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag, b_frag,
                                                        acc[nr], 0, 0, 0);
    }
  }

#pragma unroll
  for (int r4 = 0; r4 < 4; ++r4) {
    const int row = wave * 16 + lg * 4 + r4;  // C row = token within tile
    if (row >= len) continue;
#pragma unroll
    for (int nr = 0; nr < NR; ++nr) {
      const int rk = nr * 16 + li;
      if (rk < R) u[(size_t)(t0 + row) * R + rk] = f32_to_bf16(acc[nr][r4]);
    }
  }
}

template <int R>
__global__ __launch_bounds__(256) void sgmv_expand_tiled_kernel(
    uint16_t* __restrict__ y,        // [T, O], accumulated
    const uint16_t* __restrict__ u,  // [T, R]
    const uint16_t* __restrict__ b,  // [S, O, R]
    const int* __restrict__ tile_t0, const int* __restrict__ tile_len,
    const int* __restrict__ tile_slot, const float scaling, const int T,
    const int O) {
  constexpr int KC = (R + 31) / 32;  // mfma K=32 chunks over the rank dim
  const int o0 = blockIdx.x * 64;
  const int tile = blockIdx.y;
  const int t0 = tile_t0[tile];
  const int len = tile_len[tile];
  const int s = tile_slot[tile];
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int li = lane & 15;
  const int lg = lane >> 4;

  // a fragments: u rows (token = t0 + wave*16 + li), zero-padded past R
  const int trow = wave * 16 + li;
  const uint16_t* urow = u + (size_t)(t0 + min(trow, len - 1)) * R;
  sgmv_bf16x8 a_frag[KC];
#pragma unroll
  for (int kc = 0; kc < KC; ++kc) {
    const int k = kc * 32 + lg * 8;
    u16x8 raw;
    if (k < R) {
      raw = *reinterpret_cast<const u16x8*>(urow + k);
    } else {
#pragma unroll
      for (int e = 0; e < 8; ++e) raw[e] = 0;
    }
    a_frag[kc] = *reinterpret_cast<sgmv_bf16x8*>(&raw);
  }

  sgmv_f32x4 acc[4];  // 4 out chunks of 16
#pragma unroll
  for (int oc = 0; oc < 4; ++oc) {
    acc[oc] = sgmv_f32x4{0.f, 0.f, 0.f, 0.f};
    const uint16_t* brow = b + ((size_t)s * O + o0 + oc * 16 + li) * R;
#pragma unroll
    for (int kc = 0; kc < KC; ++kc) {
      const int k = kc * 32 + lg * 8;
      u16x8 raw;
      if (k < R) {
        raw = *reinterpret_cast<const u16x8*>(brow + k);
      } else {
#pragma unroll
        for (int e = 0; e < 8; ++e) raw[e] = 0;
      }
      sgmv_bf16x8 b_frag = *reinterpret_cast<sgmv_bf16x8*>(&raw);
      acc[oc] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag[kc], b_frag,
                                                        acc[oc], 0, 0, 0);
    }
  }

#pragma unroll
  for (int r4 = 0; r4 < 4; ++r4) {
    const int row = wave * 16 + lg * 4 + r4;
    if (row >= len) continue;
    uint16_t* yp = y + (size_t)(t0 + row) * O + o0;
#pragma unroll
    for (int oc = 0; oc < 4; ++oc) {
      const int col = oc * 16 + li;
      yp[col] = f32_to_bf16(bf16_to_f32(yp[col]) + scaling * acc[oc][r4]);
    }
  }
}

// ---- slot-grouped "unit" kernels (r02) -------------------------------------
//
// r02 steady-state profile (profiles/r02_kernel_stats.md): the four SGMV
// kernels above are 27% of all kernel time, dominated by re-reading adapter
// weights per token (expand: B traffic = T*O*R; at T=256 tokens over 32
// adapters each B is re-read ~8x).  The host sorts the batch's tokens by
// adapter slot into UNITS of <= 32 same-slot tokens (an indirection array,
// static-shaped for the hipGraph path).
//
// v1 units kernels (register-resident B, thread = (unit, 8 outs) with a
// serial token loop) were a MEASURED DEAD END: 31.0 vs 52.6 req/s e2e
// (gpurun bench_units vs bench_nounits) — they traded occupancy for
// locality (decode: 384-512 waves on a 256-CU chip) and the adapter set is
// L2-resident anyway, so the binding constraint was LATENCY, exactly what
// the v2 kernels' design note says.  Kept below for the record.
//
// expand_lds (the shipping units path) keeps FULL per-token parallelism and
// amortizes B through LDS: workgroup = (unit, 256-output chunk); the B tile
// (256 rows x R, a CONTIGUOUS 256*R-element span of the row-major stack) is
// cooperatively staged in LDS once, then 8-token x 32-output-chunk threads
// sweep the unit's tokens in passes of 8, each doing one y RMW against the
// LDS tile.  B traffic drops T/cap-fold with the same thread count as the
// per-token kernel.

template <int R>
__global__ __launch_bounds__(256) void sgmv_expand_lds_kernel(
    uint16_t* __restrict__ y, const uint16_t* __restrict__ u,
    const uint16_t* __restrict__ b, const int* __restrict__ order,
    const int* __restrict__ unit_t0, const int* __restrict__ unit_len,
    const int* __restrict__ unit_slot, const float scaling, const int T,
    const int O) {
  const int unit = blockIdx.y;
  const int len = unit_len[unit];
  const int s = unit_slot[unit];
  if (len <= 0 || s < 0) return;
  const int t0 = unit_t0[unit];
  const int o0 = blockIdx.x * 256;
  const int tid = threadIdx.x;

  // Stage B rows [o0, o0+256) x R TRANSPOSED: bs[r][o].  The row-major
  // [o][r] layout put thread oc's inner-loop reads at byte offset oc*2R*8 —
  // a multiple of 256 B for R=16, i.e. every lane of a half-wave on the
  // SAME LDS bank (32-way conflict on every B read; the kernel sat at
  // ~1 TB/s, 5x over its traffic floor).  Transposed, lane oc reads the
  // 16 B span bs[r*256 + oc*8] — consecutive lanes, consecutive banks,
  // conflict-free.  Staging: thread tid transposes output row o0+tid with
  // b16 writes of stride 256 elements (2-way write conflicts, once per WG).
  __shared__ uint16_t bs[R * 256];
  {
    const uint16_t* brow = b + ((size_t)s * O + o0 + tid) * R;
#pragma unroll
    for (int rc = 0; rc < R / 8; ++rc) {
      u16x8 v = *reinterpret_cast<const u16x8*>(brow + rc * 8);
#pragma unroll
      for (int e = 0; e < 8; ++e) bs[(rc * 8 + e) * 256 + tid] = v[e];
    }
  }
  __syncthreads();

  const int jt = tid >> 5;   // token lane 0..7 within a pass
  const int oc = tid & 31;   // 8-output chunk 0..31 within the 256 block
  for (int j0 = 0; j0 < len; j0 += 8) {
    const int j = j0 + jt;
    if (j >= len) break;
    const int t = order[t0 + j];
    float uv[R];
#pragma unroll
    for (int rc = 0; rc < R / 8; ++rc) {
      f32x8 v = bf16x8_to_f32x8(
          *reinterpret_cast<const u16x8*>(u + (size_t)t * R + rc * 8));
#pragma unroll
      for (int e = 0; e < 8; ++e) uv[rc * 8 + e] = v[e];
    }
    uint16_t* yp = y + (size_t)t * O + o0 + oc * 8;
    u16x8 yv = *reinterpret_cast<u16x8*>(yp);
    f32x8 acc;
#pragma unroll
    for (int e = 0; e < 8; ++e) acc[e] = 0.0f;
#pragma unroll
    for (int r = 0; r < R; ++r) {
      f32x8 bv = bf16x8_to_f32x8(
          *reinterpret_cast<const u16x8*>(bs + (size_t)r * 256 + oc * 8));
      const float us = uv[r];
#pragma unroll
      for (int e = 0; e < 8; ++e) acc[e] += bv[e] * us;
    }
    f32x8 out;
#pragma unroll
    for (int e = 0; e < 8; ++e) out[e] = bf16_to_f32(yv[e]) + acc[e] * scaling;
    *reinterpret_cast<u16x8*>(yp) = f32x8_to_bf16x8(out);
  }
}

__global__ void sgmv_shrink_units_kernel(
    uint16_t* __restrict__ u, const uint16_t* __restrict__ x,
    const uint16_t* __restrict__ a, const int* __restrict__ order,
    const int* __restrict__ unit_t0, const int* __restrict__ unit_len,
    const int* __restrict__ unit_slot, const int U, const int R,
    const int H) {
  const int wid = blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  const int unit = wid / R;
  if (unit >= U) return;
  const int len = unit_len[unit];
  const int s = unit_slot[unit];
  if (len <= 0 || s < 0) return;
  const int r = wid - unit * R;
  const int t0 = unit_t0[unit];
  const int lane = threadIdx.x & 63;
  const int tg = lane >> 3;   // token subgroup 0..7
  const int li8 = lane & 7;   // H stripe 0..7
  const int chunks = H / 8;
  const uint16_t* arow = a + ((size_t)s * R + r) * H;

  for (int j0 = 0; j0 < len; j0 += 8) {
    const int jj = j0 + tg;
    const int t = jj < len ? order[t0 + jj] : -1;
    float acc = 0.0f;
    if (t >= 0) {
      const uint16_t* xrow = x + (size_t)t * H;
#pragma unroll 4
      for (int c = li8; c < chunks; c += 8) {
        f32x8 xv =
            bf16x8_to_f32x8(*reinterpret_cast<const u16x8*>(xrow + c * 8));
        f32x8 av =
            bf16x8_to_f32x8(*reinterpret_cast<const u16x8*>(arow + c * 8));
#pragma unroll
        for (int e = 0; e < 8; ++e) acc += xv[e] * av[e];
      }
    }
#pragma unroll
    for (int st = 4; st >= 1; st >>= 1) acc += __shfl_xor(acc, st, 64);
    if (li8 == 0 && t >= 0) u[(size_t)t * R + r] = f32_to_bf16(acc);
  }
}

template <int R>
__global__ void sgmv_expand_units_kernel(
    uint16_t* __restrict__ y, const uint16_t* __restrict__ u,
    const uint16_t* __restrict__ b, const int* __restrict__ order,
    const int* __restrict__ unit_t0, const int* __restrict__ unit_len,
    const int* __restrict__ unit_slot, const float scaling, const int U,
    const int O) {
  const long total = (long)U * (O / 8);
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < total;
       idx += gridDim.x * (long)blockDim.x) {
    const int oc = idx % (O / 8);
    const int unit = idx / (O / 8);
    const int len = unit_len[unit];
    const int s = unit_slot[unit];
    if (len <= 0 || s < 0) continue;
    const int t0 = unit_t0[unit];
    // B rows for these 8 outputs: loaded ONCE for the whole unit
    const uint16_t* brow = b + ((size_t)s * O + oc * 8) * R;
    u16x8 breg[8 * (R / 8)];
#pragma unroll
    for (int e = 0; e < 8; ++e)
#pragma unroll
      for (int rc = 0; rc < R / 8; ++rc)
        breg[e * (R / 8) + rc] =
            *reinterpret_cast<const u16x8*>(brow + (size_t)e * R + rc * 8);
    for (int j = 0; j < len; ++j) {
      const int t = order[t0 + j];
      float uv[R];
#pragma unroll
      for (int rc = 0; rc < R / 8; ++rc) {
        f32x8 v = bf16x8_to_f32x8(
            *reinterpret_cast<const u16x8*>(u + (size_t)t * R + rc * 8));
#pragma unroll
        for (int e = 0; e < 8; ++e) uv[rc * 8 + e] = v[e];
      }
      u16x8 yv = *reinterpret_cast<u16x8*>(y + (size_t)t * O + oc * 8);
      f32x8 acc;
#pragma unroll
      for (int e = 0; e < 8; ++e) acc[e] = bf16_to_f32(yv[e]);
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        float aa = 0.0f;
#pragma unroll
        for (int rc = 0; rc < R / 8; ++rc) {
          f32x8 bv = bf16x8_to_f32x8(breg[e * (R / 8) + rc]);
#pragma unroll
          for (int q = 0; q < 8; ++q) aa += bv[q] * uv[rc * 8 + q];
        }
        acc[e] += aa * scaling;
      }
      *reinterpret_cast<u16x8*>(y + (size_t)t * O + oc * 8) =
          f32x8_to_bf16x8(acc);
    }
  }
}

// v2 kernels above stay the fallback path (no unit metadata: more waves
// per unit work, no serial reload chains); the _wide variants win only on
// PREFILL batches, whose tokens are sequence-contiguous so a token group
// shares one adapter — the caller says which regime this is (`wide`).
// Measured (profiles/r01_c256_breakdown.md): wide kernels on a 256-token
// DECODE batch are 3-8x slower than v2 (per-token B reloads, half the waves).

extern "C" {

void lig_sgmv_shrink_launch(uint16_t* u, const uint16_t* x, const uint16_t* a,
                            const int* slots, int T, int R, int H, int wide,
                            hipStream_t stream) {
  dim3 block(256);
  if (wide) {
    const long waves = (long)((T + 7) / 8) * R;
    long grid = (waves + 3) / 4;
    if (grid > 1048576) grid = 1048576;
    hipLaunchKernelGGL(sgmv_shrink_wide_kernel, dim3((int)grid), block, 0,
                       stream, u, x, a, slots, T, R, H);
    return;
  }
  // one wave per (token, rank-row)
  const long waves = (long)T * R;
  long grid = (waves + 3) / 4;
  if (grid > 1048576) grid = 1048576;
  hipLaunchKernelGGL(sgmv_shrink_kernel, dim3((int)grid), block, 0, stream, u,
                     x, a, slots, T, R, H);
}

void lig_sgmv_expand_launch(uint16_t* y, const uint16_t* u, const uint16_t* b,
                            const int* slots, float scaling, int T, int R,
                            int O, int wide, hipStream_t stream) {
  const long total = wide ? (long)((T + 3) / 4) * (O / 8) : (long)T * (O / 8);
  int grid = (int)((total + 255) / 256);
  if (grid > 65535) grid = 65535;
  if (grid < 1) grid = 1;
#define LAUNCH_E(RR)                                                        \
  do {                                                                      \
    if (wide)                                                               \
      hipLaunchKernelGGL((sgmv_expand_wide_kernel<RR>), dim3(grid),         \
                         dim3(256), 0, stream, y, u, b, slots, scaling, T,  \
                         O);                                                \
    else                                                                    \
      hipLaunchKernelGGL((sgmv_expand_kernel<RR>), dim3(grid), dim3(256),   \
                         0, stream, y, u, b, slots, scaling, T, O);         \
  } while (0)
  switch (R) {
    case 8: LAUNCH_E(8); break;
    case 16: LAUNCH_E(16); break;
    case 32: LAUNCH_E(32); break;
    case 64: LAUNCH_E(64); break;
    default: break;  // caller validates R % 8 == 0 && R <= 64
  }
#undef LAUNCH_E
}

void lig_sgmv_expand_lds_launch(uint16_t* y, const uint16_t* u,
                                const uint16_t* b, const int* order,
                                const int* unit_t0, const int* unit_len,
                                const int* unit_slot, float scaling, int U,
                                int T, int R, int O, hipStream_t stream) {
  dim3 grid(O / 256, U);
#define LAUNCH_EL(RR)                                                        \
  hipLaunchKernelGGL((sgmv_expand_lds_kernel<RR>), grid, dim3(256), 0,       \
                     stream, y, u, b, order, unit_t0, unit_len, unit_slot,   \
                     scaling, T, O)
  switch (R) {
    case 8: LAUNCH_EL(8); break;
    case 16: LAUNCH_EL(16); break;
    case 32: LAUNCH_EL(32); break;
    case 64: LAUNCH_EL(64); break;
    default: break;
  }
#undef LAUNCH_EL
}

void lig_sgmv_shrink_units_launch(uint16_t* u, const uint16_t* x,
                                  const uint16_t* a, const int* order,
                                  const int* unit_t0, const int* unit_len,
                                  const int* unit_slot, int U, int R, int H,
                                  hipStream_t stream) {
  const long waves = (long)U * R;
  long grid = (waves + 3) / 4;  // 4 waves per 256-thread block
  if (grid > 1048576) grid = 1048576;
  if (grid < 1) grid = 1;
  hipLaunchKernelGGL(sgmv_shrink_units_kernel, dim3((int)grid), dim3(256), 0,
                     stream, u, x, a, order, unit_t0, unit_len, unit_slot, U,
                     R, H);
}

void lig_sgmv_expand_units_launch(uint16_t* y, const uint16_t* u,
                                  const uint16_t* b, const int* order,
                                  const int* unit_t0, const int* unit_len,
                                  const int* unit_slot, float scaling, int U,
                                  int R, int O, hipStream_t stream) {
  const long total = (long)U * (O / 8);
  int grid = (int)((total + 255) / 256);
  if (grid > 65535) grid = 65535;
  if (grid < 1) grid = 1;
#define LAUNCH_EU(RR)                                                       \
  hipLaunchKernelGGL((sgmv_expand_units_kernel<RR>), dim3(grid), dim3(256), \
                     0, stream, y, u, b, order, unit_t0, unit_len,          \
                     unit_slot, scaling, U, O)
  switch (R) {
    case 8: LAUNCH_EU(8); break;
    case 16: LAUNCH_EU(16); break;
    case 32: LAUNCH_EU(32); break;
    case 64: LAUNCH_EU(64); break;
    default: break;
  }
#undef LAUNCH_EU
}

void lig_sgmv_shrink_tiled_launch(uint16_t* u, const uint16_t* x,
                                  const uint16_t* a, const int* tile_t0,
                                  const int* tile_len, const int* tile_slot,
                                  int ntiles, int T, int R, int H,
                                  hipStream_t stream) {
#define LAUNCH_ST(RR)                                                       \
  hipLaunchKernelGGL((sgmv_shrink_tiled_kernel<RR>), dim3(ntiles),          \
                     dim3(256), 0, stream, u, x, a, tile_t0, tile_len,      \
                     tile_slot, T, H)
  switch (R) {
    case 8: LAUNCH_ST(8); break;
    case 16: LAUNCH_ST(16); break;
    case 32: LAUNCH_ST(32); break;
    case 64: LAUNCH_ST(64); break;
    default: break;
  }
#undef LAUNCH_ST
}

void lig_sgmv_expand_tiled_launch(uint16_t* y, const uint16_t* u,
                                  const uint16_t* b, const int* tile_t0,
                                  const int* tile_len, const int* tile_slot,
                                  int ntiles, float scaling, int T, int R,
                                  int O, hipStream_t stream) {
#define LAUNCH_ET(RR)                                                       \
  hipLaunchKernelGGL((sgmv_expand_tiled_kernel<RR>), dim3(O / 64, ntiles),  \
                     dim3(256), 0, stream, y, u, b, tile_t0, tile_len,      \
                     tile_slot, scaling, T, O)
  switch (R) {
    case 8: LAUNCH_ET(8); break;
    case 16: LAUNCH_ET(16); break;
    case 32: LAUNCH_ET(32); break;
    case 64: LAUNCH_ET(64); break;
    default: break;
  }
#undef LAUNCH_ET
}

}  // extern "C"
