// Paged-attention PREFILL kernel — flash-style MFMA tiles over the paged KV
// cache, bf16, gfx950 / CDNA4.
//
// The compute-bound half of serving (TTFT): per chunk of prompt tokens the
// score/PV work is GEMM-shaped and belongs on MFMA (guide App. B "Fused
// attention prefill", §3).  v1 structure (correctness-first, single-buffered):
//
//   grid = (q_tiles, num_q_heads); workgroup = 4 waves.
//   A workgroup owns a 64-row query tile of ONE (sequence, q-head):
//   wave w computes rows [w*16, w*16+16) with mfma_f32_16x16x32_bf16.
//
//   Per 32-key KV tile:
//     - cooperative staging: K rows gathered from the paged cache into
//       K_lds[32][128]; V gathered TRANSPOSED into VT_lds[128][32] so the PV
//       MFMA's B-fragment reads are contiguous 16 B ds_reads;
//     - swapped QK^T (S^T = K·Q^T): the same per-lane Q registers serve as
//       the MFMA B operand, and each lane ends holding 8 scores for ONE
//       query row (col = lane&15) — the row-softmax reduce is 8 in-lane ops
//       + 2 xor-shuffles (16, 32), no LDS (guide attention ladder);
//     - causal + chunk-edge masking, online softmax with running (m, l);
//     - P goes through a per-wave LDS slab to re-shape from MFMA C-layout
//       into the A-fragment layout for PV (v1 of the guide's
//       cvt_pk+permlane32_swap in-register exchange);
//     - PV: acc_o[db] = mfma(P_frag, VT_frag(db), acc_o[db]), 8 dim-blocks.
//
// Fragment layouts verified on hardware by csrc/mfma_probe.hip.
// Capability parity: flash prefill the reference delegates to vLLM
// (SURVEY.md §2.3).

#include "common.h"

typedef __bf16 bf16x8_t __attribute__((ext_vector_type(8)));
typedef float f32x4_t __attribute__((ext_vector_type(4)));

#define KVBLK 32
#define TQ 64  // query rows per workgroup (4 waves x 16)

template <int HEAD_DIM>
__global__ __launch_bounds__(256) void paged_attention_prefill_kernel(
    uint16_t* __restrict__ out,            // [Tp, Hq, D]
    const uint16_t* __restrict__ q,        // [Tp, Hq, D]
    const uint16_t* __restrict__ k_cache,  // [B, Hkv, BS, D]
    const uint16_t* __restrict__ v_cache,
    const int* __restrict__ tile_seq,         // [ntiles]
    const int* __restrict__ tile_qstart,      // [ntiles]
    const int* __restrict__ query_start_locs, // [S+1]
    const int* __restrict__ context_lens,     // [S]
    const int* __restrict__ block_tables,     // [S, max_blocks]
    const float scale,
    const int num_q_heads,
    const int num_kv_heads,
    const int block_size,
    const int max_blocks) {
  constexpr int D = HEAD_DIM;
  constexpr int DB = D / 16;  // output dim blocks
  constexpr int KB = D / 32;  // MFMA K-steps over head_dim

  struct Shared {
    uint16_t K[KVBLK][D];
    uint16_t VT[D][KVBLK];
    uint16_t P[4][16][KVBLK];
  };
  __shared__ Shared sm;

  const int tile = blockIdx.x;
  const int h = blockIdx.y;
  const int s = tile_seq[tile];
  const int qt = tile_qstart[tile];
  const int group = num_q_heads / num_kv_heads;
  const int kvh = h / group;

  const int lo = query_start_locs[s];
  const int n = query_start_locs[s + 1] - lo;
  const int ctx = context_lens[s];
  const int* bt = block_tables + (size_t)s * max_blocks;

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int li = lane & 15;        // MFMA i/j index
  const int lg = lane >> 4;        // MFMA k-group index (0..3)

  // this lane's softmax query row (chunk-local) — col index of S^T
  const int qrow_sm = qt + wave * 16 + li;
  const bool row_valid = qrow_sm < n;
  const int qpos = ctx - n + qrow_sm;  // global key position of this row

  // Q fragments (also the B operand of the swapped QK^T)
  bf16x8_t q_frag[KB];
  {
    const int qr = row_valid ? qrow_sm : 0;
    const uint16_t* qp = q + ((size_t)(lo + qr) * num_q_heads + h) * D;
#pragma unroll
    for (int kb = 0; kb < KB; ++kb) {
      u16x8 raw = *reinterpret_cast<const u16x8*>(qp + kb * 32 + lg * 8);
      q_frag[kb] = *reinterpret_cast<bf16x8_t*>(&raw);
    }
  }

  float m = -INFINITY;
  float l = 0.0f;
  f32x4_t acc_o[DB];
#pragma unroll
  for (int db = 0; db < DB; ++db) acc_o[db] = f32x4_t{0.f, 0.f, 0.f, 0.f};

  // keys this tile's rows may attend to
  const int kv_end_tile = min(ctx, ctx - n + qt + TQ);

  for (int kv0 = 0; kv0 < kv_end_tile; kv0 += KVBLK) {
    // ---- cooperative staging ------------------------------------------------
    // K: [KVBLK][D] chunks of 8; V: same source, transposed scatter.
    const int chunks = KVBLK * (D / 8);
    for (int c = tid; c < chunks; c += 256) {
      const int row = c / (D / 8);
      const int dchunk = c % (D / 8);
      const int t = kv0 + row;
      u16x8 kval, vval;
      if (t < ctx) {
        const size_t base =
            (((size_t)bt[t / block_size] * num_kv_heads + kvh) * block_size +
             (t % block_size)) * D + dchunk * 8;
        kval = *reinterpret_cast<const u16x8*>(k_cache + base);
        vval = *reinterpret_cast<const u16x8*>(v_cache + base);
      } else {
#pragma unroll
        for (int e = 0; e < 8; ++e) { kval[e] = 0; vval[e] = 0; }
      }
      *reinterpret_cast<u16x8*>(&sm.K[row][dchunk * 8]) = kval;
#pragma unroll
      for (int e = 0; e < 8; ++e) sm.VT[dchunk * 8 + e][row] = vval[e];
    }
    __syncthreads();

    // ---- S^T = K · Q^T (per wave) ------------------------------------------
    f32x4_t acc_s[2];
#pragma unroll
    for (int kt = 0; kt < 2; ++kt) {
      acc_s[kt] = f32x4_t{0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kb = 0; kb < KB; ++kb) {
        u16x8 raw =
            *reinterpret_cast<const u16x8*>(&sm.K[kt * 16 + li][kb * 32 + lg * 8]);
        bf16x8_t a_frag = *reinterpret_cast<bf16x8_t*>(&raw);
        acc_s[kt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_frag, q_frag[kb], acc_s[kt], 0, 0, 0);
      }
    }

    // ---- mask + online softmax (lane owns row qrow_sm) ---------------------
    float sc[8];
    float rowmax = -INFINITY;
#pragma unroll
    for (int kt = 0; kt < 2; ++kt) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int key = kv0 + kt * 16 + lg * 4 + r;
        float v = acc_s[kt][r] * scale;
        if (!row_valid || key > qpos || key >= ctx) v = -INFINITY;
        sc[kt * 4 + r] = v;
        rowmax = fmaxf(rowmax, v);
      }
    }
    rowmax = stride_reduce_max<16, 4>(rowmax);
    const float m_new = fmaxf(m, rowmax);
    float alpha;
    float psum = 0.0f;
    float p[8];
    if (m_new == -INFINITY) {
      alpha = 1.0f;
#pragma unroll
      for (int i = 0; i < 8; ++i) p[i] = 0.0f;
    } else {
      alpha = (m == -INFINITY) ? 0.0f : __expf(m - m_new);
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        p[i] = (sc[i] == -INFINITY) ? 0.0f : __expf(sc[i] - m_new);
        psum += p[i];
      }
      m = m_new;
    }
    psum = stride_reduce_sum<16, 4>(psum);
    l = l * alpha + psum;
    // acc_o's rows follow the MFMA C layout (row = lg*4+r), while the
    // per-lane softmax stats belong to row li — shuffle each row's alpha
    // from the lane that owns it (alpha is replicated across lg groups, so
    // lane (lg*4+r) holds row (lg*4+r)'s value).
    float alpha_r[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) alpha_r[r] = __shfl(alpha, lg * 4 + r, 64);
#pragma unroll
    for (int db = 0; db < DB; ++db) {
#pragma unroll
      for (int r = 0; r < 4; ++r) acc_o[db][r] *= alpha_r[r];
    }

    // ---- P: C-layout -> A-fragment layout via per-wave LDS slab ------------
#pragma unroll
    for (int kt = 0; kt < 2; ++kt) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        sm.P[wave][li][kt * 16 + lg * 4 + r] = f32_to_bf16(p[kt * 4 + r]);
      }
    }
    // same-wave LDS read-after-write: ordered by lgkmcnt waits the compiler
    // inserts; no workgroup barrier needed (the slab is wave-private).
    u16x8 praw = *reinterpret_cast<const u16x8*>(&sm.P[wave][li][lg * 8]);
    bf16x8_t p_frag = *reinterpret_cast<bf16x8_t*>(&praw);

    // ---- PV: acc_o[db] += P · V --------------------------------------------
#pragma unroll
    for (int db = 0; db < DB; ++db) {
      u16x8 vraw =
          *reinterpret_cast<const u16x8*>(&sm.VT[db * 16 + li][lg * 8]);
      bf16x8_t v_frag = *reinterpret_cast<bf16x8_t*>(&vraw);
      acc_o[db] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(p_frag, v_frag,
                                                          acc_o[db], 0, 0, 0);
    }
    __syncthreads();  // protect K/VT before next tile's staging
  }

  // ---- epilogue -------------------------------------------------------------
  // shuffle each output row's l from its owner lane FIRST (all lanes active —
  // __shfl under divergence would read inactive lanes), then write.
  float inv_l[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const float lr = __shfl(l, lg * 4 + r, 64);
    inv_l[r] = lr > 0.f ? 1.0f / lr : 0.0f;
  }
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int orow = qt + wave * 16 + lg * 4 + r;  // C-layout row = M = qrow
    if (orow >= n) continue;
    uint16_t* op = out + ((size_t)(lo + orow) * num_q_heads + h) * D;
#pragma unroll
    for (int db = 0; db < DB; ++db) {
      op[db * 16 + li] = f32_to_bf16(acc_o[db][r] * inv_l[r]);
    }
  }
}

extern "C" void lig_paged_attention_prefill_launch(
    uint16_t* out, const uint16_t* q, const uint16_t* k_cache,
    const uint16_t* v_cache, const int* tile_seq, const int* tile_qstart,
    const int* query_start_locs, const int* context_lens,
    const int* block_tables, float scale, int ntiles, int num_q_heads,
    int num_kv_heads, int head_dim, int block_size, int max_blocks,
    hipStream_t stream) {
  dim3 grid(ntiles, num_q_heads);
  dim3 block(256);
#define LAUNCH_P(HD)                                                       \
  hipLaunchKernelGGL((paged_attention_prefill_kernel<HD>), grid, block, 0, \
                     stream, out, q, k_cache, v_cache, tile_seq,           \
                     tile_qstart, query_start_locs, context_lens,          \
                     block_tables, scale, num_q_heads, num_kv_heads,       \
                     block_size, max_blocks)
  switch (head_dim) {
    case 64: LAUNCH_P(64); break;
    case 128: LAUNCH_P(128); break;
    default: break;  // caller validates
  }
#undef LAUNCH_P
}
