// Paged-attention DECODE kernel (single query token per sequence), bf16,
// gfx950 / CDNA4.
//
// This is the serving hot path's memory-bound half: per decoded token the
// kernel streams the sequence's entire paged KV history once.  Design
// (guide App. B "Attention decode" + §2 coalescing rules):
//
//   grid  = (num_seqs, num_kv_heads); block = GROUP waves (GROUP = Hq/Hkv).
//   Wave w owns q-head (kv_head*GROUP + w).  The GROUP waves of a workgroup
//   run on one CU and stream the SAME K/V lines, so the per-CU vector L1
//   absorbs the GQA re-read (one HBM fetch per line).
//
//   Lane map (64 = 8 tokens x 8 lanes): lane = tok*8 + dchunk; each lane
//   loads head_dim/8 contiguous dims of one token's K row — a token row
//   (head_dim*2 B) is covered by 8 lanes in full 16/32 B chunks, and the 8
//   token rows of one iteration are contiguous within a KV block, so a wave
//   iteration issues fully-coalesced loads of 8 rows.
//
//   Scores: per-lane partial dot -> xor-shuffle reduce over the 8 lanes of a
//   token group (1,2,4) -> online softmax with running (m, l) replicated
//   across the wave (token-group max/sum via xor 8,16,32).  V accumulates
//   into head_dim/8 f32 per lane; one cross-token xor reduce at the end.
//
// Cache layout: [num_blocks, num_kv_heads, block_size, head_dim] bf16.
// Capability parity: the paged-attention the reference delegates to vLLM
// (SURVEY.md §2.3 "Paged-attention prefill + decode").

#include "common.h"

typedef __bf16 bf16x2_t __attribute__((ext_vector_type(2)));

// h-th bf16 pair of a packed u16x8 (folds to subregister access once the
// caller's loop is unrolled)
static __device__ __forceinline__ bf16x2_t bf16pair(const u16x8& v, int h) {
  union {
    u16x8 w;
    bf16x2_t p[4];
  } u;
  u.w = v;
  return u.p[h];
}

// SPLIT=true: flash-decoding — blockIdx.z selects a context slice; the
// un-normalized partial (oacc, m, l) goes to scratch fp32
// [N, Hq, nsplits, D+2] and a combine kernel merges the slices.  Doubles
// wave-level parallelism for long contexts (ROADMAP #1).
// MAX_THREADS: real blocks are GROUP*64 threads (GQA group <= 16 -> <=1024),
// but Llama/Mixtral run GROUP=4 (256 threads); compiling the hot
// instantiation with launch_bounds(256) lifts the 1024-thread VGPR ceiling
// (110 VGPRs -> 4 waves/SIMD at LB=1024) so the scheduler can trade
// registers for occupancy per the real block size.
// MIN_BLOCKS (launch_bounds 2nd arg): workgroups the compiler must fit per
// CU.  The 256-thread (GQA group 4) instantiations pin 5 -> 5 waves/SIMD,
// forcing VGPRs <= 96 (the allocator otherwise settles at 98 = 4 waves;
// the packed-bf16 dot + fused V-convert brought it down from 110).
template <int HEAD_DIM, bool SPLIT = false, int MAX_THREADS = 1024,
          int MIN_BLOCKS = 1>
__global__ __launch_bounds__(MAX_THREADS, MIN_BLOCKS) void
paged_attention_decode_kernel(
    uint16_t* __restrict__ out,            // [N, Hq, D]
    const uint16_t* __restrict__ q,        // [N, Hq, D]
    const uint16_t* __restrict__ k_cache,  // [B, Hkv, BS, D]
    const uint16_t* __restrict__ v_cache,
    const int* __restrict__ block_tables,  // [N, max_blocks]
    const int* __restrict__ context_lens,  // [N]
    const float scale,
    const int num_q_heads,
    const int num_kv_heads,
    const int block_size,
    const int max_blocks,
    float* __restrict__ scratch = nullptr,  // [N, Hq, nsplits, D+2]
    const int nsplits = 1) {
  constexpr int DL = HEAD_DIM / 8;  // dims per lane
  const int seq = blockIdx.x;
  const int kvh = blockIdx.y;
  const int wave = threadIdx.x / WAVE_SIZE;
  const int lane = threadIdx.x % WAVE_SIZE;
  const int group = num_q_heads / num_kv_heads;
  const int qh = kvh * group + wave;

  const int ctx = context_lens[seq];
  const int tok = lane >> 3;     // 0..7: token within the 8-token iteration
  const int dchunk = lane & 7;   // 0..7: which head_dim/8 slice

  // q fragment: PACKED bf16 for DL>=8 (v_dot2c_f32_bf16 consumes bf16
  // pairs directly — halves the q registers vs an f32 copy and removes the
  // per-iteration K conversion buffer; the kernel is VGPR-occupancy-bound:
  // 106 VGPRs = 4 waves/SIMD, <=96 buys a 5th); f32 for the small DL<8
  // instantiations (HEAD_DIM 16, test-only shapes).
  constexpr int NQ = DL >= 8 ? DL / 8 : 1;
  u16x8 qpk[NQ];
  float qreg[DL < 8 ? DL : 1];
  {
    const uint16_t* qp =
        q + ((size_t)seq * num_q_heads + qh) * HEAD_DIM + dchunk * DL;
    if constexpr (DL >= 8) {
#pragma unroll
      for (int i = 0; i < DL / 8; ++i)
        qpk[i] = *reinterpret_cast<const u16x8*>(qp + i * 8);
    } else {
#pragma unroll
      for (int i = 0; i < DL; ++i) qreg[i] = bf16_to_f32(qp[i]);
    }
  }

  float m = -INFINITY;
  float l = 0.0f;
  float oacc[DL];
#pragma unroll
  for (int i = 0; i < DL; ++i) oacc[i] = 0.0f;

  const int* bt = block_tables + (size_t)seq * max_blocks;
  const int iters = (ctx + 7) / 8;
  int it0 = 0, it1 = iters;
  if constexpr (SPLIT) {
    const int span = (iters + nsplits - 1) / nsplits;
    it0 = blockIdx.z * span;
    it1 = min(iters, it0 + span);
  }

  // Software-pipelined K/V: the next iteration's vector loads are issued
  // BEFORE this iteration's dot/softmax/accumulate chain (whose xor-shuffle
  // reduces serialize ~10 cross-lane ops), so HBM latency hides behind
  // compute.  TWO prefetch buffers as SEPARATE NAMED arrays selected by a
  // manually 2-unrolled loop: a runtime-indexed buf[2][NV] spills to scratch
  // (measured 6x slower than no prefetch).  A THIRD stage was a MEASURED
  // DEAD END: +34 VGPRs (110 -> 144) dropped occupancy 4 -> 3 waves/SIMD
  // and decode bandwidth fell 2.6 -> 1.9 TB/s (gpurun bench_attention A/B);
  // occupancy beats pipeline depth here.
  constexpr int NV = DL >= 8 ? DL / 8 : 1;  // u16x8 vectors per lane
  u16x8 kraw0[NV], vraw0[NV], kraw1[NV], vraw1[NV];
  bool valid0 = false, valid1 = false;

#define ISSUE_LOAD(KBUF, VBUF, VALID, IT)                                    \
  do {                                                                       \
    const int t_ = (IT) * 8 + tok;                                           \
    (VALID) = (IT) < it1 && t_ < ctx;                                        \
    if (VALID) {                                                             \
      const int blk_ = bt[t_ / block_size];                                  \
      const size_t base_ = (((size_t)blk_ * num_kv_heads + kvh) *            \
                                block_size + (t_ % block_size)) * HEAD_DIM + \
                           dchunk * DL;                                      \
      if constexpr (DL >= 8) {                                               \
        _Pragma("unroll") for (int i_ = 0; i_ < NV; ++i_) {                  \
          (KBUF)[i_] =                                                       \
              *reinterpret_cast<const u16x8*>(k_cache + base_ + i_ * 8);     \
          (VBUF)[i_] =                                                       \
              *reinterpret_cast<const u16x8*>(v_cache + base_ + i_ * 8);     \
        }                                                                    \
      } else {                                                               \
        _Pragma("unroll") for (int i_ = 0; i_ < DL; ++i_) {                  \
          (KBUF)[0][i_] = k_cache[base_ + i_];                               \
          (VBUF)[0][i_] = v_cache[base_ + i_];                               \
        }                                                                    \
      }                                                                      \
    }                                                                        \
  } while (0)

#define STEP_BODY(KBUF, VBUF, VALID)                                         \
  do {                                                                       \
    const bool valid = (VALID);                                              \
    float s = 0.0f;                                                          \
    if constexpr (DL >= 8) {                                                 \
      /* packed-bf16 dot: no f32 K buffer; an invalid lane's stale K     */ \
      /* registers feed a dot whose result is discarded by the select    */ \
      /* below (s := -INF), so no zeroing pass is needed.                */ \
      _Pragma("unroll") for (int i = 0; i < NQ; ++i) {                       \
        _Pragma("unroll") for (int h = 0; h < 4; ++h) {                      \
          s = __builtin_amdgcn_fdot2_f32_bf16(                               \
              bf16pair(qpk[i], h), bf16pair((KBUF)[i], h), s, false);        \
        }                                                                    \
      }                                                                      \
    } else {                                                                 \
      _Pragma("unroll") for (int i = 0; i < DL; ++i)                         \
          s += qreg[i] * (valid ? bf16_to_f32((KBUF)[0][i]) : 0.0f);         \
    }                                                                        \
    s = group_reduce_sum<8>(s);                                              \
    s = valid ? s * scale : -INFINITY;                                       \
    /* PER-TOKEN-STREAM online softmax: the 8 lanes of this token group    */\
    /* keep their own (m, l, oacc) for the tokens they stream (tok, tok+8, */\
    /* ...).  The old per-iteration cross-token reduces (stride_reduce_max */\
    /* + stride_reduce_sum = 6 serialized shuffles EVERY 8-token chunk)    */\
    /* were the kernel's latency chain; the 8 states merge ONCE after the  */\
    /* K loop instead.                                                     */\
    const float new_m = fmaxf(m, s);                                         \
    float p, alpha;                                                          \
    if (new_m == -INFINITY) {                                                \
      p = 0.0f;                                                              \
      alpha = 1.0f;                                                          \
    } else {                                                                 \
      alpha = (m == -INFINITY) ? 0.0f : __expf(m - new_m);                   \
      p = (s == -INFINITY) ? 0.0f : __expf(s - new_m);                       \
      m = new_m;                                                             \
    }                                                                        \
    l = l * alpha + p;                                                       \
    /* V accumulate with inline conversion (no vf[DL] buffer — VGPRs).   */ \
    /* Invalid lanes force the converted value to 0 so stale V registers */ \
    /* cannot inject NaN through p * v (p is 0 there, but 0*NaN = NaN).  */ \
    if constexpr (DL >= 8) {                                                 \
      _Pragma("unroll") for (int i = 0; i < NV; ++i) {                       \
        f32x8 v8 = bf16x8_to_f32x8((VBUF)[i]);                               \
        _Pragma("unroll") for (int j = 0; j < 8; ++j)                        \
            oacc[i * 8 + j] =                                                \
                oacc[i * 8 + j] * alpha + p * (valid ? v8[j] : 0.0f);        \
      }                                                                      \
    } else {                                                                 \
      _Pragma("unroll") for (int i = 0; i < DL; ++i)                         \
          oacc[i] = oacc[i] * alpha +                                        \
                    p * (valid ? bf16_to_f32((VBUF)[0][i]) : 0.0f);          \
    }                                                                        \
  } while (0)

  ISSUE_LOAD(kraw0, vraw0, valid0, it0);
  for (int it = it0; it < it1; it += 2) {
    ISSUE_LOAD(kraw1, vraw1, valid1, it + 1);
    STEP_BODY(kraw0, vraw0, valid0);
    if (it + 1 >= it1) break;
    ISSUE_LOAD(kraw0, vraw0, valid0, it + 2);
    STEP_BODY(kraw1, vraw1, valid1);
  }
#undef ISSUE_LOAD
#undef STEP_BODY

  // Merge the 8 per-token-stream softmax states (lanes stride-8 apart share
  // dchunk): rescale every stream to the global max, then sum l and oacc.
  {
    const float gm = stride_reduce_max<8, 8>(m);
    const float w = (m == -INFINITY) ? 0.0f : __expf(m - gm);
    l = stride_reduce_sum<8, 8>(l * w);
#pragma unroll
    for (int i = 0; i < DL; ++i) oacc[i] = stride_reduce_sum<8, 8>(oacc[i] * w);
    m = gm;
  }

  if constexpr (SPLIT) {
    // un-normalized partial out; m and l are wave-uniform after the
    // stride reduces above
    if (tok == 0) {
      float* sp = scratch +
          (((size_t)seq * num_q_heads + qh) * nsplits + blockIdx.z) *
              (HEAD_DIM + 2);
#pragma unroll
      for (int i = 0; i < DL; ++i) sp[dchunk * DL + i] = oacc[i];
      if (dchunk == 0) {
        sp[HEAD_DIM] = m;
        sp[HEAD_DIM + 1] = l;
      }
    }
    return;
  }

  if (tok == 0) {
    const float inv_l = (l > 0.0f) ? 1.0f / l : 0.0f;
    uint16_t* op =
        out + ((size_t)seq * num_q_heads + qh) * HEAD_DIM + dchunk * DL;
    if constexpr (DL >= 8) {
#pragma unroll
      for (int i = 0; i < DL; i += 8) {
        f32x8 v8;
#pragma unroll
        for (int j = 0; j < 8; ++j) v8[j] = oacc[i + j] * inv_l;
        *reinterpret_cast<u16x8*>(op + i) = f32x8_to_bf16x8(v8);
      }
    } else {
#pragma unroll
      for (int i = 0; i < DL; ++i) op[i] = f32_to_bf16(oacc[i] * inv_l);
    }
  }
}

template <int HEAD_DIM>
__global__ void paged_attention_decode_combine_kernel(
    uint16_t* __restrict__ out,           // [N, Hq, D]
    const float* __restrict__ scratch,    // [N, Hq, nsplits, D+2]
    const int num_q_heads, const int nsplits) {
  // one wave per (seq, q-head); lane covers D/64 dims
  const int seq = blockIdx.x;
  const int qh = blockIdx.y;
  const int lane = threadIdx.x;
  constexpr int DPL = HEAD_DIM >= 64 ? HEAD_DIM / 64 : 1;
  const float* base =
      scratch + ((size_t)seq * num_q_heads + qh) * (size_t)nsplits *
                    (HEAD_DIM + 2);
  float gm = -INFINITY;
  for (int sI = 0; sI < nsplits; ++sI)
    gm = fmaxf(gm, base[(size_t)sI * (HEAD_DIM + 2) + HEAD_DIM]);
  float gl = 0.0f;
  float acc[DPL];
#pragma unroll
  for (int i = 0; i < DPL; ++i) acc[i] = 0.0f;
  for (int sI = 0; sI < nsplits; ++sI) {
    const float* sp = base + (size_t)sI * (HEAD_DIM + 2);
    const float ms = sp[HEAD_DIM];
    if (ms == -INFINITY) continue;
    const float w = __expf(ms - gm);
    gl += sp[HEAD_DIM + 1] * w;
#pragma unroll
    for (int i = 0; i < DPL; ++i) {
      const int d = lane * DPL + i;
      if (d < HEAD_DIM) acc[i] += sp[d] * w;
    }
  }
  const float inv = gl > 0.0f ? 1.0f / gl : 0.0f;
  uint16_t* op = out + ((size_t)seq * num_q_heads + qh) * HEAD_DIM;
#pragma unroll
  for (int i = 0; i < DPL; ++i) {
    const int d = lane * DPL + i;
    if (d < HEAD_DIM) op[d] = f32_to_bf16(acc[i] * inv);
  }
}

extern "C" void lig_paged_attention_decode_split_launch(
    uint16_t* out, const uint16_t* q, const uint16_t* k_cache,
    const uint16_t* v_cache, const int* block_tables,
    const int* context_lens, float* scratch, float scale, int num_seqs,
    int num_q_heads, int num_kv_heads, int head_dim, int block_size,
    int max_blocks, int nsplits, hipStream_t stream) {
  const int group = num_q_heads / num_kv_heads;
  dim3 grid(num_seqs, num_kv_heads, nsplits);
  dim3 block(group * WAVE_SIZE);
#define LAUNCH_S(HD)                                                        \
  do {                                                                      \
    if (group <= 4)                                                         \
      hipLaunchKernelGGL((paged_attention_decode_kernel<HD, true, 256, 5>),    \
                         grid, block, 0, stream, out, q, k_cache, v_cache,  \
                         block_tables, context_lens, scale, num_q_heads,    \
                         num_kv_heads, block_size, max_blocks, scratch,     \
                         nsplits);                                          \
    else                                                                    \
      hipLaunchKernelGGL((paged_attention_decode_kernel<HD, true>), grid,   \
                         block, 0, stream, out, q, k_cache, v_cache,        \
                         block_tables, context_lens, scale, num_q_heads,    \
                         num_kv_heads, block_size, max_blocks, scratch,     \
                         nsplits);                                          \
    hipLaunchKernelGGL((paged_attention_decode_combine_kernel<HD>),         \
                       dim3(num_seqs, num_q_heads), dim3(WAVE_SIZE), 0,     \
                       stream, out, scratch, num_q_heads, nsplits);         \
  } while (0)
  switch (head_dim) {
    case 16: LAUNCH_S(16); break;
    case 64: LAUNCH_S(64); break;
    case 128: LAUNCH_S(128); break;
    default: break;
  }
#undef LAUNCH_S
}

extern "C" void lig_paged_attention_decode_launch(
    uint16_t* out, const uint16_t* q, const uint16_t* k_cache,
    const uint16_t* v_cache, const int* block_tables,
    const int* context_lens, float scale, int num_seqs, int num_q_heads,
    int num_kv_heads, int head_dim, int block_size, int max_blocks,
    hipStream_t stream) {
  const int group = num_q_heads / num_kv_heads;
  dim3 grid(num_seqs, num_kv_heads);
  dim3 block(group * WAVE_SIZE);
#define LAUNCH(HD)                                                          \
  do {                                                                      \
    if (group <= 4)                                                         \
      hipLaunchKernelGGL((paged_attention_decode_kernel<HD, false, 256, 5>),   \
                         grid, block, 0, stream, out, q, k_cache, v_cache,  \
                         block_tables, context_lens, scale, num_q_heads,    \
                         num_kv_heads, block_size, max_blocks);             \
    else                                                                    \
      hipLaunchKernelGGL((paged_attention_decode_kernel<HD>), grid, block,  \
                         0, stream, out, q, k_cache, v_cache, block_tables, \
                         context_lens, scale, num_q_heads, num_kv_heads,    \
                         block_size, max_blocks);                           \
  } while (0)
  switch (head_dim) {
    case 16: LAUNCH(16); break;
    case 64: LAUNCH(64); break;
    case 128: LAUNCH(128); break;
    default:
      // unsupported head_dim — surfaced as hipErrorInvalidValue by the caller
      break;
  }
#undef LAUNCH
}
