// RMSNorm kernels (bf16), gfx950.
//
// Memory-bound elementwise+reduction: one workgroup (256 threads = 4 waves)
// per token row, bf16x8 (16 B) vectorized loads per lane (guide App. B:
// scalar bf16 ~2x slower), f32 accumulation, LDS cross-wave reduce.
//
// fused_add_rms_norm fuses the residual add into the same pass (one read of
// each operand, one write of each result) — the reference's engines delegate
// this to vLLM; here it is owned natively.

#include "common.h"

// One row per workgroup.  hidden must be a multiple of 8.
template <bool FUSED_ADD>
__global__ void rms_norm_kernel(
    uint16_t* __restrict__ out,          // [T, H] bf16
    const uint16_t* __restrict__ input,  // [T, H]
    uint16_t* __restrict__ residual,     // [T, H] in/out (FUSED_ADD only)
    const uint16_t* __restrict__ weight, // [H]
    const float eps,
    const int hidden) {
  const int row = blockIdx.x;
  const int tid = threadIdx.x;
  const int nthreads = blockDim.x;
  const uint16_t* in_row = input + (size_t)row * hidden;
  uint16_t* res_row = FUSED_ADD ? residual + (size_t)row * hidden : nullptr;
  uint16_t* out_row = out + (size_t)row * hidden;

  // Pass 1: sum of squares (and residual add, kept in registers when the row
  // fits: 4096 hidden / 256 threads = 2 x bf16x8 per thread).
  float ss = 0.0f;
  const int chunks = hidden / 8;
  // register stash for up to 4 chunks/thread (hidden <= 8192 at 256 threads)
  f32x8 stash[4];
  int nstash = 0;
  const bool stashable = chunks <= nthreads * 4;

  for (int c = tid; c < chunks; c += nthreads) {
    u16x8 hv = *reinterpret_cast<const u16x8*>(in_row + c * 8);
    f32x8 v = bf16x8_to_f32x8(hv);
    if (FUSED_ADD) {
      u16x8 hr = *reinterpret_cast<const u16x8*>(res_row + c * 8);
      f32x8 r = bf16x8_to_f32x8(hr);
#pragma unroll
      for (int i = 0; i < 8; ++i) v[i] += r[i];
      *reinterpret_cast<u16x8*>(res_row + c * 8) = f32x8_to_bf16x8(v);
    }
#pragma unroll
    for (int i = 0; i < 8; ++i) ss += v[i] * v[i];
    if (stashable && nstash < 4) stash[nstash++] = v;
  }

  // wave reduce + LDS cross-wave reduce
  ss = wave_reduce_sum(ss);
  __shared__ float lds_sum[16];
  const int wave = tid / WAVE_SIZE;
  const int lane = tid % WAVE_SIZE;
  const int nwaves = nthreads / WAVE_SIZE;
  if (lane == 0) lds_sum[wave] = ss;
  __syncthreads();
  float total = 0.0f;
  for (int w = 0; w < nwaves; ++w) total += lds_sum[w];
  const float inv_rms = rsqrtf(total / hidden + eps);

  // Pass 2: scale + weight; rows were stashed in registers (no re-read).
  int si = 0;
  for (int c = tid; c < chunks; c += nthreads) {
    f32x8 v;
    if (stashable) {
      v = stash[si++];
    } else {
      u16x8 hv = *reinterpret_cast<const u16x8*>(
          (FUSED_ADD ? res_row : in_row) + c * 8);
      v = bf16x8_to_f32x8(hv);
    }
    u16x8 hw = *reinterpret_cast<const u16x8*>(weight + c * 8);
    f32x8 w = bf16x8_to_f32x8(hw);
#pragma unroll
    for (int i = 0; i < 8; ++i) v[i] = v[i] * inv_rms * w[i];
    *reinterpret_cast<u16x8*>(out_row + c * 8) = f32x8_to_bf16x8(v);
  }
}

extern "C" {

void lig_rms_norm_launch(uint16_t* out, const uint16_t* input,
                         const uint16_t* weight, float eps, int num_tokens,
                         int hidden, hipStream_t stream) {
  dim3 grid(num_tokens);
  dim3 block(256);
  hipLaunchKernelGGL((rms_norm_kernel<false>), grid, block, 0, stream, out,
                     input, nullptr, weight, eps, hidden);
}

void lig_fused_add_rms_norm_launch(uint16_t* out, const uint16_t* input,
                                   uint16_t* residual, const uint16_t* weight,
                                   float eps, int num_tokens, int hidden,
                                   hipStream_t stream) {
  dim3 grid(num_tokens);
  dim3 block(256);
  hipLaunchKernelGGL((rms_norm_kernel<true>), grid, block, 0, stream, out,
                     input, residual, weight, eps, hidden);
}

}  // extern "C"
