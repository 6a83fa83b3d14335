// Fused dynamic row-wise fp8 (OCP e4m3) quantization, bf16 -> fp8, gfx950.
//
// The fp8 serving mode's activation quant via torch elementwise ops costs
// ~4 kernel launches and 3 full passes over x per projection GEMM (abs,
// amax-reduce, scale-divide, dtype cast) — measured to outweigh the fp8
// GEMM's weight-traffic saving (models/quant.py MEASURED STATUS).  This
// kernel does the whole thing in ONE launch and 2 read passes (rows are
// L2-hot between passes):
//
//   workgroup per row (256 threads); pass 1: per-thread amax over u16x8
//   chunks -> wave xor-reduce -> cross-wave LDS reduce; pass 2: scale,
//   saturate-convert to e4m3 (__hip_cvt_float_to_fp8, RNE), pack 8 bytes
//   per store.
//
// gfx950 is OCP fp8 (e4m3fn) — NOT the gfx94x fnuz encoding; matches
// torch.float8_e4m3fn bit-for-bit.

#include <hip/hip_fp8.h>

#include "common.h"

__global__ __launch_bounds__(256) void quant_fp8_rowwise_kernel(
    uint8_t* __restrict__ xq,        // [T, K] e4m3 bytes
    float* __restrict__ xs,          // [T] scales (amax/448)
    const uint16_t* __restrict__ x,  // [T, K] bf16
    const int K) {
  const int row = blockIdx.x;
  const uint16_t* xr = x + (size_t)row * K;
  const int tid = threadIdx.x;
  const int chunks = K / 8;

  float amax = 0.0f;
  for (int c = tid; c < chunks; c += 256) {
    f32x8 v = bf16x8_to_f32x8(*reinterpret_cast<const u16x8*>(xr + c * 8));
#pragma unroll
    for (int e = 0; e < 8; ++e) amax = fmaxf(amax, fabsf(v[e]));
  }
#pragma unroll
  for (int st = 1; st < 64; st <<= 1)
    amax = fmaxf(amax, __shfl_xor(amax, st, 64));
  __shared__ float wmax[4];
  if ((tid & 63) == 0) wmax[tid >> 6] = amax;
  __syncthreads();
  amax = fmaxf(fmaxf(wmax[0], wmax[1]), fmaxf(wmax[2], wmax[3]));
  amax = fmaxf(amax, 1e-6f);  // matches models/quant.py's clamp
  const float inv = 448.0f / amax;
  if (tid == 0) xs[row] = amax / 448.0f;

  uint8_t* qr = xq + (size_t)row * K;
  for (int c = tid; c < chunks; c += 256) {
    f32x8 v = bf16x8_to_f32x8(*reinterpret_cast<const u16x8*>(xr + c * 8));
    uint64_t pk = 0;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      const float q = fminf(fmaxf(v[e] * inv, -448.0f), 448.0f);
      const __hip_fp8_storage_t b =
          __hip_cvt_float_to_fp8(q, __HIP_SATFINITE, __HIP_E4M3);
      pk |= (uint64_t)b << (8 * e);
    }
    *reinterpret_cast<uint64_t*>(qr + c * 8) = pk;
  }
}

extern "C" void lig_quant_fp8_rowwise_launch(uint8_t* xq, float* xs,
                                             const uint16_t* x, int T, int K,
                                             hipStream_t stream) {
  if (T == 0) return;
  hipLaunchKernelGGL(quant_fp8_rowwise_kernel, dim3(T), dim3(256), 0, stream,
                     xq, xs, x, K);
}
