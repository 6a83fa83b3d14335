// SwiGLU activation: out = silu(x[:, :d]) * x[:, d:], bf16, gfx950.
//
// Memory-bound elementwise; bf16x8 (16 B/lane) vectorized grid-stride loop
// (guide App. B).  Fusing the gate/up split here means the preceding
// gate_up GEMM writes one contiguous [T, 2d] buffer and this kernel is the
// only extra HBM pass.

#include "common.h"

__global__ void silu_and_mul_kernel(uint16_t* __restrict__ out,      // [T, d]
                                    const uint16_t* __restrict__ x,  // [T, 2d]
                                    const int num_tokens, const int d) {
  const int chunks = d / 8;
  const long total = (long)num_tokens * chunks;
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < total;
       idx += gridDim.x * (long)blockDim.x) {
    const int c = idx % chunks;
    const long t = idx / chunks;
    const uint16_t* gate = x + (size_t)t * 2 * d + c * 8;
    const uint16_t* up = gate + d;
    f32x8 g = bf16x8_to_f32x8(*reinterpret_cast<const u16x8*>(gate));
    f32x8 u = bf16x8_to_f32x8(*reinterpret_cast<const u16x8*>(up));
    f32x8 o;
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      const float gi = g[i];
      o[i] = gi / (1.0f + __expf(-gi)) * u[i];
    }
    *reinterpret_cast<u16x8*>(out + (size_t)t * d + c * 8) = f32x8_to_bf16x8(o);
  }
}

extern "C" void lig_silu_and_mul_launch(uint16_t* out, const uint16_t* x,
                                        int num_tokens, int d,
                                        hipStream_t stream) {
  const long total = (long)num_tokens * (d / 8);
  const int block = 256;
  int grid = (int)((total + block - 1) / block);
  if (grid > 65535) grid = 65535;
  if (grid < 1) grid = 1;
  hipLaunchKernelGGL(silu_and_mul_kernel, dim3(grid), dim3(block), 0, stream,
                     out, x, num_tokens, d);
}
