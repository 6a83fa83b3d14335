// MFMA layout probe: one 16x16x32 bf16 tile computed by one wave.
//
// Validates (on hardware, vs torch.matmul) the fragment layout assumptions
// the attention kernels build on (guide §3; no ISA doc in this image):
//   A[16][32]: lane l holds A[i=l&15][k=(l>>4)*8 + v], v=0..7  (bf16x8)
//   B[32][16]: lane l holds B[k=(l>>4)*8 + v][j=l&15]
//   C[16][16]: lane l holds C[row=(l>>4)*4 + r][col=l&15], r=0..3 (f32x4)

#include "common.h"

typedef __bf16 bf16x8_t __attribute__((ext_vector_type(8)));
typedef float f32x4_t __attribute__((ext_vector_type(4)));

__global__ void mfma_probe_kernel(float* __restrict__ out,      // [16,16]
                                  const uint16_t* __restrict__ a,  // [16,32]
                                  const uint16_t* __restrict__ b)  // [32,16]
{
  const int lane = threadIdx.x & 63;
  const int i = lane & 15;
  const int kbase = (lane >> 4) * 8;

  bf16x8_t av, bv;
#pragma unroll
  for (int v = 0; v < 8; ++v) {
    uint16_t ar = a[i * 32 + kbase + v];
    uint16_t br = b[(kbase + v) * 16 + i];
    av[v] = *reinterpret_cast<__bf16*>(&ar);
    bv[v] = *reinterpret_cast<__bf16*>(&br);
  }
  f32x4_t acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(av, bv, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    out[((lane >> 4) * 4 + r) * 16 + (lane & 15)] = acc[r];
  }
}

extern "C" void lig_mfma_probe_launch(float* out, const uint16_t* a,
                                      const uint16_t* b, hipStream_t stream) {
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, stream, out, a, b);
}
