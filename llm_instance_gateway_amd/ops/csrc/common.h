// Common device helpers for the gfx950 (MI355X / CDNA4) kernels.
//
// Written for CDNA4 only: wave64, MFMA, 160 KiB LDS per CU, HBM3E.
// No CUDA-compat shims, no multi-arch dispatch.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <cstdint>

#define WAVE_SIZE 64

#define LIG_DEVICE __device__ __forceinline__

// ---- bf16 <-> f32 -----------------------------------------------------------

LIG_DEVICE float bf16_to_f32(uint16_t h) {
  union {
    uint32_t u;
    float f;
  } v;
  v.u = static_cast<uint32_t>(h) << 16;
  return v.f;
}

LIG_DEVICE uint16_t f32_to_bf16(float f) {
  union {
    float f;
    uint32_t u;
  } v;
  v.f = f;
  // round-to-nearest-even
  uint32_t rounding_bias = ((v.u >> 16) & 1u) + 0x7FFFu;
  return static_cast<uint16_t>((v.u + rounding_bias) >> 16);
}

// 8 bf16 = 16 B: the unit of a fully-coalesced per-lane load (guide §2:
// 16 B/lane x 64 lanes = 1 KiB per instruction).
typedef uint16_t u16x8 __attribute__((ext_vector_type(8)));
typedef float f32x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

LIG_DEVICE f32x8 bf16x8_to_f32x8(u16x8 h) {
  f32x8 out;
#pragma unroll
  for (int i = 0; i < 8; ++i) out[i] = bf16_to_f32(h[i]);
  return out;
}

LIG_DEVICE u16x8 f32x8_to_bf16x8(f32x8 f) {
  u16x8 out;
#pragma unroll
  for (int i = 0; i < 8; ++i) out[i] = f32_to_bf16(f[i]);
  return out;
}

// ---- wave reductions --------------------------------------------------------
// Butterfly over the full 64-lane wave; result replicated in every lane.

LIG_DEVICE float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE_SIZE);
  return v;
}

LIG_DEVICE float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE_SIZE));
  return v;
}

// Partial butterfly: reduce within groups of `width` lanes (width = power of 2).
template <int WIDTH>
LIG_DEVICE float group_reduce_sum(float v) {
#pragma unroll
  for (int off = WIDTH / 2; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE_SIZE);
  return v;
}

// Reduce ACROSS groups: lanes i, i+STRIDE, i+2*STRIDE ... combined.
template <int STRIDE, int N>
LIG_DEVICE float stride_reduce_sum(float v) {
#pragma unroll
  for (int off = STRIDE; off < STRIDE * N; off <<= 1) v += __shfl_xor(v, off, WAVE_SIZE);
  return v;
}

template <int STRIDE, int N>
LIG_DEVICE float stride_reduce_max(float v) {
#pragma unroll
  for (int off = STRIDE; off < STRIDE * N; off <<= 1)
    v = fmaxf(v, __shfl_xor(v, off, WAVE_SIZE));
  return v;
}

LIG_DEVICE int cdiv(int a, int b) { return (a + b - 1) / b; }
