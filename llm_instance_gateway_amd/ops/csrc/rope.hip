// Rotary position embedding (Llama / NeoX half-rotation), bf16, gfx950.
//
// cos/sin are precomputed on HOST into a [max_pos, head_dim] f32 cache
// (first half cos, second half sin) — on-device sinf/cosf would turn this
// memory-bound op VALU-bound (guide App. B).  Applied in-place to q and k.
//
// Thread mapping: one thread per (token, head, dim-pair-pair): each thread
// rotates 2 adjacent pairs (float2-width loads on the f32 cache, 2x uint32
// on the bf16 halves).

#include "common.h"

__global__ void rope_kernel(
    const int64_t* __restrict__ positions,  // [T]
    uint16_t* __restrict__ q,               // [T, HQ*D]
    uint16_t* __restrict__ k,               // [T, HK*D]
    const float* __restrict__ cos_sin,      // [P, D] (D/2 cos | D/2 sin)
    const int num_tokens,
    const int num_q_heads,
    const int num_k_heads,
    const int head_dim) {
  const int half = head_dim / 2;
  // pairs processed 2-at-a-time
  const int pair2_per_head = half / 2;
  const int total = num_tokens * (num_q_heads + num_k_heads) * pair2_per_head;

  for (int idx = blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += gridDim.x * blockDim.x) {
    const int p2 = idx % pair2_per_head;
    const int h = (idx / pair2_per_head) % (num_q_heads + num_k_heads);
    const int t = idx / (pair2_per_head * (num_q_heads + num_k_heads));
    const int64_t pos = positions[t];
    const float* cs = cos_sin + pos * head_dim + p2 * 2;

    uint16_t* base;
    if (h < num_q_heads) {
      base = q + ((size_t)t * num_q_heads + h) * head_dim;
    } else {
      base = k + ((size_t)t * num_k_heads + (h - num_q_heads)) * head_dim;
    }
    const float c0 = cs[0], c1 = cs[1];
    const float s0 = cs[half], s1 = cs[half + 1];

    uint16_t* x1p = base + p2 * 2;
    uint16_t* x2p = base + half + p2 * 2;
    uint32_t x1u = *reinterpret_cast<uint32_t*>(x1p);
    uint32_t x2u = *reinterpret_cast<uint32_t*>(x2p);
    float x1a = bf16_to_f32(x1u & 0xffff), x1b = bf16_to_f32(x1u >> 16);
    float x2a = bf16_to_f32(x2u & 0xffff), x2b = bf16_to_f32(x2u >> 16);

    float o1a = x1a * c0 - x2a * s0;
    float o1b = x1b * c1 - x2b * s1;
    float o2a = x2a * c0 + x1a * s0;
    float o2b = x2b * c1 + x1b * s1;

    *reinterpret_cast<uint32_t*>(x1p) =
        (uint32_t)f32_to_bf16(o1a) | ((uint32_t)f32_to_bf16(o1b) << 16);
    *reinterpret_cast<uint32_t*>(x2p) =
        (uint32_t)f32_to_bf16(o2a) | ((uint32_t)f32_to_bf16(o2b) << 16);
  }
}

extern "C" void lig_rope_launch(const int64_t* positions, uint16_t* q,
                                uint16_t* k, const float* cos_sin,
                                int num_tokens, int num_q_heads,
                                int num_k_heads, int head_dim,
                                hipStream_t stream) {
  const int half2 = (head_dim / 2) / 2;
  const long total = (long)num_tokens * (num_q_heads + num_k_heads) * half2;
  const int block = 256;
  int grid = (int)((total + block - 1) / block);
  if (grid > 65535) grid = 65535;
  if (grid < 1) grid = 1;
  hipLaunchKernelGGL(rope_kernel, dim3(grid), dim3(block), 0, stream,
                     positions, q, k, cos_sin, num_tokens, num_q_heads,
                     num_k_heads, head_dim);
}
