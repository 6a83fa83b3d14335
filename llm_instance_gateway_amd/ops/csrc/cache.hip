// Paged KV-cache scatter (reshape_and_cache), bf16, gfx950.
//
// Writes the freshly-computed K/V rows of a step's flat token batch into the
// paged cache at slot_mapping[t] (slot = block * block_size + offset).
// Cache layout: [num_blocks, num_kv_heads, block_size, head_dim] — a token's
// (head, :) row is contiguous (head_dim*2 bytes), so per-(token,head) writes
// are fully coalesced 16 B/lane chunks.

#include "common.h"

__global__ void reshape_and_cache_kernel(
    const uint16_t* __restrict__ k,    // [T, HK*D]
    const uint16_t* __restrict__ v,    // [T, HK*D]
    uint16_t* __restrict__ k_cache,    // [B, HK, BS, D]
    uint16_t* __restrict__ v_cache,
    const int64_t* __restrict__ slot_mapping,  // [T]
    const int num_tokens,
    const int num_kv_heads,
    const int head_dim,
    const int block_size) {
  // one thread per (token, head, 8-elem chunk), for k and v both
  const int chunks_per_head = head_dim / 8;
  const long total = (long)num_tokens * num_kv_heads * chunks_per_head;
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < total;
       idx += gridDim.x * (long)blockDim.x) {
    const int c = idx % chunks_per_head;
    const int h = (idx / chunks_per_head) % num_kv_heads;
    const long t = idx / (chunks_per_head * (long)num_kv_heads);
    const int64_t slot = slot_mapping[t];
    if (slot < 0) continue;  // padding token
    const int64_t blk = slot / block_size;
    const int off = (int)(slot % block_size);
    const size_t src = ((size_t)t * num_kv_heads + h) * head_dim + c * 8;
    const size_t dst =
        (((size_t)blk * num_kv_heads + h) * block_size + off) * head_dim + c * 8;
    *reinterpret_cast<u16x8*>(k_cache + dst) =
        *reinterpret_cast<const u16x8*>(k + src);
    *reinterpret_cast<u16x8*>(v_cache + dst) =
        *reinterpret_cast<const u16x8*>(v + src);
  }
}

extern "C" void lig_reshape_and_cache_launch(
    const uint16_t* k, const uint16_t* v, uint16_t* k_cache,
    uint16_t* v_cache, const int64_t* slot_mapping, int num_tokens,
    int num_kv_heads, int head_dim, int block_size, hipStream_t stream) {
  const long total = (long)num_tokens * num_kv_heads * (head_dim / 8);
  const int block = 256;
  int grid = (int)((total + block - 1) / block);
  if (grid > 65535) grid = 65535;
  if (grid < 1) grid = 1;
  hipLaunchKernelGGL(reshape_and_cache_kernel, dim3(grid), dim3(block), 0,
                     stream, k, v, k_cache, v_cache, slot_mapping, num_tokens,
                     num_kv_heads, head_dim, block_size);
}
