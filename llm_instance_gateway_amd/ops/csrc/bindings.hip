// Python bindings for the gfx950 kernels (torch extension, compiled by hipcc).
//
// Convention: every op takes the current HIP stream as an int64 handle
// (Python side passes torch.cuda.current_stream().cuda_stream) so the kernels
// serialize correctly with the library GEMMs on torch's stream without
// depending on the HIP/CUDA stream-masquerading ATen headers.

#include <torch/extension.h>

#include "common.h"

// LIG_DEBUG (make build-debug): host-side validation of device index tensors
// BEFORE a kernel dereferences them — a D2H copy per check, so debug-only.
#ifdef LIG_DEBUG
static void lig_check_range(const torch::Tensor& idx, long lo, long hi,
                            const char* what) {
  if (idx.numel() == 0) return;
  auto c = idx.to(torch::kCPU);
  const long mn = c.min().item<long>();
  const long mx = c.max().item<long>();
  TORCH_CHECK(mn >= lo && mx < hi, "LIG_DEBUG: ", what, " out of range [",
              lo, ", ", hi, "): min=", mn, " max=", mx);
}
#define LIG_CHECK_RANGE(t, lo, hi, what) lig_check_range(t, lo, hi, what)
#else
#define LIG_CHECK_RANGE(t, lo, hi, what) \
  do {                                   \
  } while (0)
#endif

extern "C" {
void lig_rms_norm_launch(uint16_t*, const uint16_t*, const uint16_t*, float,
                         int, int, hipStream_t);
void lig_fused_add_rms_norm_launch(uint16_t*, const uint16_t*, uint16_t*,
                                   const uint16_t*, float, int, int,
                                   hipStream_t);
void lig_rope_launch(const int64_t*, uint16_t*, uint16_t*, const float*, int,
                     int, int, int, hipStream_t);
void lig_reshape_and_cache_launch(const uint16_t*, const uint16_t*, uint16_t*,
                                  uint16_t*, const int64_t*, int, int, int,
                                  int, hipStream_t);
void lig_silu_and_mul_launch(uint16_t*, const uint16_t*, int, int,
                             hipStream_t);
void lig_paged_attention_decode_launch(uint16_t*, const uint16_t*,
                                       const uint16_t*, const uint16_t*,
                                       const int*, const int*, float, int,
                                       int, int, int, int, int, hipStream_t);
void lig_paged_attention_decode_split_launch(
    uint16_t*, const uint16_t*, const uint16_t*, const uint16_t*,
    const int*, const int*, float*, float, int, int, int, int, int, int,
    int, hipStream_t);
void lig_mfma_probe_launch(float*, const uint16_t*, const uint16_t*,
                           hipStream_t);
void lig_sgmv_shrink_launch(uint16_t*, const uint16_t*, const uint16_t*,
                            const int*, int, int, int, int, hipStream_t);
void lig_sgmv_expand_launch(uint16_t*, const uint16_t*, const uint16_t*,
                            const int*, float, int, int, int, int,
                            hipStream_t);
void lig_skinny_gemm_launch(uint16_t*, const uint16_t*, const uint16_t*,
                            int, int, int, hipStream_t);
void lig_quant_fp8_rowwise_launch(uint8_t*, float*, const uint16_t*, int,
                                  int, hipStream_t);
void lig_sgmv_shrink_units_launch(uint16_t*, const uint16_t*, const uint16_t*,
                                  const int*, const int*, const int*,
                                  const int*, int, int, int, hipStream_t);
void lig_sgmv_expand_lds_launch(uint16_t*, const uint16_t*, const uint16_t*,
                                const int*, const int*, const int*,
                                const int*, float, int, int, int, int,
                                hipStream_t);
void lig_sgmv_expand_units_launch(uint16_t*, const uint16_t*, const uint16_t*,
                                  const int*, const int*, const int*,
                                  const int*, float, int, int, int,
                                  hipStream_t);
void lig_sgmv_shrink_tiled_launch(uint16_t*, const uint16_t*, const uint16_t*,
                                  const int*, const int*, const int*, int,
                                  int, int, int, hipStream_t);
void lig_sgmv_expand_tiled_launch(uint16_t*, const uint16_t*, const uint16_t*,
                                  const int*, const int*, const int*, int,
                                  float, int, int, int, hipStream_t);
void lig_paged_attention_prefill_launch(
    uint16_t*, const uint16_t*, const uint16_t*, const uint16_t*, const int*,
    const int*, const int*, const int*, const int*, float, int, int, int,
    int, int, int, hipStream_t);
}

namespace {

inline uint16_t* bf16_ptr(torch::Tensor& t) {
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, "expected bf16 tensor");
  TORCH_CHECK(t.is_contiguous(), "expected contiguous tensor");
  TORCH_CHECK(t.is_cuda(), "expected device tensor");
  return reinterpret_cast<uint16_t*>(t.data_ptr());
}

inline const uint16_t* bf16_cptr(const torch::Tensor& t) {
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, "expected bf16 tensor");
  TORCH_CHECK(t.is_contiguous(), "expected contiguous tensor");
  TORCH_CHECK(t.is_cuda(), "expected device tensor");
  return reinterpret_cast<const uint16_t*>(t.data_ptr());
}

inline hipStream_t as_stream(int64_t s) {
  return reinterpret_cast<hipStream_t>(s);
}

void rms_norm(torch::Tensor out, torch::Tensor input, torch::Tensor weight,
              double eps, int64_t stream) {
  const int T = input.size(0);
  const int H = input.size(1);
  TORCH_CHECK(H % 8 == 0, "hidden must be a multiple of 8");
  lig_rms_norm_launch(bf16_ptr(out), bf16_cptr(input), bf16_cptr(weight),
                      (float)eps, T, H, as_stream(stream));
}

void fused_add_rms_norm(torch::Tensor out, torch::Tensor input,
                        torch::Tensor residual, torch::Tensor weight,
                        double eps, int64_t stream) {
  const int T = input.size(0);
  const int H = input.size(1);
  TORCH_CHECK(H % 8 == 0, "hidden must be a multiple of 8");
  lig_fused_add_rms_norm_launch(bf16_ptr(out), bf16_cptr(input),
                                bf16_ptr(residual), bf16_cptr(weight),
                                (float)eps, T, H, as_stream(stream));
}

void rotary_embedding(torch::Tensor positions, torch::Tensor q,
                      torch::Tensor k, torch::Tensor cos_sin,
                      int64_t head_dim, int64_t stream) {
  TORCH_CHECK(positions.scalar_type() == torch::kInt64);
  TORCH_CHECK(cos_sin.scalar_type() == torch::kFloat32);
  TORCH_CHECK(head_dim % 4 == 0, "head_dim must be a multiple of 4");
  const int T = q.size(0);
  const int hq = q.size(1) / head_dim;
  const int hk = k.size(1) / head_dim;
  lig_rope_launch(positions.data_ptr<int64_t>(), bf16_ptr(q), bf16_ptr(k),
                  cos_sin.data_ptr<float>(), T, hq, hk, (int)head_dim,
                  as_stream(stream));
}

void reshape_and_cache(torch::Tensor k, torch::Tensor v,
                       torch::Tensor k_cache, torch::Tensor v_cache,
                       torch::Tensor slot_mapping, int64_t stream) {
  TORCH_CHECK(slot_mapping.scalar_type() == torch::kInt64);
  const int T = k.size(0);
  const int Hkv = k_cache.size(1);
  const int BS = k_cache.size(2);
  const int D = k_cache.size(3);
  TORCH_CHECK(D % 8 == 0, "head_dim must be a multiple of 8");
  lig_reshape_and_cache_launch(bf16_cptr(k), bf16_cptr(v), bf16_ptr(k_cache),
                               bf16_ptr(v_cache),
                               slot_mapping.data_ptr<int64_t>(), T, Hkv, D, BS,
                               as_stream(stream));
}

void silu_and_mul(torch::Tensor out, torch::Tensor x, int64_t stream) {
  const int T = x.size(0);
  const int d = x.size(1) / 2;
  TORCH_CHECK(d % 8 == 0, "intermediate must be a multiple of 8");
  lig_silu_and_mul_launch(bf16_ptr(out), bf16_cptr(x), T, d,
                          as_stream(stream));
}

void paged_attention_decode(torch::Tensor out, torch::Tensor q,
                            torch::Tensor k_cache, torch::Tensor v_cache,
                            torch::Tensor block_tables,
                            torch::Tensor context_lens, double scale,
                            int64_t stream) {
  TORCH_CHECK(block_tables.scalar_type() == torch::kInt32);
  TORCH_CHECK(context_lens.scalar_type() == torch::kInt32);
  const int N = q.size(0);
  const int Hq = q.size(1);
  const int D = q.size(2);
  const int Hkv = k_cache.size(1);
  const int BS = k_cache.size(2);
  const int max_blocks = block_tables.size(1);
  TORCH_CHECK(D == 16 || D == 64 || D == 128, "head_dim must be 16/64/128");
  TORCH_CHECK(Hq % Hkv == 0 && Hq / Hkv <= 16, "bad GQA group size");
  if (N == 0) return;
  LIG_CHECK_RANGE(block_tables, 0, k_cache.size(0), "decode block_tables");
  LIG_CHECK_RANGE(context_lens, 0, (long)block_tables.size(1) * BS + 1,
                  "decode context_lens");
  lig_paged_attention_decode_launch(
      bf16_ptr(out), bf16_cptr(q), bf16_cptr(k_cache), bf16_cptr(v_cache),
      block_tables.data_ptr<int>(), context_lens.data_ptr<int>(),
      (float)scale, N, Hq, Hkv, D, BS, max_blocks, as_stream(stream));
}

// Flash-decoding variant: context split across nsplits workgroups with an
// fp32 partial scratch [N, Hq, nsplits, D+2] and a combine pass (ROADMAP #1;
// experimental — not yet on the default dispatch path).
void paged_attention_decode_split(torch::Tensor out, torch::Tensor q,
                                  torch::Tensor k_cache,
                                  torch::Tensor v_cache,
                                  torch::Tensor block_tables,
                                  torch::Tensor context_lens,
                                  torch::Tensor scratch, double scale,
                                  int64_t nsplits, int64_t stream) {
  TORCH_CHECK(block_tables.scalar_type() == torch::kInt32);
  TORCH_CHECK(context_lens.scalar_type() == torch::kInt32);
  TORCH_CHECK(scratch.scalar_type() == torch::kFloat32);
  const int N = q.size(0);
  const int Hq = q.size(1);
  const int D = q.size(2);
  const int Hkv = k_cache.size(1);
  const int BS = k_cache.size(2);
  const int max_blocks = block_tables.size(1);
  TORCH_CHECK(D == 16 || D == 64 || D == 128, "head_dim must be 16/64/128");
  TORCH_CHECK(Hq % Hkv == 0 && Hq / Hkv <= 16, "bad GQA group size");
  TORCH_CHECK(scratch.numel() >= (int64_t)N * Hq * nsplits * (D + 2),
              "scratch too small");
  if (N == 0) return;
  lig_paged_attention_decode_split_launch(
      bf16_ptr(out), bf16_cptr(q), bf16_cptr(k_cache), bf16_cptr(v_cache),
      block_tables.data_ptr<int>(), context_lens.data_ptr<int>(),
      scratch.data_ptr<float>(), (float)scale, N, Hq, Hkv, D, BS,
      max_blocks, (int)nsplits, as_stream(stream));
}

void paged_attention_prefill(torch::Tensor out, torch::Tensor q,
                             torch::Tensor k_cache, torch::Tensor v_cache,
                             torch::Tensor tile_seq, torch::Tensor tile_qstart,
                             torch::Tensor query_start_locs,
                             torch::Tensor context_lens,
                             torch::Tensor block_tables, double scale,
                             int64_t stream) {
  TORCH_CHECK(tile_seq.scalar_type() == torch::kInt32);
  TORCH_CHECK(tile_qstart.scalar_type() == torch::kInt32);
  TORCH_CHECK(query_start_locs.scalar_type() == torch::kInt32);
  TORCH_CHECK(context_lens.scalar_type() == torch::kInt32);
  TORCH_CHECK(block_tables.scalar_type() == torch::kInt32);
  const int Hq = q.size(1);
  const int D = q.size(2);
  const int Hkv = k_cache.size(1);
  const int BS = k_cache.size(2);
  const int ntiles = tile_seq.size(0);
  const int max_blocks = block_tables.size(1);
  TORCH_CHECK(D == 64 || D == 128, "prefill MFMA kernel needs head_dim 64/128");
  if (ntiles == 0) return;
  lig_paged_attention_prefill_launch(
      bf16_ptr(out), bf16_cptr(q), bf16_cptr(k_cache), bf16_cptr(v_cache),
      tile_seq.data_ptr<int>(), tile_qstart.data_ptr<int>(),
      query_start_locs.data_ptr<int>(), context_lens.data_ptr<int>(),
      block_tables.data_ptr<int>(), (float)scale, ntiles, Hq, Hkv, D, BS,
      max_blocks, as_stream(stream));
}

void sgmv_shrink(torch::Tensor u, torch::Tensor x, torch::Tensor a_stack,
                 torch::Tensor slots, int64_t stream) {
  TORCH_CHECK(slots.scalar_type() == torch::kInt32);
  const int T = x.size(0);
  const int H = x.size(1);
  const int R = a_stack.size(1);
  TORCH_CHECK(a_stack.size(2) == H, "A stack in-features mismatch");
  TORCH_CHECK(H % 8 == 0 && R <= 64);
  if (T == 0) return;
  lig_sgmv_shrink_launch(bf16_ptr(u), bf16_cptr(x), bf16_cptr(a_stack),
                         slots.data_ptr<int>(), T, R, H, T >= 256,
                         as_stream(stream));
}

// one call = shrink + expand (halves Python dispatch; u is a caller-owned
// workspace so no per-call allocation)
void sgmv_apply(torch::Tensor y, torch::Tensor x, torch::Tensor a_stack,
                torch::Tensor b_stack, torch::Tensor u_ws,
                torch::Tensor slots, double scaling, bool use_wide,
                int64_t stream) {
  TORCH_CHECK(slots.scalar_type() == torch::kInt32);
  const int T = x.size(0);
  const int H = x.size(1);
  const int R = a_stack.size(1);
  const int O = y.size(1);
  TORCH_CHECK(u_ws.size(0) >= T && u_ws.size(1) == R, "u workspace too small");
  TORCH_CHECK(R % 8 == 0 && R <= 64, "rank must be 8/16/32/64");
  if (T == 0) return;
  LIG_CHECK_RANGE(slots, -1, a_stack.size(0), "sgmv slots");
  hipStream_t st = as_stream(stream);
  lig_sgmv_shrink_launch(bf16_ptr(u_ws), bf16_cptr(x), bf16_cptr(a_stack),
                         slots.data_ptr<int>(), T, R, H, use_wide, st);
  lig_sgmv_expand_launch(bf16_ptr(y), bf16_cptr(u_ws), bf16_cptr(b_stack),
                         slots.data_ptr<int>(), (float)scaling, T, R, O,
                         use_wide, st);
}

// Slot-grouped units (r02): shrink via the proven v2 kernels (occupancy-
// first), expand via the LDS-staged unit kernel — full per-token thread
// parallelism with B read once per <=32-token unit instead of per token
// (profiles/r02_kernel_stats.md: the per-token kernels were 27% of all
// kernel time).  Static shapes (padded empty units) so the hipGraph decode
// path can bake it.  Requires O % 256 == 0 (every serving module dim);
// caller falls back to sgmv_apply otherwise.
void sgmv_apply_units(torch::Tensor y, torch::Tensor x, torch::Tensor a_stack,
                      torch::Tensor b_stack, torch::Tensor u_ws,
                      torch::Tensor slots, torch::Tensor order,
                      torch::Tensor unit_t0, torch::Tensor unit_len,
                      torch::Tensor unit_slot, double scaling, bool use_wide,
                      int64_t stream) {
  TORCH_CHECK(slots.scalar_type() == torch::kInt32);
  TORCH_CHECK(order.scalar_type() == torch::kInt32);
  TORCH_CHECK(unit_t0.scalar_type() == torch::kInt32);
  TORCH_CHECK(unit_len.scalar_type() == torch::kInt32);
  TORCH_CHECK(unit_slot.scalar_type() == torch::kInt32);
  const int T = x.size(0);
  const int H = x.size(1);
  const int R = a_stack.size(1);
  const int O = y.size(1);
  const int U = unit_t0.size(0);
  TORCH_CHECK(u_ws.size(0) >= T && u_ws.size(1) == R, "u workspace too small");
  TORCH_CHECK(R % 8 == 0 && R <= 64, "rank must be 8/16/32/64");
  TORCH_CHECK(H % 8 == 0 && O % 256 == 0, "units expand needs O%256==0");
  if (T == 0 || U == 0) return;
  LIG_CHECK_RANGE(slots, -1, a_stack.size(0), "sgmv slots");
  LIG_CHECK_RANGE(order, 0, T, "sgmv unit order");
  LIG_CHECK_RANGE(unit_t0, 0, T, "sgmv unit_t0");
  LIG_CHECK_RANGE(unit_len, 0, T + 1, "sgmv unit_len");
  LIG_CHECK_RANGE(unit_slot, -1, b_stack.size(0), "sgmv unit_slot");
  hipStream_t st = as_stream(stream);
  lig_sgmv_shrink_launch(bf16_ptr(u_ws), bf16_cptr(x), bf16_cptr(a_stack),
                         slots.data_ptr<int>(), T, R, H, use_wide, st);
  lig_sgmv_expand_lds_launch(
      bf16_ptr(y), bf16_cptr(u_ws), bf16_cptr(b_stack),
      order.data_ptr<int>(), unit_t0.data_ptr<int>(),
      unit_len.data_ptr<int>(), unit_slot.data_ptr<int>(), (float)scaling, U,
      T, R, O, st);
}

// Prefill LoRA: one shrink+expand over host-built 64-token segment tiles
// (MFMA path; engine/model_runner.py builds the tiles).
void sgmv_apply_tiled(torch::Tensor y, torch::Tensor x, torch::Tensor a_stack,
                      torch::Tensor b_stack, torch::Tensor u_ws,
                      torch::Tensor tile_t0, torch::Tensor tile_len,
                      torch::Tensor tile_slot, double scaling,
                      int64_t stream) {
  const int T = x.size(0);
  const int H = x.size(1);
  const int R = a_stack.size(1);
  const int O = y.size(1);
  const int ntiles = tile_t0.size(0);
  TORCH_CHECK(tile_t0.scalar_type() == torch::kInt32);
  TORCH_CHECK(u_ws.size(0) >= T && u_ws.size(1) == R, "u workspace too small");
  TORCH_CHECK(R % 8 == 0 && R <= 64, "rank must be 8/16/32/64");
  TORCH_CHECK(H % 32 == 0 && O % 64 == 0, "tiled sgmv needs H%32==0, O%64==0");
  if (T == 0 || ntiles == 0) return;
  hipStream_t st = as_stream(stream);
  lig_sgmv_shrink_tiled_launch(bf16_ptr(u_ws), bf16_cptr(x),
                               bf16_cptr(a_stack), tile_t0.data_ptr<int>(),
                               tile_len.data_ptr<int>(),
                               tile_slot.data_ptr<int>(), ntiles, T, R, H, st);
  lig_sgmv_expand_tiled_launch(bf16_ptr(y), bf16_cptr(u_ws),
                               bf16_cptr(b_stack), tile_t0.data_ptr<int>(),
                               tile_len.data_ptr<int>(),
                               tile_slot.data_ptr<int>(), ntiles,
                               (float)scaling, T, R, O, st);
}

void skinny_gemm(torch::Tensor c, torch::Tensor a, torch::Tensor w,
                 int64_t stream) {
  const int M = a.size(0);
  const int K = a.size(1);
  const int N = w.size(0);
  TORCH_CHECK(w.size(1) == K && c.size(0) == M && c.size(1) == N,
              "skinny_gemm shape mismatch");
  TORCH_CHECK(K % 32 == 0 && N % 64 == 0, "skinny_gemm needs K%32==0, N%64==0");
  TORCH_CHECK(a.is_contiguous() && w.is_contiguous() && c.is_contiguous());
  if (M == 0) return;
  lig_skinny_gemm_launch(bf16_ptr(c), bf16_cptr(a), bf16_cptr(w), M, N, K,
                         as_stream(stream));
}

void sgmv_expand(torch::Tensor y, torch::Tensor u, torch::Tensor b_stack,
                 torch::Tensor slots, double scaling, int64_t stream) {
  TORCH_CHECK(slots.scalar_type() == torch::kInt32);
  const int T = u.size(0);
  const int R = u.size(1);
  const int O = y.size(1);
  TORCH_CHECK(b_stack.size(1) == O && b_stack.size(2) == R, "B stack mismatch");
  TORCH_CHECK(O % 8 == 0 && R % 8 == 0 && R <= 64, "rank must be 8/16/32/64");
  if (T == 0) return;
  lig_sgmv_expand_launch(bf16_ptr(y), bf16_cptr(u), bf16_cptr(b_stack),
                         slots.data_ptr<int>(), (float)scaling, T, R, O,
                         T >= 256, as_stream(stream));
}

void quant_fp8_rowwise(torch::Tensor xq, torch::Tensor xs, torch::Tensor x,
                       int64_t stream) {
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16, "x must be bf16");
  TORCH_CHECK(xq.element_size() == 1, "xq must be an 8-bit (fp8) tensor");
  TORCH_CHECK(xs.scalar_type() == torch::kFloat32, "xs must be fp32");
  TORCH_CHECK(x.dim() == 2 && xq.sizes() == x.sizes(), "shape mismatch");
  const int T = x.size(0);
  const int K = x.size(1);
  TORCH_CHECK(K % 8 == 0, "K must be a multiple of 8");
  TORCH_CHECK(xs.numel() == T, "xs must have one scale per row");
  TORCH_CHECK(x.is_contiguous() && xq.is_contiguous() && xs.is_contiguous());
  TORCH_CHECK(xq.is_cuda() && xs.is_cuda());
  lig_quant_fp8_rowwise_launch(reinterpret_cast<uint8_t*>(xq.data_ptr()),
                               xs.data_ptr<float>(), bf16_cptr(x), T, K,
                               as_stream(stream));
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "MI355X (gfx950) native kernels for llm_instance_gateway_amd";
  m.def("rms_norm", &rms_norm);
  m.def("fused_add_rms_norm", &fused_add_rms_norm);
  m.def("rotary_embedding", &rotary_embedding);
  m.def("reshape_and_cache", &reshape_and_cache);
  m.def("silu_and_mul", &silu_and_mul);
  m.def("paged_attention_decode", &paged_attention_decode);
  m.def("paged_attention_decode_split", &paged_attention_decode_split);
  m.def("paged_attention_prefill", &paged_attention_prefill);
  m.def("sgmv_shrink", &sgmv_shrink);
  m.def("sgmv_apply", &sgmv_apply);
  m.def("sgmv_apply_units", &sgmv_apply_units);
  m.def("skinny_gemm", &skinny_gemm);
  m.def("sgmv_apply_tiled", &sgmv_apply_tiled);
  m.def("sgmv_expand", &sgmv_expand);
  m.def("quant_fp8_rowwise", &quant_fp8_rowwise);
  m.def("mfma_probe", [](torch::Tensor out, torch::Tensor a, torch::Tensor b,
                         int64_t stream) {
    TORCH_CHECK(out.scalar_type() == torch::kFloat32);
    lig_mfma_probe_launch(out.data_ptr<float>(), bf16_cptr(a), bf16_cptr(b),
                          as_stream(stream));
  });
}
