// Python bindings for the arks_amd gfx950 kernels (torch extension).
// Host-side validation lives here; kernels are in the *.hip files.
#include <torch/extension.h>

#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#include <cstring>
#include <string>
#include <vector>

namespace py = pybind11;

// --- kernel launchers (see the .hip files) ---
extern "C" {
void arks_rmsnorm(void* out, const void* input, const void* weight, float eps,
                  int rows, int hidden, hipStream_t stream);
void arks_fused_add_rmsnorm(void* out, const void* input, void* residual,
                            const void* weight, float eps, int rows, int hidden,
                            hipStream_t stream);
void arks_silu_mul(void* out, const void* gate_up, int64_t rows, int d,
                   hipStream_t stream);
void arks_rope_inplace(const void* positions, void* q, void* k,
                       const void* cos_sin, int num_tokens, int head_dim,
                       int num_q_heads, int num_kv_heads, int64_t q_stride,
                       int64_t k_stride, hipStream_t stream);
void arks_reshape_and_cache(const void* k, const void* v, void* k_cache,
                            void* v_cache, const void* slot_mapping,
                            int num_tokens, int num_kv_heads, int head_dim,
                            int block_size, int64_t kv_stride,
                            hipStream_t stream);
void arks_reshape_and_cache_fp8(const void* k, const void* v, void* k_cache,
                                void* v_cache, const void* slot_mapping,
                                int num_tokens, int num_kv_heads, int head_dim,
                                int block_size, int64_t kv_stride,
                                hipStream_t stream);
void arks_rope_and_cache(const void* positions, void* q, void* k,
                         const void* v, void* k_cache, void* v_cache,
                         const void* slot_mapping, const void* cos_sin,
                         int num_tokens, int head_dim, int num_q_heads,
                         int num_kv_heads, int block_size, int64_t q_stride,
                         int64_t k_stride, int kv_fp8, hipStream_t stream);
void arks_attn_decode_paged(void* out, void* part_out, const void* q,
                            const void* k_cache, const void* v_cache,
                            const void* block_tables, const void* seq_lens,
                            float scale, int num_seqs, int num_q_heads,
                            int num_kv_heads, int head_dim, int max_blocks,
                            int nparts, int64_t q_stride, int kv_fp8,
                            int window, hipStream_t stream);
void arks_attn_prefill_varlen(void* out, const void* q, const void* k,
                              const void* v, const void* cu_seqlens,
                              const void* tile_info, int ntiles, float scale,
                              int num_q_heads, int num_kv_heads, int head_dim,
                              int64_t q_stride, int64_t kv_stride, int window,
                              hipStream_t stream);
void arks_attn_extend_paged(void* out, const void* q, const void* k_cache,
                            const void* v_cache, const void* block_tables,
                            const void* kv_lens, const void* cu_seqlens_q,
                            const void* tile_info, int ntiles, float scale,
                            int num_q_heads, int num_kv_heads, int head_dim,
                            int max_blocks, int64_t q_stride, int kv_fp8,
                            int window, hipStream_t stream);
void arks_attn_extend_paged2(void* out, const void* q, const void* k_cache,
                             const void* v_cache, const void* block_tables,
                             const void* kv_lens, const void* cu_seqlens_q,
                             const void* tile_info, void* part_ws, int ntiles,
                             float scale, int num_q_heads, int num_kv_heads,
                             int head_dim, int max_blocks, int64_t q_stride,
                             int window, hipStream_t stream);
void arks_attn_extend2_combine(void* out, const void* part_ws,
                               const void* combine_table,
                               const void* cu_seqlens_q, int n_split,
                               float scale, int num_q_heads, int head_dim,
                               hipStream_t stream);
void arks_quant_fp8_rows(void* out, void* inv_scale, const void* x, int rows,
                         int cols, hipStream_t stream);
void arks_rmsnorm_fp8(void* out, void* inv_scale, const void* input,
                      void* residual, const void* weight, float eps, int rows,
                      int hidden, int fused_add, hipStream_t stream);
void arks_silu_mul_fp8(void* out, void* inv_scale, const void* gate_up,
                       int rows, int d, hipStream_t stream);
void arks_skinny_gemm(void* part, void* out, const void* a, const void* w,
                      const void* bias, int m_rows, int n_total, int k_total,
                      int k_per_split, int nsplits, int64_t a_stride,
                      hipStream_t stream);
void arks_skinny_gemm_v(void* part, void* out, const void* a, const void* w,
                        const void* bias, int m_rows, int n_total, int k_total,
                        int k_per_split, int nsplits, int64_t a_stride,
                        int variant, bool fuse_silu, hipStream_t stream);
void arks_greedy_sample(void* out, const void* logits, int rows, int vocab,
                        hipStream_t stream);
void arks_gumbel_sample(void* out, const void* logits, const void* temperatures,
                        const void* uniform, int rows, int vocab,
                        hipStream_t stream);
void arks_ar_seq_inc(void* seq, hipStream_t stream);
void arks_one_shot_allreduce(void* out, const void* src, void* mail_ptrs[8],
                             void* flag_ptrs[8], void* seq, int rank,
                             int world, int64_t n, hipStream_t stream);
void arks_moe_topk(void* weights, void* ids, const void* logits, int T,
                   int E, int k, int renorm, hipStream_t stream);
void arks_moe_mix(void* out, const void* y, const void* weights,
                  const void* ids, int T, int H, int k, int expert_base,
                  int n_local, hipStream_t stream);
void arks_moe_mix_rows(void* out, const void* y, const void* weights,
                       const void* rows, int T, int H, int k,
                       hipStream_t stream);
void arks_mfma_probe(void* d, const void* a, const void* b, hipStream_t stream);
void arks_mfma_probe32(void* d, const void* a, const void* b, hipStream_t stream);
void arks_tr16_probe(void* out, int stride_bytes, hipStream_t stream);
}

namespace {

hipStream_t current_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

// KV caches may be bf16 or fp8 e4m3 (kv_cache_dtype="fp8"); returns fp8?
bool check_kv_cache(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
  if (t.scalar_type() == torch::kFloat8_e4m3fn) return true;
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, name, " must be bf16/fp8");
  return false;
}

void check_bf16_contig(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, name, " must be bf16");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

// bf16 tensor whose trailing dims are contiguous; dim-0 rows may be strided
// (a view into the fused QKV buffer).
void check_bf16_rowstrided(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, name, " must be bf16");
  TORCH_CHECK(t.stride(-1) == 1, name, " last dim must be contiguous");
  if (t.dim() == 3) {
    TORCH_CHECK(t.stride(1) == t.size(2), name, " inner dims must be dense");
  }
}

void rmsnorm(torch::Tensor out, torch::Tensor input, torch::Tensor weight,
             double eps) {
  check_bf16_contig(out, "out");
  check_bf16_contig(input, "input");
  check_bf16_contig(weight, "weight");
  const int hidden = input.size(-1);
  TORCH_CHECK(hidden % 8 == 0, "hidden must be a multiple of 8");
  const int rows = input.numel() / hidden;
  arks_rmsnorm(out.data_ptr(), input.data_ptr(), weight.data_ptr(), (float)eps,
               rows, hidden, current_stream());
}

void fused_add_rmsnorm(torch::Tensor out, torch::Tensor input,
                       torch::Tensor residual, torch::Tensor weight,
                       double eps) {
  check_bf16_contig(out, "out");
  check_bf16_contig(input, "input");
  check_bf16_contig(residual, "residual");
  const int hidden = input.size(-1);
  TORCH_CHECK(hidden % 8 == 0, "hidden must be a multiple of 8");
  const int rows = input.numel() / hidden;
  arks_fused_add_rmsnorm(out.data_ptr(), input.data_ptr(), residual.data_ptr(),
                         weight.data_ptr(), (float)eps, rows, hidden,
                         current_stream());
}

void silu_mul(torch::Tensor out, torch::Tensor gate_up) {
  check_bf16_contig(out, "out");
  check_bf16_contig(gate_up, "gate_up");
  const int d = out.size(-1);
  TORCH_CHECK(gate_up.size(-1) == 2 * d, "gate_up last dim must be 2*d");
  TORCH_CHECK(d % 8 == 0, "d must be a multiple of 8");
  const int64_t rows = out.numel() / d;
  arks_silu_mul(out.data_ptr(), gate_up.data_ptr(), rows, d, current_stream());
}

void rope_inplace(torch::Tensor positions, torch::Tensor q, torch::Tensor k,
                  torch::Tensor cos_sin, int64_t head_dim) {
  check_bf16_rowstrided(q, "q");
  check_bf16_rowstrided(k, "k");
  TORCH_CHECK(q.dim() == 2 && k.dim() == 2, "q/k must be 2-D [T, H*D]");
  TORCH_CHECK(positions.scalar_type() == torch::kInt64, "positions must be i64");
  TORCH_CHECK(cos_sin.scalar_type() == torch::kFloat32, "cos_sin must be f32");
  TORCH_CHECK(head_dim % 4 == 0, "head_dim must be a multiple of 4");
  const int num_tokens = q.size(0);
  const int num_q_heads = q.size(-1) / head_dim;
  const int num_kv_heads = k.size(-1) / head_dim;
  arks_rope_inplace(positions.data_ptr(), q.data_ptr(), k.data_ptr(),
                    cos_sin.data_ptr(), num_tokens, (int)head_dim, num_q_heads,
                    num_kv_heads, q.stride(0), k.stride(0), current_stream());
}

void reshape_and_cache(torch::Tensor k, torch::Tensor v, torch::Tensor k_cache,
                       torch::Tensor v_cache, torch::Tensor slot_mapping) {
  check_bf16_rowstrided(k, "k");
  check_bf16_rowstrided(v, "v");
  const bool fp8 = check_kv_cache(k_cache, "k_cache");
  check_kv_cache(v_cache, "v_cache");
  TORCH_CHECK(slot_mapping.scalar_type() == torch::kInt64);
  TORCH_CHECK(k.stride(0) == v.stride(0), "k/v must share row stride");
  const int num_tokens = k.size(0);
  const int num_kv_heads = k_cache.size(1);
  const int block_size = k_cache.size(2);
  const int head_dim = k_cache.size(3);
  TORCH_CHECK(head_dim % 8 == 0);
  auto fn = fp8 ? arks_reshape_and_cache_fp8 : arks_reshape_and_cache;
  fn(k.data_ptr(), v.data_ptr(), k_cache.data_ptr(),
     v_cache.data_ptr(), slot_mapping.data_ptr(),
     num_tokens, num_kv_heads, head_dim, block_size,
     k.stride(0), current_stream());
}

void rope_and_cache(torch::Tensor positions, torch::Tensor q,
                    torch::Tensor k, torch::Tensor v, torch::Tensor k_cache,
                    torch::Tensor v_cache, torch::Tensor slot_mapping,
                    torch::Tensor cos_sin, int64_t head_dim) {
  TORCH_CHECK(positions.scalar_type() == torch::kInt64);
  TORCH_CHECK(slot_mapping.scalar_type() == torch::kInt64);
  TORCH_CHECK(cos_sin.scalar_type() == torch::kFloat32 &&
              cos_sin.is_contiguous());
  check_bf16_rowstrided(q, "q");
  check_bf16_rowstrided(k, "k");
  check_bf16_rowstrided(v, "v");
  const bool kv_fp8 = check_kv_cache(k_cache, "k_cache");
  check_kv_cache(v_cache, "v_cache");
  TORCH_CHECK(k.stride(0) == v.stride(0), "k/v must share row stride");
  const int T = q.size(0);
  const int nq = (int)(q.numel() / std::max<int64_t>(T, 1) / head_dim);
  const int nkv = (int)(k.numel() / std::max<int64_t>(T, 1) / head_dim);
  const int block_size = k_cache.size(2);
  arks_rope_and_cache(positions.data_ptr(), q.data_ptr(), k.data_ptr(),
                      v.data_ptr(), k_cache.data_ptr(), v_cache.data_ptr(),
                      slot_mapping.data_ptr(), cos_sin.data_ptr(), T,
                      (int)head_dim, nq, nkv, block_size, q.stride(0),
                      k.stride(0), kv_fp8 ? 1 : 0, current_stream());
}

void attention_decode_paged(torch::Tensor out, torch::Tensor q,
                            torch::Tensor k_cache, torch::Tensor v_cache,
                            torch::Tensor block_tables, torch::Tensor seq_lens,
                            double scale, torch::Tensor part_out,
                            int64_t nparts, int64_t window) {
  check_bf16_contig(out, "out");
  check_bf16_rowstrided(q, "q");
  const bool kv_fp8 = check_kv_cache(k_cache, "k_cache");
  check_kv_cache(v_cache, "v_cache");
  TORCH_CHECK(block_tables.scalar_type() == torch::kInt32);
  TORCH_CHECK(seq_lens.scalar_type() == torch::kInt32);
  const int num_seqs = q.size(0);
  const int num_q_heads = q.size(1);
  const int head_dim = q.size(2);
  const int num_kv_heads = k_cache.size(1);
  TORCH_CHECK(k_cache.size(2) == 16, "KV block size must be 16");
  TORCH_CHECK(head_dim == 64 || head_dim == 128, "head_dim must be 64 or 128");
  const int gq = num_q_heads / num_kv_heads;
  TORCH_CHECK(gq >= 1 && gq <= 8 && gq * num_kv_heads == num_q_heads,
              "q/kv head ratio must be integral and <= 8");
  const int max_blocks = block_tables.size(1);
  if (nparts > 1) {
    TORCH_CHECK(part_out.scalar_type() == torch::kFloat32 &&
                part_out.numel() >=
                    (int64_t)num_seqs * num_kv_heads * nparts * gq *
                        (head_dim + 2),
                "part_out workspace too small");
  }
  arks_attn_decode_paged(out.data_ptr(),
                         nparts > 1 ? part_out.data_ptr() : nullptr,
                         q.data_ptr(), k_cache.data_ptr(), v_cache.data_ptr(),
                         block_tables.data_ptr(), seq_lens.data_ptr(),
                         (float)scale, num_seqs, num_q_heads, num_kv_heads,
                         head_dim, max_blocks, (int)nparts, q.stride(0),
                         kv_fp8 ? 1 : 0, (int)window, current_stream());
}

void attention_prefill_varlen(torch::Tensor out, torch::Tensor q,
                              torch::Tensor k, torch::Tensor v,
                              torch::Tensor cu_seqlens, torch::Tensor tile_info,
                              double scale, int64_t window) {
  check_bf16_contig(out, "out");
  check_bf16_rowstrided(q, "q");
  check_bf16_rowstrided(k, "k");
  check_bf16_rowstrided(v, "v");
  TORCH_CHECK(k.stride(0) == v.stride(0), "k/v must share row stride");
  TORCH_CHECK(cu_seqlens.scalar_type() == torch::kInt32);
  TORCH_CHECK(tile_info.scalar_type() == torch::kInt32);
  TORCH_CHECK(tile_info.dim() == 2 && tile_info.size(1) == 2);
  const int num_q_heads = q.size(1);
  const int head_dim = q.size(2);
  const int num_kv_heads = k.size(1);
  TORCH_CHECK(head_dim == 64 || head_dim == 128, "head_dim must be 64 or 128");
  const int ntiles = tile_info.size(0);
  arks_attn_prefill_varlen(out.data_ptr(), q.data_ptr(), k.data_ptr(),
                           v.data_ptr(), cu_seqlens.data_ptr(),
                           tile_info.data_ptr(), ntiles, (float)scale,
                           num_q_heads, num_kv_heads, head_dim, q.stride(0),
                           k.stride(0), (int)window, current_stream());
}

void attention_extend_paged(torch::Tensor out, torch::Tensor q,
                            torch::Tensor k_cache, torch::Tensor v_cache,
                            torch::Tensor block_tables, torch::Tensor kv_lens,
                            torch::Tensor cu_seqlens_q, torch::Tensor tile_info,
                            double scale, int64_t window) {
  check_bf16_contig(out, "out");
  check_bf16_rowstrided(q, "q");
  const bool kv_fp8 = check_kv_cache(k_cache, "k_cache");
  check_kv_cache(v_cache, "v_cache");
  TORCH_CHECK(block_tables.scalar_type() == torch::kInt32);
  TORCH_CHECK(kv_lens.scalar_type() == torch::kInt32);
  TORCH_CHECK(cu_seqlens_q.scalar_type() == torch::kInt32);
  TORCH_CHECK(tile_info.scalar_type() == torch::kInt32);
  TORCH_CHECK(tile_info.dim() == 2 && tile_info.size(1) == 2);
  const int num_q_heads = q.size(1);
  const int head_dim = q.size(2);
  const int num_kv_heads = k_cache.size(1);
  TORCH_CHECK(k_cache.size(2) == 16, "KV block size must be 16");
  TORCH_CHECK(head_dim == 64 || head_dim == 128, "head_dim must be 64 or 128");
  const int ntiles = tile_info.size(0);
  const int max_blocks = block_tables.size(1);
  arks_attn_extend_paged(out.data_ptr(), q.data_ptr(), k_cache.data_ptr(),
                         v_cache.data_ptr(), block_tables.data_ptr(),
                         kv_lens.data_ptr(), cu_seqlens_q.data_ptr(),
                         tile_info.data_ptr(), ntiles, (float)scale,
                         num_q_heads, num_kv_heads, head_dim, max_blocks,
                         q.stride(0), kv_fp8 ? 1 : 0, (int)window,
                         current_stream());
}

// 8-wave 32x32-MFMA ladder (attn_extend2.hip): 256-row q tiles, bf16 KV.
void attention_extend_paged2(torch::Tensor out, torch::Tensor q,
                             torch::Tensor k_cache, torch::Tensor v_cache,
                             torch::Tensor block_tables, torch::Tensor kv_lens,
                             torch::Tensor cu_seqlens_q,
                             torch::Tensor tile_info, torch::Tensor part_ws,
                             double scale, int64_t window) {
  check_bf16_contig(out, "out");
  check_bf16_rowstrided(q, "q");
  check_bf16_contig(k_cache, "k_cache");
  check_bf16_contig(v_cache, "v_cache");
  TORCH_CHECK(block_tables.scalar_type() == torch::kInt32);
  TORCH_CHECK(kv_lens.scalar_type() == torch::kInt32);
  TORCH_CHECK(cu_seqlens_q.scalar_type() == torch::kInt32);
  TORCH_CHECK(tile_info.scalar_type() == torch::kInt32);
  TORCH_CHECK(tile_info.dim() == 2 && tile_info.size(1) == 4,
              "tile_info rows are (seq, q0, part, nparts)");
  const int num_q_heads = q.size(1);
  const int head_dim = q.size(2);
  const int num_kv_heads = k_cache.size(1);
  TORCH_CHECK(k_cache.size(2) == 16, "KV block size must be 16");
  TORCH_CHECK(head_dim == 64 || head_dim == 128, "head_dim must be 64 or 128");
  const int ntiles = tile_info.size(0);
  const int max_blocks = block_tables.size(1);
  void* ws = nullptr;
  if (part_ws.numel() > 0) {
    // sized by the host: split rows sort first, so only their slabs exist
    TORCH_CHECK(part_ws.scalar_type() == torch::kFloat32);
    ws = part_ws.data_ptr();
  }
  arks_attn_extend_paged2(out.data_ptr(), q.data_ptr(), k_cache.data_ptr(),
                          v_cache.data_ptr(), block_tables.data_ptr(),
                          kv_lens.data_ptr(), cu_seqlens_q.data_ptr(),
                          tile_info.data_ptr(), ws, ntiles, (float)scale,
                          num_q_heads, num_kv_heads, head_dim, max_blocks,
                          q.stride(0), (int)window, current_stream());
}

void attention_extend2_combine(torch::Tensor out, torch::Tensor part_ws,
                               torch::Tensor combine_table,
                               torch::Tensor cu_seqlens_q, double scale) {
  check_bf16_rowstrided(out, "out");
  TORCH_CHECK(part_ws.scalar_type() == torch::kFloat32);
  TORCH_CHECK(combine_table.scalar_type() == torch::kInt32 &&
              combine_table.dim() == 2 && combine_table.size(1) == 4);
  const int num_q_heads = out.size(1);
  const int head_dim = out.size(2);
  arks_attn_extend2_combine(out.data_ptr(), part_ws.data_ptr(),
                            combine_table.data_ptr(), cu_seqlens_q.data_ptr(),
                            combine_table.size(0), (float)scale, num_q_heads,
                            head_dim, current_stream());
}

void skinny_gemm(torch::Tensor out, torch::Tensor part, torch::Tensor a,
                 torch::Tensor w, c10::optional<torch::Tensor> bias,
                 int64_t k_per_split, int64_t nsplits) {
  check_bf16_contig(out, "out");
  check_bf16_rowstrided(a, "a");
  check_bf16_contig(w, "w");
  const int m = a.size(0), k = a.size(1), n = w.size(0);
  TORCH_CHECK(m >= 1 && m <= 64, "skinny_gemm needs 1 <= M <= 64");
  TORCH_CHECK(n % 64 == 0 && k % 32 == 0, "N%64==0 and K%32==0 required");
  TORCH_CHECK(w.size(1) == k && out.size(0) == m && out.size(1) == n);
  const void* bias_ptr = nullptr;
  if (bias.has_value()) {
    check_bf16_contig(*bias, "bias");
    TORCH_CHECK(bias->numel() == n);
    bias_ptr = bias->data_ptr();
  }
  const int mrows = ((m + 15) / 16) * 16;
  if (nsplits > 1) {
    TORCH_CHECK(part.scalar_type() == torch::kFloat32 &&
                part.numel() >= (int64_t)nsplits * n * mrows,
                "skinny_gemm workspace too small");
  }
  arks_skinny_gemm(nsplits > 1 ? part.data_ptr() : nullptr, out.data_ptr(),
                   a.data_ptr(), w.data_ptr(), bias_ptr, m, n, k,
                   (int)k_per_split, (int)nsplits, a.stride(0),
                   current_stream());
}

// Variant/fused entry (MT fixed at 4, i.e. M <= 64). fuse_silu: `a` is the
// gate_up output [M, >=2K] (gate cols [0,K), up cols [K,2K)).
void skinny_gemm_v(torch::Tensor out, torch::Tensor part, torch::Tensor a,
                   torch::Tensor w, c10::optional<torch::Tensor> bias,
                   int64_t k_per_split, int64_t nsplits, int64_t variant,
                   bool fuse_silu) {
  check_bf16_contig(out, "out");
  check_bf16_rowstrided(a, "a");
  check_bf16_contig(w, "w");
  const int m = a.size(0), k = w.size(1), n = w.size(0);
  TORCH_CHECK(m >= 1 && m <= 64, "skinny_gemm_v needs 1 <= M <= 64");
  TORCH_CHECK(n % 64 == 0 && k % 32 == 0, "N%64==0 and K%32==0 required");
  TORCH_CHECK(a.size(1) >= (fuse_silu ? 2 * k : k), "A too narrow");
  TORCH_CHECK(out.size(0) == m && out.size(1) == n);
  TORCH_CHECK(0 <= variant && variant <= 8, "variant in 0..8");
  const void* bias_ptr = nullptr;
  if (bias.has_value()) {
    check_bf16_contig(*bias, "bias");
    TORCH_CHECK(bias->numel() == n);
    bias_ptr = bias->data_ptr();
  }
  if (nsplits > 1) {
    TORCH_CHECK(part.scalar_type() == torch::kFloat32 &&
                part.numel() >= (int64_t)nsplits * n * 64,
                "skinny_gemm_v workspace too small");
  }
  arks_skinny_gemm_v(nsplits > 1 ? part.data_ptr() : nullptr, out.data_ptr(),
                     a.data_ptr(), w.data_ptr(), bias_ptr, m, n, k,
                     (int)k_per_split, (int)nsplits, a.stride(0),
                     (int)variant, fuse_silu, current_stream());
}

void rmsnorm_fp8(torch::Tensor out, torch::Tensor inv_scale,
                 torch::Tensor input, c10::optional<torch::Tensor> residual,
                 torch::Tensor weight, double eps) {
  check_bf16_contig(input, "input");
  check_bf16_contig(weight, "weight");
  TORCH_CHECK(out.scalar_type() == torch::kFloat8_e4m3fn && out.is_contiguous());
  TORCH_CHECK(inv_scale.scalar_type() == torch::kFloat32);
  const int hidden = input.size(-1);
  TORCH_CHECK(hidden % 8 == 0 && hidden <= 8192,
              "rmsnorm_fp8 requires hidden % 8 == 0 and hidden <= 8192");
  const int rows = input.numel() / hidden;
  void* res_ptr = nullptr;
  if (residual.has_value()) {
    check_bf16_contig(*residual, "residual");
    res_ptr = residual->data_ptr();
  }
  arks_rmsnorm_fp8(out.data_ptr(), inv_scale.data_ptr(), input.data_ptr(),
                   res_ptr, weight.data_ptr(), (float)eps, rows, hidden,
                   residual.has_value() ? 1 : 0, current_stream());
}

void silu_mul_fp8(torch::Tensor out, torch::Tensor inv_scale,
                  torch::Tensor gate_up) {
  check_bf16_contig(gate_up, "gate_up");
  TORCH_CHECK(out.scalar_type() == torch::kFloat8_e4m3fn && out.is_contiguous());
  TORCH_CHECK(inv_scale.scalar_type() == torch::kFloat32);
  const int d = out.size(-1);
  TORCH_CHECK(gate_up.size(-1) == 2 * d && d % 8 == 0);
  TORCH_CHECK((int64_t)d * 2 <= 160 * 1024, "row too large for LDS staging");
  const int rows = out.numel() / d;
  arks_silu_mul_fp8(out.data_ptr(), inv_scale.data_ptr(), gate_up.data_ptr(),
                    rows, d, current_stream());
}

void quant_fp8_rows(torch::Tensor out, torch::Tensor inv_scale,
                    torch::Tensor x) {
  check_bf16_contig(x, "x");
  TORCH_CHECK(out.scalar_type() == torch::kFloat8_e4m3fn && out.is_contiguous());
  TORCH_CHECK(inv_scale.scalar_type() == torch::kFloat32);
  TORCH_CHECK(x.size(1) % 8 == 0, "cols must be a multiple of 8");
  arks_quant_fp8_rows(out.data_ptr(), inv_scale.data_ptr(), x.data_ptr(),
                      x.size(0), x.size(1), current_stream());
}

void greedy_sample(torch::Tensor out, torch::Tensor logits) {
  check_bf16_contig(logits, "logits");
  TORCH_CHECK(out.scalar_type() == torch::kInt64);
  arks_greedy_sample(out.data_ptr(), logits.data_ptr(), logits.size(0),
                     logits.size(1), current_stream());
}

void gumbel_sample(torch::Tensor out, torch::Tensor logits,
                   torch::Tensor temperatures, torch::Tensor uniform) {
  check_bf16_contig(logits, "logits");
  TORCH_CHECK(out.scalar_type() == torch::kInt64);
  TORCH_CHECK(temperatures.scalar_type() == torch::kFloat32);
  TORCH_CHECK(uniform.scalar_type() == torch::kFloat32);
  arks_gumbel_sample(out.data_ptr(), logits.data_ptr(), temperatures.data_ptr(),
                     uniform.data_ptr(), logits.size(0), logits.size(1),
                     current_stream());
}

// D[16,16] = A[16,32] @ B[32,16], all bf16 in / f32 out. Verifies the MFMA
// fragment layout assumption on hardware.
void tr16_probe(torch::Tensor out, int64_t stride_bytes) {
  TORCH_CHECK(out.scalar_type() == torch::kInt16 && out.numel() >= 256);
  arks_tr16_probe(out.data_ptr(), (int)stride_bytes, current_stream());
}

void mfma_probe32(torch::Tensor d, torch::Tensor a, torch::Tensor b) {
  check_bf16_contig(a, "a");
  check_bf16_contig(b, "b");
  TORCH_CHECK(d.scalar_type() == torch::kFloat32 && d.is_contiguous());
  TORCH_CHECK(a.size(0) == 32 && a.size(1) == 16);
  TORCH_CHECK(b.size(0) == 16 && b.size(1) == 32);
  arks_mfma_probe32(d.data_ptr(), a.data_ptr(), b.data_ptr(), current_stream());
}

// --- p2p one-shot all-reduce plumbing (allreduce.hip) ---
// IPC-shared buffers are raw hipMalloc allocations: a handle for a torch
// caching-allocator tensor maps the UNDERLYING block, so the opened peer
// pointer would miss the tensor's offset inside it.
py::tuple ipc_alloc(int64_t nbytes) {
  void* ptr = nullptr;
  auto err = hipMalloc(&ptr, (size_t)nbytes);
  TORCH_CHECK(err == hipSuccess, "hipMalloc: ", hipGetErrorString(err));
  (void)hipMemset(ptr, 0, (size_t)nbytes);
  hipIpcMemHandle_t h;
  err = hipIpcGetMemHandle(&h, ptr);
  TORCH_CHECK(err == hipSuccess, "hipIpcGetMemHandle: ",
              hipGetErrorString(err));
  return py::make_tuple(
      reinterpret_cast<int64_t>(ptr),
      py::bytes(reinterpret_cast<const char*>(&h), sizeof(h)));
}

void ipc_alloc_free(int64_t ptr) {
  (void)hipFree(reinterpret_cast<void*>(ptr));
}

// test hook: arm a raw flag buffer with huge values (0x7f per byte) so the
// single-GPU sequential harness never spins
void arm_flags(int64_t ptr, int64_t n) {
  (void)hipMemset(reinterpret_cast<void*>(ptr), 0x7f, (size_t)n * 8);
}

py::bytes ipc_handle(torch::Tensor t) {
  TORCH_CHECK(t.is_cuda(), "ipc_handle wants a CUDA tensor");
  hipIpcMemHandle_t h;
  auto err = hipIpcGetMemHandle(&h, t.data_ptr());
  TORCH_CHECK(err == hipSuccess, "hipIpcGetMemHandle: ",
              hipGetErrorString(err));
  return py::bytes(reinterpret_cast<const char*>(&h), sizeof(h));
}

int64_t ipc_open(py::bytes handle) {
  std::string s = handle;
  TORCH_CHECK(s.size() == sizeof(hipIpcMemHandle_t), "bad handle size");
  hipIpcMemHandle_t h;
  memcpy(&h, s.data(), sizeof(h));
  void* ptr = nullptr;
  auto err = hipIpcOpenMemHandle(&ptr, h, hipIpcMemLazyEnablePeerAccess);
  TORCH_CHECK(err == hipSuccess, "hipIpcOpenMemHandle: ",
              hipGetErrorString(err));
  return reinterpret_cast<int64_t>(ptr);
}

void ipc_close(int64_t ptr) {
  (void)hipIpcCloseMemHandle(reinterpret_cast<void*>(ptr));
}

void one_shot_allreduce(torch::Tensor out, torch::Tensor src,
                        std::vector<int64_t> mail,
                        std::vector<int64_t> flags, torch::Tensor seq,
                        int64_t rank) {  // mail/flags: ipc_alloc'd raw ptrs
  check_bf16_contig(out, "out");
  check_bf16_contig(src, "src");
  TORCH_CHECK(seq.is_cuda() && seq.scalar_type() == torch::kInt64);
  const int world = (int)mail.size();
  TORCH_CHECK(world == (int)flags.size());
  TORCH_CHECK(world == 1 || world == 2 || world == 4 || world == 8,
              "world must be 1/2/4/8");
  const int64_t n = src.numel();
  TORCH_CHECK(n % 8 == 0, "numel must be a multiple of 8");
  TORCH_CHECK(out.numel() == n);
  void* mp[8] = {};
  void* fp[8] = {};
  for (int p = 0; p < world; ++p) {
    mp[p] = reinterpret_cast<void*>(mail[p]);
    fp[p] = reinterpret_cast<void*>(flags[p]);
  }
  auto stream = current_stream();
  arks_ar_seq_inc(seq.data_ptr(), stream);
  arks_one_shot_allreduce(out.data_ptr(), src.data_ptr(), mp, fp,
                          seq.data_ptr(), (int)rank, world, n, stream);
}

void moe_topk(torch::Tensor weights, torch::Tensor ids,
              torch::Tensor logits, int64_t k, int64_t renorm) {
  TORCH_CHECK(logits.is_cuda() && logits.is_contiguous() &&
              logits.scalar_type() == torch::kFloat32, "logits f32 contig");
  TORCH_CHECK(weights.scalar_type() == torch::kFloat32 &&
              ids.scalar_type() == torch::kInt32);
  const int T = logits.size(0), E = logits.size(1);
  TORCH_CHECK(k >= 1 && k <= 16 && E <= 1024 && k <= E,
              "need 1 <= k <= min(16, E), E <= 1024");
  arks_moe_topk(weights.data_ptr(), ids.data_ptr(), logits.data_ptr(), T, E,
                (int)k, (int)renorm, current_stream());
}

void moe_mix(torch::Tensor out, torch::Tensor y, torch::Tensor weights,
             torch::Tensor ids, int64_t expert_base, int64_t n_local) {
  check_bf16_contig(out, "out");
  check_bf16_contig(y, "y");
  TORCH_CHECK(weights.scalar_type() == torch::kFloat32 &&
              ids.scalar_type() == torch::kInt32);
  const int T = out.size(0), H = out.size(1);
  const int k = weights.size(1);
  TORCH_CHECK(k <= 16 && H % 2 == 0);
  TORCH_CHECK(y.size(1) == T && y.size(2) == H && y.size(0) == n_local);
  arks_moe_mix(out.data_ptr(), y.data_ptr(), weights.data_ptr(),
               ids.data_ptr(), T, H, k, (int)expert_base, (int)n_local,
               current_stream());
}

void moe_mix_rows(torch::Tensor out, torch::Tensor y,
                  torch::Tensor weights, torch::Tensor rows) {
  check_bf16_contig(out, "out");
  check_bf16_contig(y, "y");
  TORCH_CHECK(weights.scalar_type() == torch::kFloat32 &&
              rows.scalar_type() == torch::kInt32);
  const int T = out.size(0), H = out.size(1);
  const int k = weights.size(1);
  TORCH_CHECK(k <= 16 && H % 2 == 0);
  TORCH_CHECK(rows.size(0) == T && rows.size(1) == k);
  arks_moe_mix_rows(out.data_ptr(), y.data_ptr(), weights.data_ptr(),
                    rows.data_ptr(), T, H, k, current_stream());
}

void mfma_probe(torch::Tensor d, torch::Tensor a, torch::Tensor b) {
  check_bf16_contig(a, "a");
  check_bf16_contig(b, "b");
  TORCH_CHECK(d.scalar_type() == torch::kFloat32 && d.is_contiguous());
  TORCH_CHECK(a.size(0) == 16 && a.size(1) == 32);
  TORCH_CHECK(b.size(0) == 32 && b.size(1) == 16);
  arks_mfma_probe(d.data_ptr(), a.data_ptr(), b.data_ptr(), current_stream());
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rmsnorm", &rmsnorm);
  m.def("fused_add_rmsnorm", &fused_add_rmsnorm);
  m.def("silu_mul", &silu_mul);
  m.def("rope_inplace", &rope_inplace);
  m.def("reshape_and_cache", &reshape_and_cache);
  m.def("rope_and_cache", &rope_and_cache);
  m.def("attention_decode_paged", &attention_decode_paged);
  m.def("attention_prefill_varlen", &attention_prefill_varlen);
  m.def("attention_extend_paged", &attention_extend_paged);
  m.def("attention_extend_paged2", &attention_extend_paged2);
  m.def("attention_extend2_combine", &attention_extend2_combine);
  m.def("quant_fp8_rows", &quant_fp8_rows);
  m.def("rmsnorm_fp8", &rmsnorm_fp8);
  m.def("silu_mul_fp8", &silu_mul_fp8);
  m.def("skinny_gemm", &skinny_gemm);
  m.def("skinny_gemm_v", &skinny_gemm_v);
  m.def("greedy_sample", &greedy_sample);
  m.def("gumbel_sample", &gumbel_sample);
  m.def("ipc_alloc", &ipc_alloc);
  m.def("arm_flags", &arm_flags);
  m.def("ipc_alloc_free", &ipc_alloc_free);
  m.def("ipc_handle", &ipc_handle);
  m.def("ipc_open", &ipc_open);
  m.def("ipc_close", &ipc_close);
  m.def("one_shot_allreduce", &one_shot_allreduce);
  m.def("moe_topk", &moe_topk);
  m.def("moe_mix", &moe_mix);
  m.def("moe_mix_rows", &moe_mix_rows);
  m.def("mfma_probe", &mfma_probe);
  m.def("mfma_probe32", &mfma_probe32);
  m.def("tr16_probe", &tr16_probe);
}
