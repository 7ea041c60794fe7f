// Paged KV-cache scatter: write the new tokens' K/V into their assigned slots.
// Cache layout: [num_blocks, num_kv_heads, block_size, head_dim] bf16
// slot = block_id * block_size + offset  (flat, per token)
#include "common.h"

#include <hip/hip_fp8.h>

namespace arks {

__global__ void reshape_and_cache_kernel(
    const bf16* __restrict__ k,  // [T, Hkv, D], token rows strided
    const bf16* __restrict__ v,
    bf16* __restrict__ k_cache,  // [B, Hkv, block_size, D]
    bf16* __restrict__ v_cache,
    const int64_t* __restrict__ slot_mapping,  // [T]
    const int num_kv_heads, const int head_dim, const int block_size,
    const int64_t kv_stride) {
  const int token = blockIdx.x;
  const int64_t slot = slot_mapping[token];
  if (slot < 0) return;  // padding slot
  const int64_t block_id = slot / block_size;
  const int offset = (int)(slot % block_size);
  const int nvec = num_kv_heads * head_dim / 8;
  const bf16* k_src = k + (int64_t)token * kv_stride;
  const bf16* v_src = v + (int64_t)token * kv_stride;
  for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
    const int h = (i * 8) / head_dim;
    const int d = (i * 8) % head_dim;
    const int64_t dst = ((block_id * num_kv_heads + h) * block_size + offset) *
                            head_dim + d;
    *reinterpret_cast<ushort8*>(k_cache + dst) =
        *reinterpret_cast<const ushort8*>(k_src + i * 8);
    *reinterpret_cast<ushort8*>(v_cache + dst) =
        *reinterpret_cast<const ushort8*>(v_src + i * 8);
  }
}

__global__ void reshape_and_cache_fp8_kernel(
    const bf16* __restrict__ k,  // [T, Hkv, D], token rows strided
    const bf16* __restrict__ v,
    uint8_t* __restrict__ k_cache,  // [B, Hkv, block_size, D] e4m3
    uint8_t* __restrict__ v_cache,
    const int64_t* __restrict__ slot_mapping,  // [T]
    const int num_kv_heads, const int head_dim, const int block_size,
    const int64_t kv_stride) {
  const int token = blockIdx.x;
  const int64_t slot = slot_mapping[token];
  if (slot < 0) return;  // padding slot
  const int64_t block_id = slot / block_size;
  const int offset = (int)(slot % block_size);
  const int nvec = num_kv_heads * head_dim / 8;
  const bf16* k_src = k + (int64_t)token * kv_stride;
  const bf16* v_src = v + (int64_t)token * kv_stride;
  for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
    const int h = (i * 8) / head_dim;
    const int d = (i * 8) % head_dim;
    const int64_t dst = ((block_id * num_kv_heads + h) * block_size + offset) *
                            head_dim + d;
    ushort8 kv8 = *reinterpret_cast<const ushort8*>(k_src + i * 8);
    ushort8 vv8 = *reinterpret_cast<const ushort8*>(v_src + i * 8);
    uchar8 ko, vo;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      __hip_fp8_e4m3 kq(bf16_bits_to_float(kv8[e]));
      __hip_fp8_e4m3 vq(bf16_bits_to_float(vv8[e]));
      ko[e] = kq.__x;
      vo[e] = vq.__x;
    }
    *reinterpret_cast<uchar8*>(k_cache + dst) = ko;
    *reinterpret_cast<uchar8*>(v_cache + dst) = vo;
  }
}

}  // namespace arks

using namespace arks;

extern "C" void arks_reshape_and_cache_fp8(
    const void* k, const void* v, void* k_cache, void* v_cache,
    const void* slot_mapping, int num_tokens, int num_kv_heads, int head_dim,
    int block_size, int64_t kv_stride, hipStream_t stream) {
  if (num_tokens == 0) return;
  int threads = std::min(256, num_kv_heads * head_dim / 8);
  hipLaunchKernelGGL(reshape_and_cache_fp8_kernel, dim3(num_tokens),
                     dim3(threads), 0, stream, (const bf16*)k, (const bf16*)v,
                     (uint8_t*)k_cache, (uint8_t*)v_cache,
                     (const int64_t*)slot_mapping, num_kv_heads, head_dim,
                     block_size, kv_stride);
}

extern "C" void arks_reshape_and_cache(const void* k, const void* v,
                                       void* k_cache, void* v_cache,
                                       const void* slot_mapping, int num_tokens,
                                       int num_kv_heads, int head_dim,
                                       int block_size, int64_t kv_stride,
                                       hipStream_t stream) {
  if (num_tokens == 0) return;
  int threads = std::min(256, num_kv_heads * head_dim / 8);
  hipLaunchKernelGGL(reshape_and_cache_kernel, dim3(num_tokens), dim3(threads),
                     0, stream, (const bf16*)k, (const bf16*)v, (bf16*)k_cache,
                     (bf16*)v_cache, (const int64_t*)slot_mapping, num_kv_heads,
                     head_dim, block_size, kv_stride);
}
