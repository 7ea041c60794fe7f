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

namespace arks {

// ---------------------------------------------------------------------------
// Fused RoPE + cache write: rotate q/k in place AND scatter the rotated k
// (plus v) into the paged cache in one launch — the separate
// rope_kernel + reshape_and_cache pair cost a kernel boundary and a full
// re-read of k every layer on the decode path. The rotating thread already
// holds the rotated k values in registers, so the cache store is free.
// Thread space: [q heads | kv heads] x pair-couples for the rotation, then
// the v copy strided over the same block.
// ---------------------------------------------------------------------------
template <bool FP8>
__global__ void rope_cache_kernel(
    const int64_t* __restrict__ positions,
    bf16* __restrict__ q,        // [T, Hq*D] rows strided
    bf16* __restrict__ k,        // [T, Hkv*D] rows strided
    const bf16* __restrict__ v,  // [T, Hkv*D] rows strided (k_stride)
    void* __restrict__ k_cache,  // [B, Hkv, block_size, D] bf16 or e4m3
    void* __restrict__ v_cache,
    const int64_t* __restrict__ slot_mapping,  // [T] (-1 = no cache write)
    const float* __restrict__ cos_sin,         // [max_pos, D]
    const int head_dim, const int num_q_heads, const int num_kv_heads,
    const int block_size, const int64_t q_stride, const int64_t k_stride) {
  const int token = blockIdx.x;
  const int half = head_dim / 2;
  const int64_t pos = positions[token];
  const float* cs = cos_sin + pos * head_dim;
  const int64_t slot = slot_mapping[token];
  const bool cache = slot >= 0;
  const int64_t block_id = cache ? slot / block_size : 0;
  const int offset = cache ? (int)(slot % block_size) : 0;
  const int64_t cbase =
      (block_id * num_kv_heads) * (int64_t)block_size * head_dim +
      (int64_t)offset * head_dim;
  const int64_t hstride = (int64_t)block_size * head_dim;

  const int total_heads = num_q_heads + num_kv_heads;
  const int pairs2 = half / 2;
  for (int idx = threadIdx.x; idx < total_heads * pairs2; idx += blockDim.x) {
    const int h = idx / pairs2;
    const int p2 = (idx % pairs2) * 2;
    const bool is_q = h < num_q_heads;
    bf16* base = is_q
                     ? q + (int64_t)token * q_stride + (int64_t)h * head_dim
                     : k + (int64_t)token * k_stride +
                           (int64_t)(h - num_q_heads) * head_dim;
    ushort2v x1 = *reinterpret_cast<const ushort2v*>(base + p2);
    ushort2v x2 = *reinterpret_cast<const ushort2v*>(base + half + p2);
    ushort2v o1, o2;
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      const float c = cs[p2 + j];
      const float s = cs[half + p2 + j];
      const float a = bf16_bits_to_float(x1[j]);
      const float b = bf16_bits_to_float(x2[j]);
      o1[j] = float_to_bf16_bits(a * c - b * s);
      o2[j] = float_to_bf16_bits(b * c + a * s);
    }
    *reinterpret_cast<ushort2v*>(base + p2) = o1;
    *reinterpret_cast<ushort2v*>(base + half + p2) = o2;
    if (!is_q && cache) {
      const int64_t dst = cbase + (int64_t)(h - num_q_heads) * hstride;
      if constexpr (FP8) {
        uint8_t* kc = reinterpret_cast<uint8_t*>(k_cache);
#pragma unroll
        for (int j = 0; j < 2; ++j) {
          // convert from the bf16-ROUNDED values so the cache is
          // bit-identical to the separate rope + reshape_and_cache path
          __hip_fp8_e4m3 a(bf16_bits_to_float(o1[j]));
          __hip_fp8_e4m3 b(bf16_bits_to_float(o2[j]));
          kc[dst + p2 + j] = a.__x;
          kc[dst + half + p2 + j] = b.__x;
        }
      } else {
        bf16* kc = reinterpret_cast<bf16*>(k_cache);
        *reinterpret_cast<ushort2v*>(kc + dst + p2) = o1;
        *reinterpret_cast<ushort2v*>(kc + dst + half + p2) = o2;
      }
    }
  }
  if (!cache) return;
  // v: plain copy into the page (vectorized 8)
  const bf16* v_src = v + (int64_t)token * k_stride;
  const int nvec = num_kv_heads * head_dim / 8;
  for (int i = threadIdx.x; i < nvec; i += blockDim.x) {
    const int h = (i * 8) / head_dim;
    const int d = (i * 8) % head_dim;
    const int64_t dst = cbase + (int64_t)h * hstride + d;
    ushort8 vv8 = *reinterpret_cast<const ushort8*>(v_src + i * 8);
    if constexpr (FP8) {
      uchar8 vo;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        __hip_fp8_e4m3 vq(bf16_bits_to_float(vv8[e]));
        vo[e] = vq.__x;
      }
      *reinterpret_cast<uchar8*>(reinterpret_cast<uint8_t*>(v_cache) + dst) = vo;
    } else {
      *reinterpret_cast<ushort8*>(reinterpret_cast<bf16*>(v_cache) + dst) = vv8;
    }
  }
}

}  // namespace arks

extern "C" void arks_rope_and_cache(
    const void* positions, void* q, void* k, const void* v, void* k_cache,
    void* v_cache, const void* slot_mapping, const void* cos_sin,
    int num_tokens, int head_dim, int num_q_heads, int num_kv_heads,
    int block_size, int64_t q_stride, int64_t k_stride, int kv_fp8,
    hipStream_t stream) {
  if (num_tokens == 0) return;
  dim3 grid(num_tokens), block(256);
  if (kv_fp8) {
    hipLaunchKernelGGL((rope_cache_kernel<true>), grid, block, 0, stream,
                       (const int64_t*)positions, (bf16*)q, (bf16*)k,
                       (const bf16*)v, k_cache, v_cache,
                       (const int64_t*)slot_mapping, (const float*)cos_sin,
                       head_dim, num_q_heads, num_kv_heads, block_size,
                       q_stride, k_stride);
  } else {
    hipLaunchKernelGGL((rope_cache_kernel<false>), grid, block, 0, stream,
                       (const int64_t*)positions, (bf16*)q, (bf16*)k,
                       (const bf16*)v, k_cache, v_cache,
                       (const int64_t*)slot_mapping, (const float*)cos_sin,
                       head_dim, num_q_heads, num_kv_heads, block_size,
                       q_stride, k_stride);
  }
}
