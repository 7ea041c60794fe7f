// Paged "extend" attention on MFMA (gfx950): varlen causal attention where
// each sequence's NEW q tokens attend over its FULL KV history read from the
// paged cache (cached prefix + the new tokens, which reshape_and_cache wrote
// to their pages before this kernel runs on the same stream).
//
// This is the prefix-caching / chunked-prefill half of the runtime-slot
// contract (SGLang's radix-cache "extend" phase, delegated
// by the reference to external images at arksapplication_controller.go:956-969).
//
// Structure matches attn_prefill.hip: one workgroup = 4 waves = a 64-row
// Q tile of one (seq, q-head); K/V iterated in 32-key LDS tiles (K rows
// XOR-swizzled, V transposed). The only differences:
//   * K/V rows are gathered from [nblocks, Hkv, 16, D] cache pages via the
//     sequence's block table (two page lookups per 32-key tile);
//   * the causal bound is offset by the cached-prefix length:
//     q row r (index among the new tokens) sits at global position
//     kv_off + r, kv_off = kv_len - q_len, so keys kg <= kv_off + r are
//     visible.
#include "common.h"

#include <cfloat>
#include <hip/hip_fp8.h>

namespace arks {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) __bf16 ext_bf16x4v;
typedef __attribute__((ext_vector_type(4))) float f32x4;

__device__ __forceinline__ f32x4 ext_mfma16x16x32(bf16x8 a, bf16x8 b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

__device__ __forceinline__ int ext_frag_k(int a, int jj) { return 8 * a + jj; }

constexpr int EXT_QTILE_WAVE = 16;
constexpr int EXT_NUM_WAVES = 4;
constexpr int EXT_QTILE = EXT_QTILE_WAVE * EXT_NUM_WAVES;  // 64 q rows
constexpr int EXT_KTILE = 32;   // keys per LDS tile (= 2 KV pages)
constexpr int EXT_VT_PAD = 40;  // padded VT row length
constexpr int PAGE = 16;        // KV page (block) size in tokens

__device__ __forceinline__ uint16_t ext_fp8_to_bf16(uint8_t b) {
  __hip_fp8_e4m3 q;
  q.__x = b;
  return float_to_bf16_bits((float)q);
}

template <int HEAD_DIM, bool KV_FP8>
__global__ __launch_bounds__(256) void attn_extend_kernel(
    bf16* __restrict__ out,            // [Tq, Hq, D]
    const bf16* __restrict__ q,        // [Tq, Hq, D] (packed new tokens)
    const bf16* __restrict__ k_cache,  // [nblocks, Hkv, 16, D]
    const bf16* __restrict__ v_cache,  // [nblocks, Hkv, 16, D]
    const int* __restrict__ block_tables,  // [num_seqs, max_blocks]
    const int* __restrict__ kv_lens,       // [num_seqs] total kv per seq
    const int* __restrict__ cu_seqlens_q,  // [num_seqs + 1]
    const int* __restrict__ tile_info,     // [ntiles, 2] = (seq_idx, q0)
    const float scale, const int num_q_heads, const int num_kv_heads,
    const int max_blocks, const int64_t q_stride, const int window) {
  constexpr int CHUNKS = HEAD_DIM / 16;
  constexpr int STEPS = HEAD_DIM / 32;

  const int h = blockIdx.x;
  const int kvh = h / (num_q_heads / num_kv_heads);
  const int seq_idx = tile_info[blockIdx.y * 2];
  const int q0 = tile_info[blockIdx.y * 2 + 1];
  const int q_start = cu_seqlens_q[seq_idx];
  const int q_len = cu_seqlens_q[seq_idx + 1] - q_start;
  const int kv_len = kv_lens[seq_idx];
  const int kv_off = kv_len - q_len;  // cached-prefix length
  const int* __restrict__ bt = block_tables + (int64_t)seq_idx * max_blocks;

  const int tid = threadIdx.x;
  const int wave = tid / WAVE_SIZE;
  const int lane = tid % WAVE_SIZE;
  const int lq = lane % 16;
  const int la = lane / 16;

  __shared__ __attribute__((aligned(16))) bf16 k_lds[EXT_KTILE][HEAD_DIM];
  // V as a tr16 image (see attn_prefill.hip): [cb][kb][4 key x 16 col]
  // 96-elem blocks; staged with b128 stores, consumed with
  // ds_read_b64_tr_b16 B-fragments.
  __shared__ __attribute__((aligned(16))) bf16 vt_img[HEAD_DIM / 16][8][96];
  __shared__ __attribute__((aligned(16)))
      bf16 p_lds[EXT_NUM_WAVES][EXT_QTILE_WAVE][EXT_VT_PAD];

  // Q fragments: registers, loaded once. Wave w covers rows q0+16w..+15.
  const int my_qrow = q0 + wave * EXT_QTILE_WAVE + lq;
  const bool qrow_valid = my_qrow < q_len;
  bf16x8 qfrag[STEPS];
  {
    const int64_t qbase =
        (int64_t)(q_start + (qrow_valid ? my_qrow : 0)) * q_stride +
        (int64_t)h * HEAD_DIM;
#pragma unroll
    for (int st = 0; st < STEPS; ++st) {
      ushort8 u = *reinterpret_cast<const ushort8*>(q + qbase + st * 32 +
                                                    ext_frag_k(la, 0));
      qfrag[st] = *reinterpret_cast<bf16x8*>(&u);
    }
  }

  float m_run = -FLT_MAX;
  float l_run = 0.f;
  f32x4 oacc[CHUNKS];
#pragma unroll
  for (int c = 0; c < CHUNKS; ++c) oacc[c] = {0.f, 0.f, 0.f, 0.f};

  // Keys visible to this workgroup: strictly below kmax.
  const int kmax = min(kv_len, kv_off + q0 + EXT_QTILE);
  const int ntiles = (kmax + EXT_KTILE - 1) / EXT_KTILE;
  // Sliding window: start at the first tile any row of this q-tile sees.
  const int j0 =
      (window > 0) ? max(0, (kv_off + q0 - window + 1) / EXT_KTILE) : 0;

  for (int j = j0; j < ntiles; ++j) {
    const int key_base = j * EXT_KTILE;
    // Page base offsets for this tile's two 16-key pages.
    const int64_t page_elems = (int64_t)num_kv_heads * PAGE * HEAD_DIM;
    // window-dropped pages carry block-table entry -1; their tokens are
    // masked, so the page id is sanitized to 0
    const int pg0 = max(0, bt[key_base / PAGE]);
    const int pg1 =
        (key_base + PAGE < kmax) ? max(0, bt[key_base / PAGE + 1]) : pg0;
    const int64_t pbase0 =
        (int64_t)pg0 * page_elems + (int64_t)kvh * PAGE * HEAD_DIM;
    const int64_t pbase1 =
        (int64_t)pg1 * page_elems + (int64_t)kvh * PAGE * HEAD_DIM;
    {
      const int nvec = EXT_KTILE * HEAD_DIM / 8;
      for (int i = tid; i < nvec; i += 256) {
        const int key = i / (HEAD_DIM / 8);
        const int col8 = (i % (HEAD_DIM / 8)) * 8;
        const int kg = key_base + key;
        ushort8 kv{}, vv{};
        if (kg < kmax) {
          const int64_t src = (key < PAGE ? pbase0 : pbase1) +
                              (int64_t)(kg % PAGE) * HEAD_DIM + col8;
          if constexpr (KV_FP8) {
            uchar8 k8 = *reinterpret_cast<const uchar8*>(
                reinterpret_cast<const uint8_t*>(k_cache) + src);
            uchar8 v8 = *reinterpret_cast<const uchar8*>(
                reinterpret_cast<const uint8_t*>(v_cache) + src);
#pragma unroll
            for (int e = 0; e < 8; ++e) {
              kv[e] = ext_fp8_to_bf16(k8[e]);
              vv[e] = ext_fp8_to_bf16(v8[e]);
            }
          } else {
            kv = *reinterpret_cast<const ushort8*>(k_cache + src);
            vv = *reinterpret_cast<const ushort8*>(v_cache + src);
          }
        }
        const int row_byte = col8 * 2;
        const int swz = row_byte ^ ((key & 7) << 4);
        *reinterpret_cast<ushort8*>(reinterpret_cast<char*>(&k_lds[key][0]) +
                                    swz) = kv;
        *reinterpret_cast<ushort8*>(
            &vt_img[col8 >> 4][key >> 2][(key & 3) * 16 + (col8 & 15)]) = vv;
      }
    }
    __syncthreads();

    // QK^T: A = K from LDS, B = Q registers (see attn_prefill.hip).
    f32x4 sc[2];
    sc[0] = {0.f, 0.f, 0.f, 0.f};
    sc[1] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int sub = 0; sub < 2; ++sub) {
      const int key = sub * 16 + lq;
#pragma unroll
      for (int st = 0; st < STEPS; ++st) {
        const int col_byte = (st * 32 + ext_frag_k(la, 0)) * 2;
        const int swz = col_byte ^ ((key & 7) << 4);
        ushort8 u = *reinterpret_cast<const ushort8*>(
            reinterpret_cast<const char*>(&k_lds[key][0]) + swz);
        sc[sub] =
            ext_mfma16x16x32(*reinterpret_cast<bf16x8*>(&u), qfrag[st], sc[sub]);
      }
    }

    // Masked online softmax; causal bound offset by the cached prefix.
    float p[8];
    float tile_max = -FLT_MAX;
    bool msk[8];
#pragma unroll
    for (int sub = 0; sub < 2; ++sub) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int kg = key_base + sub * 16 + 4 * la + r;
        const int i = sub * 4 + r;
        msk[i] = qrow_valid && (kg <= kv_off + my_qrow) && (kg < kmax) &&
                 (window <= 0 || kg > kv_off + my_qrow - window);
        p[i] = msk[i] ? sc[sub][r] * scale : -FLT_MAX;
        tile_max = fmaxf(tile_max, p[i]);
      }
    }
    tile_max = fmaxf(tile_max, __shfl_xor(tile_max, 16, WAVE_SIZE));
    tile_max = fmaxf(tile_max, __shfl_xor(tile_max, 32, WAVE_SIZE));

    const float m_new = fmaxf(m_run, tile_max);
    float alpha = 1.f;
    float psum = 0.f;
    if (m_new > -FLT_MAX) {
      alpha = __expf(m_run - m_new);
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        p[i] = msk[i] ? __expf(p[i] - m_new) : 0.f;
        psum += p[i];
      }
    } else {
#pragma unroll
      for (int i = 0; i < 8; ++i) p[i] = 0.f;
    }
    psum += __shfl_xor(psum, 16, WAVE_SIZE);
    psum += __shfl_xor(psum, 32, WAVE_SIZE);
    l_run = l_run * alpha + psum;
    m_run = m_new;

    float row_alpha[4];
#pragma unroll
    for (int r = 0; r < 4; ++r)
      row_alpha[r] = __shfl(alpha, 4 * la + r, WAVE_SIZE);
#pragma unroll
    for (int c = 0; c < CHUNKS; ++c) {
#pragma unroll
      for (int r = 0; r < 4; ++r) oacc[c][r] *= row_alpha[r];
    }

#pragma unroll
    for (int sub = 0; sub < 2; ++sub) {
      ushort4v pk;
#pragma unroll
      for (int r = 0; r < 4; ++r) pk[r] = float_to_bf16_bits(p[sub * 4 + r]);
      *reinterpret_cast<ushort4v*>(&p_lds[wave][lq][sub * 16 + 4 * la]) = pk;
    }
    __syncthreads();

    ushort8 pa =
        *reinterpret_cast<const ushort8*>(&p_lds[wave][lq][ext_frag_k(la, 0)]);
#pragma unroll
    for (int c = 0; c < CHUNKS; ++c) {
      ext_bf16x4v v1 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
          (__attribute__((address_space(3))) ext_bf16x4v*)(
              reinterpret_cast<char*>(&vt_img[c][2 * la][0]) + lq * 8));
      ext_bf16x4v v2 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
          (__attribute__((address_space(3))) ext_bf16x4v*)(
              reinterpret_cast<char*>(&vt_img[c][2 * la + 1][0]) + lq * 8));
      bf16x8 vb;
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        vb[e] = v1[e];
        vb[e + 4] = v2[e];
      }
      oacc[c] = ext_mfma16x16x32(*reinterpret_cast<bf16x8*>(&pa), vb, oacc[c]);
    }
    __syncthreads();
  }

  float row_inv[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const float lr = __shfl(l_run, 4 * la + r, WAVE_SIZE);
    row_inv[r] = lr > 0.f ? 1.f / lr : 0.f;
  }
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int row = q0 + wave * EXT_QTILE_WAVE + 4 * la + r;
    if (row >= q_len) continue;
    const int64_t obase = ((int64_t)(q_start + row) * num_q_heads + h) * HEAD_DIM;
#pragma unroll
    for (int c = 0; c < CHUNKS; ++c) {
      out[obase + c * 16 + lq] = float_to_bf16_bits(oacc[c][r] * row_inv[r]);
    }
  }
}

}  // namespace arks

using namespace arks;

extern "C" void arks_attn_extend_paged(
    void* out, const void* q, const void* k_cache, const void* v_cache,
    const void* block_tables, const void* kv_lens, const void* cu_seqlens_q,
    const void* tile_info, int ntiles, float scale, int num_q_heads,
    int num_kv_heads, int head_dim, int max_blocks, int64_t q_stride,
    int kv_fp8, int window, hipStream_t stream) {
  dim3 grid(num_q_heads, ntiles), block(256);
  if (head_dim == 128 && kv_fp8) {
    hipLaunchKernelGGL((attn_extend_kernel<128, true>), grid, block, 0, stream,
                       (bf16*)out, (const bf16*)q, (const bf16*)k_cache,
                       (const bf16*)v_cache, (const int*)block_tables,
                       (const int*)kv_lens, (const int*)cu_seqlens_q,
                       (const int*)tile_info, scale, num_q_heads, num_kv_heads,
                       max_blocks, q_stride, window);
  } else if (head_dim == 64 && kv_fp8) {
    hipLaunchKernelGGL((attn_extend_kernel<64, true>), grid, block, 0, stream,
                       (bf16*)out, (const bf16*)q, (const bf16*)k_cache,
                       (const bf16*)v_cache, (const int*)block_tables,
                       (const int*)kv_lens, (const int*)cu_seqlens_q,
                       (const int*)tile_info, scale, num_q_heads, num_kv_heads,
                       max_blocks, q_stride, window);
  } else if (head_dim == 128) {
    hipLaunchKernelGGL((attn_extend_kernel<128, false>), grid, block, 0, stream,
                       (bf16*)out, (const bf16*)q, (const bf16*)k_cache,
                       (const bf16*)v_cache, (const int*)block_tables,
                       (const int*)kv_lens, (const int*)cu_seqlens_q,
                       (const int*)tile_info, scale, num_q_heads, num_kv_heads,
                       max_blocks, q_stride, window);
  } else if (head_dim == 64) {
    hipLaunchKernelGGL((attn_extend_kernel<64, false>), grid, block, 0, stream,
                       (bf16*)out, (const bf16*)q, (const bf16*)k_cache,
                       (const bf16*)v_cache, (const int*)block_tables,
                       (const int*)kv_lens, (const int*)cu_seqlens_q,
                       (const int*)tile_info, scale, num_q_heads, num_kv_heads,
                       max_blocks, q_stride, window);
  }
}
