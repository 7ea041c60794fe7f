// Elementwise / normalization kernels for the decode hot path (bf16, gfx950).
// All memory-bound: vectorized 16 B/lane loads (guide G13), grid-stride, f32
// accumulation. Ops: rmsnorm, fused residual-add + rmsnorm, silu_mul (SwiGLU),
// neox-style RoPE (in-place on q and k).
#include "common.h"

namespace arks {

// ---------------------------------------------------------------------------
// RMSNorm: one workgroup (256 threads = 4 waves) per row.
// hidden must be a multiple of 8 (bf16x8 vector loads).
// ---------------------------------------------------------------------------
template <bool FUSED_ADD>
__global__ void rmsnorm_kernel(bf16* __restrict__ out,
                               const bf16* __restrict__ input,
                               bf16* __restrict__ residual,  // null unless FUSED_ADD
                               const bf16* __restrict__ weight,
                               const float eps, const int hidden) {
  constexpr int BLOCK = 256;
  const int row = blockIdx.x;
  const bf16* in_row = input + (int64_t)row * hidden;
  bf16* res_row = FUSED_ADD ? residual + (int64_t)row * hidden : nullptr;
  bf16* out_row = out + (int64_t)row * hidden;

  const int nvec = hidden / 8;
  float ss = 0.f;
  // Pass 1: (optionally add residual and store it), accumulate sum of squares.
  for (int i = threadIdx.x; i < nvec; i += BLOCK) {
    ushort8 x = *reinterpret_cast<const ushort8*>(in_row + i * 8);
    if constexpr (FUSED_ADD) {
      ushort8 r = *reinterpret_cast<const ushort8*>(res_row + i * 8);
      ushort8 s;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float v = bf16_bits_to_float(x[j]) + bf16_bits_to_float(r[j]);
        s[j] = float_to_bf16_bits(v);
        // Accumulate on the bf16-rounded value so pass 2's re-read matches.
        float vb = bf16_bits_to_float(s[j]);
        ss += vb * vb;
      }
      *reinterpret_cast<ushort8*>(res_row + i * 8) = s;
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float v = bf16_bits_to_float(x[j]);
        ss += v * v;
      }
    }
  }

  __shared__ float red[BLOCK / WAVE_SIZE];
  float total = block_reduce_sum<BLOCK>(ss, red);
  const float rrms = rsqrtf(total / (float)hidden + eps);

  // Pass 2: normalize (re-read from L1/L2) and scale by weight.
  const bf16* src_row = FUSED_ADD ? res_row : in_row;
  for (int i = threadIdx.x; i < nvec; i += BLOCK) {
    ushort8 x = *reinterpret_cast<const ushort8*>(src_row + i * 8);
    ushort8 w = *reinterpret_cast<const ushort8*>(weight + i * 8);
    ushort8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float v = bf16_bits_to_float(x[j]) * rrms * bf16_bits_to_float(w[j]);
      o[j] = float_to_bf16_bits(v);
    }
    *reinterpret_cast<ushort8*>(out_row + i * 8) = o;
  }
}

// ---------------------------------------------------------------------------
// SiLU-mul: input [rows, 2*d] = [gate | up], output [rows, d].
// ---------------------------------------------------------------------------
__global__ void silu_mul_kernel(bf16* __restrict__ out,
                                const bf16* __restrict__ gate_up,
                                const int d, const int64_t total_vec) {
  // Grid-stride over row-major output in 8-element vectors.
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       idx < total_vec; idx += (int64_t)gridDim.x * blockDim.x) {
    const int64_t row = idx / (d / 8);
    const int col8 = (int)(idx % (d / 8)) * 8;
    const bf16* g = gate_up + row * (2 * (int64_t)d) + col8;
    const bf16* u = g + d;
    ushort8 gv = *reinterpret_cast<const ushort8*>(g);
    ushort8 uv = *reinterpret_cast<const ushort8*>(u);
    ushort8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float x = bf16_bits_to_float(gv[j]);
      float silu = x / (1.f + __expf(-x));
      o[j] = float_to_bf16_bits(silu * bf16_bits_to_float(uv[j]));
    }
    *reinterpret_cast<ushort8*>(out + row * d + col8) = o;
  }
}

// ---------------------------------------------------------------------------
// RoPE (neox / rotate-half), in place on q [T, Hq*D] and k [T, Hkv*D].
// cos_sin: [max_pos, D] f32, first half cos, second half sin (host-precomputed
// — guide App. B: never sinf/cosf on device for RoPE).
// One workgroup per token; threads cover (head, pair) space, 2 pairs each.
// ---------------------------------------------------------------------------
__global__ void rope_kernel(const int64_t* __restrict__ positions,
                            bf16* __restrict__ q, bf16* __restrict__ k,
                            const float* __restrict__ cos_sin,
                            const int head_dim, const int num_q_heads,
                            const int num_kv_heads, const int64_t q_stride,
                            const int64_t k_stride) {
  const int token = blockIdx.x;
  const int half = head_dim / 2;
  const int64_t pos = positions[token];
  const float* cs = cos_sin + pos * head_dim;
  const int total_heads = num_q_heads + num_kv_heads;
  // Each thread handles 2 consecutive rotary pairs (4 bf16 values).
  const int pairs2 = half / 2;  // pair-couples per head
  for (int idx = threadIdx.x; idx < total_heads * pairs2; idx += blockDim.x) {
    const int h = idx / pairs2;
    const int p2 = (idx % pairs2) * 2;  // first pair index of the couple
    bf16* base = (h < num_q_heads)
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
  }
}

}  // namespace arks

// ------------------------------- launchers ---------------------------------
using namespace arks;

extern "C" {

void arks_rmsnorm(void* out, const void* input, const void* weight, float eps,
                  int rows, int hidden, hipStream_t stream) {
  dim3 grid(rows), block(256);
  hipLaunchKernelGGL((rmsnorm_kernel<false>), grid, block, 0, stream,
                     (bf16*)out, (const bf16*)input, nullptr,
                     (const bf16*)weight, eps, hidden);
}

void arks_fused_add_rmsnorm(void* out, const void* input, void* residual,
                            const void* weight, float eps, int rows, int hidden,
                            hipStream_t stream) {
  dim3 grid(rows), block(256);
  hipLaunchKernelGGL((rmsnorm_kernel<true>), grid, block, 0, stream, (bf16*)out,
                     (const bf16*)input, (bf16*)residual, (const bf16*)weight,
                     eps, hidden);
}

void arks_silu_mul(void* out, const void* gate_up, int64_t rows, int d,
                   hipStream_t stream) {
  int64_t total_vec = rows * (d / 8);
  int blocks = (int)std::min<int64_t>((total_vec + 255) / 256, 2048);
  if (blocks == 0) blocks = 1;
  hipLaunchKernelGGL(silu_mul_kernel, dim3(blocks), dim3(256), 0, stream,
                     (bf16*)out, (const bf16*)gate_up, d, total_vec);
}

void arks_rope_inplace(const void* positions, void* q, void* k,
                       const void* cos_sin, int num_tokens, int head_dim,
                       int num_q_heads, int num_kv_heads, int64_t q_stride,
                       int64_t k_stride, hipStream_t stream) {
  dim3 grid(num_tokens), block(256);
  hipLaunchKernelGGL(rope_kernel, grid, block, 0, stream,
                     (const int64_t*)positions, (bf16*)q, (bf16*)k,
                     (const float*)cos_sin, head_dim, num_q_heads,
                     num_kv_heads, q_stride, k_stride);
}

}  // extern "C"
