// Skinny-M GEMM for the decode hot path (gfx950): C[M,N] = A[M,K] @ W[N,K]^T,
// M <= 64 (one decode token per sequence). At these shapes the GEMM is pure
// weight streaming (arithmetic intensity M/2 flop/byte), but hipBLASLt's
// tuned algos reach only 1.3-2.3 TB/s on the N<=4608 shapes
// (profiles/r01_p2). This kernel streams W at near-HBM rate:
//
//   grid = (N/64, nsplits): a workgroup owns a 64-row N-tile and a K-range.
//   W is read exactly once, 16 B/lane (guide G13), double-buffered in
//   REGISTERS so the next chunk's loads issue during the current chunk's
//   MFMAs. A (the activations, tiny, L2-hot) is staged through ping-pong LDS
//   buffers so the staging barrier is never on the critical path (PMC v1:
//   67% SQ_WAIT_ANY from the single-buffer round-trip — gpurun_out/pmc).
//   Swapped operands (A-frag = W rows, B-frag = activations) put the C
//   fragment at [n, m], n = 4*la+r, col = lane%16 (same trick as the
//   attention kernels, guide common-mistake #6).
//   nsplits > 1 (small N) writes f32 partials [split, N, M]; a combine
//   kernel reduces them in fixed order (deterministic, no atomics).
//
// N must be a multiple of 64 and K of 32 (true for every Qwen/Llama shape;
// the Python wrapper falls back to hipBLASLt otherwise).
#include "common.h"

#include <cfloat>

namespace arks {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

__device__ __forceinline__ f32x4 sk_mfma(bf16x8 a, bf16x8 b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

constexpr int SK_NT = 64;   // N rows per workgroup (16 per wave)
constexpr int SK_KC = 128;  // K per LDS staging buffer (ping-pong pair)
constexpr int SK_AP = 8;    // a_lds row pad (bf16): rows stay 16B-aligned
constexpr int SK_KSTEPS = SK_KC / 32;

// MT = number of 16-row M tiles (M <= 16*MT, MT in 1..4).
template <int MT>
__global__ __launch_bounds__(256) void skinny_gemm_kernel(
    float* __restrict__ part,   // [nsplits, N, 16*MT] (nsplits > 1)
    bf16* __restrict__ out,     // [M, N] (nsplits == 1)
    const bf16* __restrict__ a, // [M, K] (row stride a_stride)
    const bf16* __restrict__ w, // [N, K] row-major
    const bf16* __restrict__ bias,  // [N] or null (applied when nsplits==1)
    const int m_rows, const int n_total, const int k_total,
    const int k_per_split, const int nsplits, const int64_t a_stride) {
  constexpr int MROWS = 16 * MT;
  const int n0 = blockIdx.x * SK_NT;
  const int kb = blockIdx.y * k_per_split;
  const int ke = min(k_total, kb + k_per_split);

  const int tid = threadIdx.x;
  const int wave = tid / WAVE_SIZE;
  const int lane = tid % WAVE_SIZE;
  const int lq = lane % 16;
  const int la = lane / 16;

  // Ping-pong A buffers; the direct-path C transpose reuses the same LDS
  // after the main loop (union keeps the footprint at ~2x17 KB -> 4 WG/CU).
  __shared__ __attribute__((aligned(16))) union {
    bf16 a_buf[2][64][SK_KC + SK_AP];
    float c_buf[SK_NT][64 + 1];
  } lds;

  f32x4 acc[MT];
#pragma unroll
  for (int t = 0; t < MT; ++t) acc[t] = {0.f, 0.f, 0.f, 0.f};

  const bf16* wrow = w + (int64_t)(n0 + wave * 16 + lq) * k_total;

  // Cooperative A staging of one chunk into buffer `buf` (no barrier here).
  auto stage_a = [&](int kc, int buf) {
    const int kw = min(SK_KC, ke - kc);
    for (int i = tid; i < MROWS * (kw / 8); i += 256) {
      const int row = i / (kw / 8);
      const int col8 = (i % (kw / 8)) * 8;
      ushort8 v{};
      if (row < m_rows)
        v = *reinterpret_cast<const ushort8*>(a + (int64_t)row * a_stride +
                                              kc + col8);
      *reinterpret_cast<ushort8*>(&lds.a_buf[buf][row][col8]) = v;
    }
  };

  // W fragments, register-double-buffered.
  ushort8 wreg[SK_KSTEPS];
  auto load_w = [&](int kc, ushort8* dst) {
    const int kw = min(SK_KC, ke - kc);
#pragma unroll
    for (int s = 0; s < SK_KSTEPS; ++s)
      dst[s] = (s * 32 < kw)
                   ? *reinterpret_cast<const ushort8*>(wrow + kc + s * 32 +
                                                       8 * la)
                   : ushort8{};
  };

  load_w(kb, wreg);
  stage_a(kb, 0);
  __syncthreads();

  int buf = 0;
  for (int kc = kb; kc < ke; kc += SK_KC, buf ^= 1) {
    const int kw = min(SK_KC, ke - kc);  // multiple of 32
    const int kn = kc + SK_KC;
    ushort8 wnext[SK_KSTEPS];
    if (kn < ke) {
      load_w(kn, wnext);   // issues during this chunk's MFMAs
      stage_a(kn, buf ^ 1);
    }
#pragma unroll
    for (int s = 0; s < SK_KSTEPS; ++s) {
      if (s * 32 >= kw) break;
      bf16x8 wfrag = *reinterpret_cast<bf16x8*>(&wreg[s]);
#pragma unroll
      for (int t = 0; t < MT; ++t) {
        ushort8 af = *reinterpret_cast<const ushort8*>(
            &lds.a_buf[buf][t * 16 + lq][s * 32 + 8 * la]);
        acc[t] = sk_mfma(wfrag, *reinterpret_cast<bf16x8*>(&af), acc[t]);
      }
    }
#pragma unroll
    for (int s = 0; s < SK_KSTEPS; ++s) wreg[s] = wnext[s];
    __syncthreads();  // next buffer's staging writes have had the whole
                      // compute phase to land; also fences buf reuse
  }

  if (nsplits > 1) {
    // f32 partials, [split][n][m]: lanes with the same la write 64 B
    // contiguous (m is the fast axis).
    float* p = part + ((int64_t)blockIdx.y * n_total + n0) * MROWS;
#pragma unroll
    for (int t = 0; t < MT; ++t) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int n = wave * 16 + 4 * la + r;
        p[n * MROWS + t * 16 + lq] = acc[t][r];
      }
    }
    return;
  }

  // Direct epilogue: transpose C^T through LDS so the [M, N] store is
  // row-contiguous (128 B per m-row per workgroup).
  __syncthreads();  // all a_buf reads done before the union flips to c_buf
#pragma unroll
  for (int t = 0; t < MT; ++t) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      lds.c_buf[wave * 16 + 4 * la + r][t * 16 + lq] = acc[t][r];
    }
  }
  __syncthreads();
  for (int i = tid; i < m_rows * SK_NT; i += 256) {
    const int m = i / SK_NT;
    const int n = i % SK_NT;
    float v = lds.c_buf[n][m];
    if (bias != nullptr) v += bf16_bits_to_float(bias[n0 + n]);
    out[(int64_t)m * n_total + n0 + n] = float_to_bf16_bits(v);
  }
}

// Reduce the split partials: out[m][n] = sum_s part[s][n][m] (+bias[n]).
// One workgroup per 16-column N-tile (4x the parallelism of v1, which was
// 93% latency-parked — gpurun_out/pmc); each thread owns a float4 of m for
// one n, so the per-split reads are 16 B and independent across splits.
template <int MT>
__global__ __launch_bounds__(256) void skinny_combine_kernel(
    bf16* __restrict__ out, const float* __restrict__ part,
    const bf16* __restrict__ bias, const int m_rows, const int n_total,
    const int nsplits) {
  constexpr int MROWS = 16 * MT;
  constexpr int NT = 16;  // n per workgroup
  const int n0 = blockIdx.x * NT;
  const int tid = threadIdx.x;
  __shared__ float c_lds[NT][MROWS + 1];
  // thread -> (n, m4): MROWS/4 quads per n
  for (int i = tid; i < NT * (MROWS / 4); i += 256) {
    const int n = i / (MROWS / 4);
    const int m4 = (i % (MROWS / 4)) * 4;
    float4v s{0.f, 0.f, 0.f, 0.f};
    for (int sp = 0; sp < nsplits; ++sp) {
      const float4v v = *reinterpret_cast<const float4v*>(
          part + ((int64_t)sp * n_total + n0 + n) * MROWS + m4);
#pragma unroll
      for (int e = 0; e < 4; ++e) s[e] += v[e];
    }
    *reinterpret_cast<float4v*>(&c_lds[n][m4]) = s;
  }
  __syncthreads();
  for (int i = tid; i < m_rows * NT; i += 256) {
    const int m = i / NT;
    const int n = i % NT;
    float v = c_lds[n][m];
    if (bias != nullptr) v += bf16_bits_to_float(bias[n0 + n]);
    out[(int64_t)m * n_total + n0 + n] = float_to_bf16_bits(v);
  }
}

}  // namespace arks

using namespace arks;

extern "C" void arks_skinny_gemm(void* part, void* out, const void* a,
                                 const void* w, const void* bias, int m_rows,
                                 int n_total, int k_total, int k_per_split,
                                 int nsplits, int64_t a_stride,
                                 hipStream_t stream) {
  const int mt = (m_rows + 15) / 16;
  dim3 grid(n_total / SK_NT, nsplits), block(256);
  dim3 cgrid(n_total / 16), cblock(256);
#define SK_LAUNCH(MT)                                                         \
  do {                                                                        \
    hipLaunchKernelGGL((skinny_gemm_kernel<MT>), grid, block, 0, stream,      \
                       (float*)part, (bf16*)out, (const bf16*)a,              \
                       (const bf16*)w, (const bf16*)bias, m_rows, n_total,    \
                       k_total, k_per_split, nsplits, a_stride);              \
    if (nsplits > 1) {                                                        \
      hipLaunchKernelGGL((skinny_combine_kernel<MT>), cgrid, cblock, 0,       \
                         stream, (bf16*)out, (const float*)part,              \
                         (const bf16*)bias, m_rows, n_total, nsplits);        \
    }                                                                         \
  } while (0)
  switch (mt) {
    case 1: SK_LAUNCH(1); break;
    case 2: SK_LAUNCH(2); break;
    case 3: SK_LAUNCH(3); break;
    default: SK_LAUNCH(4); break;
  }
#undef SK_LAUNCH
}
