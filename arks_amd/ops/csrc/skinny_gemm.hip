// Skinny-M GEMM for the decode hot path (gfx950): C[M,N] = A[M,K] @ W[N,K]^T,
// M <= 64 (one decode token per sequence). At these shapes the GEMM is pure
// weight streaming (arithmetic intensity M/2 flop/byte), but hipBLASLt's
// tuned algos reach only 1.3-2.9 TB/s on the decode shapes
// (profiles/r01_p2, profiles/data/dec_top r2). This kernel streams W at
// near-HBM rate:
//
//   grid = (N/64, nsplits): a workgroup owns a 64-row N-tile and a K-range.
//   W is read exactly once, 16 B/lane (guide G13), staged PF chunks ahead
//   in registers; A (the activations, tiny, L2/LLC-hot) is staged through
//   a ring of AB LDS buffers. What actually governs throughput here (r2
//   ISA audit, profiles/r02_final.md): the staging loops must be BATCHED
//   and BRANCHLESS — with per-item guards the compiler emits
//   global_load -> s_waitcnt vmcnt(0) -> ds_write per item, and because
//   vmcnt is an issue-ordered counter each of those waits also drains
//   every in-flight W prefetch, pinning the stream at ~3 TB/s no matter
//   how deep PF is. With batched staging, PF=1/AB=3 reaches ~3.9 TB/s on
//   the down-proj shape (hipBLASLt in-graph: 2.9).
//   Swapped operands (A-frag = W rows, B-frag = activations) put the C
//   fragment at [n, m], n = 4*la+r, col = lane%16 (same trick as the
//   attention kernels, guide common-mistake #6).
//   nsplits > 1 (small N) writes f32 partials [split, N, M]; a combine
//   kernel reduces them in fixed order (deterministic, no atomics).
//
//   FUSE_SILU: A is the raw gate_up projection output [M, 2K] (gate in
//   cols [0,K), up in [K,2K), reference vLLM SiluAndMul layout); the LDS
//   staging step computes silu(gate)*up on the fly, bf16-rounded with the
//   same formula as silu_mul_kernel (elementwise.hip:89) so the fused
//   down-proj is bit-identical to silu_mul + skinny_gemm. This removes the
//   silu_mul kernel and its intermediate tensor from every decode layer.
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

constexpr int SK_NT = 64;  // N rows per workgroup (16 per wave)
constexpr int SK_AP = 8;   // a_lds row pad (bf16): rows stay 16B-aligned

// MT = number of 16-row M tiles (M <= 16*MT, MT in 1..4).
// KC = K elements per staged chunk (ping-pong pair in LDS).
// PF = W register-prefetch distance in chunks (in-flight W = PF*KC*2 B/row).
// FUSE_SILU: see header comment.
// NT: non-temporal W loads (W is streamed exactly once; keeps it out of
// the LLC so KV/activation lines survive across decode steps).
template <int MT, int KC, int PF, bool FUSE_SILU, int AB = 2, bool NT = false>
__global__ __launch_bounds__(256) void skinny_gemm_kernel(
    float* __restrict__ part,   // [nsplits, N, 16*MT] (nsplits > 1)
    bf16* __restrict__ out,     // [M, N] (nsplits == 1)
    const bf16* __restrict__ a, // [M, K] (row stride a_stride)
    const bf16* __restrict__ w, // [N, K] row-major
    const bf16* __restrict__ bias,  // [N] or null (applied when nsplits==1)
    const int m_rows, const int n_total, const int k_total,
    const int k_per_split, const int nsplits, const int64_t a_stride) {
  constexpr int MROWS = 16 * MT;
  constexpr int KSTEPS = KC / 32;
  const int n0 = blockIdx.x * SK_NT;
  const int kb = blockIdx.y * k_per_split;
  const int ke = min(k_total, kb + k_per_split);

  const int tid = threadIdx.x;
  const int wave = tid / WAVE_SIZE;
  const int lane = tid % WAVE_SIZE;
  const int lq = lane % 16;
  const int la = lane / 16;

  // Ping-pong A buffers; the direct-path C transpose reuses the same LDS
  // after the main loop.
  __shared__ __attribute__((aligned(16))) union {
    bf16 a_buf[AB][64][KC + SK_AP];
    float c_buf[SK_NT][64 + 1];
  } lds;

  f32x4 acc[MT];
#pragma unroll
  for (int t = 0; t < MT; ++t) acc[t] = {0.f, 0.f, 0.f, 0.f};

  const bf16* wrow = w + (int64_t)(n0 + wave * 16 + lq) * k_total;

  // Cooperative A staging of one chunk into buffer `buf` (no barrier here).
  // The thread->(row, col) map divides by the COMPILE-TIME chunk width, not
  // the runtime tail width: a runtime divisor compiles to a ~30-op integer
  // division sequence per index — PMC r2 (profiles/data/pmc_skinny_r2)
  // measured 2,200 VALU insts/wave (15x the MFMA count) from exactly this,
  // serializing the inter-barrier path. Tail chunks just predicate columns.
  // BRANCHLESS batched staging: all loads issue back-to-back into
  // registers, then all ds_writes. With per-item guards (`continue` /
  // row<m_rows branches) the compiler emitted load -> s_waitcnt vmcnt(0) ->
  // ds_write per item — four FULL vmem drains per chunk that serialized the
  // whole W stream (ISA audit r2, profiles/data/pmc_skinny_r2). Guards are
  // replaced by address clamps: duplicated rows / tail columns load real
  // in-bounds values that the MFMA loop never consumes.
  auto stage_a = [&](int kc, int buf) {
    constexpr int CPT = KC / 8;  // 16 B column-groups per row
    constexpr int NIT = MROWS * CPT / 256;
    ushort8 gv[NIT], uv[NIT];
#pragma unroll
    for (int it = 0; it < NIT; ++it) {
      const int i = tid + it * 256;
      const int row = min(i / CPT, m_rows - 1);
      const int kcol = min(kc + (i % CPT) * 8, k_total - 8);
      const bf16* base = a + (int64_t)row * a_stride + kcol;
      gv[it] = *reinterpret_cast<const ushort8*>(base);
      if constexpr (FUSE_SILU)
        uv[it] = *reinterpret_cast<const ushort8*>(base + k_total);
    }
#pragma unroll
    for (int it = 0; it < NIT; ++it) {
      const int i = tid + it * 256;
      const int row = i / CPT;
      const int col8 = (i % CPT) * 8;
      ushort8 v = gv[it];
      if constexpr (FUSE_SILU) {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float x = bf16_bits_to_float(gv[it][j]);
          float silu = x / (1.f + __expf(-x));
          v[j] = float_to_bf16_bits(silu * bf16_bits_to_float(uv[it][j]));
        }
      }
      *reinterpret_cast<ushort8*>(&lds.a_buf[buf][row][col8]) = v;
    }
  };

  // W fragments, register-staged PF chunks ahead (ring of PF+1 buffers,
  // shifted each chunk — the shifts are v_movs that dual-issue with MFMA).
  ushort8 wreg[PF + 1][KSTEPS];
  auto load_w = [&](int kc, ushort8 (&dst)[KSTEPS]) {
    if (kc >= ke) {
#pragma unroll
      for (int s = 0; s < KSTEPS; ++s) dst[s] = ushort8{};
      return;
    }
    const int kw = min(KC, ke - kc);
#pragma unroll
    for (int s = 0; s < KSTEPS; ++s) {
      const ushort8* src =
          reinterpret_cast<const ushort8*>(wrow + kc + s * 32 + 8 * la);
      if (s * 32 >= kw)
        dst[s] = ushort8{};
      else if constexpr (NT)
        dst[s] = __builtin_nontemporal_load(src);
      else
        dst[s] = *src;
    }
  };

  // A staging issues BEFORE the W prefetch: vmcnt is an ISSUE-ORDERED
  // counter, so a ds_write waiting on a later-issued A load would otherwise
  // drain every in-flight W prefetch each chunk — serializing the W stream
  // (measured: ~2x loss; the reordered kernel overlaps W with compute).
#pragma unroll
  for (int p = 0; p < AB - 1; ++p)
    if (kb + p * KC < ke) stage_a(kb + p * KC, p);
#pragma unroll
  for (int p = 0; p <= PF; ++p) load_w(kb + p * KC, wreg[p]);
  __syncthreads();

  int buf = 0;  // LDS slot holding the current chunk (ring of AB slots)
  for (int kc = kb; kc < ke; kc += KC) {
    const int kw = min(KC, ke - kc);  // multiple of 32
    if (kc + (AB - 1) * KC < ke) {
      int tgt = buf + (AB - 1);
      if (tgt >= AB) tgt -= AB;
      stage_a(kc + (AB - 1) * KC, tgt);  // A loads + ds_writes first (see
                                         // vmcnt note above)
    }
    ushort8 wnext[KSTEPS];
    load_w(kc + (PF + 1) * KC, wnext);  // issues during this chunk's MFMAs

#pragma unroll
    for (int s = 0; s < KSTEPS; ++s) {
      if (s * 32 >= kw) break;
      bf16x8 wfrag = *reinterpret_cast<bf16x8*>(&wreg[0][s]);
#pragma unroll
      for (int t = 0; t < MT; ++t) {
        ushort8 af = *reinterpret_cast<const ushort8*>(
            &lds.a_buf[buf][t * 16 + lq][s * 32 + 8 * la]);
        acc[t] = sk_mfma(wfrag, *reinterpret_cast<bf16x8*>(&af), acc[t]);
      }
    }
#pragma unroll
    for (int p = 0; p < PF; ++p)
#pragma unroll
      for (int s = 0; s < KSTEPS; ++s) wreg[p][s] = wreg[p + 1][s];
#pragma unroll
    for (int s = 0; s < KSTEPS; ++s) wreg[PF][s] = wnext[s];
    __syncthreads();  // staged writes have had >= one full compute phase to
                      // land; also fences slot reuse
    buf = (buf + 1 == AB) ? 0 : buf + 1;
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

// Wave-private variant — MEASURED LOSER, kept for the record (bench_down
// r2: 1.1 TB/s): each wave stages its OWN copy of the A chunk into its own
// LDS slice so there are no cross-wave barriers, but the 4x A reads and 4x
// staging instructions per W byte cost far more than the barriers did.
template <int MT, bool FUSE_SILU>
__global__ __launch_bounds__(256) void skinny_gemm_wave_kernel(
    float* __restrict__ part, bf16* __restrict__ out,
    const bf16* __restrict__ a, const bf16* __restrict__ w,
    const bf16* __restrict__ bias, const int m_rows, const int n_total,
    const int k_total, const int k_per_split, const int nsplits,
    const int64_t a_stride) {
  constexpr int MROWS = 16 * MT;
  constexpr int KC2 = 64;
  constexpr int KS2 = KC2 / 32;  // k32 steps per chunk
  const int n0 = blockIdx.x * SK_NT;
  const int kb = blockIdx.y * k_per_split;
  const int ke = min(k_total, kb + k_per_split);

  const int tid = threadIdx.x;
  const int wave = tid / WAVE_SIZE;
  const int lane = tid % WAVE_SIZE;
  const int lq = lane % 16;
  const int la = lane / 16;

  __shared__ __attribute__((aligned(16))) union {
    bf16 a_buf[4][2][64][KC2 + SK_AP];
    float c_buf[SK_NT][64 + 1];
  } lds;

  f32x4 acc[MT];
#pragma unroll
  for (int t = 0; t < MT; ++t) acc[t] = {0.f, 0.f, 0.f, 0.f};

  const bf16* wrow = w + (int64_t)(n0 + wave * 16 + lq) * k_total + 8 * la;

  // Per-wave A staging: 64 lanes cover 64 rows x KC2/8 col-groups.
  auto stage_a = [&](int kc, int buf) {
    const int kw = min(KC2, ke - kc);
    constexpr int CPT = KC2 / 8;
#pragma unroll
    for (int it = 0; it < MROWS * CPT / WAVE_SIZE; ++it) {
      const int i = lane + it * WAVE_SIZE;
      const int row = i / CPT;
      const int col8 = (i % CPT) * 8;
      if (col8 >= kw) continue;
      ushort8 v{};
      if (row < m_rows) {
        const bf16* base = a + (int64_t)row * a_stride + kc + col8;
        if constexpr (FUSE_SILU) {
          ushort8 gv = *reinterpret_cast<const ushort8*>(base);
          ushort8 uv = *reinterpret_cast<const ushort8*>(base + k_total);
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            float x = bf16_bits_to_float(gv[j]);
            float silu = x / (1.f + __expf(-x));
            v[j] = float_to_bf16_bits(silu * bf16_bits_to_float(uv[j]));
          }
        } else {
          v = *reinterpret_cast<const ushort8*>(base);
        }
      }
      *reinterpret_cast<ushort8*>(&lds.a_buf[wave][buf][row][col8]) = v;
    }
  };

  ushort8 wreg[2][KS2];
  auto load_w = [&](int kc, ushort8 (&dst)[KS2]) {
#pragma unroll
    for (int s = 0; s < KS2; ++s)
      dst[s] = (kc + s * 32 < ke)
                   ? *reinterpret_cast<const ushort8*>(wrow + kc + s * 32)
                   : ushort8{};
  };

  stage_a(kb, 0);
  load_w(kb, wreg[0]);
  if (kb + KC2 < ke) load_w(kb + KC2, wreg[1]);

  int buf = 0;
  for (int kc = kb; kc < ke; kc += KC2, buf ^= 1) {
    const int kw = min(KC2, ke - kc);
    const int kn = kc + KC2;
    if (kn < ke) stage_a(kn, buf ^ 1);
    ushort8 wnext[KS2];
    if (kn + KC2 < ke) load_w(kn + KC2, wnext);
    else {
#pragma unroll
      for (int s = 0; s < KS2; ++s) wnext[s] = ushort8{};
    }
#pragma unroll
    for (int s = 0; s < KS2; ++s) {
      if (s * 32 >= kw) break;
      bf16x8 wfrag = *reinterpret_cast<bf16x8*>(&wreg[0][s]);
#pragma unroll
      for (int t = 0; t < MT; ++t) {
        ushort8 af = *reinterpret_cast<const ushort8*>(
            &lds.a_buf[wave][buf][t * 16 + lq][s * 32 + 8 * la]);
        acc[t] = sk_mfma(wfrag, *reinterpret_cast<bf16x8*>(&af), acc[t]);
      }
    }
#pragma unroll
    for (int s = 0; s < KS2; ++s) wreg[0][s] = wreg[1][s];
#pragma unroll
    for (int s = 0; s < KS2; ++s) wreg[1][s] = wnext[s];
    // no barrier: a_buf slice is wave-private; the wave's own lgkmcnt/vmcnt
    // ordering is sufficient
  }

  if (nsplits > 1) {
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

  __syncthreads();
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

// Direct (barrier-free) variant — MEASURED LOSER, kept for the record
// (bench_down r2: 1.5 TB/s vs 3.9 staged): both W and A stream straight
// from global. The dependent A loads share the vmcnt counter with W and
// quadruple the per-wave load count, swamping the issue stream; the
// raw-read probe (scripts/bench_membw.hip) only shows 6.7 TB/s for the
// W pattern when it is the ONLY load stream.
// Rows >= m_rows are clamped to the last valid row (duplicate finite reads;
// their C columns are discarded by the epilogue/partial consumer).
template <int MT, bool FUSE_SILU>
__global__ __launch_bounds__(256) void skinny_gemm_direct_kernel(
    float* __restrict__ part, bf16* __restrict__ out,
    const bf16* __restrict__ a, const bf16* __restrict__ w,
    const bf16* __restrict__ bias, const int m_rows, const int n_total,
    const int k_total, const int k_per_split, const int nsplits,
    const int64_t a_stride) {
  constexpr int MROWS = 16 * MT;
  const int n0 = blockIdx.x * SK_NT;
  const int kb = blockIdx.y * k_per_split;
  const int ke = min(k_total, kb + k_per_split);

  const int tid = threadIdx.x;
  const int wave = tid / WAVE_SIZE;
  const int lane = tid % WAVE_SIZE;
  const int lq = lane % 16;
  const int la = lane / 16;

  f32x4 acc[MT];
#pragma unroll
  for (int t = 0; t < MT; ++t) acc[t] = {0.f, 0.f, 0.f, 0.f};

  const bf16* wrow = w + (int64_t)(n0 + wave * 16 + lq) * k_total + 8 * la;
  const bf16* arow[MT];
#pragma unroll
  for (int t = 0; t < MT; ++t)
    arow[t] = a + (int64_t)min(t * 16 + lq, m_rows - 1) * a_stride + 8 * la;

#pragma unroll 4
  for (int k = kb; k < ke; k += 32) {
    bf16x8 wfrag = *reinterpret_cast<const bf16x8*>(wrow + k);
#pragma unroll
    for (int t = 0; t < MT; ++t) {
      bf16x8 af;
      if constexpr (FUSE_SILU) {
        ushort8 gv = *reinterpret_cast<const ushort8*>(arow[t] + k);
        ushort8 uv = *reinterpret_cast<const ushort8*>(arow[t] + k_total + k);
        ushort8 v;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float x = bf16_bits_to_float(gv[j]);
          float silu = x / (1.f + __expf(-x));
          v[j] = float_to_bf16_bits(silu * bf16_bits_to_float(uv[j]));
        }
        af = *reinterpret_cast<bf16x8*>(&v);
      } else {
        af = *reinterpret_cast<const bf16x8*>(arow[t] + k);
      }
      acc[t] = sk_mfma(wfrag, af, acc[t]);
    }
  }

  if (nsplits > 1) {
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

  __shared__ float c_lds[SK_NT][64 + 1];
#pragma unroll
  for (int t = 0; t < MT; ++t) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      c_lds[wave * 16 + 4 * la + r][t * 16 + lq] = acc[t][r];
    }
  }
  __syncthreads();
  for (int i = tid; i < m_rows * SK_NT; i += 256) {
    const int m = i / SK_NT;
    const int n = i % SK_NT;
    float v = c_lds[n][m];
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
    hipLaunchKernelGGL((skinny_gemm_kernel<MT, 128, 1, false>), grid, block,  \
                       0, stream, (float*)part, (bf16*)out, (const bf16*)a,   \
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

// Variant entry for the big streaming decode shapes (down-proj), MT=4 only:
// variant selects (KC, PF); fuse_silu selects the fused SwiGLU staging (A is
// then the [M, 2K] gate_up output; a_stride is its row stride).
extern "C" void arks_skinny_gemm_v(void* part, void* out, const void* a,
                                   const void* w, const void* bias,
                                   int m_rows, int n_total, int k_total,
                                   int k_per_split, int nsplits,
                                   int64_t a_stride, int variant,
                                   bool fuse_silu, hipStream_t stream) {
  dim3 grid(n_total / SK_NT, nsplits), block(256);
  dim3 cgrid(n_total / 16), cblock(256);
#define SK_LAUNCH_V(KC, PF, FS)                                               \
  hipLaunchKernelGGL((skinny_gemm_kernel<4, KC, PF, FS>), grid, block, 0,     \
                     stream, (float*)part, (bf16*)out, (const bf16*)a,        \
                     (const bf16*)w, (const bf16*)bias, m_rows, n_total,      \
                     k_total, k_per_split, nsplits, a_stride)
#define SK_SWITCH(FS)                                                         \
  switch (variant) {                                                          \
    case 1: SK_LAUNCH_V(128, 2, FS); break;                                   \
    case 2: SK_LAUNCH_V(128, 3, FS); break;                                   \
    case 3: SK_LAUNCH_V(256, 1, FS); break;                                   \
    case 4: SK_LAUNCH_V(256, 2, FS); break;                                   \
    case 5:                                                                   \
      hipLaunchKernelGGL((skinny_gemm_direct_kernel<4, FS>), grid, block, 0,  \
                         stream, (float*)part, (bf16*)out, (const bf16*)a,    \
                         (const bf16*)w, (const bf16*)bias, m_rows, n_total,  \
                         k_total, k_per_split, nsplits, a_stride);            \
      break;                                                                  \
    case 6:                                                                   \
      hipLaunchKernelGGL((skinny_gemm_wave_kernel<4, FS>), grid, block, 0,    \
                         stream, (float*)part, (bf16*)out, (const bf16*)a,    \
                         (const bf16*)w, (const bf16*)bias, m_rows, n_total,  \
                         k_total, k_per_split, nsplits, a_stride);            \
      break;                                                                  \
    case 8:                                                                   \
      hipLaunchKernelGGL((skinny_gemm_kernel<4, 128, 2, FS, 3, true>), grid,  \
                         block, 0, stream, (float*)part, (bf16*)out,          \
                         (const bf16*)a, (const bf16*)w, (const bf16*)bias,   \
                         m_rows, n_total, k_total, k_per_split, nsplits,      \
                         a_stride);                                           \
      break;                                                                  \
    case 7:                                                                   \
      hipLaunchKernelGGL((skinny_gemm_kernel<4, 128, 2, FS, 3>), grid,        \
                         block, 0, stream, (float*)part, (bf16*)out,          \
                         (const bf16*)a, (const bf16*)w, (const bf16*)bias,   \
                         m_rows, n_total, k_total, k_per_split, nsplits,      \
                         a_stride);                                           \
      break;                                                                  \
    default: SK_LAUNCH_V(128, 1, FS); break;                                  \
  }
  if (fuse_silu) {
    SK_SWITCH(true)
  } else {
    SK_SWITCH(false)
  }
#undef SK_SWITCH
#undef SK_LAUNCH_V
  if (nsplits > 1) {
    hipLaunchKernelGGL((skinny_combine_kernel<4>), cgrid, cblock, 0, stream,
                       (bf16*)out, (const float*)part, (const bf16*)bias,
                       m_rows, n_total, nsplits);
  }
}
