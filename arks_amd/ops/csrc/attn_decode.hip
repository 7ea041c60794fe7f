// Paged decode attention for CDNA4 (gfx950), flash-decode style.
//
// Grid: (num_kv_heads, num_seqs, num_partitions). A workgroup = 256 threads
// = 4 waves processes one (sequence, kv_head, KV-partition). All G q-heads
// of the GQA group are processed together so each K/V byte is read once for
// the whole group (KV reads are the decode bottleneck: memory-bound against
// HBM3E at ~8 TB/s). The partition axis keeps >= several hundred workgroups
// in flight at small batch (256 CUs need the parallelism far more than each
// block needs more pages — profiles/r01_first_decode_profile.md).
//
// Within a wave, the 64 lanes split into four 16-lane groups; each group
// owns one token at a time and reads its K/V row 16 B/lane (16 lanes x 16 B
// = one contiguous 256 B row at head_dim=128 — coalesced). Online softmax in
// f32 registers per group; partials merge via __shfl_xor across groups, then
// across waves via LDS. With >1 partition each workgroup emits an f32
// partial (m, l, acc) and attn_decode_combine reduces them.
//
// Cache layout: [num_blocks, num_kv_heads, BLOCK_SIZE, head_dim] bf16.
// q may be a strided row view into the fused QKV projection (q_stride).
// Fills the runtime-slot contract of scitix/arks natively.
#include "common.h"

#include <cfloat>

namespace arks {

constexpr int KV_BLOCK_SIZE = 16;  // tokens per KV page (matches engine)

template <int HEAD_DIM, int GQ>
__global__ __launch_bounds__(256) void attn_decode_kernel(
    bf16* __restrict__ out,        // [S, Hq, D] (used when gridDim.z == 1)
    float* __restrict__ part_out,  // [S, Hkv, P, GQ, D+2] (when gridDim.z > 1)
    const bf16* __restrict__ q,    // rows of length q_stride
    const bf16* __restrict__ k_cache,  // [B, Hkv, 16, D]
    const bf16* __restrict__ v_cache,
    const int* __restrict__ block_tables,  // [S, max_blocks]
    const int* __restrict__ seq_lens,      // [S]
    const float scale, const int num_kv_heads, const int max_blocks,
    const int64_t q_stride) {
  constexpr int E = HEAD_DIM / 16;  // elems per lane (8 for D=128)
  constexpr int NUM_WAVES = 4;

  const int seq = blockIdx.y;
  const int kvh = blockIdx.x;
  const int part = blockIdx.z;
  const int nparts = gridDim.z;
  const int L = seq_lens[seq];
  const int nblocks = (L + KV_BLOCK_SIZE - 1) / KV_BLOCK_SIZE;
  const int num_q_heads = num_kv_heads * GQ;

  // Partition p owns pages [p*chunk, min((p+1)*chunk, nblocks)).
  const int chunk = (nblocks + nparts - 1) / nparts;
  const int pb_lo = part * chunk;
  const int pb_hi = min(pb_lo + chunk, nblocks);

  const int tid = threadIdx.x;
  const int wave = tid / WAVE_SIZE;
  const int lane = tid % WAVE_SIZE;
  const int grp = lane / 16;     // 16-lane group within wave
  const int lane16 = lane % 16;  // position within group

  // Q fragments (scale folded in).
  float qreg[GQ][E];
#pragma unroll
  for (int g = 0; g < GQ; ++g) {
    const bf16* qp =
        q + (int64_t)seq * q_stride + (kvh * GQ + g) * HEAD_DIM + lane16 * E;
    ushort8 qv8{};
    if constexpr (E == 8) {
      qv8 = *reinterpret_cast<const ushort8*>(qp);
    } else {
      ushort4v q4 = *reinterpret_cast<const ushort4v*>(qp);
#pragma unroll
      for (int e = 0; e < 4; ++e) qv8[e] = q4[e];
    }
#pragma unroll
    for (int e = 0; e < E; ++e) qreg[g][e] = bf16_bits_to_float(qv8[e]) * scale;
  }

  float m[GQ], lsum[GQ], acc[GQ][E];
#pragma unroll
  for (int g = 0; g < GQ; ++g) {
    m[g] = -FLT_MAX;
    lsum[g] = 0.f;
#pragma unroll
    for (int e = 0; e < E; ++e) acc[g][e] = 0.f;
  }

  const int* bt = block_tables + (int64_t)seq * max_blocks;

  for (int pb = pb_lo + wave; pb < pb_hi; pb += NUM_WAVES) {
    const int64_t phys = bt[pb];
    const bf16* kb = k_cache + ((phys * num_kv_heads + kvh) * KV_BLOCK_SIZE) *
                                   (int64_t)HEAD_DIM;
    const bf16* vb = v_cache + ((phys * num_kv_heads + kvh) * KV_BLOCK_SIZE) *
                                   (int64_t)HEAD_DIM;
#pragma unroll
    for (int it = 0; it < KV_BLOCK_SIZE / 4; ++it) {
      const int t = grp + it * 4;  // token within page
      const int tok = pb * KV_BLOCK_SIZE + t;
      const bool active = tok < L;
      float kf[E], vf[E];
      if (active) {
        const bf16* kp = kb + t * HEAD_DIM + lane16 * E;
        const bf16* vp = vb + t * HEAD_DIM + lane16 * E;
        if constexpr (E == 8) {
          ushort8 kv8 = *reinterpret_cast<const ushort8*>(kp);
          ushort8 vv8 = *reinterpret_cast<const ushort8*>(vp);
#pragma unroll
          for (int e = 0; e < 8; ++e) {
            kf[e] = bf16_bits_to_float(kv8[e]);
            vf[e] = bf16_bits_to_float(vv8[e]);
          }
        } else {
          ushort4v kv4 = *reinterpret_cast<const ushort4v*>(kp);
          ushort4v vv4 = *reinterpret_cast<const ushort4v*>(vp);
#pragma unroll
          for (int e = 0; e < E; ++e) {
            kf[e] = bf16_bits_to_float(kv4[e]);
            vf[e] = bf16_bits_to_float(vv4[e]);
          }
        }
      }
#pragma unroll
      for (int g = 0; g < GQ; ++g) {
        float s = 0.f;
        if (active) {
#pragma unroll
          for (int e = 0; e < E; ++e) s += qreg[g][e] * kf[e];
        }
        s = group16_reduce_sum(s);  // all 16 lanes get the dot product
        if (active) {
          const float mn = fmaxf(m[g], s);
          const float alpha = __expf(m[g] - mn);
          const float p = __expf(s - mn);
          lsum[g] = lsum[g] * alpha + p;
#pragma unroll
          for (int e = 0; e < E; ++e) acc[g][e] = acc[g][e] * alpha + p * vf[e];
          m[g] = mn;
        }
      }
    }
  }

  // Merge the four 16-lane groups of each wave (lanes l, l^16, l^32, l^48
  // share the same dim slice lane16*E).
#pragma unroll
  for (int off = 16; off <= 32; off <<= 1) {
#pragma unroll
    for (int g = 0; g < GQ; ++g) {
      const float om = __shfl_xor(m[g], off, WAVE_SIZE);
      const float ol = __shfl_xor(lsum[g], off, WAVE_SIZE);
      const float mn = fmaxf(m[g], om);
      const float a1 = __expf(m[g] - mn);
      const float a2 = __expf(om - mn);
      lsum[g] = lsum[g] * a1 + ol * a2;
#pragma unroll
      for (int e = 0; e < E; ++e) {
        const float oa = __shfl_xor(acc[g][e], off, WAVE_SIZE);
        acc[g][e] = acc[g][e] * a1 + oa * a2;
      }
      m[g] = mn;
    }
  }

  // Cross-wave merge via LDS: wave partials [w][g][D + 2] f32.
  __shared__ float lds[NUM_WAVES][GQ][HEAD_DIM + 2];
  if (grp == 0) {
#pragma unroll
    for (int g = 0; g < GQ; ++g) {
#pragma unroll
      for (int e = 0; e < E; ++e) lds[wave][g][lane16 * E + e] = acc[g][e];
      if (lane16 == 0) {
        lds[wave][g][HEAD_DIM] = m[g];
        lds[wave][g][HEAD_DIM + 1] = lsum[g];
      }
    }
  }
  __syncthreads();

  // Head g is finalized by wave g % NUM_WAVES, lanes 0..15.
  if (grp == 0) {
    for (int g = wave; g < GQ; g += NUM_WAVES) {
      float mf = -FLT_MAX, lf = 0.f, af[E];
#pragma unroll
      for (int e = 0; e < E; ++e) af[e] = 0.f;
#pragma unroll
      for (int w = 0; w < NUM_WAVES; ++w) {
        const float mw = lds[w][g][HEAD_DIM];
        const float lw = lds[w][g][HEAD_DIM + 1];
        const float mn = fmaxf(mf, mw);
        const float a1 = __expf(mf - mn);
        const float a2 = __expf(mw - mn);
        lf = lf * a1 + lw * a2;
#pragma unroll
        for (int e = 0; e < E; ++e)
          af[e] = af[e] * a1 + lds[w][g][lane16 * E + e] * a2;
        mf = mn;
      }
      if (nparts == 1) {
        const float inv = lf > 0.f ? 1.f / lf : 0.f;
        bf16* op = out +
                   ((int64_t)seq * num_q_heads + kvh * GQ + g) * HEAD_DIM +
                   lane16 * E;
        if constexpr (E == 8) {
          ushort8 o8;
#pragma unroll
          for (int e = 0; e < 8; ++e) o8[e] = float_to_bf16_bits(af[e] * inv);
          *reinterpret_cast<ushort8*>(op) = o8;
        } else {
          ushort4v o4;
#pragma unroll
          for (int e = 0; e < E; ++e) o4[e] = float_to_bf16_bits(af[e] * inv);
          *reinterpret_cast<ushort4v*>(op) = o4;
        }
      } else {
        float* pp = part_out +
                    ((((int64_t)seq * num_kv_heads + kvh) * nparts + part) * GQ + g) *
                        (HEAD_DIM + 2);
#pragma unroll
        for (int e = 0; e < E; ++e) pp[lane16 * E + e] = af[e];
        if (lane16 == 0) {
          pp[HEAD_DIM] = mf;
          pp[HEAD_DIM + 1] = lf;
        }
      }
    }
  }
}

// Combine partials: one wave per (seq, q_head); lanes 0..15 hold dim slices.
template <int HEAD_DIM>
__global__ __launch_bounds__(64) void attn_decode_combine_kernel(
    bf16* __restrict__ out,             // [S, Hq, D]
    const float* __restrict__ part_out,  // [S, Hkv, P, GQ, D+2]
    const int* __restrict__ seq_lens, const int num_q_heads,
    const int num_kv_heads, const int nparts) {
  constexpr int E = HEAD_DIM / 16;
  const int seq = blockIdx.x;
  const int h = blockIdx.y;
  const int gq = num_q_heads / num_kv_heads;
  const int kvh = h / gq;
  const int g = h % gq;
  const int lane = threadIdx.x % WAVE_SIZE;
  const int lane16 = lane % 16;
  if (lane >= 16) return;

  const int L = seq_lens[seq];
  const int nblocks = (L + KV_BLOCK_SIZE - 1) / KV_BLOCK_SIZE;
  const int chunk = (nblocks + nparts - 1) / nparts;
  const int used_parts = min(nparts, (nblocks + chunk - 1) / chunk);

  float mf = -FLT_MAX, lf = 0.f, af[E];
#pragma unroll
  for (int e = 0; e < E; ++e) af[e] = 0.f;
  for (int p = 0; p < used_parts; ++p) {
    const float* pp = part_out +
                      ((((int64_t)seq * num_kv_heads + kvh) * nparts + p) * gq + g) *
                          (HEAD_DIM + 2);
    const float mw = pp[HEAD_DIM];
    const float lw = pp[HEAD_DIM + 1];
    const float mn = fmaxf(mf, mw);
    const float a1 = __expf(mf - mn);
    const float a2 = __expf(mw - mn);
    lf = lf * a1 + lw * a2;
#pragma unroll
    for (int e = 0; e < E; ++e)
      af[e] = af[e] * a1 + pp[lane16 * E + e] * a2;
    mf = mn;
  }
  const float inv = lf > 0.f ? 1.f / lf : 0.f;
  bf16* op = out + ((int64_t)seq * num_q_heads + h) * HEAD_DIM + lane16 * E;
  if constexpr (E == 8) {
    ushort8 o8;
#pragma unroll
    for (int e = 0; e < 8; ++e) o8[e] = float_to_bf16_bits(af[e] * inv);
    *reinterpret_cast<ushort8*>(op) = o8;
  } else {
    ushort4v o4;
#pragma unroll
    for (int e = 0; e < E; ++e) o4[e] = float_to_bf16_bits(af[e] * inv);
    *reinterpret_cast<ushort4v*>(op) = o4;
  }
}

template <int HEAD_DIM>
void launch_decode_gq(bf16* out, float* part_out, const bf16* q,
                      const bf16* kc, const bf16* vc, const int* bt,
                      const int* sl, float scale, int num_seqs,
                      int num_kv_heads, int gq, int max_blocks, int nparts,
                      int64_t q_stride, hipStream_t stream) {
  dim3 grid(num_kv_heads, num_seqs, nparts), block(256);
#define ARKS_CASE(G)                                                          \
  case G:                                                                     \
    hipLaunchKernelGGL((attn_decode_kernel<HEAD_DIM, G>), grid, block, 0,     \
                       stream, out, part_out, q, kc, vc, bt, sl, scale,       \
                       num_kv_heads, max_blocks, q_stride);                   \
    break;
  switch (gq) {
    ARKS_CASE(1)
    ARKS_CASE(2)
    ARKS_CASE(3)
    ARKS_CASE(4)
    ARKS_CASE(5)
    ARKS_CASE(6)
    ARKS_CASE(7)
    ARKS_CASE(8)
    default:
      break;  // validated host-side
  }
#undef ARKS_CASE
  if (nparts > 1) {
    dim3 cgrid(num_seqs, num_kv_heads * gq), cblock(64);
    hipLaunchKernelGGL((attn_decode_combine_kernel<HEAD_DIM>), cgrid, cblock,
                       0, stream, out, part_out, sl, num_kv_heads * gq,
                       num_kv_heads, nparts);
  }
}

}  // namespace arks

using namespace arks;

extern "C" void arks_attn_decode_paged(void* out, void* part_out, const void* q,
                                       const void* k_cache, const void* v_cache,
                                       const void* block_tables,
                                       const void* seq_lens, float scale,
                                       int num_seqs, int num_q_heads,
                                       int num_kv_heads, int head_dim,
                                       int max_blocks, int nparts,
                                       int64_t q_stride, hipStream_t stream) {
  const int gq = num_q_heads / num_kv_heads;
  if (head_dim == 128) {
    launch_decode_gq<128>((bf16*)out, (float*)part_out, (const bf16*)q,
                          (const bf16*)k_cache, (const bf16*)v_cache,
                          (const int*)block_tables, (const int*)seq_lens,
                          scale, num_seqs, num_kv_heads, gq, max_blocks,
                          nparts, q_stride, stream);
  } else if (head_dim == 64) {
    launch_decode_gq<64>((bf16*)out, (float*)part_out, (const bf16*)q,
                         (const bf16*)k_cache, (const bf16*)v_cache,
                         (const int*)block_tables, (const int*)seq_lens, scale,
                         num_seqs, num_kv_heads, gq, max_blocks, nparts,
                         q_stride, stream);
  }
}
