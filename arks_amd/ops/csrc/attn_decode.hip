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
#include <hip/hip_fp8.h>

namespace arks {

constexpr int KV_BLOCK_SIZE = 16;  // tokens per KV page (matches engine)

// MFMA fragment types (shared convention with attn_prefill.hip; layout
// verified on hardware by tests/test_ops_gpu.py::test_mfma_probe).
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) __bf16 dec_bf16x4v;
typedef __attribute__((ext_vector_type(4))) float f32x4;

__device__ __forceinline__ f32x4 dec_mfma(bf16x8 a, bf16x8 b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

__device__ __forceinline__ int dec_frag_k(int a, int jj) { return 8 * a + jj; }

constexpr int DEC_CHUNK_TOK = 32;  // tokens per wave iteration (2 KV pages)
constexpr int DEC_VT_PAD = 40;     // padded token axis in V^T / P LDS tiles

// v3 (MFMA): one workgroup per (seq, kv_head, partition); the GQA head
// group (GQ <= 16 q-heads padded to a 16-row tile) is the MFMA M axis, so
// QK^T for all heads of the group over a 16-token page is ONE mfma chain
// instead of GQ serial 16-lane shuffle reductions (v2 spent ~112 cyc/token
// on shuffles at GQ=7 and reached only 3.0 TB/s of KV stream). Each of the
// 4 waves owns a 32-token chunk (wave-strided over the partition): K feeds
// the A-operand straight from cache pages (16 B/lane), V is transposed
// through a per-wave LDS tile for the PV B-operand, P goes through a
// per-wave LDS tile exactly as in attn_prefill.hip. Column-wise online
// softmax: lane lq owns head lq, so the per-tile reduction is two
// shfl_xor's for ALL heads at once. Wave partials merge flash-style in LDS
// (the V/P tiles are re-used as the merge buffer), and the partition
// output/partial format is unchanged from v2 (combine kernel below).
__device__ __forceinline__ uint16_t dec_fp8_to_bf16(uint8_t b) {
  __hip_fp8_e4m3 q;
  q.__x = b;
  return float_to_bf16_bits((float)q);
}

template <int HEAD_DIM, int GQ, bool KV_FP8>
__global__ __launch_bounds__(256) void attn_decode_kernel(
    bf16* __restrict__ out,        // [S, Hq, D] (used when gridDim.z == 1)
    float* __restrict__ part_out,  // [S, Hkv, P, GQ, D+2] (when gridDim.z > 1)
    const bf16* __restrict__ q,    // rows of length q_stride
    const bf16* __restrict__ k_cache,  // [B, Hkv, 16, D]
    const bf16* __restrict__ v_cache,
    const int* __restrict__ block_tables,  // [S, max_blocks]
    const int* __restrict__ seq_lens,      // [S]
    const float scale, const int num_kv_heads, const int max_blocks,
    const int64_t q_stride, const int window) {
  constexpr int NUM_WAVES = 4;
  constexpr int STEPS = HEAD_DIM / 32;   // QK^T contraction steps
  constexpr int CHUNKS = HEAD_DIM / 16;  // PV output dim chunks
  constexpr int E = HEAD_DIM / 16;       // dim elems per lane16 slice

  const int seq = blockIdx.y;
  const int kvh = blockIdx.x;
  const int part = blockIdx.z;
  const int nparts = gridDim.z;
  const int L = seq_lens[seq];
  const int nblocks = (L + KV_BLOCK_SIZE - 1) / KV_BLOCK_SIZE;
  const int num_q_heads = num_kv_heads * GQ;

  const int chunk = (nblocks + nparts - 1) / nparts;  // pages per partition
  const int pb_lo = part * chunk;
  const int pb_hi = min(pb_lo + chunk, nblocks);
  const int tid = threadIdx.x;
  const int wave = tid / WAVE_SIZE;
  const int lane = tid % WAVE_SIZE;
  const int lq = lane % 16;
  const int la = lane / 16;

  // Per-wave LDS region: V^T tile + P tile during the main loop, merge
  // buffer afterwards (union; a wave touches only its own region between
  // the loop and the final barrier).
  // V tr16 image: [cb = D/16][kb = 8][96] (64 data + 32 pad) bf16
  constexpr int VT_BYTES = (HEAD_DIM / 16) * 8 * 96 * 2;
  constexpr int P_BYTES = 16 * DEC_VT_PAD * 2;
  constexpr int MERGE_BYTES = 16 * (HEAD_DIM + 2) * 4;
  constexpr int WAVE_BYTES0 =
      VT_BYTES + P_BYTES > MERGE_BYTES ? VT_BYTES + P_BYTES : MERGE_BYTES;
  constexpr int WAVE_BYTES = (WAVE_BYTES0 + 15) & ~15;
  __shared__ __attribute__((aligned(16))) char smem[NUM_WAVES][WAVE_BYTES];
  bf16* vt_lds = reinterpret_cast<bf16*>(smem[wave]);              // [D][40]
  bf16* p_lds = reinterpret_cast<bf16*>(smem[wave] + VT_BYTES);    // [16][40]
  float* merge_lds = reinterpret_cast<float*>(smem[wave]);         // [16][D+2]

  // Q fragments: B-operand [32k x 16 heads] per step; rows g >= GQ zero.
  bf16x8 qfrag[STEPS];
  {
    const bool valid = lq < GQ;
    const bf16* qp = q + (int64_t)seq * q_stride +
                     (kvh * GQ + (valid ? lq : 0)) * HEAD_DIM;
#pragma unroll
    for (int st = 0; st < STEPS; ++st) {
      ushort8 u{};
      if (valid)
        u = *reinterpret_cast<const ushort8*>(qp + st * 32 + dec_frag_k(la, 0));
      qfrag[st] = *reinterpret_cast<bf16x8*>(&u);
    }
  }

  float m_run = -FLT_MAX;  // per head lq (replicated over la)
  float l_run = 0.f;
  f32x4 oacc[CHUNKS];
#pragma unroll
  for (int c = 0; c < CHUNKS; ++c) oacc[c] = {0.f, 0.f, 0.f, 0.f};

  const int* bt = block_tables + (int64_t)seq * max_blocks;
  const int64_t page_elems = (int64_t)num_kv_heads * KV_BLOCK_SIZE * HEAD_DIM;
  const int64_t head_off = (int64_t)kvh * KV_BLOCK_SIZE * HEAD_DIM;

  // 32-token chunks, wave-strided over the partition's pages. Software
  // pipeline: chunk i+1's K rides in registers and its loads issue during
  // chunk i's MFMAs; V loads issue at the top of the chunk and their
  // LDS-transpose happens after QK+softmax have covered the latency. The
  // LDS producer->consumer wait is lgkmcnt-only so outstanding K/V global
  // prefetches keep flowing (a full s_waitcnt(0) here serialized every
  // chunk: v3 capped at ~3 TB/s of KV stream).
  // Sliding window: tokens below L-window are masked; their pages may
  // have been dropped by the allocator (block-table entry -1), so page ids
  // are sanitized to 0 (any read from them is masked out).
  //
  // Block-table ids are software-pipelined TWO iterations ahead: the K/V
  // addresses depend on bt[pb], so loading it in the consuming iteration
  // emits global_load_dword -> s_waitcnt vmcnt(0) before every tile — a
  // dependent pointer-chase that drains all in-flight K/V prefetches per
  // page (ISA audit r2, same class of stall as the skinny-GEMM staging).
  auto page_base = [&](int id) {
    return (int64_t)(id < 0 ? 0 : id) * page_elems + head_off;
  };
  auto raw_bt = [&](int pb_) { return pb_ < pb_hi ? bt[pb_] : 0; };
  auto load_k = [&](int pb, ushort8 (*dst)[STEPS], int64_t pbase0,
                    int64_t pbase1) {
    const bool have_p1 = pb + 1 < pb_hi;
#pragma unroll
    for (int sub = 0; sub < 2; ++sub) {
      const int64_t pbase = sub == 0 ? pbase0 : pbase1;
#pragma unroll
      for (int st = 0; st < STEPS; ++st) {
        ushort8 u{};
        if (sub == 0 || have_p1) {
          const int64_t src =
              pbase + (int64_t)lq * HEAD_DIM + st * 32 + dec_frag_k(la, 0);
          if constexpr (KV_FP8) {
            uchar8 u8 = *reinterpret_cast<const uchar8*>(
                reinterpret_cast<const uint8_t*>(k_cache) + src);
#pragma unroll
            for (int e = 0; e < 8; ++e) u[e] = dec_fp8_to_bf16(u8[e]);
          } else {
            u = *reinterpret_cast<const ushort8*>(k_cache + src);
          }
        }
        dst[sub][st] = u;
      }
    }
  };

  constexpr int NV = DEC_CHUNK_TOK * HEAD_DIM / 8 / WAVE_SIZE;
  // Window lower bound: skip 2-page chunks wholly below it (chunk grid
  // stays 2-page aligned to pb_lo so the wave stride is unchanged).
  int pb_first = pb_lo;
  if (window > 0 && L > window) {
    const int win_lo_page = (L - window) / KV_BLOCK_SIZE;
    if (win_lo_page > pb_lo)
      pb_first = pb_lo + ((win_lo_page - pb_lo) / 2) * 2;
  }
  ushort8 kreg[2][STEPS];
  const int pb_w0 = pb_first + wave * 2;
  // bt pipeline: cbt = this iteration's pages, nbt = next iteration's
  // (already in flight); nnbt issues at each loop top for the one after.
  int cbt0 = raw_bt(pb_w0), cbt1 = raw_bt(pb_w0 + 1);
  int nbt0 = raw_bt(pb_w0 + NUM_WAVES * 2);
  int nbt1 = raw_bt(pb_w0 + NUM_WAVES * 2 + 1);
  if (pb_w0 < pb_hi)
    load_k(pb_w0, kreg, page_base(cbt0), page_base(cbt1));

  for (int pb = pb_w0; pb < pb_hi; pb += NUM_WAVES * 2) {
    const int pnn = pb + 2 * NUM_WAVES * 2;
    const int nnbt0 = raw_bt(pnn), nnbt1 = raw_bt(pnn + 1);
    const int64_t pbase0 = page_base(cbt0);
    const int64_t pbase1 = page_base(cbt1);
    const int tok0 = pb * KV_BLOCK_SIZE;  // first token of the chunk
    // valid tokens here: bounded by L AND by the partition's page range
    const int kmax =
        min(min(L - tok0, (pb_hi - pb) * KV_BLOCK_SIZE), DEC_CHUNK_TOK);

    // ---- V for this chunk into registers (zero-padded); transposed to
    // LDS only after QK+softmax (in-order LDS keeps it behind the previous
    // chunk's PV reads).
    ushort8 vv[NV];
#pragma unroll
    for (int j = 0; j < NV; ++j) {
      const int i = lane + WAVE_SIZE * j;
      const int t = i / (HEAD_DIM / 8);
      const int col8 = (i % (HEAD_DIM / 8)) * 8;
      ushort8 v{};
      if (t < kmax) {
        const int64_t src = (t < KV_BLOCK_SIZE ? pbase0 : pbase1) +
                            (int64_t)(t & (KV_BLOCK_SIZE - 1)) * HEAD_DIM +
                            col8;
        if constexpr (KV_FP8) {
          uchar8 v8 = *reinterpret_cast<const uchar8*>(
              reinterpret_cast<const uint8_t*>(v_cache) + src);
#pragma unroll
          for (int e = 0; e < 8; ++e) v[e] = dec_fp8_to_bf16(v8[e]);
        } else {
          v = *reinterpret_cast<const ushort8*>(v_cache + src);
        }
      }
      vv[j] = v;
    }

    // ---- prefetch next chunk's K while this chunk computes.
    ushort8 knext[2][STEPS];
    const int pnext = pb + NUM_WAVES * 2;
    if (pnext < pb_hi)
      load_k(pnext, knext, page_base(nbt0), page_base(nbt1));

    // ---- QK^T: A-frag = K registers, B-frag = Q registers.
    f32x4 sc[2];
    sc[0] = {0.f, 0.f, 0.f, 0.f};
    sc[1] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int sub = 0; sub < 2; ++sub) {
#pragma unroll
      for (int st = 0; st < STEPS; ++st) {
        sc[sub] = dec_mfma(*reinterpret_cast<bf16x8*>(&kreg[sub][st]),
                           qfrag[st], sc[sub]);
      }
    }

    // ---- masked column-wise online softmax (head = lq; tokens in lanes).
    float p[8];
    float tile_max = -FLT_MAX;
#pragma unroll
    for (int sub = 0; sub < 2; ++sub) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int t = sub * 16 + 4 * la + r;
        const int i = sub * 4 + r;
        const bool vis =
            t < kmax && (window <= 0 || tok0 + t >= L - window);
        p[i] = vis ? sc[sub][r] * scale : -FLT_MAX;
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
        p[i] = (p[i] > -FLT_MAX) ? __expf(p[i] - m_new) : 0.f;
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

    // rescale O rows (head = 4*la + r in the PV fragment).
    float row_alpha[4];
#pragma unroll
    for (int r = 0; r < 4; ++r)
      row_alpha[r] = __shfl(alpha, 4 * la + r, WAVE_SIZE);
#pragma unroll
    for (int c = 0; c < CHUNKS; ++c) {
#pragma unroll
      for (int r = 0; r < 4; ++r) oacc[c][r] *= row_alpha[r];
    }

    // ---- store V into the tr16 image (b128; after the previous PV reads,
    // in order within the wave).
#pragma unroll
    for (int j = 0; j < NV; ++j) {
      const int i = lane + WAVE_SIZE * j;
      const int t = i / (HEAD_DIM / 8);
      const int col8 = (i % (HEAD_DIM / 8)) * 8;
      *reinterpret_cast<ushort8*>(
          &vt_lds[((col8 >> 4) * 8 + (t >> 2)) * 96 + (t & 3) * 16 +
                  (col8 & 15)]) = vv[j];
    }

    // ---- P to LDS (bf16) then PV (contract the full 32-token chunk).
#pragma unroll
    for (int sub = 0; sub < 2; ++sub) {
      ushort4v pk;
#pragma unroll
      for (int r = 0; r < 4; ++r) pk[r] = float_to_bf16_bits(p[sub * 4 + r]);
      *reinterpret_cast<ushort4v*>(&p_lds[lq * DEC_VT_PAD + sub * 16 + 4 * la]) =
          pk;
    }
    // Intra-wave LDS producer->consumer (lanes of one wave): wait on the
    // LDS counter ONLY — global prefetches must stay outstanding.
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    ushort8 pa = *reinterpret_cast<const ushort8*>(
        &p_lds[lq * DEC_VT_PAD + dec_frag_k(la, 0)]);
#pragma unroll
    for (int c = 0; c < CHUNKS; ++c) {
      dec_bf16x4v v1 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
          (__attribute__((address_space(3))) dec_bf16x4v*)(
              reinterpret_cast<char*>(&vt_lds[(c * 8 + 2 * la) * 96]) +
              lq * 8));
      dec_bf16x4v v2 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
          (__attribute__((address_space(3))) dec_bf16x4v*)(
              reinterpret_cast<char*>(&vt_lds[(c * 8 + 2 * la + 1) * 96]) +
              lq * 8));
      bf16x8 vb;
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        vb[e] = v1[e];
        vb[e + 4] = v2[e];
      }
      oacc[c] = dec_mfma(*reinterpret_cast<bf16x8*>(&pa), vb, oacc[c]);
    }
#pragma unroll
    for (int sub = 0; sub < 2; ++sub)
#pragma unroll
      for (int st = 0; st < STEPS; ++st) kreg[sub][st] = knext[sub][st];
    cbt0 = nbt0;
    cbt1 = nbt1;
    nbt0 = nnbt0;
    nbt1 = nnbt1;
  }

  // ---- merge the 4 wave partials (flash-style) and emit.
  __syncthreads();  // everyone done with vt/p before the union flips
  {
    // wave writes its [16 head][D] partial + per-head (m, l)
#pragma unroll
    for (int c = 0; c < CHUNKS; ++c) {
#pragma unroll
      for (int r = 0; r < 4; ++r)
        merge_lds[(4 * la + r) * (HEAD_DIM + 2) + c * 16 + lq] = oacc[c][r];
    }
    if (la == 0) {
      merge_lds[lq * (HEAD_DIM + 2) + HEAD_DIM] = m_run;
      merge_lds[lq * (HEAD_DIM + 2) + HEAD_DIM + 1] = l_run;
    }
  }
  __syncthreads();

  // 256 threads = 16 heads x 16 dim slices; head g merged over the 4 waves.
  {
    const int g = tid / 16;
    const int lane16 = tid % 16;
    if (g < GQ) {
      float mf = -FLT_MAX, lf = 0.f, af[E];
#pragma unroll
      for (int e = 0; e < E; ++e) af[e] = 0.f;
      for (int w = 0; w < NUM_WAVES; ++w) {
        const float* wl = reinterpret_cast<const float*>(smem[w]);
        const float mw = wl[g * (HEAD_DIM + 2) + HEAD_DIM];
        const float lw = wl[g * (HEAD_DIM + 2) + HEAD_DIM + 1];
        const float mn = fmaxf(mf, mw);
        const float a1 = __expf(mf - mn);
        const float a2 = __expf(mw - mn);
        lf = lf * a1 + lw * a2;
#pragma unroll
        for (int e = 0; e < E; ++e)
          af[e] = af[e] * a1 + wl[g * (HEAD_DIM + 2) + lane16 * E + e] * a2;
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
                    ((((int64_t)seq * num_kv_heads + kvh) * nparts + part) * GQ +
                     g) * (HEAD_DIM + 2);
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
                      int64_t q_stride, bool kv_fp8, int window,
                      hipStream_t stream) {
  dim3 grid(num_kv_heads, num_seqs, nparts), block(256);
#define ARKS_CASE(G)                                                          \
  case G:                                                                     \
    if (kv_fp8) {                                                             \
      hipLaunchKernelGGL((attn_decode_kernel<HEAD_DIM, G, true>), grid,       \
                         block, 0, stream, out, part_out, q, kc, vc, bt, sl,  \
                         scale, num_kv_heads, max_blocks, q_stride, window);  \
    } else {                                                                  \
      hipLaunchKernelGGL((attn_decode_kernel<HEAD_DIM, G, false>), grid,      \
                         block, 0, stream, out, part_out, q, kc, vc, bt, sl,  \
                         scale, num_kv_heads, max_blocks, q_stride, window);  \
    }                                                                         \
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
                                       int64_t q_stride, int kv_fp8, int window,
                                       hipStream_t stream) {
  const int gq = num_q_heads / num_kv_heads;
  if (head_dim == 128) {
    launch_decode_gq<128>((bf16*)out, (float*)part_out, (const bf16*)q,
                          (const bf16*)k_cache, (const bf16*)v_cache,
                          (const int*)block_tables, (const int*)seq_lens,
                          scale, num_seqs, num_kv_heads, gq, max_blocks,
                          nparts, q_stride, kv_fp8 != 0, window, stream);
  } else if (head_dim == 64) {
    launch_decode_gq<64>((bf16*)out, (float*)part_out, (const bf16*)q,
                         (const bf16*)k_cache, (const bf16*)v_cache,
                         (const int*)block_tables, (const int*)seq_lens, scale,
                         num_seqs, num_kv_heads, gq, max_blocks, nparts,
                         q_stride, kv_fp8 != 0, window, stream);
  }
}
