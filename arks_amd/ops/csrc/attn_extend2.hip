// Paged "extend" attention, 8-wave 32x32-MFMA ladder (gfx950).
//
// Second-generation prefill/extend kernel: replaces the 4-wave 16x16
// attn_extend for large q tiles. Structure follows the CDNA4 attention
// ladder (guide "Fused attention prefill"): one workgroup = 8 waves = a
// 256-row Q tile of one (seq, q-head); each wave owns 32 q rows. K/V are
// iterated in 64-key tiles, double-buffered in dynamic LDS with register
// staging (issue-early / write-late, T14), K rows XOR-swizzled for
// ds_read_b128, V as a tr16 image read with ds_read_b64_tr_b16.
//
// QK^T uses swapped operands on v_mfma_f32_32x32x16_bf16 (A = K-tile,
// B = Q) so each lane's C fragment holds 16 scores for ONE q-row
// (col = lane&31); the online softmax is then fully in-register: row max =
// in-lane fmax tree + one cross-half shuffle, P stays in VGPRs and is
// packed into the PV A-fragment with v_cvt_pk_bf16_f32 + permlane32_swap
// (16 cvt_pk + 8 swaps per 64-key tile) — no P round trip through LDS and
// one barrier per KV tile. O-accumulator rescale is skipped while the
// running max is within RESCALE_THR of the tile max (defer-max, T13).
//
// Sliding-window attention: `window > 0` restricts visibility to the last
// `window` key positions per query (Mistral-style); the KV loop starts at
// the first tile any of this workgroup's rows can see.
//
// This is the prefix-caching / chunked-prefill attention of the runtime
// slot (the reference delegates it to vLLM/SGLang images
// at arksapplication_controller.go:941-1002).
#include "common.h"

#include <cfloat>

namespace arks {

typedef __attribute__((ext_vector_type(8))) __bf16 e2_bf16x8;
typedef __attribute__((ext_vector_type(4))) __bf16 e2_bf16x4;
typedef __attribute__((ext_vector_type(16))) float f32x16;

__device__ __forceinline__ f32x16 e2_mfma32(e2_bf16x8 a, e2_bf16x8 b, f32x16 c) {
  return __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
}

constexpr int E2_WAVES = 8;
constexpr int E2_QTILE = 256;      // q rows per workgroup (32 per wave)
constexpr int E2_KVBLK = 64;       // keys per LDS tile
constexpr int E2_PAGE = 16;        // KV page size (tokens)
constexpr float E2_LOG2E = 1.44269504088896340736f;
constexpr float RESCALE_THR = 8.f; // defer-max threshold (exp2 domain)

// v_cvt_pk_bf16_f32: pack two f32 into one dword of two bf16 (RNE).
__device__ __forceinline__ uint32_t cvt_pk_bf16(float lo, float hi) {
  uint32_t r;
  asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(lo), "v"(hi));
  return r;
}

// V-image geometry: 4-key x 16-col blocks padded to 80 elements (160 B) so
// the cooperative b128 staging writes hit all 32 write-banks (stride-160
// spreads the 8-lane contiguous write groups; packed-128 is 4-way
// conflicted) while the tr16 read pairs overlap by only 8 of 64 banks.
constexpr int VBLK_ELEMS = 80;

template <int HEAD_DIM>
__global__ __launch_bounds__(512) void attn_extend2_kernel(
    bf16* __restrict__ out,            // [Tq, Hq, D]
    const bf16* __restrict__ q,        // [Tq, Hq, D] (packed new tokens)
    const bf16* __restrict__ k_cache,  // [nblocks, Hkv, 16, D]
    const bf16* __restrict__ v_cache,  // [nblocks, Hkv, 16, D]
    const int* __restrict__ block_tables,  // [num_seqs, max_blocks]
    const int* __restrict__ kv_lens,       // [num_seqs]
    const int* __restrict__ cu_seqlens_q,  // [num_seqs + 1]
    // [ntiles, 4] = (seq_idx, q0, part_idx, nparts): nparts > 1 splits the
    // KV range flash-decode-style; partials go to `part_ws` and a combine
    // kernel merges them (load balance: causal q-tiles differ 4x in KV
    // depth and small batches leave the chip half idle otherwise)
    const int* __restrict__ tile_info,
    float* __restrict__ part_ws,  // [ntiles, Hq, QTILE, D+2] f32 slabs
    const float scale, const int num_q_heads, const int num_kv_heads,
    const int max_blocks, const int64_t q_stride, const int window) {
  constexpr int STEPS = HEAD_DIM / 16;   // QK^T K-contraction steps
  constexpr int DTILES = HEAD_DIM / 32;  // 32-dim output tiles
  constexpr int CB = HEAD_DIM / 16;      // 16-col blocks per key row

  const int h = blockIdx.x;
  const int kvh = h / (num_q_heads / num_kv_heads);
  const int seq_idx = tile_info[blockIdx.y * 4];
  const int q0 = tile_info[blockIdx.y * 4 + 1];
  const int part = tile_info[blockIdx.y * 4 + 2];
  const int nparts = tile_info[blockIdx.y * 4 + 3];
  const int q_start = cu_seqlens_q[seq_idx];
  const int q_len = cu_seqlens_q[seq_idx + 1] - q_start;
  const int kv_len = kv_lens[seq_idx];
  const int kv_off = kv_len - q_len;  // cached-prefix length
  const int* __restrict__ bt = block_tables + (int64_t)seq_idx * max_blocks;

  const int tid = threadIdx.x;
  const int wave = tid / WAVE_SIZE;
  const int lane = tid % WAVE_SIZE;
  const int col = lane & 31;   // q-row within wave (QK^T) / dim col (PV)
  const int half = lane >> 5;  // lane half

  // Dynamic LDS: K tiles (XOR-swizzled 2*HEAD_DIM-byte rows) then the V
  // tr16 image, both double-buffered. 73.7 KiB at D=128 (> the 64 KiB
  // static limit; the launcher raises the dynamic-LDS cap).
  extern __shared__ __attribute__((aligned(16))) char lds_raw[];
  bf16* k_lds = reinterpret_cast<bf16*>(lds_raw);  // [2][KVBLK][HEAD_DIM]
  bf16* v_img = k_lds + 2 * E2_KVBLK * HEAD_DIM;   // [2][16][CB][80]
  constexpr int KBUF = E2_KVBLK * HEAD_DIM;
  constexpr int VBUF = 16 * CB * VBLK_ELEMS;

  // --- Q fragments (registers, loaded once): lane's q-row = q0+32w+col,
  // frag s holds dims 16s+8*half .. +7.
  const int my_qrow = q0 + wave * 32 + col;
  const bool qrow_valid = my_qrow < q_len;
  const int my_pos = kv_off + my_qrow;  // global kv position of this q row
  e2_bf16x8 qfrag[STEPS];
  {
    const int64_t qbase =
        (int64_t)(q_start + (qrow_valid ? my_qrow : 0)) * q_stride +
        (int64_t)h * HEAD_DIM + 8 * half;
#pragma unroll
    for (int s = 0; s < STEPS; ++s) {
      ushort8 u = *reinterpret_cast<const ushort8*>(q + qbase + s * 16);
      qfrag[s] = *reinterpret_cast<e2_bf16x8*>(&u);
    }
  }

  // Online-softmax state (per lane for q-row `my_qrow`, replicated across
  // the two halves). The running max is kept in RAW score units (scale is
  // folded into the exp2 argument); masked scores use a sentinel so far
  // below any real score that exp2(fma(sentinel, scale2, -m*scale2))
  // underflows to 0 with no per-element select.
  constexpr float MASKED = -3e38f;
  float m_run = -1e30f;
  float l_run = 0.f;
  f32x16 oacc[DTILES];
#pragma unroll
  for (int n = 0; n < DTILES; ++n) oacc[n] = {};

  const int kmax = min(kv_len, kv_off + q0 + E2_QTILE);
  const int all_tiles = (kmax + E2_KVBLK - 1) / E2_KVBLK;
  // Sliding window: the first key visible to any row of this tile.
  const int jw =
      (window > 0) ? max(0, (kv_off + q0 - window + 1) / E2_KVBLK) : 0;
  // this part's contiguous slice of the KV tile range
  const int span = all_tiles - jw;
  const int per = (span + nparts - 1) / nparts;
  const int j0 = jw + part * per;
  const int ntiles = min(all_tiles, j0 + per);
  const float scale2 = scale * E2_LOG2E;

  // Cooperative staging: 512 threads x 2 vec8 cover one 64x128 tile (one
  // vec8 per thread at D=64).
  constexpr int NVEC = E2_KVBLK * HEAD_DIM / 8;
  constexpr int VPT = NVEC / 512;
  const int64_t page_elems = (int64_t)num_kv_heads * E2_PAGE * HEAD_DIM;
  const int64_t head_off = (int64_t)kvh * E2_PAGE * HEAD_DIM;

  // Branchy per-element guards de-pipeline global loads (guide §5 trap (c)),
  // so full in-range tiles take an unguarded path; only the final (ragged)
  // tile pays the per-element branch.
  auto load_tile = [&](int j, ushort8* kr, ushort8* vr) {
    const int key_base = j * E2_KVBLK;
    if (key_base + E2_KVBLK <= kmax) {
#pragma unroll
      for (int vv = 0; vv < VPT; ++vv) {
        const int e = tid + 512 * vv;
        const int key = e / (HEAD_DIM / 8);
        const int col8 = (e % (HEAD_DIM / 8)) * 8;
        const int kg = key_base + key;
        const int64_t pg = max(0, bt[kg / E2_PAGE]);
        const int64_t src = pg * page_elems + head_off +
                            (int64_t)(kg % E2_PAGE) * HEAD_DIM + col8;
        kr[vv] = *reinterpret_cast<const ushort8*>(k_cache + src);
        vr[vv] = *reinterpret_cast<const ushort8*>(v_cache + src);
      }
      return;
    }
#pragma unroll
    for (int vv = 0; vv < VPT; ++vv) {
      const int e = tid + 512 * vv;
      const int key = e / (HEAD_DIM / 8);
      const int col8 = (e % (HEAD_DIM / 8)) * 8;
      const int kg = key_base + key;
      ushort8 kv{}, vv8{};
      if (kg < kmax) {
        // window-dropped pages are -1 in the block table; sanitized to 0
        // (their tokens are masked out)
        const int64_t pg = max(0, bt[kg / E2_PAGE]);
        const int64_t src = pg * page_elems + head_off +
                            (int64_t)(kg % E2_PAGE) * HEAD_DIM + col8;
        kv = *reinterpret_cast<const ushort8*>(k_cache + src);
        vv8 = *reinterpret_cast<const ushort8*>(v_cache + src);
      }
      kr[vv] = kv;
      vr[vv] = vv8;
    }
  };

  auto store_tile = [&](int buf, const ushort8* kr, const ushort8* vr) {
#pragma unroll
    for (int vv = 0; vv < VPT; ++vv) {
      const int e = tid + 512 * vv;
      const int key = e / (HEAD_DIM / 8);
      const int col8 = (e % (HEAD_DIM / 8)) * 8;
      // K: XOR-swizzled byte offset within the 2*HEAD_DIM-byte row.
      const int swz = (col8 * 2) ^ ((key & 7) << 4);
      *reinterpret_cast<ushort8*>(
          reinterpret_cast<char*>(k_lds + buf * KBUF + key * HEAD_DIM) + swz) =
          kr[vv];
      // V image: block (kb = key/4, cb = col8/16), row key%4, col col8%16.
      bf16* blk = v_img + buf * VBUF + ((key >> 2) * CB + (col8 >> 4)) * VBLK_ELEMS;
      *reinterpret_cast<ushort8*>(blk + (key & 3) * 16 + (col8 & 15)) = vr[vv];
    }
  };

  {
    ushort8 kr0[VPT], vr0[VPT];
    load_tile(j0, kr0, vr0);
    store_tile(0, kr0, vr0);
  }
  __syncthreads();

  int buf = 0;
  for (int j = j0; j < ntiles; ++j, buf ^= 1) {
    const int key_base = j * E2_KVBLK;
    // ---- prefetch tile j+1 into registers (overlaps with the MFMAs).
    ushort8 krn[VPT], vrn[VPT];
    if (j + 1 < ntiles) load_tile(j + 1, krn, vrn);

    // ---- QK^T (swapped): A = K rows (32 keys), B = Q. C[key][qrow=col].
    // The two subtiles interleave so consecutive MFMAs hit independent
    // accumulators (the 32x32 dependent-accumulator latency exceeds the
    // 32-cycle issue interval).
    f32x16 sc[2];
    sc[0] = {};
    sc[1] = {};
    const char* krow0 = reinterpret_cast<const char*>(
        k_lds + buf * KBUF + col * HEAD_DIM);
    const char* krow1 = krow0 + 32 * HEAD_DIM * 2;
    const int swz_mask = (col & 7) << 4;
#pragma unroll
    for (int s = 0; s < STEPS; ++s) {
      const int off = ((s * 16 + 8 * half) * 2) ^ swz_mask;
      ushort8 u0 = *reinterpret_cast<const ushort8*>(krow0 + off);
      ushort8 u1 = *reinterpret_cast<const ushort8*>(krow1 + off);
      sc[0] = e2_mfma32(*reinterpret_cast<e2_bf16x8*>(&u0), qfrag[s], sc[0]);
      sc[1] = e2_mfma32(*reinterpret_cast<e2_bf16x8*>(&u1), qfrag[s], sc[1]);
    }

    // ---- Masked softmax, fully in-register (exp2 domain). Lane's 32
    // scores are all for q-row `my_qrow`; C reg 4g+r of subtile `sub` is
    // key key_base + 32*sub + 8g + 4*half + r. Tiles fully visible to
    // every row of this wave (the bulk of the causal region) skip the 32
    // per-key compares; the reductions are trees, not 31-deep chains.
    float p[32];
    const bool full_vis =
        (q0 + wave * 32 + 31 < q_len) &&
        (key_base + E2_KVBLK - 1 <= kv_off + q0 + wave * 32) &&
        (key_base + E2_KVBLK <= kmax) &&
        (window <= 0 || key_base > kv_off + q0 + wave * 32 + 31 - window);
    if (full_vis) {
#pragma unroll
      for (int sub = 0; sub < 2; ++sub)
#pragma unroll
        for (int r16 = 0; r16 < 16; ++r16)
          p[sub * 16 + r16] = sc[sub][r16];
    } else {
#pragma unroll
      for (int sub = 0; sub < 2; ++sub) {
        const int kb_s = key_base + 32 * sub + 4 * half;
#pragma unroll
        for (int r16 = 0; r16 < 16; ++r16) {
          const int kg = kb_s + 8 * (r16 >> 2) + (r16 & 3);
          bool ok = qrow_valid && kg <= my_pos && kg < kmax;
          if (window > 0) ok = ok && (kg > my_pos - window);
          p[sub * 16 + r16] = ok ? sc[sub][r16] : MASKED;
        }
      }
    }
    float mt[16];
#pragma unroll
    for (int i = 0; i < 16; ++i) mt[i] = fmaxf(p[i], p[i + 16]);
#pragma unroll
    for (int w = 8; w > 0; w >>= 1)
#pragma unroll
      for (int i = 0; i < w; ++i) mt[i] = fmaxf(mt[i], mt[i + w]);
    const float tmax = fmaxf(mt[0], __shfl_xor(mt[0], 32, WAVE_SIZE));

    // Defer-max (T13): only rescale O when the tile max meaningfully
    // exceeds the running max (threshold in raw units; exp2(THR)=256
    // stays comfortably finite).
    const float thr_raw = RESCALE_THR / scale2;
    if (!__all(tmax - m_run <= thr_raw)) {
      const float m_new = fmaxf(m_run, tmax);
      const float alpha = __builtin_amdgcn_exp2f((m_run - m_new) * scale2);
      l_run *= alpha;
      m_run = m_new;
      // O rows live across regs (C layout): fetch each row's alpha from
      // the lane that owns that q-row.
#pragma unroll
      for (int r16 = 0; r16 < 16; ++r16) {
        const int row = 8 * (r16 >> 2) + 4 * half + (r16 & 3);
        const float a_r = __shfl(alpha, 32 * (lane >> 5) + row, WAVE_SIZE);
#pragma unroll
        for (int n = 0; n < DTILES; ++n) oacc[n][r16] *= a_r;
      }
    }

    // p = exp2((s - m) * scale2) as one fma + exp2; MASKED entries
    // underflow to exactly 0 (|MASKED*scale2| >> |m*scale2|).
    const float nms = -m_run * scale2;
#pragma unroll
    for (int i = 0; i < 32; ++i)
      p[i] = __builtin_amdgcn_exp2f(__builtin_fmaf(p[i], scale2, nms));
    float st[16];
#pragma unroll
    for (int i = 0; i < 16; ++i) st[i] = p[i] + p[i + 16];
#pragma unroll
    for (int w = 8; w > 0; w >>= 1)
#pragma unroll
      for (int i = 0; i < w; ++i) st[i] += st[i + w];
    l_run += st[0] + __shfl_xor(st[0], 32, WAVE_SIZE);

    // ---- P -> bf16 PV A-fragments: per subtile, 8 cvt_pk words
    // W[g][0] = pk(p[4g], p[4g+1]), W[g][1] = pk(p[4g+2], p[4g+3]); the
    // permlane32_swap pairs (W[2t][*], W[2t+1][*]) then give each half its
    // step-t fragment from both halves' key sets.
    uint32_t afrag[4][4];  // [k-step t][4 dwords = 8 bf16]
#pragma unroll
    for (int sub = 0; sub < 2; ++sub) {
      uint32_t w0[4], w1[4];
#pragma unroll
      for (int g = 0; g < 4; ++g) {
        const int b = sub * 16 + 4 * g;
        w0[g] = cvt_pk_bf16(p[b + 0], p[b + 1]);
        w1[g] = cvt_pk_bf16(p[b + 2], p[b + 3]);
      }
#pragma unroll
      for (int tau = 0; tau < 2; ++tau) {
        auto r0 = __builtin_amdgcn_permlane32_swap(w0[2 * tau], w0[2 * tau + 1],
                                                   false, false);
        auto r1 = __builtin_amdgcn_permlane32_swap(w1[2 * tau], w1[2 * tau + 1],
                                                   false, false);
        const int t = sub * 2 + tau;
        afrag[t][0] = r0[0];
        afrag[t][1] = r1[0];
        afrag[t][2] = r0[1];
        afrag[t][3] = r1[1];
      }
    }

    // ---- PV: O[qrow][dim] += P[qrow][key] V[key][dim]; B fragments via
    // tr16 reads of the V image (per step t: keys 16t+8*half+0..7 at
    // column 16*(colhalf)+l&15 -> blocks kb = 4t+2*half (+1), cb per group).
    const int g4 = lane >> 4;  // 16-lane tr16 group
    bf16* vbase = v_img + buf * VBUF;
    // t outer / n inner: consecutive MFMAs accumulate into the DTILES
    // independent oacc registers instead of chaining one.
#pragma unroll
    for (int t = 0; t < 4; ++t) {
      const int kb = 4 * t + 2 * (g4 >> 1);
#pragma unroll
      for (int n = 0; n < DTILES; ++n) {
        const int cb = 2 * n + (g4 & 1);
        e2_bf16x4 v1 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
            (__attribute__((address_space(3))) e2_bf16x4*)(
                reinterpret_cast<char*>(vbase + (kb * CB + cb) * VBLK_ELEMS) +
                (lane & 15) * 8));
        e2_bf16x4 v2 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
            (__attribute__((address_space(3))) e2_bf16x4*)(
                reinterpret_cast<char*>(vbase + ((kb + 1) * CB + cb) * VBLK_ELEMS) +
                (lane & 15) * 8));
        e2_bf16x8 vb;
#pragma unroll
        for (int e = 0; e < 4; ++e) {
          vb[e] = v1[e];
          vb[e + 4] = v2[e];
        }
        oacc[n] = e2_mfma32(*reinterpret_cast<e2_bf16x8*>(&afrag[t][0]), vb,
                            oacc[n]);
      }
    }

    // ---- write-late: scatter the prefetched tile into the idle buffer.
    if (j + 1 < ntiles) store_tile(buf ^ 1, krn, vrn);
    __syncthreads();
  }

  // ---- Epilogue. O reg 4g+r of dim-tile n is q-row 8g+4*half+r, dim
  // 32n+col; l_run for row `row` lives in lane row (either half).
  if (nparts > 1) {
    // partial: UNNORMALIZED O + per-row (m, l) into this part's slab
    // [QTILE rows][D+2] f32 (combined by attn_extend2_combine)
    float* slab = part_ws +
        ((int64_t)blockIdx.y * num_q_heads + h) * E2_QTILE * (HEAD_DIM + 2);
#pragma unroll
    for (int r16 = 0; r16 < 16; ++r16) {
      const int row = 8 * (r16 >> 2) + 4 * half + (r16 & 3);
      float* wr = slab + (int64_t)(wave * 32 + row) * (HEAD_DIM + 2) + col;
#pragma unroll
      for (int n = 0; n < DTILES; ++n) wr[32 * n] = oacc[n][r16];
    }
    if (half == 0) {
      // lane `col` owns q-row `col` of this wave: write its m/l
      float* wr = slab + (int64_t)(wave * 32 + col) * (HEAD_DIM + 2);
      wr[HEAD_DIM] = m_run;
      wr[HEAD_DIM + 1] = l_run;
    }
    return;
  }
#pragma unroll
  for (int r16 = 0; r16 < 16; ++r16) {
    const int row = 8 * (r16 >> 2) + 4 * half + (r16 & 3);
    const int qrow = q0 + wave * 32 + row;
    const float lr = __shfl(l_run, 32 * (lane >> 5) + row, WAVE_SIZE);
    if (qrow >= q_len) continue;
    const float inv = lr > 0.f ? 1.f / lr : 0.f;
    const int64_t obase =
        ((int64_t)(q_start + qrow) * num_q_heads + h) * HEAD_DIM + col;
#pragma unroll
    for (int n = 0; n < DTILES; ++n) {
      out[obase + 32 * n] = float_to_bf16_bits(oacc[n][r16] * inv);
    }
  }
}

// Merge the flash partials of split tiles: grid (Hq, n_split, row-groups
// of 8) — the row-group axis exists purely for parallelism (a single
// (head, tile) block is latency-bound on the strided part reads);
// combine_table rows = (first_y, nparts, seq, q0). Standard flash merge
// in f32: m* = max m_p; out = sum_p exp(m_p - m*) O_p / sum_p
// exp(m_p - m*) l_p (m is base-2: exp2).
template <int HEAD_DIM>
__global__ __launch_bounds__(256) void attn_extend2_combine_kernel(
    bf16* __restrict__ out, const float* __restrict__ part_ws,
    const int* __restrict__ combine_table,
    const int* __restrict__ cu_seqlens_q, const float scale,
    const int num_q_heads) {
  const int h = blockIdx.x;
  const int ci = blockIdx.y;
  const int first_y = combine_table[ci * 4];
  const int nparts = combine_table[ci * 4 + 1];
  const int seq_idx = combine_table[ci * 4 + 2];
  const int q0 = combine_table[ci * 4 + 3];
  const int q_start = cu_seqlens_q[seq_idx];
  const int q_len = cu_seqlens_q[seq_idx + 1] - q_start;
  const float scale2 = scale * E2_LOG2E;
  const int64_t slab_stride = (int64_t)num_q_heads * E2_QTILE * (HEAD_DIM + 2);
  const float* base = part_ws + (int64_t)first_y * slab_stride +
                      (int64_t)h * E2_QTILE * (HEAD_DIM + 2);
  // this block's 8-row group; 256 threads = 256/HEAD_DIM rows per pass
  const int row0 = blockIdx.z * 8;
  for (int row = row0 + threadIdx.x / HEAD_DIM; row < row0 + 8;
       row += 256 / HEAD_DIM) {
    const int qrow = q0 + row;
    if (qrow >= q_len) continue;
    const int d = threadIdx.x % HEAD_DIM;
    const float* r0 = base + (int64_t)row * (HEAD_DIM + 2);
    float m_star = -1e30f;
    for (int p = 0; p < nparts; ++p)
      m_star = fmaxf(m_star, r0[p * slab_stride + HEAD_DIM]);
    float l_tot = 0.f;
    float acc = 0.f;
    for (int p = 0; p < nparts; ++p) {
      const float mp = r0[p * slab_stride + HEAD_DIM];
      const float lp = r0[p * slab_stride + HEAD_DIM + 1];
      const float a = __builtin_amdgcn_exp2f((mp - m_star) * scale2);
      l_tot += a * lp;
      acc += a * r0[p * slab_stride + d];
    }
    const float inv = l_tot > 0.f ? 1.f / l_tot : 0.f;
    out[((int64_t)(q_start + qrow) * num_q_heads + h) * HEAD_DIM + d] =
        float_to_bf16_bits(acc * inv);
  }
}

}  // namespace arks

using namespace arks;

extern "C" void arks_attn_extend_paged2(
    void* out, const void* q, const void* k_cache, const void* v_cache,
    const void* block_tables, const void* kv_lens, const void* cu_seqlens_q,
    const void* tile_info, void* part_ws, int ntiles, float scale,
    int num_q_heads, int num_kv_heads, int head_dim, int max_blocks,
    int64_t q_stride, int window, hipStream_t stream) {
  dim3 grid(num_q_heads, ntiles), block(512);
  static bool attr_set[2] = {false, false};
  auto launch = [&](auto kern, int lds_bytes, int idx) {
    if (!attr_set[idx]) {
      (void)hipFuncSetAttribute(reinterpret_cast<const void*>(kern),
                          hipFuncAttributeMaxDynamicSharedMemorySize,
                          lds_bytes);
      attr_set[idx] = true;
    }
    hipLaunchKernelGGL(kern, grid, block, lds_bytes, stream, (bf16*)out,
                       (const bf16*)q, (const bf16*)k_cache,
                       (const bf16*)v_cache, (const int*)block_tables,
                       (const int*)kv_lens, (const int*)cu_seqlens_q,
                       (const int*)tile_info, (float*)part_ws, scale,
                       num_q_heads, num_kv_heads, max_blocks, q_stride,
                       window);
  };
  if (head_dim == 128) {
    constexpr int LDS = (2 * 64 * 128 + 2 * 16 * 8 * VBLK_ELEMS) * 2;
    launch(attn_extend2_kernel<128>, LDS, 0);
  } else if (head_dim == 64) {
    constexpr int LDS = (2 * 64 * 64 + 2 * 16 * 4 * VBLK_ELEMS) * 2;
    launch(attn_extend2_kernel<64>, LDS, 1);
  }
}

extern "C" void arks_attn_extend2_combine(
    void* out, const void* part_ws, const void* combine_table,
    const void* cu_seqlens_q, int n_split, float scale, int num_q_heads,
    int head_dim, hipStream_t stream) {
  dim3 grid(num_q_heads, n_split, E2_QTILE / 8), block(256);
  if (head_dim == 128) {
    hipLaunchKernelGGL((attn_extend2_combine_kernel<128>), grid, block, 0,
                       stream, (bf16*)out, (const float*)part_ws,
                       (const int*)combine_table, (const int*)cu_seqlens_q,
                       scale, num_q_heads);
  } else if (head_dim == 64) {
    hipLaunchKernelGGL((attn_extend2_combine_kernel<64>), grid, block, 0,
                       stream, (bf16*)out, (const float*)part_ws,
                       (const int*)combine_table, (const int*)cu_seqlens_q,
                       scale, num_q_heads);
  }
}
