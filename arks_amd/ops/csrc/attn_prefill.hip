// Flash-style varlen causal prefill attention on MFMA (gfx950).
//
// Grid: (num_q_heads, num_q_tiles). A workgroup = 256 threads = 4 waves
// handles one 64-row Q tile of one sequence for one q-head; each wave owns 16
// Q rows. K/V are iterated in 32-key tiles staged in LDS (K XOR-swizzled
// against the D=128 row-major bank conflict — guide §6 G4; V stored
// transposed so the PV B-fragment is a contiguous ds_read_b128).
//
// QK^T uses swapped operands (A = K-tile, B = Q-tile) so each lane's 16x16x32
// C-fragment holds scores for ONE q-row (col = lane&15) — the online-softmax
// row reduction is then 8 regs + 2 shuffles, never a serial-lane loop
// (guide common-mistake #6). P is staged through a per-wave padded LDS buffer
// to re-shape for the PV A-fragment.
//
// This kernel is the prefill half of the runtime-slot contract described in
// SURVEY.md §2.4 (vLLM's flash prefill at reference
// arksapplication_controller.go:941-955 delegated it to an external image).
#include "common.h"

#include <cfloat>

namespace arks {

// gfx950 MFMA fragment types.
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4v;
typedef __attribute__((ext_vector_type(4))) float f32x4;

__device__ __forceinline__ f32x4 mfma16x16x32(bf16x8 a, bf16x8 b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

// A/B fragment k-index for lane quarter `a` (= lane/16), element jj in 0..7.
// Layout hypothesis H1: contiguous 8 elements per lane. Verified on hardware
// by tests/test_ops_gpu.py::test_mfma_probe.
__device__ __forceinline__ int frag_k(int a, int jj) { return 8 * a + jj; }

constexpr int QTILE_WAVE = 16;  // q rows per wave
constexpr int NUM_WAVES = 4;
constexpr int QTILE = QTILE_WAVE * NUM_WAVES;  // 64 q rows per workgroup
constexpr int KTILE = 32;                      // keys per LDS tile
constexpr int VT_PAD = 40;                     // padded VT row length (keys)

template <int HEAD_DIM>
__global__ __launch_bounds__(256) void attn_prefill_kernel(
    bf16* __restrict__ out,      // [T, Hq, D]
    const bf16* __restrict__ q,  // [T, Hq, D]
    const bf16* __restrict__ k,  // [T, Hkv, D]
    const bf16* __restrict__ v,  // [T, Hkv, D]
    const int* __restrict__ cu_seqlens,  // [num_seqs + 1]
    const int* __restrict__ tile_info,   // [ntiles, 2] = (seq_idx, q0)
    const float scale, const int num_q_heads, const int num_kv_heads,
    const int64_t q_stride, const int64_t kv_stride, const int window) {
  constexpr int CHUNKS = HEAD_DIM / 16;  // dim chunks for PV output
  constexpr int STEPS = HEAD_DIM / 32;   // K-contraction steps for QK^T

  const int h = blockIdx.x;
  const int kvh = h / (num_q_heads / num_kv_heads);
  const int seq_idx = tile_info[blockIdx.y * 2];
  const int q0 = tile_info[blockIdx.y * 2 + 1];
  const int seq_start = cu_seqlens[seq_idx];
  const int seq_len = cu_seqlens[seq_idx + 1] - seq_start;

  const int tid = threadIdx.x;
  const int wave = tid / WAVE_SIZE;
  const int lane = tid % WAVE_SIZE;
  const int lq = lane % 16;   // q-row (QK^T) / dim-col (PV) within fragment
  const int la = lane / 16;   // lane quarter

  // LDS: ping-pong K tiles (XOR-swizzled rows) + transposed V tiles, and a
  // per-wave P buffer. 16-B alignment required for the ushort8 (b128)
  // accesses — guide §6 G17. Double buffering + register staging makes the
  // tile loop a single-barrier software pipeline: tile j+1's global loads
  // land in registers during tile j's MFMAs, are scattered to the idle
  // buffer, and one __syncthreads flips the buffers (v1 took two barriers
  // per tile and stalled on the staging round-trip, ~118 TF effective).
  __shared__ __attribute__((aligned(16))) bf16 k_lds[2][KTILE][HEAD_DIM];
  // V tile as a tr16 image: [d-chunk cb][key-block kb][4 key x 16 col]
  // blocks of 64 bf16 padded to 96 so the two 16-lane regions of one
  // 32-lane ds_read_b64_tr_b16 group land on disjoint bank halves
  // (stride 192 B = 48 dwords; probe: gpurun_out/probe_layouts.log).
  // Each lane then receives the 4 keys of its column — B-fragment shape —
  // and V staging becomes two b128 stores instead of 16 scalar b16 writes
  // (the hot block of the v2 disassembly).
  __shared__ __attribute__((aligned(16))) bf16 vt_img[2][HEAD_DIM / 16][8][96];
  __shared__ __attribute__((aligned(16))) bf16 p_lds[NUM_WAVES][QTILE_WAVE][VT_PAD];

  // --- Q fragments: registers, loaded once. Wave w covers rows q0+16w..+15.
  const int my_qrow = q0 + wave * QTILE_WAVE + lq;  // row in sequence
  const bool qrow_valid = my_qrow < seq_len;
  bf16x8 qfrag[STEPS];
  {
    const int64_t qbase =
        (int64_t)(seq_start + (qrow_valid ? my_qrow : 0)) * q_stride +
        (int64_t)h * HEAD_DIM;
#pragma unroll
    for (int st = 0; st < STEPS; ++st) {
      ushort8 u = *reinterpret_cast<const ushort8*>(q + qbase + st * 32 +
                                                    frag_k(la, 0));
      qfrag[st] = *reinterpret_cast<bf16x8*>(&u);
    }
  }

  // Online softmax state: per lane for q-row `my_qrow` (replicated over the 4
  // lanes sharing lq).
  float m_run = -FLT_MAX;
  float l_run = 0.f;
  f32x4 oacc[CHUNKS];
#pragma unroll
  for (int c = 0; c < CHUNKS; ++c) oacc[c] = {0.f, 0.f, 0.f, 0.f};

  // Keys visible to this workgroup: strictly below kmax.
  const int kmax = min(seq_len, q0 + QTILE);
  const int ntiles = (kmax + KTILE - 1) / KTILE;
  // Sliding window: first tile any row of this q-tile can see.
  const int j0 = (window > 0) ? max(0, (q0 - window + 1) / KTILE) : 0;

  // Per-thread cooperative staging share: tile elements i = tid + 256*v.
  constexpr int NVEC = KTILE * HEAD_DIM / 8;      // vec8 per tile
  constexpr int VPT = NVEC / 256;                 // vec8 per thread (2 at D=128)

  auto load_tile = [&](int j, ushort8* kr, ushort8* vr) {
    const int key_base = j * KTILE;
#pragma unroll
    for (int vv8 = 0; vv8 < VPT; ++vv8) {
      const int i = tid + 256 * vv8;
      const int key = i / (HEAD_DIM / 8);
      const int col8 = (i % (HEAD_DIM / 8)) * 8;
      const int kg = key_base + key;
      ushort8 kv{}, vv{};
      if (kg < kmax) {
        const int64_t src =
            (int64_t)(seq_start + kg) * kv_stride + (int64_t)kvh * HEAD_DIM + col8;
        kv = *reinterpret_cast<const ushort8*>(k + src);
        vv = *reinterpret_cast<const ushort8*>(v + src);
      }
      kr[vv8] = kv;
      vr[vv8] = vv;
    }
  };

  auto store_tile = [&](int buf, const ushort8* kr, const ushort8* vr) {
#pragma unroll
    for (int vv8 = 0; vv8 < VPT; ++vv8) {
      const int i = tid + 256 * vv8;
      const int key = i / (HEAD_DIM / 8);
      const int col8 = (i % (HEAD_DIM / 8)) * 8;
      // K: swizzle byte offset within the 2*HEAD_DIM-byte row.
      const int row_byte = col8 * 2;
      const int swz = row_byte ^ ((key & 7) << 4);
      *reinterpret_cast<ushort8*>(
          reinterpret_cast<char*>(&k_lds[buf][key][0]) + swz) = kr[vv8];
      // V into the tr16 image: one b128 store per vec8.
      const int kb = key >> 2, r = key & 3;
      const int cb = col8 >> 4, i0 = col8 & 15;
      *reinterpret_cast<ushort8*>(&vt_img[buf][cb][kb][r * 16 + i0]) = vr[vv8];
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
    const int key_base = j * KTILE;
    // ---- prefetch tile j+1 into registers (global loads overlap MFMAs).
    ushort8 krn[VPT], vrn[VPT];
    if (j + 1 < ntiles) load_tile(j + 1, krn, vrn);

    // ---- QK^T: two 16-key subtiles; A = K from LDS, B = Q registers.
    f32x4 sc[2];
    sc[0] = {0.f, 0.f, 0.f, 0.f};
    sc[1] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int sub = 0; sub < 2; ++sub) {
      const int key = sub * 16 + lq;
#pragma unroll
      for (int st = 0; st < STEPS; ++st) {
        const int col_byte = (st * 32 + frag_k(la, 0)) * 2;
        const int swz = col_byte ^ ((key & 7) << 4);
        ushort8 u = *reinterpret_cast<const ushort8*>(
            reinterpret_cast<const char*>(&k_lds[buf][key][0]) + swz);
        sc[sub] = mfma16x16x32(*reinterpret_cast<bf16x8*>(&u), qfrag[st], sc[sub]);
      }
    }

    // ---- Masked online softmax. Lane holds 8 scores for q-row `my_qrow`,
    // keys key_base + sub*16 + 4*la + r.
    float p[8];
    float tile_max = -FLT_MAX;
    bool msk[8];
#pragma unroll
    for (int sub = 0; sub < 2; ++sub) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int kg = key_base + sub * 16 + 4 * la + r;
        const int i = sub * 4 + r;
        msk[i] = qrow_valid && (kg <= my_qrow) && (kg < kmax) &&
                 (window <= 0 || kg > my_qrow - window);
        p[i] = msk[i] ? sc[sub][r] * scale : -FLT_MAX;
        tile_max = fmaxf(tile_max, p[i]);
      }
    }
    // Row max across the 4 lanes sharing this q-row.
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

    // Rescale O accumulator. oacc rows are q-rows 4*la + r — fetch each row's
    // alpha from the lane that owns it (lane 4*la + r has lq == that row).
    float row_alpha[4];
#pragma unroll
    for (int r = 0; r < 4; ++r)
      row_alpha[r] = __shfl(alpha, 4 * la + r, WAVE_SIZE);
#pragma unroll
    for (int c = 0; c < CHUNKS; ++c) {
#pragma unroll
      for (int r = 0; r < 4; ++r) oacc[c][r] *= row_alpha[r];
    }

    // ---- P to LDS (bf16), then PV.
#pragma unroll
    for (int sub = 0; sub < 2; ++sub) {
      ushort4v pk;
#pragma unroll
      for (int r = 0; r < 4; ++r) pk[r] = float_to_bf16_bits(p[sub * 4 + r]);
      *reinterpret_cast<ushort4v*>(&p_lds[wave][lq][sub * 16 + 4 * la]) = pk;
    }
    // P is a per-wave buffer: an LDS-counter wait orders the read below
    // behind the writes without a workgroup barrier.
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

    // PV: one MFMA per 16-dim chunk contracts the full 32-key tile.
    ushort8 pa =
        *reinterpret_cast<const ushort8*>(&p_lds[wave][lq][frag_k(la, 0)]);
#pragma unroll
    for (int c = 0; c < CHUNKS; ++c) {
      // B-fragment via two hardware transpose reads: keys 8*la..+3, +4..7
      // at column lq of chunk c.
      bf16x4v v1 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
          (__attribute__((address_space(3))) bf16x4v*)(
              reinterpret_cast<char*>(&vt_img[buf][c][2 * la][0]) + lq * 8));
      bf16x4v v2 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
          (__attribute__((address_space(3))) bf16x4v*)(
              reinterpret_cast<char*>(&vt_img[buf][c][2 * la + 1][0]) + lq * 8));
      bf16x8 vb;
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        vb[e] = v1[e];
        vb[e + 4] = v2[e];
      }
      oacc[c] = mfma16x16x32(*reinterpret_cast<bf16x8*>(&pa), vb, oacc[c]);
    }
    // ---- scatter the prefetched tile into the idle buffer and flip.
    if (j + 1 < ntiles) store_tile(buf ^ 1, krn, vrn);
    __syncthreads();
  }

  // ---- Epilogue: normalize and store. oacc[c][r] is q-row 4*la+r, dim
  // c*16+lq. Each needed l_run lives in lane 4*la+r.
  float row_inv[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const float lr = __shfl(l_run, 4 * la + r, WAVE_SIZE);
    row_inv[r] = lr > 0.f ? 1.f / lr : 0.f;
  }
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int row = q0 + wave * QTILE_WAVE + 4 * la + r;
    if (row >= seq_len) continue;
    const int64_t obase = ((int64_t)(seq_start + row) * num_q_heads + h) * HEAD_DIM;
#pragma unroll
    for (int c = 0; c < CHUNKS; ++c) {
      out[obase + c * 16 + lq] = float_to_bf16_bits(oacc[c][r] * row_inv[r]);
    }
  }
}

// Single-wave probe: D[16,16] = A[16,32] @ B[32,16] using the fragment
// layout conventions of the kernels above. Lets the GPU test suite verify the
// lane->element mapping against torch.matmul before trusting the attention
// kernel (guide §3: always check with asymmetric operands).
__global__ void mfma_probe_kernel(float* __restrict__ d,
                                  const bf16* __restrict__ a,
                                  const bf16* __restrict__ b) {
  const int lane = threadIdx.x % WAVE_SIZE;
  const int lq = lane % 16;
  const int la = lane / 16;
  bf16x8 af, bf_;
#pragma unroll
  for (int jj = 0; jj < 8; ++jj) {
    // A[row=lq][k], B[k][col=lq]
    ushort au = a[lq * 32 + frag_k(la, jj)];
    ushort bu = b[frag_k(la, jj) * 16 + lq];
    af[jj] = *reinterpret_cast<__bf16*>(&au);
    bf_[jj] = *reinterpret_cast<__bf16*>(&bu);
  }
  f32x4 c = {0.f, 0.f, 0.f, 0.f};
  c = mfma16x16x32(af, bf_, c);
#pragma unroll
  for (int r = 0; r < 4; ++r) d[(4 * la + r) * 16 + lq] = c[r];
}

}  // namespace arks

using namespace arks;

// 32x32x16 probe: D[32,32] = A[32,16] @ B[16,32] using the layout
// hypothesis A[row=l%32][k=8*(l/32)+j], B[k=8*(l/32)+j][col=l%32],
// C[col=l%32][rows 4*(l/32) + 8*g + r] per f32x4 acc g. Verified on
// hardware by tests/test_ops_gpu.py::test_mfma_probe_32x32 before any
// kernel trusts it (guide §3).
__global__ void mfma_probe32_kernel(float* __restrict__ d,
                                    const bf16* __restrict__ a,
                                    const bf16* __restrict__ b) {
  typedef __attribute__((ext_vector_type(16))) float f32x16;
  const int lane = threadIdx.x % WAVE_SIZE;
  const int col = lane % 32;
  const int half = lane / 32;
  bf16x8 af, bf_;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    ushort au = a[col * 16 + 8 * half + j];   // A[row=col][k]
    ushort bu = b[(8 * half + j) * 32 + col]; // B[k][col]
    af[j] = *reinterpret_cast<__bf16*>(&au);
    bf_[j] = *reinterpret_cast<__bf16*>(&bu);
  }
  f32x16 c = {};
  c = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, bf_, c, 0, 0, 0);
#pragma unroll
  for (int g = 0; g < 4; ++g) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = 8 * g + 4 * half + r;
      d[row * 32 + col] = c[g * 4 + r];
    }
  }
}


// tr16 probe: LDS[i] = i (u16), each lane does ds_read_b64_tr_b16 at
// address lane*8 + row_sel*stride; dumps its 4 values so the host can
// derive the exact lane->element mapping empirically before any kernel
// uses the transpose read (T10 pitfall: misaligned tr reads return
// wrong data silently).
__global__ void tr16_probe_kernel(uint16_t* __restrict__ out,
                                  const int stride_bytes) {
  typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4v;
  __shared__ uint16_t lds[4096];
  const int lane = threadIdx.x;
  for (int i = lane; i < 4096; i += 64) lds[i] = (uint16_t)i;
  __syncthreads();
  char* base = reinterpret_cast<char*>(&lds[0]);
  auto addr = (__attribute__((address_space(3))) bf16x4v*)(
      base + (int64_t)lane * stride_bytes);
  bf16x4v v = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(addr);
  union { bf16x4v bv; uint16_t u[4]; } cv;
  cv.bv = v;
#pragma unroll
  for (int j = 0; j < 4; ++j) out[lane * 4 + j] = cv.u[j];
}

extern "C" void arks_tr16_probe(void* out, int stride_bytes,
                                hipStream_t stream) {
  hipLaunchKernelGGL(tr16_probe_kernel, dim3(1), dim3(64), 0, stream,
                     (uint16_t*)out, stride_bytes);
}

extern "C" void arks_mfma_probe32(void* d, const void* a, const void* b,
                                  hipStream_t stream) {
  hipLaunchKernelGGL(mfma_probe32_kernel, dim3(1), dim3(64), 0, stream,
                     (float*)d, (const bf16*)a, (const bf16*)b);
}

extern "C" void arks_mfma_probe(void* d, const void* a, const void* b,
                                hipStream_t stream) {
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, stream, (float*)d,
                     (const bf16*)a, (const bf16*)b);
}

extern "C" void arks_attn_prefill_varlen(
    void* out, const void* q, const void* k, const void* v,
    const void* cu_seqlens, const void* tile_info, int ntiles, float scale,
    int num_q_heads, int num_kv_heads, int head_dim, int64_t q_stride,
    int64_t kv_stride, int window, hipStream_t stream) {
  dim3 grid(num_q_heads, ntiles), block(256);
  if (head_dim == 128) {
    hipLaunchKernelGGL((attn_prefill_kernel<128>), grid, block, 0, stream,
                       (bf16*)out, (const bf16*)q, (const bf16*)k,
                       (const bf16*)v, (const int*)cu_seqlens,
                       (const int*)tile_info, scale, num_q_heads, num_kv_heads,
                       q_stride, kv_stride, window);
  } else if (head_dim == 64) {
    hipLaunchKernelGGL((attn_prefill_kernel<64>), grid, block, 0, stream,
                       (bf16*)out, (const bf16*)q, (const bf16*)k,
                       (const bf16*)v, (const int*)cu_seqlens,
                       (const int*)tile_info, scale, num_q_heads, num_kv_heads,
                       q_stride, kv_stride, window);
  }
}
