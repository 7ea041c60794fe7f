// Dynamic per-row (per-token) fp8 e4m3 quantization for W8A8 serving GEMMs.
//
// gfx950 fp8 is OCP e4m3fn (NOT the MI300X fnuz encoding — guide §4), so the
// output bytes match torch.float8_e4m3fn and feed torch._scaled_mm /
// hipBLASLt directly. One workgroup per row: pass 1 reduces |x|max with
// 16 B/lane vector loads, pass 2 re-reads (L1/L2-resident rows at serving
// sizes), scales to the e4m3 range (+-448) and converts with the hardware
// cvt instructions via __hip_fp8_e4m3.
#include "common.h"

#include <hip/hip_fp8.h>

namespace arks {

typedef __attribute__((ext_vector_type(8))) uint8_t uchar8;

__global__ void quant_fp8_rows_kernel(uint8_t* __restrict__ out,
                                      float* __restrict__ inv_scale,
                                      const bf16* __restrict__ x,
                                      const int cols) {
  constexpr int BLOCK = 256;
  const int row = blockIdx.x;
  const bf16* in_row = x + (int64_t)row * cols;
  uint8_t* out_row = out + (int64_t)row * cols;
  const int nvec = cols / 8;

  float amax = 0.f;
  for (int i = threadIdx.x; i < nvec; i += BLOCK) {
    ushort8 v = *reinterpret_cast<const ushort8*>(in_row + i * 8);
#pragma unroll
    for (int j = 0; j < 8; ++j) amax = fmaxf(amax, fabsf(bf16_bits_to_float(v[j])));
  }
  __shared__ float red[BLOCK / WAVE_SIZE];
  {  // block max via the shared scratch (same shape as block_reduce_sum)
    const int wave = threadIdx.x / WAVE_SIZE;
    float w = wave_reduce_max(amax);
    if (threadIdx.x % WAVE_SIZE == 0) red[wave] = w;
    __syncthreads();
    amax = fmaxf(fmaxf(red[0], red[1]), fmaxf(red[2], red[3]));
  }
  const float a = fmaxf(amax, 1e-6f);
  const float scale = 448.f / a;       // quant multiplier
  if (threadIdx.x == 0) inv_scale[row] = a / 448.f;  // dequant scale for the GEMM

  for (int i = threadIdx.x; i < nvec; i += BLOCK) {
    ushort8 v = *reinterpret_cast<const ushort8*>(in_row + i * 8);
    uchar8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      __hip_fp8_e4m3 q(bf16_bits_to_float(v[j]) * scale);
      o[j] = q.__x;
    }
    *reinterpret_cast<uchar8*>(out_row + i * 8) = o;
  }
}

}  // namespace arks

using namespace arks;

extern "C" void arks_quant_fp8_rows(void* out, void* inv_scale, const void* x,
                                    int rows, int cols, hipStream_t stream) {
  hipLaunchKernelGGL(quant_fp8_rows_kernel, dim3(rows), dim3(256), 0, stream,
                     (uint8_t*)out, (float*)inv_scale, (const bf16*)x, cols);
}

namespace arks {

// ---------------------------------------------------------------------------
// Fused RMSNorm -> fp8 e4m3 (+ optional residual add, same contract as
// rmsnorm_kernel<FUSED_ADD> in elementwise.hip): one pass loads x (and w)
// into registers, so the normalized row's amax is known before any output
// is written and the whole op stays single-read single-write. Requires
// hidden <= 8192 (<= 4 vec8 per thread); the Python wrapper falls back to
// rmsnorm + quant_fp8_rows beyond that.
// ---------------------------------------------------------------------------
template <bool FUSED_ADD>
__global__ void rmsnorm_fp8_kernel(uint8_t* __restrict__ out,
                                   float* __restrict__ inv_scale,
                                   const bf16* __restrict__ input,
                                   bf16* __restrict__ residual,
                                   const bf16* __restrict__ weight,
                                   const float eps, const int hidden) {
  constexpr int BLOCK = 256;
  constexpr int MAXV = 4;  // vec8 per thread
  const int row = blockIdx.x;
  const bf16* in_row = input + (int64_t)row * hidden;
  bf16* res_row = FUSED_ADD ? residual + (int64_t)row * hidden : nullptr;
  uint8_t* out_row = out + (int64_t)row * hidden;
  const int nvec = hidden / 8;

  float xv[MAXV][8];
  float ss = 0.f;
  int nv = 0;
  for (int i = threadIdx.x; i < nvec; i += BLOCK, ++nv) {
    ushort8 x = *reinterpret_cast<const ushort8*>(in_row + i * 8);
    if constexpr (FUSED_ADD) {
      ushort8 r = *reinterpret_cast<const ushort8*>(res_row + i * 8);
      ushort8 s;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float v = bf16_bits_to_float(x[j]) + bf16_bits_to_float(r[j]);
        s[j] = float_to_bf16_bits(v);
        const float vb = bf16_bits_to_float(s[j]);
        xv[nv][j] = vb;
        ss += vb * vb;
      }
      *reinterpret_cast<ushort8*>(res_row + i * 8) = s;
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float v = bf16_bits_to_float(x[j]);
        xv[nv][j] = v;
        ss += v * v;
      }
    }
  }

  __shared__ float red[BLOCK / WAVE_SIZE];
  float total = block_reduce_sum<BLOCK>(ss, red);
  const float rrms = rsqrtf(total / (float)hidden + eps);

  // normalized values + row amax (weights read once)
  float amax = 0.f;
  int k = 0;
  for (int i = threadIdx.x; i < nvec; i += BLOCK, ++k) {
    ushort8 w = *reinterpret_cast<const ushort8*>(weight + i * 8);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float y = xv[k][j] * rrms * bf16_bits_to_float(w[j]);
      xv[k][j] = y;
      amax = fmaxf(amax, fabsf(y));
    }
  }
  {
    const int wave = threadIdx.x / WAVE_SIZE;
    float wmax = wave_reduce_max(amax);
    __syncthreads();  // red[] reuse after block_reduce_sum
    if (threadIdx.x % WAVE_SIZE == 0) red[wave] = wmax;
    __syncthreads();
    amax = fmaxf(fmaxf(red[0], red[1]), fmaxf(red[2], red[3]));
  }
  const float a = fmaxf(amax, 1e-6f);
  const float scale = 448.f / a;
  if (threadIdx.x == 0) inv_scale[row] = a / 448.f;

  k = 0;
  for (int i = threadIdx.x; i < nvec; i += BLOCK, ++k) {
    uchar8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      __hip_fp8_e4m3 q(xv[k][j] * scale);
      o[j] = q.__x;
    }
    *reinterpret_cast<uchar8*>(out_row + i * 8) = o;
  }
}

// ---------------------------------------------------------------------------
// Fused SwiGLU -> fp8: [rows, 2d] = [gate | up] -> fp8 [rows, d] + scales.
// Row staged in dynamic LDS as bf16 between the compute and quant passes
// (d*2 bytes; the wrapper bounds d so it fits).
// ---------------------------------------------------------------------------
__global__ void silu_mul_fp8_kernel(uint8_t* __restrict__ out,
                                    float* __restrict__ inv_scale,
                                    const bf16* __restrict__ gate_up,
                                    const int d) {
  constexpr int BLOCK = 256;
  extern __shared__ bf16 row_lds[];
  const int row = blockIdx.x;
  const bf16* g_row = gate_up + (int64_t)row * (2 * d);
  uint8_t* out_row = out + (int64_t)row * d;
  const int nvec = d / 8;

  float amax = 0.f;
  for (int i = threadIdx.x; i < nvec; i += BLOCK) {
    ushort8 gv = *reinterpret_cast<const ushort8*>(g_row + i * 8);
    ushort8 uv = *reinterpret_cast<const ushort8*>(g_row + d + i * 8);
    ushort8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float x = bf16_bits_to_float(gv[j]);
      const float silu = x / (1.f + __expf(-x));
      const float y = silu * bf16_bits_to_float(uv[j]);
      o[j] = float_to_bf16_bits(y);
      amax = fmaxf(amax, fabsf(bf16_bits_to_float(o[j])));
    }
    *reinterpret_cast<ushort8*>(row_lds + i * 8) = o;
  }
  __shared__ float red[BLOCK / WAVE_SIZE];
  {
    const int wave = threadIdx.x / WAVE_SIZE;
    float wmax = wave_reduce_max(amax);
    if (threadIdx.x % WAVE_SIZE == 0) red[wave] = wmax;
    __syncthreads();
    amax = fmaxf(fmaxf(red[0], red[1]), fmaxf(red[2], red[3]));
  }
  const float a = fmaxf(amax, 1e-6f);
  const float scale = 448.f / a;
  if (threadIdx.x == 0) inv_scale[row] = a / 448.f;
  for (int i = threadIdx.x; i < nvec; i += BLOCK) {
    ushort8 o = *reinterpret_cast<const ushort8*>(row_lds + i * 8);
    uchar8 q8;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      __hip_fp8_e4m3 q(bf16_bits_to_float(o[j]) * scale);
      q8[j] = q.__x;
    }
    *reinterpret_cast<uchar8*>(out_row + i * 8) = q8;
  }
}

}  // namespace arks

extern "C" void arks_rmsnorm_fp8(void* out, void* inv_scale, const void* input,
                                 void* residual, const void* weight, float eps,
                                 int rows, int hidden, int fused_add,
                                 hipStream_t stream) {
  if (fused_add) {
    hipLaunchKernelGGL((arks::rmsnorm_fp8_kernel<true>), dim3(rows), dim3(256),
                       0, stream, (uint8_t*)out, (float*)inv_scale,
                       (const arks::bf16*)input, (arks::bf16*)residual,
                       (const arks::bf16*)weight, eps, hidden);
  } else {
    hipLaunchKernelGGL((arks::rmsnorm_fp8_kernel<false>), dim3(rows), dim3(256),
                       0, stream, (uint8_t*)out, (float*)inv_scale,
                       (const arks::bf16*)input, (arks::bf16*)residual,
                       (const arks::bf16*)weight, eps, hidden);
  }
}

extern "C" void arks_silu_mul_fp8(void* out, void* inv_scale,
                                  const void* gate_up, int rows, int d,
                                  hipStream_t stream) {
  hipLaunchKernelGGL(arks::silu_mul_fp8_kernel, dim3(rows), dim3(256),
                     (size_t)d * 2, stream, (uint8_t*)out, (float*)inv_scale,
                     (const arks::bf16*)gate_up, d);
}
