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
