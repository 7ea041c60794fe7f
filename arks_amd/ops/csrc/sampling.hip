// Sampling kernels: greedy argmax and Gumbel-max categorical sampling.
//
// Gumbel-max: argmax(logits/T + G) with G ~ Gumbel(0,1) is an exact sample
// from softmax(logits/T) — one memory-bound pass over the logits, no
// normalization, no sort. Rows with T == 0 fall back to plain argmax.
// uniform noise is supplied by the caller (torch.rand on the same stream) so
// sampling stays reproducible under torch.manual_seed.
#include "common.h"

#include <cfloat>

namespace arks {

// One workgroup (256 threads) per row; each thread scans a strided slice
// keeping (best_val, best_idx); reduce via wave shuffles + LDS.
template <bool GUMBEL>
__global__ void sample_kernel(int64_t* __restrict__ out,
                              const bf16* __restrict__ logits,  // [rows, vocab]
                              const float* __restrict__ temperatures,  // null if !GUMBEL
                              const float* __restrict__ uniform,  // [rows, vocab]
                              const int vocab) {
  constexpr int BLOCK = 256;
  const int row = blockIdx.x;
  const bf16* lrow = logits + (int64_t)row * vocab;
  float inv_t = 0.f;
  bool greedy = true;
  if constexpr (GUMBEL) {
    const float t = temperatures[row];
    greedy = (t <= 0.f);
    inv_t = greedy ? 1.f : 1.f / t;
  }
  const float* urow = GUMBEL ? uniform + (int64_t)row * vocab : nullptr;

  float best = -FLT_MAX;
  int best_idx = 0;
  // Vectorized 8-wide scan. vocab need not be a multiple of 8: tail handled
  // scalar.
  const int nvec = vocab / 8;
  for (int i = threadIdx.x; i < nvec; i += BLOCK) {
    ushort8 x = *reinterpret_cast<const ushort8*>(lrow + i * 8);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float v = bf16_bits_to_float(x[j]);
      if constexpr (GUMBEL) {
        if (!greedy) {
          float u = urow[i * 8 + j];
          u = fmaxf(u, 1e-20f);
          v = v * inv_t - __logf(-__logf(u));
        }
      }
      const int idx = i * 8 + j;
      if (v > best || (v == best && idx < best_idx)) {
        best = v;
        best_idx = idx;
      }
    }
  }
  for (int idx = nvec * 8 + threadIdx.x; idx < vocab; idx += BLOCK) {
    float v = bf16_bits_to_float(lrow[idx]);
    if constexpr (GUMBEL) {
      if (!greedy) {
        float u = fmaxf(urow[idx], 1e-20f);
        v = v * inv_t - __logf(-__logf(u));
      }
    }
    if (v > best || (v == best && idx < best_idx)) {
      best = v;
      best_idx = idx;
    }
  }

  // Wave reduction on (val, idx); ties -> lowest index (matches torch argmax).
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    float ov = __shfl_xor(best, off, WAVE_SIZE);
    int oi = __shfl_xor(best_idx, off, WAVE_SIZE);
    if (ov > best || (ov == best && oi < best_idx)) {
      best = ov;
      best_idx = oi;
    }
  }
  __shared__ float sval[BLOCK / WAVE_SIZE];
  __shared__ int sidx[BLOCK / WAVE_SIZE];
  const int wave = threadIdx.x / WAVE_SIZE;
  if (threadIdx.x % WAVE_SIZE == 0) {
    sval[wave] = best;
    sidx[wave] = best_idx;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    float b = sval[0];
    int bi = sidx[0];
#pragma unroll
    for (int w = 1; w < BLOCK / WAVE_SIZE; ++w) {
      if (sval[w] > b || (sval[w] == b && sidx[w] < bi)) {
        b = sval[w];
        bi = sidx[w];
      }
    }
    out[row] = bi;
  }
}

}  // namespace arks

using namespace arks;

extern "C" {

void arks_greedy_sample(void* out, const void* logits, int rows, int vocab,
                        hipStream_t stream) {
  hipLaunchKernelGGL((sample_kernel<false>), dim3(rows), dim3(256), 0, stream,
                     (int64_t*)out, (const bf16*)logits, nullptr, nullptr,
                     vocab);
}

void arks_gumbel_sample(void* out, const void* logits, const void* temperatures,
                        const void* uniform, int rows, int vocab,
                        hipStream_t stream) {
  hipLaunchKernelGGL((sample_kernel<true>), dim3(rows), dim3(256), 0, stream,
                     (int64_t*)out, (const bf16*)logits,
                     (const float*)temperatures, (const float*)uniform, vocab);
}

}  // extern "C"
