// Common helpers for arks_amd gfx950 (CDNA4) kernels.
// Wave size is 64 on CDNA4; block sizes are multiples of 64 throughout.
#pragma once

#include <hip/hip_runtime.h>

#include <cstdint>

#define WAVE_SIZE 64

namespace arks {

// bf16 is handled as opaque 16-bit storage throughout (conversions are
// explicit bit ops below) — no dependence on hip_bf16.h struct layout.
using bf16 = uint16_t;

// Vectorized load types: 8 bf16 = 16 bytes per lane (the coalescing sweet
// spot on CDNA4 — guide G13).
typedef __attribute__((ext_vector_type(8))) uint16_t ushort8;
typedef __attribute__((ext_vector_type(4))) uint16_t ushort4v;
typedef __attribute__((ext_vector_type(2))) uint16_t ushort2v;
typedef __attribute__((ext_vector_type(4))) float float4v;
typedef __attribute__((ext_vector_type(8))) float float8v;

__device__ __forceinline__ float bf16_bits_to_float(uint16_t u) {
  union {
    uint32_t i;
    float f;
  } c;
  c.i = static_cast<uint32_t>(u) << 16;
  return c.f;
}

__device__ __forceinline__ uint16_t float_to_bf16_bits(float f) {
  union {
    float f;
    uint32_t u;
  } c;
  c.f = f;
  // Round-to-nearest-even (matches PyTorch's fp32->bf16 cast for finite x).
  const uint32_t rounded = c.u + 0x7fffu + ((c.u >> 16) & 1u);
  return static_cast<uint16_t>(rounded >> 16);
}

// Sum across all 64 lanes of a wave.
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE_SIZE);
  return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE_SIZE));
  return v;
}

// Reduce within an aligned 16-lane group (lanes l..l+15, l % 16 == 0).
__device__ __forceinline__ float group16_reduce_sum(float v) {
#pragma unroll
  for (int off = 8; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE_SIZE);
  return v;
}

__device__ __forceinline__ float group16_reduce_max(float v) {
#pragma unroll
  for (int off = 8; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE_SIZE));
  return v;
}

// Block-level reduce-sum over up to 16 waves, via LDS. Every thread returns
// the total. `scratch` must hold >= num_waves floats.
template <int BLOCK_THREADS>
__device__ __forceinline__ float block_reduce_sum(float v, float* scratch) {
  constexpr int NUM_WAVES = BLOCK_THREADS / WAVE_SIZE;
  const int wave = threadIdx.x / WAVE_SIZE;
  const int lane = threadIdx.x % WAVE_SIZE;
  v = wave_reduce_sum(v);
  if (lane == 0) scratch[wave] = v;
  __syncthreads();
  float total = 0.f;
#pragma unroll
  for (int w = 0; w < NUM_WAVES; ++w) total += scratch[w];
  return total;
}

__host__ __forceinline__ int ceil_div(int a, int b) { return (a + b - 1) / b; }

// OCP e4m3 <-> bf16 bit conversions for the fp8 KV cache (scale 1.0; the
// e4m3 range of +-448 covers K/V magnitudes). Uses the gfx950 hardware cvt
// via __hip_fp8_e4m3 on device.
typedef __attribute__((ext_vector_type(8))) uint8_t uchar8;

}  // namespace arks

#define HIP_CHECK_KERNEL()                                    \
  do {                                                        \
    hipError_t err_ = hipGetLastError();                      \
    if (err_ != hipSuccess) {                                 \
      printf("HIP kernel launch error: %s\n",                 \
             hipGetErrorString(err_));                        \
    }                                                         \
  } while (0)
