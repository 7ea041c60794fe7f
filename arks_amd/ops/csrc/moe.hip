// MoE routing + mixing kernels (gfx950).
//
// The dense-decode MoE path (models/llama_family.py MoEMLP.forward) is
// weight-HBM-bound in its two expert bmms, but the torch glue around them
// (softmax -> topk -> renorm -> scatter, then float casts + einsum for the
// weighted mix) costs ~10 small kernels per layer — several ms per decode
// step at batch 64-256. These two kernels replace that glue:
//
//   moe_topk:  probs = softmax(logits[T,E]); per row select top-k
//              (ties -> lower expert id, matching torch.topk), optional
//              renormalization; emits (weights[T,k] f32, ids[T,k] i32).
//   moe_mix:   out[t,h] = sum_i w[t,i] * y[ids[t,i], t, h] — the weighted
//              gather-mix over the dense per-expert outputs y[E,T,H].
//
// Reference parity note: the reference delegates MoE serving to external
// images; these kernels are runtime-slot internals.
#include "common.h"

#include <cfloat>

namespace arks {

// One wave per token row: softmax over E (<= 1024) then k iterative
// argmax passes. E is a runtime arg; lanes stride the row.
template <int MAXK>
__global__ __launch_bounds__(64) void moe_topk_kernel(
    float* __restrict__ weights,    // [T, k]
    int* __restrict__ ids,          // [T, k]
    const float* __restrict__ logits,  // [T, E]
    const int E, const int k, const int renorm) {
  const int t = blockIdx.x;
  const int lane = threadIdx.x;
  const float* row = logits + (int64_t)t * E;

  // max + sum for softmax (over the full row)
  float m = -FLT_MAX;
  for (int e = lane; e < E; e += 64) m = fmaxf(m, row[e]);
  m = wave_reduce_max(m);
  float s = 0.f;
  for (int e = lane; e < E; e += 64) s += __expf(row[e] - m);
  s = wave_reduce_sum(s);

  // k argmax passes over the logits (same order as over probs).
  // Selected entries are masked per-lane via a small local bitmask.
  uint32_t taken[16];  // supports E <= 1024 (16 strided slots per lane)
#pragma unroll
  for (int i = 0; i < 16; ++i) taken[i] = 0;
  float wsum = 0.f;
  float wk[MAXK];
  int idk[MAXK];
  for (int j = 0; j < k; ++j) {
    float best = -FLT_MAX;
    int bi = E;
    for (int e = lane, slot = 0; e < E; e += 64, ++slot) {
      if (taken[slot >> 5] & (1u << (slot & 31))) continue;
      const float v = row[e];
      if (v > best || (v == best && e < bi)) {
        best = v;
        bi = e;
      }
    }
    // wave argmax: prefer larger value, then smaller index
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      const float ov = __shfl_xor(best, off, WAVE_SIZE);
      const int oi = __shfl_xor(bi, off, WAVE_SIZE);
      if (ov > best || (ov == best && oi < bi)) {
        best = ov;
        bi = oi;
      }
    }
    if ((bi % 64) == lane) {
      const int slot = bi / 64;
      taken[slot >> 5] |= 1u << (slot & 31);
    }
    const float w = __expf(best - m) / s;
    wk[j] = w;
    idk[j] = bi;
    wsum += w;
  }
  const float inv = renorm ? 1.f / wsum : 1.f;
  for (int j = lane; j < k; j += 64) {
    weights[(int64_t)t * k + j] = wk[j] * inv;
    ids[(int64_t)t * k + j] = idk[j];
  }
}

// Weighted mix of per-expert dense outputs: one block per token, lanes
// stride the hidden dim; k gathered rows from y[E,T,H].
__global__ __launch_bounds__(256) void moe_mix_kernel(
    bf16* __restrict__ out,          // [T, H]
    const bf16* __restrict__ y,      // [E_local, T, H]
    const float* __restrict__ weights,  // [T, k] (global expert ids)
    const int* __restrict__ ids,        // [T, k]
    const int T, const int H, const int k,
    const int expert_base, const int n_local) {
  const int t = blockIdx.x;
  // k-indexed state in LDS (runtime-k register arrays would spill).
  // Experts outside this rank's [expert_base, +n_local) slice contribute 0
  // (their partial sums come from other TP ranks' all-reduce).
  __shared__ float w[16];
  __shared__ int64_t base[16];
  if (threadIdx.x < k) {
    const int j = threadIdx.x;
    const int le = ids[(int64_t)t * k + j] - expert_base;
    const bool mine = le >= 0 && le < n_local;
    w[j] = mine ? weights[(int64_t)t * k + j] : 0.f;
    base[j] = ((int64_t)(mine ? le : 0) * T + t) * H;
  }
  __syncthreads();
  for (int h = threadIdx.x * 2; h < H; h += 256 * 2) {
    float a0 = 0.f, a1 = 0.f;
    for (int j = 0; j < k; ++j) {
      const uint32_t u = *reinterpret_cast<const uint32_t*>(y + base[j] + h);
      a0 += w[j] * bf16_bits_to_float((uint16_t)u);
      a1 += w[j] * bf16_bits_to_float((uint16_t)(u >> 16));
    }
    uint32_t packed = (uint32_t)float_to_bf16_bits(a0) |
                      ((uint32_t)float_to_bf16_bits(a1) << 16);
    *reinterpret_cast<uint32_t*>(out + (int64_t)t * H + h) = packed;
  }
}

// Row-indexed variant for the sparse/grouped prefill path: y is a flat
// [R, H] buffer (padded grouped-bmm block + overflow segments); rows[t,k]
// addresses each token's k expert outputs directly. Replaces torch
// index_add_ scatters (36% of Qwen3-30B prefill GPU time) with one
// bandwidth-bound gather; zero-weight slots point at row 0.
__global__ __launch_bounds__(256) void moe_mix_rows_kernel(
    bf16* __restrict__ out,          // [T, H]
    const bf16* __restrict__ y,      // [R, H]
    const float* __restrict__ weights,  // [T, k]
    const int* __restrict__ rows,       // [T, k]
    const int T, const int H, const int k) {
  const int t = blockIdx.x;
  __shared__ float w[16];
  __shared__ int64_t base[16];
  if (threadIdx.x < k) {
    const int j = threadIdx.x;
    w[j] = weights[(int64_t)t * k + j];
    base[j] = (int64_t)rows[(int64_t)t * k + j] * H;
  }
  __syncthreads();
  for (int h = threadIdx.x * 2; h < H; h += 256 * 2) {
    float a0 = 0.f, a1 = 0.f;
    for (int j = 0; j < k; ++j) {
      const uint32_t u = *reinterpret_cast<const uint32_t*>(y + base[j] + h);
      a0 += w[j] * bf16_bits_to_float((uint16_t)u);
      a1 += w[j] * bf16_bits_to_float((uint16_t)(u >> 16));
    }
    uint32_t packed = (uint32_t)float_to_bf16_bits(a0) |
                      ((uint32_t)float_to_bf16_bits(a1) << 16);
    *reinterpret_cast<uint32_t*>(out + (int64_t)t * H + h) = packed;
  }
}

}  // namespace arks

using namespace arks;

extern "C" void arks_moe_topk(void* weights, void* ids, const void* logits,
                              int T, int E, int k, int renorm,
                              hipStream_t stream) {
  dim3 grid(T), block(64);
  hipLaunchKernelGGL((moe_topk_kernel<16>), grid, block, 0, stream,
                     (float*)weights, (int*)ids, (const float*)logits, E, k,
                     renorm);
}

extern "C" void arks_moe_mix_rows(void* out, const void* y,
                                  const void* weights, const void* rows,
                                  int T, int H, int k, hipStream_t stream) {
  dim3 grid(T), block(256);
  hipLaunchKernelGGL(moe_mix_rows_kernel, grid, block, 0, stream, (bf16*)out,
                     (const bf16*)y, (const float*)weights, (const int*)rows,
                     T, H, k);
}

extern "C" void arks_moe_mix(void* out, const void* y, const void* weights,
                             const void* ids, int T, int H, int k,
                             int expert_base, int n_local,
                             hipStream_t stream) {
  dim3 grid(T), block(256);
  hipLaunchKernelGGL(moe_mix_kernel, grid, block, 0, stream, (bf16*)out,
                     (const bf16*)y, (const float*)weights, (const int*)ids,
                     T, H, k, expert_base, n_local);
}
