// One-shot peer-to-peer all-reduce over xGMI (gfx950).
//
// RCCL's ring all-reduce is per-link bound on MI355X (7 p2p xGMI links
// x ~153 GB/s, no switch) and latency-heavy for the small per-token
// tensors of tensor-parallel decode. For payloads <= ~1 MiB
// this kernel does the classic small-message one-shot instead: every rank
// PUSHES its contribution into a slot of every peer's mailbox (p2p stores
// over xGMI), signals per-block flags with system-scope release, and each
// rank then reduces its own (local-HBM) mailbox. 2 hops of latency, no
// ring serialization.
//
// Graph-safety: the sequence number lives in DEVICE memory and a tiny
// increment kernel precedes each all-reduce, so hipGraph replays of a
// captured decode step produce monotonically increasing sequence values
// and flag comparisons stay correct across replays.
//
// Deadlock containment: spin waits are bounded; on timeout the kernel
// raises a device trap (loud abort) instead of hanging the GPU. The
// Python wrapper additionally validates the whole path against
// dist.all_reduce at init and falls back to RCCL on any mismatch.
#include "common.h"

namespace arks {

constexpr int AR_MAX_WORLD = 8;
constexpr int AR_MAX_BLOCKS = 64;
// mailbox capacity per (parity, rank) slot — must match parallel/p2p.py
constexpr int64_t AR_MAX_ELEMS = 1 << 19;

struct ArPtrs {
  // peer p's mailbox base ([world][nelem] bf16) and flag base
  // ([world * AR_MAX_BLOCKS] u64), IPC-mapped into this process
  void* mail[AR_MAX_WORLD];
  unsigned long long* flags[AR_MAX_WORLD];
};

__global__ void ar_seq_inc_kernel(unsigned long long* seq) {
  if (threadIdx.x == 0) {
    ++(*seq);
  }
}

typedef __attribute__((ext_vector_type(4))) uint32_t uint4v;

template <int WORLD>
__global__ __launch_bounds__(256) void one_shot_allreduce_kernel(
    bf16* __restrict__ out, const bf16* __restrict__ src, const ArPtrs ptrs,
    const unsigned long long* __restrict__ seq, const int rank,
    const int64_t n) {
  const unsigned long long s = *seq;
  // Parity double-buffering closes the consecutive-call race: rank B's
  // call N+1 may start pushing while rank A still READS call N (B only
  // waited for A's phase-1 flag, not A's reduce). Alternating mailbox
  // halves by sequence parity makes the earliest same-slot reuse call
  // N+2 — which transitively orders after A's call-N reduce (B's N+2
  // waits A's N+1 flag; A posts it after its N reduce in stream order).
  // Flags stay single-buffered: they are monotonic and waited with >=.
  const int64_t pbase = (int64_t)(s & 1) * AR_MAX_WORLD * AR_MAX_ELEMS;
  const int nblk = gridDim.x;
  // per-block range stays 16-byte aligned (n is a multiple of 8)
  const int64_t per_blk = (((n + nblk - 1) / nblk) + 7) & ~(int64_t)7;
  const int64_t i0 = blockIdx.x * per_blk;
  const int64_t i1 = min(n, i0 + per_blk);

  // phase 1: push my chunk into slot [rank] of every rank's mailbox
  // (vectorized 8 bf16 per store; n is padded to 8 by the wrapper)
  for (int64_t i = i0 + threadIdx.x * 8; i < i1; i += 256 * 8) {
    const uint4v v = *reinterpret_cast<const uint4v*>(src + i);
#pragma unroll
    for (int p = 0; p < WORLD; ++p) {
      bf16* m =
          reinterpret_cast<bf16*>(ptrs.mail[p]) + pbase + (int64_t)rank * n;
      *reinterpret_cast<uint4v*>(m + i) = v;
    }
  }
  // publish: system-scope release so peer L2s see the payload before the
  // flag (xGMI is cache-coherent at system scope on fence)
  __threadfence_system();
  if (threadIdx.x < WORLD) {
    __hip_atomic_store(
        &ptrs.flags[threadIdx.x][(int64_t)rank * AR_MAX_BLOCKS + blockIdx.x],
        s, __ATOMIC_RELEASE, __HIP_MEMORY_SCOPE_SYSTEM);
  }

  // phase 2: wait for every rank's chunk-b flag in MY flag buffer.
  // Relaxed polling (an acquire per poll would thrash L1 — guide §6 G16),
  // one acquire fence after; bounded spin -> trap instead of a GPU hang.
  if (threadIdx.x < WORLD) {
    const unsigned long long* f =
        &ptrs.flags[rank][(int64_t)threadIdx.x * AR_MAX_BLOCKS + blockIdx.x];
    long guard = 0;
    while (__hip_atomic_load(f, __ATOMIC_RELAXED,
                             __HIP_MEMORY_SCOPE_SYSTEM) < s) {
      if (++guard > (1L << 31)) {
        __builtin_trap();  // peer missing: loud abort, never a silent hang
      }
    }
  }
  __syncthreads();
  __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "");  // system-scope acquire

  // phase 3: reduce my (local) mailbox across the world slots
  const bf16* mine = reinterpret_cast<const bf16*>(ptrs.mail[rank]) + pbase;
  for (int64_t i = i0 + threadIdx.x * 8; i < i1; i += 256 * 8) {
    float acc[8] = {};
#pragma unroll
    for (int p = 0; p < WORLD; ++p) {
      const uint4v v =
          *reinterpret_cast<const uint4v*>(mine + (int64_t)p * n + i);
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        acc[2 * e] += bf16_bits_to_float((uint16_t)v[e]);
        acc[2 * e + 1] += bf16_bits_to_float((uint16_t)(v[e] >> 16));
      }
    }
    uint4v o;
#pragma unroll
    for (int e = 0; e < 4; ++e) {
      o[e] = (uint32_t)float_to_bf16_bits(acc[2 * e]) |
             ((uint32_t)float_to_bf16_bits(acc[2 * e + 1]) << 16);
    }
    *reinterpret_cast<uint4v*>(out + i) = o;
  }
}

}  // namespace arks

using namespace arks;

extern "C" void arks_ar_seq_inc(void* seq, hipStream_t stream) {
  hipLaunchKernelGGL(ar_seq_inc_kernel, dim3(1), dim3(64), 0, stream,
                     (unsigned long long*)seq);
}

extern "C" void arks_one_shot_allreduce(void* out, const void* src,
                                        void* mail_ptrs[8],
                                        void* flag_ptrs[8], void* seq,
                                        int rank, int world, int64_t n,
                                        hipStream_t stream) {
  ArPtrs ptrs{};
  for (int p = 0; p < world; ++p) {
    ptrs.mail[p] = mail_ptrs[p];
    ptrs.flags[p] = (unsigned long long*)flag_ptrs[p];
  }
  // enough blocks to engage the links without breaking residency
  // guarantees for the spin (all blocks must be co-resident)
  int64_t want = (n + 256 * 8 - 1) / (256 * 8);
  int blocks = (int)(want > AR_MAX_BLOCKS ? AR_MAX_BLOCKS : want);
  if (blocks < 1) blocks = 1;
  dim3 grid(blocks), block(256);
#define ARKS_AR_CASE(W)                                                     \
  case W:                                                                   \
    hipLaunchKernelGGL((one_shot_allreduce_kernel<W>), grid, block, 0,      \
                       stream, (bf16*)out, (const bf16*)src, ptrs,          \
                       (const unsigned long long*)seq, rank, n);            \
    break;
  switch (world) {
    ARKS_AR_CASE(1)
    ARKS_AR_CASE(2)
    ARKS_AR_CASE(4)
    ARKS_AR_CASE(8)
    default:
      break;  // validated host-side
  }
#undef ARKS_AR_CASE
}
