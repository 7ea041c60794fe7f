// Raw HBM read-rate probe for the decode GEMM question (r2): is ~3 TB/s a
// property of the skinny kernels, or of ANY ~136 MB read on this part?
// Reads a buffer with (a) linear 16B/lane pattern, (b) the skinny kernel's
// row-strided pattern (16 rows/wave, 64B per row per inst), at several
// sizes. Build: hipcc --offload-arch=gfx950 -O3 scripts/bench_membw.hip -o
// /tmp/bench_membw
#include <hip/hip_runtime.h>
#include <cstdio>
#include <vector>

typedef __attribute__((ext_vector_type(4))) unsigned int uint4v;

__global__ void read_linear(const uint4v* __restrict__ src, size_t n4,
                            uint4v* __restrict__ sink) {
  uint4v acc = {0, 0, 0, 0};
  for (size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x; i < n4;
       i += (size_t)gridDim.x * blockDim.x)
    acc ^= src[i];
  if (acc.x == 0xdeadbeef) sink[threadIdx.x] = acc;  // never true
}

// Skinny-W pattern: rows of length k_total u16, each wave owns 16 rows
// (lane%16) and streams 64 B per row per "inst group" (lane/16 subdivides
// the 64 B), grid.y = splits of K.
__global__ void read_rows(const unsigned short* __restrict__ src, int n_rows,
                          int k_total, int k_per_split,
                          uint4v* __restrict__ sink) {
  const int wave = threadIdx.x / 64, lane = threadIdx.x % 64;
  const int lq = lane % 16, la = lane / 16;
  const int row = blockIdx.x * 64 + wave * 16 + lq;
  const int kb = blockIdx.y * k_per_split;
  const int ke = min(k_total, kb + k_per_split);
  const unsigned short* p = src + (size_t)row * k_total;
  uint4v acc = {0, 0, 0, 0};
  for (int k = kb; k < ke; k += 32)
    acc ^= *reinterpret_cast<const uint4v*>(p + k + 8 * la);
  if (acc.x == 0xdeadbeef) sink[threadIdx.x] = acc;
}

static void bench(const char* name, void (*launch)(), int iters) {
  launch();  // warm
  (void)hipDeviceSynchronize();
  hipEvent_t t0, t1;
  (void)hipEventCreate(&t0);
  (void)hipEventCreate(&t1);
  (void)hipEventRecord(t0, 0);
  for (int i = 0; i < iters; ++i) launch();
  (void)hipEventRecord(t1, 0);
  (void)hipEventSynchronize(t1);
  float ms = 0;
  (void)hipEventElapsedTime(&ms, t0, t1);
  printf("%s", name);
  printf("  %.1f us\n", ms * 1000 / iters);
  (void)hipEventDestroy(t0);
  (void)hipEventDestroy(t1);
}

int main() {
  struct Shape { int n, k; const char* label; };
  // down (3584x18944), o (3584x3584), gate_up-sized (37888x3584), 1 GB
  std::vector<Shape> shapes = {{3584, 18944, "down 136MB"},
                               {3584, 3584, "o     26MB"},
                               {37888, 3584, "gu   272MB"},
                               {14336, 37888, "big  1.1GB"}};
  for (auto& s : shapes) {
    size_t bytes = (size_t)s.n * s.k * 2;
    unsigned short* buf;
    uint4v* sink;
    (void)hipMalloc(&buf, bytes);
    (void)hipMalloc(&sink, 4096);
    (void)hipMemset(buf, 1, bytes);
    size_t n4 = bytes / 16;
    for (int wgs : {1024, 4096}) {
      char nm[128];
      snprintf(nm, sizeof nm, "%-11s linear  wgs=%4d", s.label, wgs);
      static const unsigned short* g_buf;
      static uint4v* g_sink;
      static size_t g_n4;
      static int g_wgs;
      g_buf = buf; g_sink = sink; g_n4 = n4; g_wgs = wgs;
      auto fn = +[]() {
        hipLaunchKernelGGL(read_linear, dim3(g_wgs), dim3(256), 0, 0,
                           (const uint4v*)g_buf, g_n4, g_sink);
      };
      double us_guess = bytes / 6.3e6;  // rough @6.3 TB/s in us
      int iters = us_guess > 100 ? 50 : 200;
      // time and print TB/s
      (void)us_guess;
      hipEvent_t t0, t1;
      fn(); (void)hipDeviceSynchronize();
      (void)hipEventCreate(&t0); (void)hipEventCreate(&t1);
      (void)hipEventRecord(t0, 0);
      for (int i = 0; i < iters; ++i) fn();
      (void)hipEventRecord(t1, 0); (void)hipEventSynchronize(t1);
      float ms = 0; (void)hipEventElapsedTime(&ms, t0, t1);
      double us = ms * 1000.0 / iters;
      printf("%s  %7.1f us  %5.2f TB/s\n", nm, us, bytes / us / 1e6);
      (void)hipEventDestroy(t0); (void)hipEventDestroy(t1);
    }
    for (int nsplits : {1, 16}) {
      int kps = ((s.k / nsplits + 31) / 32) * 32;
      int nsp = (s.k + kps - 1) / kps;
      char nm[128];
      snprintf(nm, sizeof nm, "%-11s rows    ns =%4d", s.label, nsp);
      hipEvent_t t0, t1;
      dim3 grid(s.n / 64, nsp);
      auto launch = [&]() {
        hipLaunchKernelGGL(read_rows, grid, dim3(256), 0, 0, buf, s.n, s.k,
                           kps, sink);
      };
      launch(); (void)hipDeviceSynchronize();
      (void)hipEventCreate(&t0); (void)hipEventCreate(&t1);
      int iters = 200;
      (void)hipEventRecord(t0, 0);
      for (int i = 0; i < iters; ++i) launch();
      (void)hipEventRecord(t1, 0); (void)hipEventSynchronize(t1);
      float ms = 0; (void)hipEventElapsedTime(&ms, t0, t1);
      double us = ms * 1000.0 / iters;
      printf("%s  %7.1f us  %5.2f TB/s\n", nm, us, bytes / us / 1e6);
      (void)hipEventDestroy(t0); (void)hipEventDestroy(t1);
    }
    (void)hipFree(buf);
    (void)hipFree(sink);
  }
  return 0;
}
