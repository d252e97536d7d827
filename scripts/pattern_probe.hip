// Read-pattern probe: 16-row-strided vs contiguous weight streaming.
#include <hip/hip_runtime.h>
#include <cstdio>
typedef __attribute__((__vector_size__(4 * sizeof(unsigned int)))) unsigned int u4;

// A: wave reads 16 rows (stride K*2 bytes) x kslice cols, 16B/lane units
__global__ __launch_bounds__(256) void probe_strided(
    const u4* __restrict__ w, unsigned int* out, long rowq, long ksq,
    int kslice_q) {
  const int wid = threadIdx.x / 64, lane = threadIdx.x & 63;
  const long row = (long)(blockIdx.x * 4 + wid) * 16 + (lane & 15);
  const int kgrp = lane >> 4;
  const u4* base = w + row * rowq + (long)blockIdx.y * kslice_q;
  u4 acc = {0, 0, 0, 0};
  for (int j = kgrp; j + 28 < kslice_q; j += 32) {
#pragma unroll
    for (int u = 0; u < 8; ++u)
      acc ^= __builtin_nontemporal_load(&base[j + u * 4]);
  }
  if (out) out[threadIdx.x] = acc[0] ^ acc[1] ^ acc[2] ^ acc[3];
}

// B: same waves read the same total bytes contiguously
__global__ __launch_bounds__(256) void probe_linear(
    const u4* __restrict__ w, unsigned int* out, long per_wave_q) {
  const int wv = blockIdx.x * 4 + threadIdx.x / 64;
  const int lane = threadIdx.x & 63;
  const u4* base = w + (long)wv * per_wave_q;
  u4 acc = {0, 0, 0, 0};
  for (long j = lane; j + 64 * 7 < per_wave_q; j += 64 * 8) {
#pragma unroll
    for (int u = 0; u < 8; ++u)
      acc ^= __builtin_nontemporal_load(&base[j + u * 64]);
  }
  if (out) out[threadIdx.x] = acc[0] ^ acc[1] ^ acc[2] ^ acc[3];
}

int main() {
  const long N = 6144, K = 4096;
  const long bytes = N * K * 2;
  void* w;
  (void)hipMalloc(&w, bytes);
  (void)hipMemset(w, 1, bytes);
  const int splitk = 8;
  const int kslice_q = (int)(K / splitk / 8);  // uint4 units per slice
  dim3 gridA((unsigned)(N / 64), splitk);
  const long nwaves = (N / 64) * splitk * 4;
  const long per_wave_q = bytes / 16 / nwaves;
  for (int rep = 0; rep < 2; ++rep) {
    hipEvent_t a, b;
    (void)hipEventCreate(&a); (void)hipEventCreate(&b);
    (void)hipEventRecord(a, 0);
    for (int i = 0; i < 50; ++i)
      hipLaunchKernelGGL(probe_strided, gridA, 256, 0, 0, (const u4*)w,
                         (unsigned int*)nullptr, K / 8, 0L, kslice_q);
    (void)hipEventRecord(b, 0);
    (void)hipEventSynchronize(b);
    float ms = 0;
    (void)hipEventElapsedTime(&ms, a, b);
    printf("strided: %.1f us  %.2f TB/s\n", ms * 20,
           (double)bytes * 50 / (ms / 1e3) / 1e12);
    (void)hipEventRecord(a, 0);
    for (int i = 0; i < 50; ++i)
      hipLaunchKernelGGL(probe_linear, dim3((unsigned)(nwaves / 4)), 256, 0,
                         0, (const u4*)w, (unsigned int*)nullptr, per_wave_q);
    (void)hipEventRecord(b, 0);
    (void)hipEventSynchronize(b);
    (void)hipEventElapsedTime(&ms, a, b);
    printf("linear : %.1f us  %.2f TB/s\n", ms * 20,
           (double)bytes * 50 / (ms / 1e3) / 1e12);
  }
  return 0;
}
