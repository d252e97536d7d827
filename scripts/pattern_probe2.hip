// Incremental probe: strided W stream + {nothing | LDS reads | MFMA | both}
#include <hip/hip_runtime.h>
#include <cstdio>
typedef __attribute__((__vector_size__(4 * sizeof(unsigned int)))) unsigned int u4;
typedef __attribute__((__vector_size__(8 * sizeof(short)))) short bf16x8_t;
typedef __attribute__((__vector_size__(4 * sizeof(float)))) float f32x4_t;

__device__ int xswz_p(int row, int byte_in_row) {
  return row * (512 * 2) + (byte_in_row ^ ((row & 15) << 4));
}

// 0 stream only, 1 +lds, 2 +mfma, 3 +both, 4 both+realstage,
// 5 both+swzreads, 6 both+atomics
template <int MODE>
__global__ __launch_bounds__(256) void probe(
    const u4* __restrict__ w, float* __restrict__ out, long K, int kslice) {
  __shared__ __align__(16) unsigned short xbuf[64 * 512];
  const int wid = threadIdx.x / 64, lane = threadIdx.x & 63;
  const int row = (blockIdx.x * 4 + wid) * 16 + (lane & 15);
  const int kgrp = lane >> 4;
  const long rowq = (long)row * (K / 8);
  const long ks = (long)blockIdx.y * kslice / 8;
  if (MODE == 4) {
    // the real staging: 16 global 16B loads into registers, then writes
    const int xr = threadIdx.x >> 2;
    const int c0 = (threadIdx.x & 3) * 8;
    const unsigned short* xs =
        reinterpret_cast<const unsigned short*>(w) + (long)(xr & 63) * K;
    uint4 v[16];
#pragma unroll
    for (int i = 0; i < 16; ++i)
      v[i] = *reinterpret_cast<const uint4*>(xs + c0 + i * 32);
#pragma unroll
    for (int i = 0; i < 16; ++i)
      *reinterpret_cast<uint4*>(reinterpret_cast<char*>(xbuf) +
                                xswz_p(xr, (c0 + i * 32) * 2)) = v[i];
  } else {
    for (int i = threadIdx.x; i < 64 * 512 / 8; i += 256)
      *reinterpret_cast<uint4*>(&xbuf[i * 8]) = uint4{1, 2, 3, 4};
  }
  __syncthreads();
  f32x4_t acc[4] = {};
  u4 junk = {0, 0, 0, 0};
  const int xr_base = (lane & 15) * 512;
  for (long k = ks + kgrp; k + 28 < ks + kslice / 8; k += 32) {
    u4 wa[8];
#pragma unroll
    for (int u = 0; u < 8; ++u)
      wa[u] = __builtin_nontemporal_load(&w[rowq + k + u * 4]);
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      if (MODE == 0) {
        junk ^= wa[u];
      } else if (MODE == 1) {
#pragma unroll
        for (int m = 0; m < 4; ++m)
          junk ^= *reinterpret_cast<const u4*>(
              &xbuf[((m * 16 + (lane & 15)) * 512 + (int)(k % 64) * 8) & 32760]);
        junk ^= wa[u];
      } else if (MODE == 2) {
#pragma unroll
        for (int m = 0; m < 4; ++m)
          acc[m] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              __builtin_bit_cast(bf16x8_t, wa[u]),
              __builtin_bit_cast(bf16x8_t, wa[u]), acc[m], 0, 0, 0);
      } else if (MODE == 5) {
        const int kc = (int)((k - ks + u * 4) * 8) % 512;
#pragma unroll
        for (int m = 0; m < 4; ++m) {
          const int xr = m * 16 + (lane & 15);
          const u4 xb = *reinterpret_cast<const u4*>(
              reinterpret_cast<const char*>(xbuf) + xswz_p(xr, kc * 2));
          acc[m] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              __builtin_bit_cast(bf16x8_t, wa[u]),
              __builtin_bit_cast(bf16x8_t, xb), acc[m], 0, 0, 0);
        }
      } else {
#pragma unroll
        for (int m = 0; m < 4; ++m) {
          const u4 xb = *reinterpret_cast<const u4*>(
              &xbuf[((m * 16 + (lane & 15)) * 512 + (int)(k % 64) * 8) & 32760]);
          acc[m] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              __builtin_bit_cast(bf16x8_t, wa[u]),
              __builtin_bit_cast(bf16x8_t, xb), acc[m], 0, 0, 0);
        }
      }
    }
  }
  float r = junk[0] + junk[1] + junk[2] + junk[3];
  if (MODE == 6) {
#pragma unroll
    for (int m = 0; m < 4; ++m)
#pragma unroll
      for (int q = 0; q < 4; ++q)
        if (out) atomicAdd(out + ((m * 16 + (lane & 15)) * 64 +
                                  blockIdx.x % 16 * 4 + q) % 1024,
                           acc[m][q]);
    return;
  }
#pragma unroll
  for (int m = 0; m < 4; ++m) r += acc[m][0];
  if (out) out[threadIdx.x] = r;
}

template <int MODE>
void run(const char* name, const u4* w, long N, long K) {
  const int splitk = 8, kslice = (int)K / splitk;
  dim3 grid((unsigned)(N / 64), splitk);
  hipEvent_t a, b;
  (void)hipEventCreate(&a); (void)hipEventCreate(&b);
  // warm
  for (int i = 0; i < 10; ++i)
    hipLaunchKernelGGL(probe<MODE>, grid, 256, 0, 0, w, (float*)nullptr, K, kslice);
  (void)hipEventRecord(a, 0);
  for (int i = 0; i < 50; ++i)
    hipLaunchKernelGGL(probe<MODE>, grid, 256, 0, 0, w, (float*)nullptr, K, kslice);
  (void)hipEventRecord(b, 0);
  (void)hipEventSynchronize(b);
  float ms = 0;
  (void)hipEventElapsedTime(&ms, a, b);
  const double bytes = (double)N * K * 2;
  printf("%s: %7.1f us  %.2f TB/s\n", name, ms * 20, bytes * 50 / (ms / 1e3) / 1e12);
}

int main() {
  const long N = 6144, K = 4096;
  void* w;
  (void)hipMalloc(&w, N * K * 2);
  (void)hipMemset(w, 1, N * K * 2);
  for (int rep = 0; rep < 2; ++rep) {
    run<0>("stream       ", (const u4*)w, N, K);
    run<3>("stream+both  ", (const u4*)w, N, K);
    run<4>("both+realstg ", (const u4*)w, N, K);
    run<5>("both+swzread ", (const u4*)w, N, K);
    run<6>("both+atomics ", (const u4*)w, N, K);
  }
  return 0;
}
