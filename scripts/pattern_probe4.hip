// What drags the GEMM stream below the 7.3 TB/s probe ceiling?
// (a) x interference: W stream + 1:2 L2-resident x reads
// (b) occupancy: same stream at 2 blocks/CU (448 blocks, half K each)
// (c) LDS round-trip: W stream + ds_write of every loaded piece
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
typedef __attribute__((__vector_size__(4 * sizeof(unsigned int)))) unsigned int u4;

template <int XRD, int DSW>
__global__ __launch_bounds__(256) void probe(const u4* __restrict__ w,
                                             const u4* __restrict__ xx,
                                             unsigned int* out, long rowq,
                                             int kq, int nslice) {
  __shared__ __align__(16) u4 slab[4][16 * 32];
  const int wid = threadIdx.x / 64, lane = threadIdx.x & 63;
  const int row_in = lane / 8, seg = lane % 8;   // 8 rows x 128B shape
  u4 acc = {0, 0, 0, 0};
  const long kper = (long)kq * nslice;
  for (int t = 0; t < 2; ++t) {
    const long row0 = (long)(blockIdx.x * 2 + t) * 64 + wid * 16;
    for (int rb = 0; rb < 16; rb += 8) {
      const u4* base = w + (row0 + rb + row_in) * rowq +
                       (long)blockIdx.y * kper;
      const u4* xb = xx + (long)(wid * 16 + rb + row_in) * rowq;
      for (long j = 0; j + 8 * 7 < kper; j += 8 * 8) {
#pragma unroll
        for (int u = 0; u < 8; ++u) {
          const u4 v = __builtin_nontemporal_load(&base[j + u * 8 + seg]);
          acc ^= v;
          if (DSW && (u & 3) == 0)
            slab[wid][lane / 2 + ((u >> 2) & 1)] = v;  // ds_write_b128
          if (XRD && (u & 1) == 0)   // one x read per 2 W reads (1:2 bytes)
            acc ^= xx[((long)(j + u * 8 + seg) & 0x3FFF)];
        }
      }
    }
  }
  if (DSW) acc ^= slab[wid][lane & 31];
  if (out) out[threadIdx.x] = acc[0] ^ acc[1] ^ acc[2] ^ acc[3];
}

int main() {
  const long N = 28672, K = 4096;
  const long bytes = N * K * 2;
  u4 *w, *xx;
  (void)hipMalloc(&w, bytes);
  (void)hipMalloc(&xx, 1 << 20);
  {
    unsigned int* h = (unsigned int*)malloc(bytes);
    srand(13);
    for (long i = 0; i < (long)(bytes / 4); ++i) h[i] = rand() * 2654435761u;
    (void)hipMemcpy(w, h, bytes, hipMemcpyHostToDevice);
    (void)hipMemcpy(xx, h, 1 << 20, hipMemcpyHostToDevice);
    free(h);
  }
#define RUN(tag, XRD, DSW, SPLIT)                                        \
  {                                                                      \
    dim3 grid((unsigned)(N / 128), SPLIT);                               \
    const int kq = (int)(K / 8 / 32 / SPLIT) * 32;                       \
    hipEvent_t a, b;                                                     \
    (void)hipEventCreate(&a); (void)hipEventCreate(&b);                  \
    for (int i = 0; i < 3; ++i)                                          \
      hipLaunchKernelGGL((probe<XRD, DSW>), grid, 256, 0, 0, w, xx,      \
                         (unsigned int*)nullptr, K / 8, kq, 1);          \
    (void)hipEventRecord(a, 0);                                          \
    for (int i = 0; i < 30; ++i)                                         \
      hipLaunchKernelGGL((probe<XRD, DSW>), grid, 256, 0, 0, w, xx,      \
                         (unsigned int*)nullptr, K / 8, kq, 1);          \
    (void)hipEventRecord(b, 0);                                          \
    (void)hipEventSynchronize(b);                                        \
    float ms = 0;                                                        \
    (void)hipEventElapsedTime(&ms, a, b);                                \
    printf("%-18s %7.1f us  %.2f TB/s\n", tag, ms / 30 * 1e3,            \
           bytes / SPLIT * SPLIT / (ms / 30 * 1e-3) / 1e12);             \
  }
  RUN("base 8rx128B", 0, 0, 1)
  RUN("+x reads 1:2", 1, 0, 1)
  RUN("+ds_write", 0, 1, 1)
  RUN("+both", 1, 1, 1)
  RUN("base sk2(2/CU)", 0, 0, 2)
  RUN("+both sk2", 1, 1, 2)
  return 0;
}
