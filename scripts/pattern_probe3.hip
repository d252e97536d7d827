// Request-granularity probe at the skinny-GEMM geometry (224 blocks, 4
// waves, nt loads, random data): (a) 16 rows x 64 B per instruction
// (the v3 fragment-shaped W read), (b) 8 rows x 128 B contiguous per
// instruction, (c) 4 rows x 256 B. Same total bytes, same rows per wave.
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
typedef __attribute__((__vector_size__(4 * sizeof(unsigned int)))) unsigned int u4;

template <int RS>  // rows per instruction: 16, 8, or 4
__global__ __launch_bounds__(256) void probe(const u4* __restrict__ w,
                                             unsigned int* out, long rowq,
                                             int kq) {
  // wave reads 32 rows (2 tiles of 16) x kq u4-units, like skinny2 G=2
  const int wid = threadIdx.x / 64, lane = threadIdx.x & 63;
  const int segs = 64 / RS;             // 16B segments per row per instr
  const int row_in = lane / segs;       // 0..RS-1
  const int seg = lane % segs;
  u4 acc = {0, 0, 0, 0};
  for (int t = 0; t < 2; ++t) {
    const long row0 = (long)(blockIdx.x * 2 + t) * 64 + wid * 16;
    for (int rb = 0; rb < 16; rb += RS) {
      const u4* base = w + (row0 + rb + row_in) * rowq;
      for (int j = 0; j + segs * 7 < kq; j += segs * 8) {
#pragma unroll
        for (int u = 0; u < 8; ++u)
          acc ^= __builtin_nontemporal_load(&base[j + u * segs + seg]);
      }
    }
  }
  if (out) out[threadIdx.x] = acc[0] ^ acc[1] ^ acc[2] ^ acc[3];
}

int main(int argc, char** argv) {
  const long N = 28672, K = 4096;
  const long bytes = N * K * 2;
  u4* w;
  (void)hipMalloc(&w, bytes);
  // random fill (device-side LCG) — memset data inflates clocks (DVFS)
  unsigned int* wi = reinterpret_cast<unsigned int*>(w);
  {
    unsigned int* h = (unsigned int*)malloc(bytes);
    srand(13);
    for (long i = 0; i < (long)(bytes / 4); ++i) h[i] = rand() * 2654435761u;
    (void)hipMemcpy(wi, h, bytes, hipMemcpyHostToDevice);
    free(h);
  }
  const int kq = (int)(K / 8);
  dim3 grid((unsigned)(N / 128), 1);
#define RUN(RS)                                                          \
  {                                                                      \
    hipEvent_t a, b;                                                     \
    (void)hipEventCreate(&a); (void)hipEventCreate(&b);                  \
    for (int i = 0; i < 3; ++i)                                          \
      hipLaunchKernelGGL(probe<RS>, grid, 256, 0, 0, w,                  \
                         (unsigned int*)nullptr, K / 8, kq);             \
    (void)hipEventRecord(a, 0);                                          \
    for (int i = 0; i < 30; ++i)                                         \
      hipLaunchKernelGGL(probe<RS>, grid, 256, 0, 0, w,                  \
                         (unsigned int*)nullptr, K / 8, kq);             \
    (void)hipEventRecord(b, 0);                                          \
    (void)hipEventSynchronize(b);                                        \
    float ms = 0;                                                        \
    (void)hipEventElapsedTime(&ms, a, b);                                \
    printf("rows/instr %2d: %7.1f us  %.2f TB/s\n", RS, ms / 30 * 1e3,   \
           bytes / (ms / 30 * 1e-3) / 1e12);                             \
  }
  RUN(16) RUN(8) RUN(4) RUN(16)
  return 0;
}
