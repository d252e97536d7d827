// Skinny decode GEMM for gfx950: out[M,N] = x[M,K] @ W[N,K]^T, M <= 64.
//
// Decode is weight-bandwidth-bound (the whole W streams through once per
// step); hipBLASLt's skinny tiles measured only 36-50% of the HBM roofline
// on the o/down/qkv shapes (profiles/r01*). This kernel streams W with one
// 16 B load per lane per MFMA directly into fragments (no LDS round trip —
// guide §5 "GEMV / M<=16 decode weights" row generalized to M<=64 via
// mfma_f32_16x16x32_bf16), reads the L2-resident x straight into B-frags,
// and split-Ks with f32 atomicAdd partials so every shape puts >=512
// workgroups on the 256-CU chip.
//
// Grid: (N/64, SPLITK); 4 waves per WG, wave w owns N rows [n0+16w, +16).
#include "common.h"
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

namespace kukeon {

typedef __attribute__((__vector_size__(8 * sizeof(short)))) short bf16x8_t;
typedef __attribute__((__vector_size__(4 * sizeof(float)))) float f32x4_t;

struct uint4_s { unsigned int x[4]; };
DEV_INLINE bf16x8_t frag_of(uint4 v) {
  uint4_s u{{v.x, v.y, v.z, v.w}};
  return __builtin_bit_cast(bf16x8_t, u);
}

__global__ void skinny_zero_kernel(float* __restrict__ ws, long n) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i * 4 + 3 < n) {
    *reinterpret_cast<float4*>(ws + i * 4) = float4{0.f, 0.f, 0.f, 0.f};
  } else {
    for (long j = i * 4; j < min(n, i * 4 + 4); ++j) ws[j] = 0.f;
  }
}

// MT = number of 16-row M subtiles (1 => M<=16, 4 => M<=64)
template <int MT, bool SPLIT>
__global__ __launch_bounds__(256) void skinny_gemm_kernel(
    unsigned short* __restrict__ out,      // [M, N] bf16 (SPLIT=false)
    float* __restrict__ ws,                // [M, N] f32  (SPLIT=true)
    const unsigned short* __restrict__ x,  // [M, K]
    const unsigned short* __restrict__ w,  // [N, K]
    int M, int N, long K, int splitk) {
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int n0 = blockIdx.x * 64 + wid * 16;     // this wave's 16 N rows
  const int row16 = lane & 15;                   // A row / B col
  const int kgrp = lane >> 4;                    // 0..3 -> k = 8*kgrp + j
  const long kchunk = (K / 32 + gridDim.y - 1) / gridDim.y;
  const long ks = (long)blockIdx.y * kchunk * 32;
  const long ke = min(K, ks + kchunk * 32);

  const unsigned short* wrow = w + (long)(n0 + row16) * K;
  f32x4_t acc[MT];
#pragma unroll
  for (int m = 0; m < MT; ++m) acc[m] = f32x4_t{0.f, 0.f, 0.f, 0.f};

  for (long k = ks; k < ke; k += 32) {
    const uint4 wa =
        *reinterpret_cast<const uint4*>(wrow + k + 8 * kgrp);
    const bf16x8_t afrag = frag_of(wa);
#pragma unroll
    for (int m = 0; m < MT; ++m) {
      const int xr = min(M - 1, m * 16 + row16);
      const uint4 xb =
          *reinterpret_cast<const uint4*>(x + (long)xr * K + k + 8 * kgrp);
      acc[m] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          afrag, frag_of(xb), acc[m], 0, 0, 0);
    }
  }

  // C layout (16x16): lane -> col = lane&15 (the M index here),
  // row = 4*(lane>>4) + r (the N index)
  const int ncol = n0 + 4 * kgrp;
#pragma unroll
  for (int m = 0; m < MT; ++m) {
    const int mrow = m * 16 + row16;
    if (mrow >= M) continue;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const long off = (long)mrow * N + ncol + r;
      if (SPLIT) {
        atomicAdd(ws + off, acc[m][r]);
      } else {
        out[off] = f2us(acc[m][r]);
      }
    }
  }
}

__global__ void skinny_cast_kernel(unsigned short* __restrict__ out,
                                   const float* __restrict__ ws, long n) {
  const long i = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  if (i + 7 < n) {
    float4 a = *reinterpret_cast<const float4*>(ws + i);
    float4 b = *reinterpret_cast<const float4*>(ws + i + 4);
    float v[8] = {a.x, a.y, a.z, a.w, b.x, b.y, b.z, b.w};
    *reinterpret_cast<uint4*>(out + i) = pack_bf16x8(v);
  } else {
    for (long j = i; j < n; ++j) out[j] = f2us(ws[j]);
  }
}

void skinny_gemm(torch::Tensor out, torch::Tensor x, torch::Tensor w,
                 torch::Tensor ws) {
  const int M = x.size(0);
  const long K = x.size(1);
  const int N = w.size(0);
  TORCH_CHECK(M >= 1 && M <= 64, "skinny_gemm: M must be in [1,64]");
  TORCH_CHECK(N % 64 == 0 && K % 32 == 0, "skinny_gemm: N%64, K%32");
  TORCH_CHECK(x.is_contiguous() && w.is_contiguous() && out.is_contiguous());
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  const int ntiles = N / 64;
  int splitk = 1;
  if (ntiles < 512) splitk = min(16, (512 + ntiles - 1) / ntiles);
  // keep each slice >= 8 k-chunks so the split overhead stays small
  splitk = max(1, min(splitk, (int)(K / 32 / 8)));
  const int MT = (M + 15) / 16;
  dim3 grid(ntiles, splitk);
  auto* op = reinterpret_cast<unsigned short*>(out.data_ptr());
  auto* xp = reinterpret_cast<const unsigned short*>(x.data_ptr());
  auto* wp = reinterpret_cast<const unsigned short*>(w.data_ptr());
  const long total = (long)M * N;

#define SK_LAUNCH(MT_)                                                       \
  if (splitk == 1) {                                                         \
    skinny_gemm_kernel<MT_, false><<<grid, 256, 0, stream>>>(                \
        op, nullptr, xp, wp, M, N, K, splitk);                               \
  } else {                                                                   \
    float* wsp = ws.data_ptr<float>();                                       \
    TORCH_CHECK(ws.numel() >= total, "skinny_gemm workspace too small");     \
    skinny_zero_kernel<<<dim3((unsigned)((total / 4 + 255) / 256)), 256, 0,  \
                         stream>>>(wsp, total);                              \
    skinny_gemm_kernel<MT_, true><<<grid, 256, 0, stream>>>(                 \
        nullptr, wsp, xp, wp, M, N, K, splitk);                              \
    skinny_cast_kernel<<<dim3((unsigned)((total / 8 + 255) / 256)), 256, 0,  \
                         stream>>>(op, wsp, total);                          \
  }
  switch (MT) {
    case 1: SK_LAUNCH(1); break;
    case 2: SK_LAUNCH(2); break;
    case 3: SK_LAUNCH(3); break;
    default: SK_LAUNCH(4); break;
  }
#undef SK_LAUNCH
  HIP_CHECK_KERNEL();
}

}  // namespace kukeon
