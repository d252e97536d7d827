// Fused RMSNorm kernels for gfx950.
//
// Memory-bound: vectorized bf16x8 (16 B/lane) loads per guide G13; one block
// per row, values register-cached between the sum-of-squares pass and the
// normalize pass so each row is read from HBM exactly once.
//
// Capability parity note: the reference (eminwux/kukeon) has no GPU ops; this
// is part of the new data plane required by BASELINE.json (hand-written CDNA4
// RMSNorm kernel).
#include "common.h"
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

namespace kukeon {

// ITERS = H / (8 * block). Register-cache the row (ITERS * 8 floats/thread).
template <int ITERS, bool FUSED_ADD>
__global__ void rmsnorm_kernel(unsigned short* __restrict__ out,
                               unsigned short* __restrict__ input,     // [T,H]
                               unsigned short* __restrict__ residual,  // [T,H] or null
                               const unsigned short* __restrict__ weight,
                               float eps, int H) {
  __shared__ float red[16];
  const long row = blockIdx.x;
  unsigned short* in_row = input + row * (long)H;
  unsigned short* res_row = FUSED_ADD ? residual + row * (long)H : nullptr;
  unsigned short* out_row = out + row * (long)H;

  float vals[ITERS][8];
  float ss = 0.f;
#pragma unroll
  for (int it = 0; it < ITERS; ++it) {
    const int base = (it * blockDim.x + threadIdx.x) * 8;
    bf16x8 v = load_bf16x8(in_row + base);
    if (FUSED_ADD) {
      bf16x8 r = load_bf16x8(res_row + base);
#pragma unroll
      for (int j = 0; j < 8; ++j) vals[it][j] = v.f(j) + r.f(j);
      // write the updated residual back (residual = residual + input)
      *reinterpret_cast<uint4*>(res_row + base) = pack_bf16x8(vals[it]);
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) vals[it][j] = v.f(j);
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) ss += vals[it][j] * vals[it][j];
  }
  ss = block_reduce(ss, red, SumOp{}, 0.f);
  const float rs = rsqrtf(ss / (float)H + eps);
#pragma unroll
  for (int it = 0; it < ITERS; ++it) {
    const int base = (it * blockDim.x + threadIdx.x) * 8;
    bf16x8 w = load_bf16x8(weight + base);
    float o[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = vals[it][j] * rs * w.f(j);
    *reinterpret_cast<uint4*>(out_row + base) = pack_bf16x8(o);
  }
}

// Small-batch (decode) variant: one WAVE per row, no cross-wave barrier —
// at T<=128 the block-per-row kernel is launch/latency-bound (64 blocks on
// 256 CUs with two __syncthreads); a wave-local reduce halves the latency.
template <int ITERS, bool FUSED_ADD>
__global__ void rmsnorm_wave_kernel(unsigned short* __restrict__ out,
                                    unsigned short* __restrict__ input,
                                    unsigned short* __restrict__ residual,
                                    const unsigned short* __restrict__ weight,
                                    float eps, int H, int T) {
  const int row = blockIdx.x * 4 + (threadIdx.x >> 6);
  if (row >= T) return;
  const int lane = threadIdx.x & 63;
  unsigned short* in_row = input + (long)row * H;
  unsigned short* res_row = FUSED_ADD ? residual + (long)row * H : nullptr;
  unsigned short* out_row = out + (long)row * H;
  float vals[ITERS][8];
  float ss = 0.f;
#pragma unroll
  for (int it = 0; it < ITERS; ++it) {
    const int base = (it * WAVE + lane) * 8;
    bf16x8 v = load_bf16x8(in_row + base);
    if (FUSED_ADD) {
      bf16x8 r = load_bf16x8(res_row + base);
#pragma unroll
      for (int j = 0; j < 8; ++j) vals[it][j] = v.f(j) + r.f(j);
      *reinterpret_cast<uint4*>(res_row + base) = pack_bf16x8(vals[it]);
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) vals[it][j] = v.f(j);
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) ss += vals[it][j] * vals[it][j];
  }
  ss = wave_sum(ss);
  const float rs = rsqrtf(ss / (float)H + eps);
#pragma unroll
  for (int it = 0; it < ITERS; ++it) {
    const int base = (it * WAVE + lane) * 8;
    bf16x8 w = load_bf16x8(weight + base);
    float o[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = vals[it][j] * rs * w.f(j);
    *reinterpret_cast<uint4*>(out_row + base) = pack_bf16x8(o);
  }
}

// Generic fallback for hidden sizes that don't match a template instance.
template <bool FUSED_ADD>
__global__ void rmsnorm_kernel_generic(unsigned short* __restrict__ out,
                                       unsigned short* __restrict__ input,
                                       unsigned short* __restrict__ residual,
                                       const unsigned short* __restrict__ weight,
                                       float eps, int H) {
  __shared__ float red[16];
  const long row = blockIdx.x;
  unsigned short* in_row = input + row * (long)H;
  unsigned short* res_row = FUSED_ADD ? residual + row * (long)H : nullptr;
  unsigned short* out_row = out + row * (long)H;
  float ss = 0.f;
  for (int i = threadIdx.x; i < H; i += blockDim.x) {
    float v = us2f(in_row[i]);
    if (FUSED_ADD) {
      v += us2f(res_row[i]);
      res_row[i] = f2us(v);
    }
    ss += v * v;
  }
  ss = block_reduce(ss, red, SumOp{}, 0.f);
  const float rs = rsqrtf(ss / (float)H + eps);
  for (int i = threadIdx.x; i < H; i += blockDim.x) {
    float v = FUSED_ADD ? us2f(res_row[i]) : us2f(in_row[i]);
    out_row[i] = f2us(v * rs * us2f(weight[i]));
  }
}

template <bool FUSED_ADD>
static void launch_rmsnorm(torch::Tensor& out, torch::Tensor& input,
                           torch::Tensor* residual, const torch::Tensor& weight,
                           double eps) {
  const int H = input.size(-1);
  const long T = input.numel() / H;
  if (T == 0) return;
  auto stream = c10::hip::getCurrentHIPStream().stream();
  dim3 grid((unsigned)T);
  const int block = 256;
  auto* o = reinterpret_cast<unsigned short*>(out.data_ptr());
  auto* in = reinterpret_cast<unsigned short*>(input.data_ptr());
  auto* res = residual ? reinterpret_cast<unsigned short*>(residual->data_ptr()) : nullptr;
  auto* w = reinterpret_cast<const unsigned short*>(weight.data_ptr());

#define RMS_CASE(N)                                                           \
  case N * 8 * 256:                                                           \
    if (T <= 128) {                                                           \
      rmsnorm_wave_kernel<N * 4, FUSED_ADD>                                   \
          <<<dim3((unsigned)((T + 3) / 4)), 256, 0, stream>>>(o, in, res, w,  \
                                                              (float)eps, H,  \
                                                              (int)T);        \
    } else {                                                                  \
      rmsnorm_kernel<N, FUSED_ADD><<<grid, block, 0, stream>>>(o, in, res, w, \
                                                               (float)eps, H);\
    }                                                                         \
    break;
  switch (H) {
    RMS_CASE(1)  // 2048
    RMS_CASE(2)  // 4096
    RMS_CASE(3)  // 6144
    RMS_CASE(4)  // 8192
    default:
      TORCH_CHECK(H % 8 == 0 || !FUSED_ADD, "hidden size must be mult of 8");
      rmsnorm_kernel_generic<FUSED_ADD>
          <<<grid, block, 0, stream>>>(o, in, res, w, (float)eps, H);
  }
#undef RMS_CASE
  HIP_CHECK_KERNEL();
}

void rmsnorm(torch::Tensor out, torch::Tensor input, torch::Tensor weight,
             double eps) {
  TORCH_CHECK(input.is_contiguous() && out.is_contiguous());
  TORCH_CHECK(input.scalar_type() == torch::kBFloat16);
  launch_rmsnorm<false>(out, input, nullptr, weight, eps);
}

// residual = residual + input; input(out, in-place) = rmsnorm(residual) * w
void fused_add_rmsnorm(torch::Tensor input, torch::Tensor residual,
                       torch::Tensor weight, double eps) {
  TORCH_CHECK(input.is_contiguous() && residual.is_contiguous());
  TORCH_CHECK(input.scalar_type() == torch::kBFloat16);
  launch_rmsnorm<true>(input, input, &residual, weight, eps);
}

}  // namespace kukeon
