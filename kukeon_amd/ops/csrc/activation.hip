// Fused SwiGLU activation: out = silu(gate) * up, bf16, vectorized 16B/lane.
// gate/up are the two halves of the fused gate_up projection output
// [T, 2*I] (gate = [:, :I], up = [:, I:]) so the MLP needs one GEMM + this.
#include "common.h"
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

namespace kukeon {

__global__ void silu_mul_kernel(unsigned short* __restrict__ out,      // [T, I]
                                const unsigned short* __restrict__ gu,  // [T, 2I]
                                long I, long total8) {
  const long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= total8) return;
  const long row = (idx * 8) / I;
  const long col = (idx * 8) % I;
  bf16x8 g = load_bf16x8(gu + row * 2 * I + col);
  bf16x8 u = load_bf16x8(gu + row * 2 * I + I + col);
  float o[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    float x = g.f(j);
    float s = x / (1.f + __expf(-x));
    o[j] = s * u.f(j);
  }
  *reinterpret_cast<uint4*>(out + row * I + col) = pack_bf16x8(o);
}

void silu_mul(torch::Tensor out, torch::Tensor gate_up) {
  TORCH_CHECK(out.is_contiguous() && gate_up.is_contiguous());
  TORCH_CHECK(gate_up.scalar_type() == torch::kBFloat16);
  const long I = out.size(-1);
  TORCH_CHECK(gate_up.size(-1) == 2 * I, "gate_up last dim must be 2*I");
  TORCH_CHECK(I % 8 == 0);
  const long total8 = out.numel() / 8;
  if (total8 == 0) return;
  auto stream = c10::hip::getCurrentHIPStream().stream();
  const int block = 256;
  const long grid = (total8 + block - 1) / block;
  silu_mul_kernel<<<dim3((unsigned)grid), block, 0, stream>>>(
      reinterpret_cast<unsigned short*>(out.data_ptr()),
      reinterpret_cast<const unsigned short*>(gate_up.data_ptr()), I, total8);
  HIP_CHECK_KERNEL();
}

}  // namespace kukeon
