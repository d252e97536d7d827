// MoE token permute/unpermute kernels for gfx950 (Mixtral path).
//
// Routing (softmax top-k) and the per-expert sort happen in torch; these
// kernels do the bandwidth-critical permute into expert-sorted order and the
// weighted combine back, both vectorized 16 B/lane.
#include "common.h"
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

namespace kukeon {

__global__ void moe_gather_kernel(unsigned short* __restrict__ out,  // [E,H]
                                  const unsigned short* __restrict__ in, // [T,H]
                                  const int* __restrict__ row_map,   // [E]
                                  int H8) {
  const long j = blockIdx.x;
  const long src = row_map[j];
  const uint4* ip = reinterpret_cast<const uint4*>(in) + src * H8;
  uint4* op = reinterpret_cast<uint4*>(out) + j * H8;
  for (int c = threadIdx.x; c < H8; c += blockDim.x) op[c] = ip[c];
}

__global__ void moe_scatter_kernel(unsigned short* __restrict__ out,   // [T,H]
                                   const unsigned short* __restrict__ in, // [E,H]
                                   const int* __restrict__ inv_map,    // [T,K]
                                   const float* __restrict__ weights,  // [T,K]
                                   int H8, int K) {
  const long t = blockIdx.x;
  uint4* op = reinterpret_cast<uint4*>(out) + t * H8;
  for (int c = threadIdx.x; c < H8; c += blockDim.x) {
    float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    for (int k = 0; k < K; ++k) {
      const float w = weights[t * K + k];
      const long src = inv_map[t * K + k];
      bf16x8 v;
      v.raw = reinterpret_cast<const uint4*>(in)[src * H8 + c];
#pragma unroll
      for (int j = 0; j < 8; ++j) acc[j] += w * v.f(j);
    }
    op[c] = pack_bf16x8(acc);
  }
}

void moe_gather_tokens(torch::Tensor out, torch::Tensor input,
                       torch::Tensor row_map) {
  const long E = out.size(0);
  if (E == 0) return;
  const int H = input.size(1);
  TORCH_CHECK(H % 8 == 0);
  TORCH_CHECK(row_map.scalar_type() == torch::kInt32);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  moe_gather_kernel<<<dim3((unsigned)E), 256, 0, stream>>>(
      reinterpret_cast<unsigned short*>(out.data_ptr()),
      reinterpret_cast<const unsigned short*>(input.data_ptr()),
      row_map.data_ptr<int>(), H / 8);
  HIP_CHECK_KERNEL();
}

void moe_scatter_tokens(torch::Tensor out, torch::Tensor input,
                        torch::Tensor inv_map, torch::Tensor weights,
                        int64_t top_k) {
  const long T = out.size(0);
  if (T == 0) return;
  const int H = out.size(1);
  TORCH_CHECK(H % 8 == 0);
  TORCH_CHECK(inv_map.scalar_type() == torch::kInt32);
  TORCH_CHECK(weights.scalar_type() == torch::kFloat32);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  moe_scatter_kernel<<<dim3((unsigned)T), 256, 0, stream>>>(
      reinterpret_cast<unsigned short*>(out.data_ptr()),
      reinterpret_cast<const unsigned short*>(input.data_ptr()),
      inv_map.data_ptr<int>(), weights.data_ptr<float>(), H / 8, (int)top_k);
  HIP_CHECK_KERNEL();
}

}  // namespace kukeon
