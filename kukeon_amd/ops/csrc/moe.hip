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

// Dense-routing router: softmax over E logits, top-K, renormalize, scatter
// into a dense [T, E] weight row — ONE launch replacing the eager
// softmax/topk/div/zeros/scatter chain (~5 kernels per MoE layer per
// decode step, ~6% of the Mixtral run in launch+tensor overhead).
// One thread per token row; E <= 32, K <= 4.
__global__ void moe_router_kernel(float* __restrict__ wdense,   // [T,E]
                                  const void* __restrict__ logits,
                                  int T, int E, int K, bool bf16_in) {
  const int t = blockIdx.x * blockDim.x + threadIdx.x;
  if (t >= T) return;
  float l[32];
  float mx = -1e30f;
  for (int e = 0; e < E; ++e) {
    l[e] = bf16_in
        ? us2f(reinterpret_cast<const unsigned short*>(
              logits)[(long)t * E + e])
        : reinterpret_cast<const float*>(logits)[(long)t * E + e];
    mx = fmaxf(mx, l[e]);
  }
  float denom = 0.f;
  for (int e = 0; e < E; ++e) {
    l[e] = __expf(l[e] - mx);
    denom += l[e];
  }
  // top-K selection by repeated max (E small); ties resolve to the
  // lowest expert id, matching torch.topk's stable order
  int ids[4];
  float vals[4];
  float picked_sum = 0.f;
  for (int k = 0; k < K; ++k) {
    int bi = -1;
    float bv = -1.f;
    for (int e = 0; e < E; ++e) {
      bool taken = false;
      for (int j = 0; j < k; ++j) taken |= (ids[j] == e);
      if (!taken && l[e] > bv) { bv = l[e]; bi = e; }
    }
    ids[k] = bi;
    vals[k] = bv;
    picked_sum += bv;
  }
  for (int e = 0; e < E; ++e) wdense[(long)t * E + e] = 0.f;
  for (int k = 0; k < K; ++k) {
    // renormalized over the picked set; the /denom cancels
    wdense[(long)t * E + ids[k]] = vals[k] / picked_sum;
  }
}

// Weighted combine of the dense-routed expert outputs: out[t] =
// sum_e w[t,e] * y[e,t] — replaces y.float()*w mul + sum(dim=0) (~4
// eager kernels + two fp32 materializations of [E,T,H]). Zero-weight
// experts are skipped, so only the K live experts' rows are read.
__global__ void moe_dense_combine_kernel(
    unsigned short* __restrict__ out,      // [T, H]
    const unsigned short* __restrict__ y,  // [E, T, H]
    const float* __restrict__ wdense,      // [T, E]
    int E, long TH8, int H8) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;  // t*H8+c
  if (i >= TH8) return;
  const long t = i / H8;
  float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  for (int e = 0; e < E; ++e) {
    const float w = wdense[t * E + e];
    if (w == 0.f) continue;
    bf16x8 v;
    v.raw = reinterpret_cast<const uint4*>(y)[(long)e * TH8 + i];
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[j] += w * v.f(j);
  }
  reinterpret_cast<uint4*>(out)[i] = pack_bf16x8(acc);
}

void moe_router_weights(torch::Tensor wdense, torch::Tensor logits,
                        long K) {
  const int T = logits.size(0);
  const int E = logits.size(1);
  const bool bf16_in = logits.scalar_type() == torch::kBFloat16;
  TORCH_CHECK(E <= 32 && K <= 4 && K <= E);
  TORCH_CHECK(bf16_in || logits.scalar_type() == torch::kFloat32);
  TORCH_CHECK(wdense.scalar_type() == torch::kFloat32);
  TORCH_CHECK(wdense.size(0) == T && wdense.size(1) == E);
  TORCH_CHECK(logits.is_contiguous() && wdense.is_contiguous());
  if (T == 0) return;
  auto stream = c10::hip::getCurrentHIPStream().stream();
  moe_router_kernel<<<dim3((unsigned)((T + 255) / 256)), 256, 0, stream>>>(
      wdense.data_ptr<float>(), logits.data_ptr(), T, E, (int)K, bf16_in);
  HIP_CHECK_KERNEL();
}

void moe_dense_combine(torch::Tensor out, torch::Tensor y,
                       torch::Tensor wdense) {
  const int E = y.size(0);
  const long T = y.size(1);
  const int H = y.size(2);
  TORCH_CHECK(H % 8 == 0);
  TORCH_CHECK(out.size(0) == T && out.size(1) == H);
  TORCH_CHECK(out.scalar_type() == torch::kBFloat16 &&
              y.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(wdense.scalar_type() == torch::kFloat32);
  TORCH_CHECK(out.is_contiguous() && y.is_contiguous() &&
              wdense.is_contiguous());
  if (T == 0) return;
  const long TH8 = T * (H / 8);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  moe_dense_combine_kernel<<<dim3((unsigned)((TH8 + 255) / 256)), 256, 0,
                             stream>>>(
      reinterpret_cast<unsigned short*>(out.data_ptr()),
      reinterpret_cast<const unsigned short*>(y.data_ptr()),
      wdense.data_ptr<float>(), E, TH8, H / 8);
  HIP_CHECK_KERNEL();
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
