// Fused top-k / top-p / temperature sampling for gfx950.
//
// Three passes per row (block per row, V ~ 128k):
//   1. row max + softmax denominator Z (with temperature folded in)
//   2. logit-threshold selection: radix walk (3 x 8-bit levels) over the
//      monotonic uint mapping of f32 logits; the top-k constraint walks a
//      count histogram, the top-p constraint walks an exp-weight histogram;
//      the final threshold is the max of both (intersection of filters).
//   3. sampling: Gumbel-max over the masked set (exact categorical draw,
//      single argmax reduction — no sorted cumsum needed); temperature<=0
//      means greedy argmax.
//
// Deterministic given (seed, step): RNG is a counter-based splitmix64 hash.
// All state lives in device tensors so the whole thing is hipGraph-capturable;
// bump_seed advances the device seed inside the captured region.
#include "common.h"
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

namespace kukeon {

DEV_INLINE unsigned int map_f32(float f) {
  unsigned int b = __builtin_bit_cast(unsigned int, f);
  return (b & 0x80000000u) ? ~b : (b | 0x80000000u);
}
DEV_INLINE float unmap_f32(unsigned int u) {
  unsigned int b = (u & 0x80000000u) ? (u & 0x7fffffffu) : ~u;
  return __builtin_bit_cast(float, b);
}

DEV_INLINE unsigned long long splitmix64(unsigned long long x) {
  x += 0x9e3779b97f4a7c15ull;
  x = (x ^ (x >> 30)) * 0xbf58476d1ce4e5b9ull;
  x = (x ^ (x >> 27)) * 0x94d049bb133111ebull;
  return x ^ (x >> 31);
}

DEV_INLINE float uniform01(unsigned long long seed, unsigned int row,
                           unsigned int idx) {
  unsigned long long h =
      splitmix64(seed ^ ((unsigned long long)row << 32) ^ idx);
  // 24 mantissa bits -> (0,1]
  return ((h >> 40) + 1.f) * (1.f / 16777217.f);
}

// ---- pass 1: max + Z ----
__global__ void sample_prep_kernel(float* __restrict__ ws,  // [B,4]
                                   const unsigned short* __restrict__ logits,
                                   const float* __restrict__ temps, int V) {
  __shared__ float red[16];
  const int r = blockIdx.x;
  const unsigned short* row = logits + (long)r * V;
  float mx = -INFINITY;
  for (int i = threadIdx.x; i < V; i += blockDim.x)
    mx = fmaxf(mx, us2f(row[i]));
  mx = block_reduce(mx, red, MaxOp{}, -INFINITY);
  const float T = temps[r];
  const float invT = T > 0.f ? 1.f / T : 1.f;
  float z = 0.f;
  for (int i = threadIdx.x; i < V; i += blockDim.x)
    z += __expf((us2f(row[i]) - mx) * invT);
  z = block_reduce(z, red, SumOp{}, 0.f);
  if (threadIdx.x == 0) {
    ws[r * 4 + 0] = mx;
    ws[r * 4 + 1] = z;
    ws[r * 4 + 2] = invT;
    ws[r * 4 + 3] = -INFINITY;  // threshold, filled by pass 2
  }
}

// ---- pass 2: radix threshold walk ----
__global__ void sample_threshold_kernel(
    float* __restrict__ ws, const unsigned short* __restrict__ logits,
    const int* __restrict__ top_k, const float* __restrict__ top_p, int V) {
  __shared__ unsigned int hist_c[256];
  __shared__ float hist_w[256];
  __shared__ unsigned int sh_prefix;
  __shared__ float sh_need;
  const int r = blockIdx.x;
  const unsigned short* row = logits + (long)r * V;
  const int k = top_k[r];
  const float p = top_p[r];
  const float mx = ws[r * 4 + 0];
  const float Z = ws[r * 4 + 1];
  const float invT = ws[r * 4 + 2];
  const bool use_k = k > 0 && k < V;
  const bool use_p = p > 0.f && p < 1.f;
  float th = -INFINITY;

  // --- top-k: count walk ---
  if (use_k) {
    unsigned int prefix = 0;
    float need = (float)k;
    for (int level = 0; level < 3; ++level) {
      const int shift = 24 - 8 * level;
      for (int i = threadIdx.x; i < 256; i += blockDim.x) hist_c[i] = 0;
      __syncthreads();
      for (int i = threadIdx.x; i < V; i += blockDim.x) {
        const unsigned int u = map_f32(us2f(row[i]));
        if (level == 0 || (u >> (shift + 8)) == prefix)
          atomicAdd(&hist_c[(u >> shift) & 0xffu], 1u);
      }
      __syncthreads();
      if (threadIdx.x == 0) {
        float cum = 0.f, nd = need;
        int sel = 0;
        for (int bin = 255; bin >= 0; --bin) {
          if (cum + (float)hist_c[bin] >= nd) { sel = bin; break; }
          cum += (float)hist_c[bin];
        }
        sh_prefix = (prefix << 8) | (unsigned)sel;
        sh_need = nd - cum;
      }
      __syncthreads();
      prefix = sh_prefix;
      need = sh_need;
      __syncthreads();
    }
    th = unmap_f32(prefix << 8);
  }

  // --- top-p: weight walk ---
  if (use_p) {
    unsigned int prefix = 0;
    float need = p * Z;
    for (int level = 0; level < 3; ++level) {
      const int shift = 24 - 8 * level;
      for (int i = threadIdx.x; i < 256; i += blockDim.x) hist_w[i] = 0.f;
      __syncthreads();
      for (int i = threadIdx.x; i < V; i += blockDim.x) {
        const float l = us2f(row[i]);
        const unsigned int u = map_f32(l);
        if (level == 0 || (u >> (shift + 8)) == prefix)
          atomicAdd(&hist_w[(u >> shift) & 0xffu], __expf((l - mx) * invT));
      }
      __syncthreads();
      if (threadIdx.x == 0) {
        float cum = 0.f, nd = need;
        int sel = 0;
        for (int bin = 255; bin >= 0; --bin) {
          if (cum + hist_w[bin] >= nd || bin == 0) { sel = bin; break; }
          cum += hist_w[bin];
        }
        sh_prefix = (prefix << 8) | (unsigned)sel;
        sh_need = nd - cum;
      }
      __syncthreads();
      prefix = sh_prefix;
      need = sh_need;
      __syncthreads();
    }
    th = fmaxf(th, unmap_f32(prefix << 8));
  }

  if (threadIdx.x == 0) ws[r * 4 + 3] = th;
}

// ---- pass 3: Gumbel-max draw over the masked set ----
__global__ void sample_draw_kernel(int* __restrict__ tokens,
                                   const unsigned short* __restrict__ logits,
                                   const float* __restrict__ ws,
                                   const float* __restrict__ temps,
                                   const unsigned long long* __restrict__ seed,
                                   int V) {
  __shared__ float red_v[16];
  __shared__ int red_i[16];
  const int r = blockIdx.x;
  const unsigned short* row = logits + (long)r * V;
  const float T = temps[r];
  const float mx = ws[r * 4 + 0];
  const float invT = ws[r * 4 + 2];
  const float th = ws[r * 4 + 3];
  const bool greedy = T <= 0.f;
  const unsigned long long sd = seed[0];
  float best = -INFINITY;
  int besti = 0;
  for (int i = threadIdx.x; i < V; i += blockDim.x) {
    const float l = us2f(row[i]);
    if (l < th) continue;
    float key;
    if (greedy) {
      key = l;
    } else {
      const float u = uniform01(sd, (unsigned)r, (unsigned)i);
      key = (l - mx) * invT - __logf(-__logf(u));
    }
    if (key > best || (key == best && i < besti)) { best = key; besti = i; }
  }
  // block argmax reduce
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    const float ov = __shfl_xor(best, off, WAVE);
    const int oi = __shfl_xor(besti, off, WAVE);
    if (ov > best || (ov == best && oi < besti)) { best = ov; besti = oi; }
  }
  if (lane == 0) { red_v[wid] = best; red_i[wid] = besti; }
  __syncthreads();
  if (threadIdx.x == 0) {
    const int nwaves = (blockDim.x + WAVE - 1) / WAVE;
    for (int w = 1; w < nwaves; ++w) {
      if (red_v[w] > best || (red_v[w] == best && red_i[w] < besti)) {
        best = red_v[w];
        besti = red_i[w];
      }
    }
    tokens[r] = besti;
  }
}

__global__ void bump_seed_kernel(unsigned long long* seed) { seed[0] += 1; }

void sample(torch::Tensor tokens, torch::Tensor logits, torch::Tensor temps,
            torch::Tensor top_k, torch::Tensor top_p, torch::Tensor seed,
            torch::Tensor workspace) {
  const int B = logits.size(0);
  if (B == 0) return;
  const int V = logits.size(1);
  TORCH_CHECK(logits.scalar_type() == torch::kBFloat16 && logits.is_contiguous());
  TORCH_CHECK(tokens.scalar_type() == torch::kInt32);
  TORCH_CHECK(workspace.numel() >= B * 4);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  const auto* lp = reinterpret_cast<const unsigned short*>(logits.data_ptr());
  float* ws = workspace.data_ptr<float>();
  sample_prep_kernel<<<B, 256, 0, stream>>>(ws, lp, temps.data_ptr<float>(), V);
  HIP_CHECK_KERNEL();
  sample_threshold_kernel<<<B, 256, 0, stream>>>(
      ws, lp, top_k.data_ptr<int>(), top_p.data_ptr<float>(), V);
  HIP_CHECK_KERNEL();
  sample_draw_kernel<<<B, 256, 0, stream>>>(
      tokens.data_ptr<int>(), lp, ws, temps.data_ptr<float>(),
      reinterpret_cast<unsigned long long*>(seed.data_ptr()), V);
  HIP_CHECK_KERNEL();
  bump_seed_kernel<<<1, 1, 0, stream>>>(
      reinterpret_cast<unsigned long long*>(seed.data_ptr()));
  HIP_CHECK_KERNEL();
}

}  // namespace kukeon
