// Fused top-k / top-p / temperature sampling for gfx950 — wide-parallel.
//
// Key observations driving the design (see profiles/r01_bench8_kernel_stats):
//  * logits are bf16, so their monotonic integer map has only 16 bits —
//    TWO 8-bit radix levels give an EXACT threshold (no 3rd pass, no ties
//    below the representable precision),
//  * one workgroup per row leaves a 256-CU chip almost idle at B<=64: every
//    pass here is split over grid (B, S) chunks of the vocab with
//    LDS-local histograms merged by global atomics,
//  * the categorical draw is Gumbel-max over the masked set (single packed
//    64-bit atomicMax, deterministic: ties resolve to the lowest index).
//
// Pipeline (7 small kernels, all hipGraph-capturable; ws row = 528 floats):
//  memset ws -> K1 max (+invT)  -> K2 level-0 count/weight hists + Z
//  -> K3 walk0 (+ hist re-zero) -> K4 level-1 hists -> K5 walk1 (threshold)
//  -> K6 masked gumbel argmax   -> K7 finalize token + bump seed.
//
// ws[b] layout (floats, reinterpreted per slot):
//  [0] row max (mapped u32)   [1] Z (f32)        [2] invT (f32)
//  [3] threshold (f32)        [4] prefix_k (u32) [5] prefix_p (u32)
//  [6..7] packed argmax (u64, 8-byte aligned: 528*4 % 8 == 0)
//  [8]  need_k (f32)          [9] need_p (f32)
//  [16..271] hist_count (u32) [272..527] hist_weight (f32)
#include "common.h"
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

namespace kukeon {

constexpr int WS_ROW = 528;

DEV_INLINE unsigned int map16(unsigned short us) {
  return (us & 0x8000u) ? (unsigned int)(~us & 0xffffu)
                        : (unsigned int)(us | 0x8000u);
}
DEV_INLINE float unmap16_f(unsigned int mu) {
  unsigned short us = (mu & 0x8000u) ? (unsigned short)(mu & 0x7fffu)
                                     : (unsigned short)(~mu & 0xffffu);
  return us2f(us);
}
DEV_INLINE unsigned int map_f32(float f) {
  unsigned int b = __builtin_bit_cast(unsigned int, f);
  return (b & 0x80000000u) ? ~b : (b | 0x80000000u);
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
  return ((h >> 40) + 1.f) * (1.f / 16777217.f);
}

struct Chunk {
  int lo, hi;
};
DEV_INLINE Chunk chunk_of(int V, int splits, int s) {
  const int per = (V + splits - 1) / splits;
  Chunk c{s * per, min(V, (s + 1) * per)};
  return c;
}

// ---- K0: zero the workspace rows. A kernel, NOT hipMemsetAsync: a
// memset captured inside a hipGraph was observed not to replay (ROCm 7.2),
// which let Z/argmax state accumulate across replays. ----
__global__ void k0_zero(float* __restrict__ ws) {
  float* wsb = ws + (long)blockIdx.x * WS_ROW;
  for (int i = threadIdx.x; i < WS_ROW; i += blockDim.x) wsb[i] = 0.f;
}

// ---- K1: row max (mapped) + invT ----
__global__ void k1_max(float* __restrict__ ws,
                       const unsigned short* __restrict__ logits,
                       const float* __restrict__ temps, int V, int splits) {
  const int b = blockIdx.x;
  const Chunk c = chunk_of(V, splits, blockIdx.y);
  const unsigned short* row = logits + (long)b * V;
  unsigned int mx = 0;
  for (int i = c.lo + threadIdx.x; i < c.hi; i += blockDim.x)
    mx = max(mx, map16(row[i]));
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    mx = max(mx, (unsigned)__shfl_xor((int)mx, off, WAVE));
  __shared__ unsigned int red[16];
  const int lane = threadIdx.x & 63, wid = threadIdx.x >> 6;
  if (lane == 0) red[wid] = mx;
  __syncthreads();
  if (threadIdx.x == 0) {
    for (int w = 1; w < (blockDim.x + 63) / 64; ++w) mx = max(mx, red[w]);
    atomicMax(reinterpret_cast<unsigned int*>(ws + (long)b * WS_ROW), mx);
    if (blockIdx.y == 0) {
      const float T = temps[b];
      ws[(long)b * WS_ROW + 2] = T > 0.f ? 1.f / T : 1.f;
    }
  }
}

// ---- K2: level-0 histograms (count + exp-weight) + Z ----
__global__ void k2_hist0(float* __restrict__ ws,
                         const unsigned short* __restrict__ logits, int V,
                         int splits) {
  __shared__ unsigned int hc[256];
  __shared__ float hw[256];
  __shared__ float redf[16];
  const int b = blockIdx.x;
  float* wsb = ws + (long)b * WS_ROW;
  const Chunk c = chunk_of(V, splits, blockIdx.y);
  const unsigned short* row = logits + (long)b * V;
  for (int i = threadIdx.x; i < 256; i += blockDim.x) {
    hc[i] = 0;
    hw[i] = 0.f;
  }
  __syncthreads();
  const float mx = unmap16_f(
      *reinterpret_cast<const unsigned int*>(wsb + 0));
  const float invT = wsb[2];
  float z = 0.f;
  for (int i = c.lo + threadIdx.x; i < c.hi; i += blockDim.x) {
    const unsigned short us = row[i];
    const unsigned int mu = map16(us);
    const float e = __expf((us2f(us) - mx) * invT);
    z += e;
    atomicAdd(&hc[mu >> 8], 1u);
    atomicAdd(&hw[mu >> 8], e);
  }
  z = block_reduce(z, redf, SumOp{}, 0.f);
  if (threadIdx.x == 0) atomicAdd(wsb + 1, z);
  __syncthreads();
  unsigned int* gc = reinterpret_cast<unsigned int*>(wsb + 16);
  float* gw = wsb + 272;
  for (int i = threadIdx.x; i < 256; i += blockDim.x) {
    if (hc[i]) atomicAdd(&gc[i], hc[i]);
    if (hw[i] != 0.f) atomicAdd(&gw[i], hw[i]);
  }
}

// ---- K3: walk level 0 for both constraints, then re-zero the hists ----
__global__ void k3_walk0(float* __restrict__ ws,
                         const int* __restrict__ top_k,
                         const float* __restrict__ top_p, int V) {
  const int b = blockIdx.x;
  float* wsb = ws + (long)b * WS_ROW;
  unsigned int* gc = reinterpret_cast<unsigned int*>(wsb + 16);
  float* gw = wsb + 272;
  if (threadIdx.x == 0) {
    const int k = top_k[b];
    if (k > 0 && k < V) {
      float cum = 0.f;
      int sel = 0;
      for (int bin = 255; bin >= 0; --bin) {
        if (cum + (float)gc[bin] >= (float)k || bin == 0) { sel = bin; break; }
        cum += (float)gc[bin];
      }
      *reinterpret_cast<unsigned int*>(wsb + 4) = (unsigned)sel;
      wsb[8] = (float)k - cum;
    } else {
      *reinterpret_cast<unsigned int*>(wsb + 4) = 0xffffffffu;
    }
  } else if (threadIdx.x == 1) {
    const float p = top_p[b];
    if (p > 0.f && p < 1.f) {
      const float need = p * wsb[1];
      float cum = 0.f;
      int sel = 0;
      for (int bin = 255; bin >= 0; --bin) {
        if (cum + gw[bin] >= need || bin == 0) { sel = bin; break; }
        cum += gw[bin];
      }
      *reinterpret_cast<unsigned int*>(wsb + 5) = (unsigned)sel;
      wsb[9] = need - cum;
    } else {
      *reinterpret_cast<unsigned int*>(wsb + 5) = 0xffffffffu;
    }
  }
  __syncthreads();
  for (int i = threadIdx.x; i < 256; i += blockDim.x) {
    gc[i] = 0;
    gw[i] = 0.f;
  }
}

// ---- K4: level-1 histograms restricted to the selected level-0 bins ----
__global__ void k4_hist1(float* __restrict__ ws,
                         const unsigned short* __restrict__ logits, int V,
                         int splits) {
  __shared__ unsigned int hc[256];
  __shared__ float hw[256];
  const int b = blockIdx.x;
  float* wsb = ws + (long)b * WS_ROW;
  const unsigned int pk = *reinterpret_cast<const unsigned int*>(wsb + 4);
  const unsigned int pp = *reinterpret_cast<const unsigned int*>(wsb + 5);
  if (pk == 0xffffffffu && pp == 0xffffffffu) return;
  const Chunk c = chunk_of(V, splits, blockIdx.y);
  const unsigned short* row = logits + (long)b * V;
  for (int i = threadIdx.x; i < 256; i += blockDim.x) {
    hc[i] = 0;
    hw[i] = 0.f;
  }
  __syncthreads();
  const float mx = unmap16_f(*reinterpret_cast<const unsigned int*>(wsb));
  const float invT = wsb[2];
  for (int i = c.lo + threadIdx.x; i < c.hi; i += blockDim.x) {
    const unsigned short us = row[i];
    const unsigned int mu = map16(us);
    const unsigned int hi8 = mu >> 8;
    if (hi8 == pk) atomicAdd(&hc[mu & 0xffu], 1u);
    if (hi8 == pp)
      atomicAdd(&hw[mu & 0xffu], __expf((us2f(us) - mx) * invT));
  }
  __syncthreads();
  unsigned int* gc = reinterpret_cast<unsigned int*>(wsb + 16);
  float* gw = wsb + 272;
  for (int i = threadIdx.x; i < 256; i += blockDim.x) {
    if (hc[i]) atomicAdd(&gc[i], hc[i]);
    if (hw[i] != 0.f) atomicAdd(&gw[i], hw[i]);
  }
}

// ---- K5: walk level 1 -> final logit threshold ----
__global__ void k5_walk1(float* __restrict__ ws, int V) {
  const int b = blockIdx.x;
  float* wsb = ws + (long)b * WS_ROW;
  if (threadIdx.x != 0) return;
  unsigned int* gc = reinterpret_cast<unsigned int*>(wsb + 16);
  float* gw = wsb + 272;
  const unsigned int pk = *reinterpret_cast<const unsigned int*>(wsb + 4);
  const unsigned int pp = *reinterpret_cast<const unsigned int*>(wsb + 5);
  float th = -INFINITY;
  if (pk != 0xffffffffu) {
    const float need = wsb[8];
    float cum = 0.f;
    int sel = 0;
    for (int bin = 255; bin >= 0; --bin) {
      if (cum + (float)gc[bin] >= need || bin == 0) { sel = bin; break; }
      cum += (float)gc[bin];
    }
    th = fmaxf(th, unmap16_f((pk << 8) | (unsigned)sel));
  }
  if (pp != 0xffffffffu) {
    const float need = wsb[9];
    float cum = 0.f;
    int sel = 0;
    for (int bin = 255; bin >= 0; --bin) {
      if (cum + gw[bin] >= need || bin == 0) { sel = bin; break; }
      cum += gw[bin];
    }
    th = fmaxf(th, unmap16_f((pp << 8) | (unsigned)sel));
  }
  wsb[3] = th;
}

// ---- K6: masked Gumbel argmax, packed 64-bit atomic merge ----
__global__ void k6_draw(float* __restrict__ ws,
                        const unsigned short* __restrict__ logits,
                        const float* __restrict__ temps,
                        const unsigned long long* __restrict__ seed, int V,
                        int splits) {
  const int b = blockIdx.x;
  float* wsb = ws + (long)b * WS_ROW;
  const Chunk c = chunk_of(V, splits, blockIdx.y);
  const unsigned short* row = logits + (long)b * V;
  const float T = temps[b];
  const bool greedy = T <= 0.f;
  const float mx = unmap16_f(*reinterpret_cast<const unsigned int*>(wsb));
  const float invT = wsb[2];
  const float th = wsb[3];
  const unsigned long long sd = seed[0];
  float best = -INFINITY;
  int besti = 0;
  for (int i = c.lo + threadIdx.x; i < c.hi; i += blockDim.x) {
    const float l = us2f(row[i]);
    if (l < th) continue;
    float key;
    if (greedy) {
      key = l;
    } else {
      key = (l - mx) * invT -
            __logf(-__logf(uniform01(sd, (unsigned)b, (unsigned)i)));
    }
    if (key > best || (key == best && i < besti)) { best = key; besti = i; }
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    const float ov = __shfl_xor(best, off, WAVE);
    const int oi = __shfl_xor(besti, off, WAVE);
    if (ov > best || (ov == best && oi < besti)) { best = ov; besti = oi; }
  }
  __shared__ float rv[16];
  __shared__ int ri[16];
  const int lane = threadIdx.x & 63, wid = threadIdx.x >> 6;
  if (lane == 0) { rv[wid] = best; ri[wid] = besti; }
  __syncthreads();
  if (threadIdx.x == 0) {
    for (int w = 1; w < (blockDim.x + 63) / 64; ++w)
      if (rv[w] > best || (rv[w] == best && ri[w] < besti)) {
        best = rv[w];
        besti = ri[w];
      }
    if (best != -INFINITY) {
      // pack: key (mapped f32) in the high word, ~idx low (ties -> min idx)
      const unsigned long long packed =
          ((unsigned long long)map_f32(best) << 32) |
          (unsigned)(~(unsigned)besti);
      atomicMax(reinterpret_cast<unsigned long long*>(wsb + 6), packed);
    }
  }
}

// ---- K7: finalize tokens + bump the seed ----
__global__ void k7_finalize(int* __restrict__ tokens,
                            const float* __restrict__ ws,
                            unsigned long long* __restrict__ seed, int B) {
  const int b = blockIdx.x * blockDim.x + threadIdx.x;
  if (b < B) {
    const unsigned long long packed =
        *reinterpret_cast<const unsigned long long*>(ws + (long)b * WS_ROW + 6);
    tokens[b] = (int)(~(unsigned)(packed & 0xffffffffu));
  }
  if (b == 0) seed[0] += 1;
}

// ---- self-advancing decode support: after sampling, feed the token back
// as the next step's input, advance the device-side cursors, and append to
// the ring buffer — so a micro-batch of decode steps replays with no host
// round-trip (one sync per micro-batch).
__global__ void decode_advance_kernel(int* __restrict__ ids,
                                      int* __restrict__ pos,
                                      int* __restrict__ seq_lens,
                                      const int* __restrict__ tokens,
                                      int* __restrict__ ring,
                                      int* __restrict__ counter, int B) {
  const int b = blockIdx.x * blockDim.x + threadIdx.x;
  if (b >= B) return;
  const int step = counter[0];
  if (seq_lens[b] > 0) {  // active rows only (padded rows stay parked)
    ids[b] = tokens[b];
    pos[b] += 1;
    seq_lens[b] += 1;
    ring[step * B + b] = tokens[b];
  }
  if (b == 0) counter[0] = step + 1;
}

void decode_advance(torch::Tensor ids, torch::Tensor pos,
                    torch::Tensor seq_lens, torch::Tensor tokens,
                    torch::Tensor ring, torch::Tensor counter) {
  const int B = ids.size(0);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  decode_advance_kernel<<<(B + 255) / 256, 256, 0, stream>>>(
      ids.data_ptr<int>(), pos.data_ptr<int>(), seq_lens.data_ptr<int>(),
      tokens.data_ptr<int>(), ring.data_ptr<int>(), counter.data_ptr<int>(),
      B);
  HIP_CHECK_KERNEL();
}

void sample(torch::Tensor tokens, torch::Tensor logits, torch::Tensor temps,
            torch::Tensor top_k, torch::Tensor top_p, torch::Tensor seed,
            torch::Tensor workspace) {
  const int B = logits.size(0);
  if (B == 0) return;
  const int V = logits.size(1);
  TORCH_CHECK(logits.scalar_type() == torch::kBFloat16 &&
              logits.is_contiguous());
  TORCH_CHECK(tokens.scalar_type() == torch::kInt32);
  TORCH_CHECK(workspace.numel() >= (long)B * WS_ROW,
              "sampling workspace must be at least [B, 528] floats");
  TORCH_CHECK(reinterpret_cast<uintptr_t>(workspace.data_ptr()) % 8 == 0);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  const auto* lp = reinterpret_cast<const unsigned short*>(logits.data_ptr());
  float* ws = workspace.data_ptr<float>();
  // fill the chip: ~512 workgroups across the split dimension
  const int splits = max(1, min(64, 512 / B));
  dim3 wide(B, splits);
  k0_zero<<<B, 256, 0, stream>>>(ws);
  k1_max<<<wide, 256, 0, stream>>>(ws, lp, temps.data_ptr<float>(), V,
                                   splits);
  k2_hist0<<<wide, 256, 0, stream>>>(ws, lp, V, splits);
  k3_walk0<<<B, 256, 0, stream>>>(ws, top_k.data_ptr<int>(),
                                  top_p.data_ptr<float>(), V);
  k4_hist1<<<wide, 256, 0, stream>>>(ws, lp, V, splits);
  k5_walk1<<<B, 64, 0, stream>>>(ws, V);
  k6_draw<<<wide, 256, 0, stream>>>(
      ws, lp, temps.data_ptr<float>(),
      reinterpret_cast<unsigned long long*>(seed.data_ptr()), V, splits);
  k7_finalize<<<(B + 255) / 256, 256, 0, stream>>>(
      tokens.data_ptr<int>(), ws,
      reinterpret_cast<unsigned long long*>(seed.data_ptr()), B);
  HIP_CHECK_KERNEL();
}

}  // namespace kukeon
