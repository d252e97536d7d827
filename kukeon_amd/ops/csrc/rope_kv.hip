// Fused rotary embedding (NEOX / Llama half-rotation style) + paged-KV-cache
// append for gfx950.
//
// Operates in place on the fused qkv projection output [T, (Hq+2*Hk)*D]
// (avoids materializing contiguous q/k/v copies), rotates q and k, and
// scatters the rotated k and raw v into the paged caches
// k_cache/v_cache [num_blocks, Hk, BLOCK, D].
//
// cos/sin are host-precomputed per guide Appendix B (trig on device turns
// memory-bound into VALU-bound): table [max_pos, D] f32 with cos in [0, D/2)
// and sin in [D/2, D).
#include "common.h"
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

namespace kukeon {

template <bool FP8>
__global__ void rope_kv_kernel(
    unsigned short* __restrict__ qkv,        // [T, (Hq+2Hk)*D]
    void* __restrict__ k_cache,              // [NB, Hk, BS, D] bf16|fp8
    void* __restrict__ v_cache,
    const float* __restrict__ cos_sin,       // [max_pos, D]
    const int* __restrict__ positions,       // [T]
    const int* __restrict__ slot_mapping,    // [T] flat slot = block*BS + off
    const int* __restrict__ block_table,     // [T, max_blocks] decode mode:
    int max_blocks,                          //   slot derived on device
    int Hq, int Hk, int D, int BS) {
  const long t = blockIdx.x;
  // grid.y splits the per-token chunk loops so small decode batches still
  // put >=256 workgroups on the chip
  const int yoff = blockIdx.y * blockDim.x + threadIdx.x;
  const int ystride = gridDim.y * blockDim.x;
  const int half = D / 2;
  const long row_stride = (long)(Hq + 2 * Hk) * D;
  unsigned short* row = qkv + t * row_stride;
  const int pos = positions[t];
  if (pos < 0) return;  // inactive decode row: no rotate, no KV write
  int slot;
  if (block_table != nullptr) {
    // decode: the device owns the sequence cursor (self-advancing graph)
    slot = block_table[t * (long)max_blocks + pos / BS] * BS + pos % BS;
  } else {
    slot = slot_mapping[t];
  }
  const float* cs = cos_sin + (long)pos * D;

  // ---- rotate q (in place) ----
  const int q_chunks = Hq * (D / 16);  // 8 pairs per chunk
  for (int c = yoff; c < q_chunks; c += ystride) {
    const int h = c / (D / 16);
    const int d0 = (c % (D / 16)) * 8;
    unsigned short* base = row + (long)h * D;
    bf16x8 x = load_bf16x8(base + d0);
    bf16x8 y = load_bf16x8(base + half + d0);
    float xo[8], yo[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float co = cs[d0 + j], si = cs[half + d0 + j];
      xo[j] = x.f(j) * co - y.f(j) * si;
      yo[j] = y.f(j) * co + x.f(j) * si;
    }
    *reinterpret_cast<uint4*>(base + d0) = pack_bf16x8(xo);
    *reinterpret_cast<uint4*>(base + half + d0) = pack_bf16x8(yo);
  }

  // ---- rotate k (in place) + append to k_cache ----
  const long cache_tok_base =
      slot >= 0 ? ((long)(slot / BS) * Hk * BS + (long)(slot % BS)) * D : 0;
  const int k_chunks = Hk * (D / 16);
  for (int c = yoff; c < k_chunks; c += ystride) {
    const int h = c / (D / 16);
    const int d0 = (c % (D / 16)) * 8;
    unsigned short* base = row + (long)(Hq + h) * D;
    bf16x8 x = load_bf16x8(base + d0);
    bf16x8 y = load_bf16x8(base + half + d0);
    float xo[8], yo[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float co = cs[d0 + j], si = cs[half + d0 + j];
      xo[j] = x.f(j) * co - y.f(j) * si;
      yo[j] = y.f(j) * co + x.f(j) * si;
    }
    uint4 px = pack_bf16x8(xo), py = pack_bf16x8(yo);
    *reinterpret_cast<uint4*>(base + d0) = px;
    *reinterpret_cast<uint4*>(base + half + d0) = py;
    if (slot >= 0) {
      const long off = cache_tok_base + (long)h * BS * D;
      if (FP8) {
        unsigned char* kc = reinterpret_cast<unsigned char*>(k_cache) + off;
        *reinterpret_cast<uint2*>(kc + d0) = pack_fp8x8(xo);
        *reinterpret_cast<uint2*>(kc + half + d0) = pack_fp8x8(yo);
      } else {
        unsigned short* kc = reinterpret_cast<unsigned short*>(k_cache) + off;
        *reinterpret_cast<uint4*>(kc + d0) = px;
        *reinterpret_cast<uint4*>(kc + half + d0) = py;
      }
    }
  }

  // ---- copy v to v_cache ----
  if (slot >= 0) {
    const int v_chunks = Hk * (D / 8);
    for (int c = yoff; c < v_chunks; c += ystride) {
      const int h = c / (D / 8);
      const int d0 = (c % (D / 8)) * 8;
      const unsigned short* base = row + (long)(Hq + Hk + h) * D;
      const long off = cache_tok_base + (long)h * BS * D;
      if (FP8) {
        bf16x8 v = load_bf16x8(base + d0);
        float f[8];
#pragma unroll
        for (int j = 0; j < 8; ++j) f[j] = v.f(j);
        unsigned char* vc = reinterpret_cast<unsigned char*>(v_cache) + off;
        *reinterpret_cast<uint2*>(vc + d0) = pack_fp8x8(f);
      } else {
        unsigned short* vc = reinterpret_cast<unsigned short*>(v_cache) + off;
        *reinterpret_cast<uint4*>(vc + d0) =
            *reinterpret_cast<const uint4*>(base + d0);
      }
    }
  }
}

void rope_kv_append(torch::Tensor qkv, torch::Tensor k_cache,
                    torch::Tensor v_cache, torch::Tensor cos_sin,
                    torch::Tensor positions, torch::Tensor slot_mapping,
                    int64_t num_q_heads, int64_t num_kv_heads,
                    int64_t head_dim, torch::Tensor block_table) {
  TORCH_CHECK(qkv.is_contiguous() && qkv.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(cos_sin.scalar_type() == torch::kFloat32);
  TORCH_CHECK(positions.scalar_type() == torch::kInt32);
  TORCH_CHECK(slot_mapping.scalar_type() == torch::kInt32);
  const long T = qkv.size(0);
  if (T == 0) return;
  const int D = (int)head_dim;
  TORCH_CHECK(D % 16 == 0, "head_dim must be a multiple of 16");
  TORCH_CHECK(qkv.size(1) == (num_q_heads + 2 * num_kv_heads) * D);
  const int BS = (int)k_cache.size(2);
  TORCH_CHECK(k_cache.size(1) == num_kv_heads && k_cache.size(3) == D);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  const bool table_mode = block_table.dim() == 2;
  const int* bt = table_mode ? block_table.data_ptr<int>() : nullptr;
  const int mb = table_mode ? (int)block_table.size(1) : 0;
  const bool fp8 = k_cache.scalar_type() == torch::kUInt8;
  const unsigned ysplit = T <= 128 ? 4 : 1;
#define RK_LAUNCH(FP8_)                                                         rope_kv_kernel<FP8_><<<dim3((unsigned)T, ysplit), 256, 0, stream>>>(              reinterpret_cast<unsigned short*>(qkv.data_ptr()), k_cache.data_ptr(),        v_cache.data_ptr(), cos_sin.data_ptr<float>(),                                positions.data_ptr<int>(), slot_mapping.data_ptr<int>(), bt, mb,              (int)num_q_heads, (int)num_kv_heads, D, BS)
  if (fp8) { RK_LAUNCH(true); } else { RK_LAUNCH(false); }
#undef RK_LAUNCH
  HIP_CHECK_KERNEL();
}

}  // namespace kukeon
