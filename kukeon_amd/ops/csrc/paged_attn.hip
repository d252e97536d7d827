// Paged decode attention for gfx950 (MI355X).
//
// One workgroup (4 waves, 256 threads) per (sequence, kv-head, kv-split).
// The whole GQA group (G q-heads sharing one kv-head) is computed by the
// workgroup so each K/V block is read from HBM exactly once and shared via
// LDS — KV bandwidth is the decode cost, never multiply it by G.
//
// Work layout per 16-token KV block:
//   - 256 threads cooperatively stage the K block (16 tok x D) into LDS
//     (XOR-swizzled by token so the 4-lane-per-token dot reads are
//     bank-conflict-free), V block staged linear (its b32 read pattern is
//     conflict-free).
//   - wave w computes heads [w*GPW, (w+1)*GPW): lane l covers token l>>2,
//     dims (l&3)*32..+31; 4-lane shfl reduce yields per-token scores;
//     online softmax (m, l) per head per wave; V accumulated with 2 dims
//     per lane (64 lanes x 2 = D=128).
//
// Split-KV (flash-decode): grid.z partitions the KV blocks; partial
// (o_unnormalized f32, m, l) go to a workspace and paged_attn_reduce
// combines them — gives >> 256 workgroups at small batch so the chip fills.
#include "common.h"
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

namespace kukeon {

typedef __attribute__((__vector_size__(2 * sizeof(short)))) short bf16x2_t;

template <int D, int BS, int GPW, bool FP8>
__global__ __launch_bounds__(256) void paged_attn_kernel(
    unsigned short* __restrict__ out,     // [B, Hq*D] bf16 (splits==1)
    float* __restrict__ tmp_out,          // [B, Hq, S, D] f32 (splits>1)
    float* __restrict__ tmp_ml,           // [B, Hq, S, 2]
    const unsigned short* __restrict__ q, // [B, q_stride] (fused qkv row)
    const void* __restrict__ k_cache,     // [NB, Hk, BS, D] bf16|fp8
    const void* __restrict__ v_cache,
    const int* __restrict__ block_table,  // [B, max_blocks]
    const int* __restrict__ seq_lens,     // [B]
    long q_stride, int Hq, int Hk, int max_blocks, int num_splits,
    float scale) {
  constexpr int NW = 4;
  const int b = blockIdx.x;
  const int hk = blockIdx.y;
  const int split = blockIdx.z;
  const int G = Hq / Hk;
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int tok = lane >> 2;        // 0..15 token within block
  const int sub = lane & 3;         // 32-dim chunk
  const int ctx = seq_lens[b];
  if (ctx <= 0) return;
  const int nblocks = (ctx + BS - 1) / BS;
  const int per_split = (nblocks + num_splits - 1) / num_splits;
  const int blk_begin = split * per_split;
  const int blk_end = min(nblocks, blk_begin + per_split);

  __shared__ unsigned short kbuf[BS * D];
  __shared__ unsigned short vbuf[BS * D];

  // ---- load Q for this wave's heads (packed bf16 pairs for v_dot2;
  // scale is applied to the reduced score instead) ----
  unsigned int qpk[GPW][16];
  const int h0 = wid * GPW;
#pragma unroll
  for (int g = 0; g < GPW; ++g) {
    const int h = h0 + g;
    if (hk * G + h0 + g < Hq && h < G) {
      const unsigned short* qp =
          q + (long)b * q_stride + (long)(hk * G + h) * D + sub * 32;
#pragma unroll
      for (int c = 0; c < 4; ++c) {
        const uint4 v = *reinterpret_cast<const uint4*>(qp + c * 8);
        qpk[g][c * 4 + 0] = v.x;
        qpk[g][c * 4 + 1] = v.y;
        qpk[g][c * 4 + 2] = v.z;
        qpk[g][c * 4 + 3] = v.w;
      }
    }
  }
  const bool active = (h0 < G);

  float m[GPW], l[GPW], o0[GPW], o1[GPW];
#pragma unroll
  for (int g = 0; g < GPW; ++g) {
    m[g] = -INFINITY; l[g] = 0.f; o0[g] = 0.f; o1[g] = 0.f;
  }

  // stage geometry: each thread owns one 16B (bf16) / 8B (fp8) chunk
  const int stg_t = threadIdx.x / (D / 8);      // token 0..15
  const int stg_d0 = (threadIdx.x % (D / 8)) * 8;
  const int stg_dsw = (((stg_d0 / 32) ^ (stg_t & 3)) * 32) + (stg_d0 % 32);
  const unsigned short* kc16 =
      reinterpret_cast<const unsigned short*>(k_cache);
  const unsigned short* vc16 =
      reinterpret_cast<const unsigned short*>(v_cache);
  const unsigned char* kc8 = reinterpret_cast<const unsigned char*>(k_cache);
  const unsigned char* vc8 = reinterpret_cast<const unsigned char*>(v_cache);
  uint4 kreg, vreg;
  uint2 kreg8, vreg8;
  if (blk_begin < blk_end) {
    {  // prologue: fetch the first block into registers
      const int pb0 = block_table[(long)b * max_blocks + blk_begin];
      const long off = (((long)pb0 * Hk + hk) * BS + stg_t) * D + stg_d0;
      if (FP8) {
        kreg8 = *reinterpret_cast<const uint2*>(kc8 + off);
        vreg8 = *reinterpret_cast<const uint2*>(vc8 + off);
      } else {
        kreg = *reinterpret_cast<const uint4*>(kc16 + off);
        vreg = *reinterpret_cast<const uint4*>(vc16 + off);
      }
    }
    for (int bi = blk_begin; bi < blk_end; ++bi) {
      // ---- write the prefetched block (fp8 converts to the bf16 LDS
      // image here, so the compute path is dtype-independent), then issue
      // the next block's loads so their HBM latency hides under this
      // block's compute (guide T14 async-stage split) ----
      __syncthreads();
      if (FP8) {
        *reinterpret_cast<uint4*>(&kbuf[stg_t * D + stg_dsw]) =
            fp8x8_to_bf16x8(kreg8);
        *reinterpret_cast<uint4*>(&vbuf[stg_t * D + stg_d0]) =
            fp8x8_to_bf16x8(vreg8);
      } else {
        *reinterpret_cast<uint4*>(&kbuf[stg_t * D + stg_dsw]) = kreg;
        *reinterpret_cast<uint4*>(&vbuf[stg_t * D + stg_d0]) = vreg;
      }
      __syncthreads();
      if (bi + 1 < blk_end) {
        const int pbn = block_table[(long)b * max_blocks + bi + 1];
        const long off = (((long)pbn * Hk + hk) * BS + stg_t) * D + stg_d0;
        if (FP8) {
          kreg8 = *reinterpret_cast<const uint2*>(kc8 + off);
          vreg8 = *reinterpret_cast<const uint2*>(vc8 + off);
        } else {
          kreg = *reinterpret_cast<const uint4*>(kc16 + off);
          vreg = *reinterpret_cast<const uint4*>(vc16 + off);
        }
      }
      if (!active) continue;
      // ---- scores ----
      const int tok_global = bi * BS + tok;
      const bool valid = tok_global < ctx;
      float s[GPW];
      {
        // packed bf16 dot: one v_dot2_f32_bf16 per 2 elems (no unpack
        // VALU). FOUR independent partial accumulators per head: a single
        // running accumulator made this a 16-deep RAW chain of dependent
        // v_dot2 (~5 cyc each), and at the ~2 resident waves/SIMD this
        // kernel runs at, that latency is exposed as issue stalls
        // (SQ_WAIT_INST_ANY was 49% of wave cycles).
        const unsigned short* kp = &kbuf[tok * D + ((sub ^ (tok & 3)) * 32)];
#pragma unroll
        for (int g = 0; g < GPW; ++g) {
          float sj[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
          for (int c = 0; c < 4; ++c) {
            const uint4 kv = *reinterpret_cast<const uint4*>(kp + c * 8);
            const unsigned int kw[4] = {kv.x, kv.y, kv.z, kv.w};
#pragma unroll
            for (int j = 0; j < 4; ++j)
              sj[j] = __builtin_amdgcn_fdot2_f32_bf16(
                  __builtin_bit_cast(bf16x2_t, kw[j]),
                  __builtin_bit_cast(bf16x2_t, qpk[g][c * 4 + j]),
                  sj[j], false);
          }
          s[g] = (sj[0] + sj[1]) + (sj[2] + sj[3]);
        }
      }
#pragma unroll
      for (int g = 0; g < GPW; ++g) {
        s[g] += __shfl_xor(s[g], 1, WAVE);
        s[g] += __shfl_xor(s[g], 2, WAVE);
        s[g] *= scale;
        if (!valid) s[g] = -INFINITY;
        // every lane now has the score of its token (replicated x4);
        // reduce across the 16 token groups
        float bmax = s[g];
        bmax = fmaxf(bmax, __shfl_xor(bmax, 4, WAVE));
        bmax = fmaxf(bmax, __shfl_xor(bmax, 8, WAVE));
        bmax = fmaxf(bmax, __shfl_xor(bmax, 16, WAVE));
        bmax = fmaxf(bmax, __shfl_xor(bmax, 32, WAVE));
        const float mn = fmaxf(m[g], bmax);
        const float alpha = (m[g] == -INFINITY) ? 0.f : __expf(m[g] - mn);
        const float p = valid ? __expf(s[g] - mn) : 0.f;
        float psum = p;
        psum += __shfl_xor(psum, 4, WAVE);
        psum += __shfl_xor(psum, 8, WAVE);
        psum += __shfl_xor(psum, 16, WAVE);
        psum += __shfl_xor(psum, 32, WAVE);
        // the xor-reduce over offsets {4,8,16,32} keeps low lane bits fixed,
        // so each token group contributes exactly once
        l[g] = l[g] * alpha + psum;
        m[g] = mn;
        // ---- V accumulate: lane owns dims (2*lane, 2*lane+1). Block-
        // local partials in two independent chains per dim (a single
        // running o0/o1 was a 16-deep dependent FMA chain — same issue-
        // stall story as the score dot above). ----
        float va0 = 0.f, va1 = 0.f, vb0 = 0.f, vb1 = 0.f;
#pragma unroll
        for (int t2 = 0; t2 < BS; t2 += 2) {
          const float pa = __shfl(p, t2 * 4, WAVE);
          const float pb = __shfl(p, (t2 + 1) * 4, WAVE);
          const unsigned int v1 =
              *reinterpret_cast<const unsigned int*>(&vbuf[t2 * D + lane * 2]);
          const unsigned int v2 = *reinterpret_cast<const unsigned int*>(
              &vbuf[(t2 + 1) * D + lane * 2]);
          va0 = fmaf(pa, us2f((unsigned short)(v1 & 0xffffu)), va0);
          va1 = fmaf(pa, us2f((unsigned short)(v1 >> 16)), va1);
          vb0 = fmaf(pb, us2f((unsigned short)(v2 & 0xffffu)), vb0);
          vb1 = fmaf(pb, us2f((unsigned short)(v2 >> 16)), vb1);
        }
        o0[g] = fmaf(o0[g], alpha, va0 + vb0);
        o1[g] = fmaf(o1[g], alpha, va1 + vb1);
      }
    }
  }

  // ---- epilogue ----
  if (!active) return;
#pragma unroll
  for (int g = 0; g < GPW; ++g) {
    const int h = hk * G + h0 + g;
    if (h0 + g >= G) continue;
    if (num_splits == 1) {
      const float inv = l[g] > 0.f ? 1.f / l[g] : 0.f;
      unsigned short r[2] = {f2us(o0[g] * inv), f2us(o1[g] * inv)};
      *reinterpret_cast<unsigned int*>(out + (long)b * Hq * D + (long)h * D +
                                       lane * 2) =
          (unsigned int)r[0] | ((unsigned int)r[1] << 16);
    } else {
      float* top = tmp_out + (((long)b * Hq + h) * num_splits + split) * D;
      top[lane * 2] = o0[g];
      top[lane * 2 + 1] = o1[g];
      if (lane == 0) {
        float* ml = tmp_ml + (((long)b * Hq + h) * num_splits + split) * 2;
        ml[0] = m[g];
        ml[1] = l[g];
      }
    }
  }
}

// Combine split-KV partials: block per (b, h); D threads (128).
template <int D>
__global__ void paged_attn_reduce_kernel(
    unsigned short* __restrict__ out,  // [B, Hq*D]
    const float* __restrict__ tmp_out, // [B, Hq, S, D]
    const float* __restrict__ tmp_ml,  // [B, Hq, S, 2]
    const int* __restrict__ seq_lens, int Hq, int num_splits, int BS) {
  const int b = blockIdx.x;
  const int h = blockIdx.y;
  const int d = threadIdx.x;
  const int ctx = seq_lens[b];
  const int nblocks = (ctx + BS - 1) / BS;
  const int per_split = (nblocks + num_splits - 1) / num_splits;
  const int used = min(num_splits, (nblocks + per_split - 1) / per_split);
  const float* ml = tmp_ml + (((long)b * Hq + h) * num_splits) * 2;
  float M = -INFINITY;
  for (int s = 0; s < used; ++s) M = fmaxf(M, ml[s * 2]);
  float L = 0.f;
  for (int s = 0; s < used; ++s)
    L += (ml[s * 2] == -INFINITY ? 0.f : __expf(ml[s * 2] - M)) * ml[s * 2 + 1];
  const float* top = tmp_out + (((long)b * Hq + h) * num_splits) * D;
  float acc = 0.f;
  for (int s = 0; s < used; ++s) {
    const float w = ml[s * 2] == -INFINITY ? 0.f : __expf(ml[s * 2] - M);
    acc += w * top[(long)s * D + d];
  }
  out[(long)b * Hq * D + (long)h * D + d] = f2us(L > 0.f ? acc / L : 0.f);
}

void paged_attention(torch::Tensor out, torch::Tensor q, torch::Tensor k_cache,
                     torch::Tensor v_cache, torch::Tensor block_table,
                     torch::Tensor seq_lens, int64_t q_offset,
                     int64_t num_splits, double scale, torch::Tensor tmp_out,
                     torch::Tensor tmp_ml) {
  const int B = q.size(0);
  if (B == 0) return;
  const int Hk = k_cache.size(1);
  const int BS = k_cache.size(2);
  const int D = k_cache.size(3);
  const int Hq = out.size(1) / D;
  const int max_blocks = block_table.size(1);
  const int G = Hq / Hk;
  TORCH_CHECK(D == 128, "only head_dim=128 supported (Llama/Mixtral family)");
  TORCH_CHECK(BS == 16, "kv block size must be 16");
  TORCH_CHECK(block_table.scalar_type() == torch::kInt32);
  TORCH_CHECK(seq_lens.scalar_type() == torch::kInt32);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  const long q_stride = q.stride(0);
  const unsigned short* qp =
      reinterpret_cast<const unsigned short*>(q.data_ptr()) + q_offset;
  dim3 grid(B, Hk, (unsigned)num_splits);
  float* tov = num_splits > 1 ? tmp_out.data_ptr<float>() : nullptr;
  float* tml = num_splits > 1 ? tmp_ml.data_ptr<float>() : nullptr;
  auto* op = reinterpret_cast<unsigned short*>(out.data_ptr());

  const bool fp8 = k_cache.scalar_type() == torch::kUInt8;
#define PA_LAUNCH(GPW, FP8_)                                                  \
  paged_attn_kernel<128, 16, GPW, FP8_><<<grid, 256, 0, stream>>>(            \
      op, tov, tml, qp, k_cache.data_ptr(), v_cache.data_ptr(),               \
      block_table.data_ptr<int>(), seq_lens.data_ptr<int>(), q_stride, Hq,    \
      Hk, max_blocks, (int)num_splits, (float)scale)
  if (G <= 4) {
    if (fp8) { PA_LAUNCH(1, true); } else { PA_LAUNCH(1, false); }
  } else if (G == 8) {
    if (fp8) { PA_LAUNCH(2, true); } else { PA_LAUNCH(2, false); }
  } else {
    TORCH_CHECK(false, "unsupported GQA group size ", G);
  }
#undef PA_LAUNCH
  HIP_CHECK_KERNEL();
  if (num_splits > 1) {
    dim3 rgrid(B, Hq);
    paged_attn_reduce_kernel<128><<<rgrid, 128, 0, stream>>>(
        op, tmp_out.data_ptr<float>(), tmp_ml.data_ptr<float>(),
        seq_lens.data_ptr<int>(), Hq, (int)num_splits, BS);
    HIP_CHECK_KERNEL();
  }
}

}  // namespace kukeon
