// Paged decode attention for gfx950 (MI355X).
//
// TWO kernels share this file: paged_attn_mfma16_kernel (the default —
// matrix-core scores/PV, wave-autonomous split slots, 86% of the HBM
// roofline; see its header further down) and the v_dot2 kernel described
// below (the fallback for GQA groups > 16 or split counts not divisible
// by 4, and the reference point the MFMA design was measured against).
//
// v_dot2 kernel: one workgroup (4 waves, 256 threads) per
// (sequence, kv-head, kv-split).
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
typedef __attribute__((__vector_size__(8 * sizeof(short)))) short bf16x8_v;
typedef __attribute__((__vector_size__(16 * sizeof(float)))) float f32x16_t;

struct u4s_ { unsigned int x[4]; };
DEV_INLINE bf16x8_v as_frag32(unsigned int w0, unsigned int w1,
                              unsigned int w2, unsigned int w3) {
  u4s_ v{{w0, w1, w2, w3}};
  return __builtin_bit_cast(bf16x8_v, v);
}
DEV_INLINE unsigned int cvt_pk_bf16d(float lo, float hi) {
  __hip_bfloat162 h = __float22bfloat162_rn(float2{lo, hi});
  unsigned int r;
  __builtin_memcpy(&r, &h, 4);
  return r;
}

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

// ===================================================================
// MFMA decode attention (wave-autonomous split-KV).
//
// The v_dot2 kernel above is issue-stall bound (PMC: SQ_WAIT_INST_ANY 49%
// of wave cycles — ~28 VALU instructions per KV token in dependent
// chains). This variant moves QK^T and PV onto the matrix cores using the
// prefill kernel's fragment machinery (swapped S^T = K.Q^T so softmax is
// lane-local; P via cvt_pk + permlane32_swap; PV as O^T = V^T.P^T with a
// transposed V image), at ~5 wave-instructions per token.
//
// Decode-specific structure: the 4 waves of a workgroup are fully
// AUTONOMOUS — each owns one of 4 consecutive tmp slots (the engine's
// split count is a multiple of 4) and walks its own KV-block range,
// staging 32-token tiles into its private LDS quarter. No barriers at
// all: while one wave waits on its stage, the other three compute, so
// stage latency hides at the CU level. All S slots merge in one reduce
// pass (every slot is written each call — empty ranges write m=-inf).
//
// The q-head dimension (G <= 32) rides the MFMA N axis; padding columns
// are garbage but PER-LANE, so nothing crosses into real heads and only
// cols < G are written back.
template <bool FP8>
__global__ __launch_bounds__(256, 2) void paged_attn_mfma_kernel(
    float* __restrict__ tmp_out,          // [B, Hq, S, D] f32
    float* __restrict__ tmp_ml,           // [B, Hq, S, 2]
    const unsigned short* __restrict__ q, // [B, q_stride]
    const void* __restrict__ k_cache,     // [NB, Hk, 16, D]
    const void* __restrict__ v_cache,
    const int* __restrict__ block_table,  // [B, max_blocks]
    const int* __restrict__ seq_lens,     // [B]
    long q_stride, int Hq, int Hk, int max_blocks, int S, float scale) {
  constexpr int D = 128;
  constexpr int BS = 16;
  constexpr int KVT = 32;
  const int b = blockIdx.x;
  const int hk = blockIdx.y;
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int hi = lane >> 5;
  const int qcol = lane & 31;
  const int G = Hq / Hk;
  const int slot = blockIdx.z * 4 + wid;
  const int ctx = seq_lens[b];
  if (ctx <= 0 || slot >= S) return;

  // per-wave private LDS quarters — no cross-wave sharing, no barriers
  __shared__ __align__(16) unsigned short kbuf[4][KVT * D];
  __shared__ __align__(16) unsigned short vtbuf[4][D * KVT];
  unsigned short* kb = kbuf[wid];
  unsigned short* vt = vtbuf[wid];

  const int nblocks = (ctx + BS - 1) / BS;
  const int per_slot = (nblocks + S - 1) / S;
  const int blk_begin = slot * per_slot;
  const int blk_end = min(nblocks, blk_begin + per_slot);

  // Q row pointer: fragments are re-read per tile from L2 (the row is
  // hot — every workgroup of this sequence reads it) instead of pinned
  // in 32 registers for the kernel's lifetime; this keeps the wave under
  // the 256-VGPR boundary for 2 waves/SIMD.
  const int qh = min(qcol, G - 1);
  const unsigned short* qp =
      q + (long)b * q_stride + (long)(hk * G + qh) * D;

  float m = -INFINITY, l_acc = 0.f;
  f32x16_t oacc[4] = {{}, {}, {}, {}};

  const unsigned short* kc16 =
      reinterpret_cast<const unsigned short*>(k_cache);
  const unsigned short* vc16 =
      reinterpret_cast<const unsigned short*>(v_cache);
  const unsigned char* kc8 = reinterpret_cast<const unsigned char*>(k_cache);
  const unsigned char* vc8 = reinterpret_cast<const unsigned char*>(v_cache);
  const int kv_hi = min(ctx, blk_end * BS);

  // per-lane staging geometry (fixed across tiles)
  const int stg_tok = lane >> 4;            // +4 per pass
  const int stg_d0 = (lane & 15) * 8;

  // T14 within the wave: tile bi's K is PREFETCHED into registers during
  // tile bi-1's compute; V loads issue right after the K LDS write and
  // fly under QK^T + softmax. Without this the wave serializes a full
  // HBM fetch per tile (measured 2.8x slower than the v_dot2 kernel).
  uint4 kpre[8];
  auto kfetch = [&](int bi_f, uint4* dst) {
#pragma unroll
    for (int pass = 0; pass < 8; ++pass) {
      const int tok = pass * 4 + stg_tok;
      const int kvpos = bi_f * BS + tok;
      const int bidx = min(kvpos / BS, nblocks - 1);
      const int pblk = block_table[(long)b * max_blocks + bidx];
      const long koff =
          (((long)pblk * Hk + hk) * BS + (kvpos % BS)) * D + stg_d0;
      if (FP8) {
        dst[pass] =
            fp8x8_to_bf16x8(*reinterpret_cast<const uint2*>(kc8 + koff));
      } else {
        dst[pass] = *reinterpret_cast<const uint4*>(kc16 + koff);
      }
    }
  };
  if (blk_begin < blk_end) kfetch(blk_begin, kpre);

  for (int bi = blk_begin; bi < blk_end; bi += 2) {
    const int kvbase = bi * BS;
    // ---- write the prefetched K tile (swizzled image) ----
#pragma unroll
    for (int pass = 0; pass < 8; ++pass) {
      const int tok = pass * 4 + stg_tok;
      const int byte = tok * 256 + ((stg_d0 * 2) ^ ((tok & 15) << 4));
      *reinterpret_cast<uint4*>(reinterpret_cast<char*>(kb) + byte) =
          kpre[pass];
    }
    // ---- issue next tile's K prefetch + this tile's V loads ----
    if (bi + 2 < blk_end) kfetch(bi + 2, kpre);
    uint4 vr[4][2];
#pragma unroll
    for (int pass = 0; pass < 4; ++pass) {
      const int kp = pass * 4 + stg_tok;              // kv pair 0..15
      const int vpos0 = kvbase + 2 * kp;
      const int vpos1 = vpos0 + 1;
      const int vb0 = min(vpos0 / BS, nblocks - 1);
      const int vb1 = min(vpos1 / BS, nblocks - 1);
      const long voff0 =
          (((long)block_table[(long)b * max_blocks + vb0] * Hk + hk) * BS +
           (vpos0 % BS)) * D + stg_d0;
      const long voff1 =
          (((long)block_table[(long)b * max_blocks + vb1] * Hk + hk) * BS +
           (vpos1 % BS)) * D + stg_d0;
      if (FP8) {
        vr[pass][0] =
            fp8x8_to_bf16x8(*reinterpret_cast<const uint2*>(vc8 + voff0));
        vr[pass][1] =
            fp8x8_to_bf16x8(*reinterpret_cast<const uint2*>(vc8 + voff1));
      } else {
        vr[pass][0] = *reinterpret_cast<const uint4*>(vc16 + voff0);
        vr[pass][1] = *reinterpret_cast<const uint4*>(vc16 + voff1);
      }
    }

    // ---- S^T = K . Q^T ----
    f32x16_t sacc = {};
#pragma unroll
    for (int st = 0; st < 8; ++st) {
      const int row = qcol;  // A row = kv token in tile
      const int byte =
          row * 256 + (((st * 16 + hi * 8) * 2) ^ ((row & 15) << 4));
      const uint4 kf = *reinterpret_cast<const uint4*>(
          reinterpret_cast<const char*>(kb) + byte);
      const uint4 qv = *reinterpret_cast<const uint4*>(qp + st * 16 + hi * 8);
      sacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
          as_frag32(kf.x, kf.y, kf.z, kf.w),
          as_frag32(qv.x, qv.y, qv.z, qv.w),
          sacc, 0, 0, 0);
    }

    // ---- online softmax over this lane's 16 kv rows (lane-local) ----
    float sv[16];
    float tmax = -INFINITY;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int kvrow = (r & 3) + 8 * (r >> 2) + 4 * hi;
      const bool valid = (kvbase + kvrow) < kv_hi;
      sv[r] = valid ? sacc[r] * scale : -INFINITY;
      tmax = fmaxf(tmax, sv[r]);
    }
    tmax = fmaxf(tmax, __shfl_xor(tmax, 32, WAVE));
    const float mn = fmaxf(m, tmax);
    const float alpha = (m == -INFINITY) ? 0.f : __expf(m - mn);
    m = mn;
    float psum = 0.f;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      sv[r] = (sv[r] == -INFINITY) ? 0.f : __expf(sv[r] - mn);
      psum += sv[r];
    }
    psum += __shfl_xor(psum, 32, WAVE);
    l_acc = l_acc * alpha + psum;
#pragma unroll
    for (int db = 0; db < 4; ++db)
#pragma unroll
      for (int r = 0; r < 16; ++r) oacc[db][r] *= alpha;

    // ---- write the V^T image (loads issued before QK have landed) ----
#pragma unroll
    for (int pass = 0; pass < 4; ++pass) {
      const int kp = pass * 4 + stg_tok;
      const int vpos0 = kvbase + 2 * kp;
      const int vpos1 = vpos0 + 1;
      uint4 r0 = vr[pass][0];
      uint4 r1 = vr[pass][1];
      if (vpos0 >= kv_hi) r0 = uint4{0, 0, 0, 0};
      if (vpos1 >= kv_hi) r1 = uint4{0, 0, 0, 0};
      const unsigned int* a0 = reinterpret_cast<const unsigned int*>(&r0);
      const unsigned int* a1 = reinterpret_cast<const unsigned int*>(&r1);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const unsigned short e0 = (a0[j >> 1] >> ((j & 1) * 16)) & 0xffffu;
        const unsigned short e1 = (a1[j >> 1] >> ((j & 1) * 16)) & 0xffffu;
        const unsigned int packed = (unsigned)e0 | ((unsigned)e1 << 16);
        const int d = stg_d0 + j;
        const int byte = d * 64 + ((4 * kp) ^ ((d & 3) << 4));
        *reinterpret_cast<unsigned int*>(
            reinterpret_cast<char*>(vt) + byte) = packed;
      }
    }

    // ---- P -> bf16 B-fragments via cvt_pk + permlane32_swap ----
    unsigned int cp[8];
#pragma unroll
    for (int j = 0; j < 8; ++j)
      cp[j] = cvt_pk_bf16d(sv[2 * j], sv[2 * j + 1]);
    unsigned int pb[2][4];
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      const int b0 = 4 * c;
      auto s1 = __builtin_amdgcn_permlane32_swap(cp[b0 + 0], cp[b0 + 2],
                                                 false, false);
      auto s2 = __builtin_amdgcn_permlane32_swap(cp[b0 + 1], cp[b0 + 3],
                                                 false, false);
      pb[c][0] = s1[0]; pb[c][1] = s2[0];
      pb[c][2] = s1[1]; pb[c][3] = s2[1];
    }

    // ---- O^T += V^T . P^T ----
#pragma unroll
    for (int db = 0; db < 4; ++db) {
      const int d = db * 32 + qcol;
#pragma unroll
      for (int c = 0; c < 2; ++c) {
        const int byte = d * 64 + (((c * 32) + hi * 16) ^ ((d & 3) << 4));
        const uint4 vf = *reinterpret_cast<const uint4*>(
            reinterpret_cast<const char*>(vt) + byte);
        oacc[db] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            as_frag32(vf.x, vf.y, vf.z, vf.w),
            as_frag32(pb[c][0], pb[c][1], pb[c][2], pb[c][3]),
            oacc[db], 0, 0, 0);
      }
    }
  }

  // ---- epilogue: unnormalized O + (m, l) into this wave's slot ----
  if (qcol >= G) return;
  const int head = hk * G + qcol;
  float* top = tmp_out + (((long)b * Hq + head) * S + slot) * D;
#pragma unroll
  for (int db = 0; db < 4; ++db) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int d = db * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
      top[d] = oacc[db][r];
    }
  }
  if (hi == 0) {
    float* ml = tmp_ml + (((long)b * Hq + head) * S + slot) * 2;
    ml[0] = m;
    ml[1] = l_acc;
  }
}

// ---- 16x16x16 (K=16) bf16 MFMA layout probe: C[16,16]=A[16,16]@B[16,16]
// assumed maps (validated on-device by test_mfma_probe16k):
//   A: lane -> row = l&15, k = 4*(l>>4)+j (j=0..3, bf16x4)
//   B: lane -> col = l&15, k = 4*(l>>4)+j
//   C: lane -> col = l&15, row = 4*(l>>4)+r
typedef __attribute__((__vector_size__(4 * sizeof(short)))) short bf16x4_v;
typedef __attribute__((__vector_size__(4 * sizeof(float)))) float f32x4_v;
struct u2s_ { unsigned int x[2]; };
DEV_INLINE bf16x4_v as_frag16(unsigned int w0, unsigned int w1) {
  u2s_ v{{w0, w1}};
  return __builtin_bit_cast(bf16x4_v, v);
}

__global__ void mfma_probe16k_kernel(float* __restrict__ c,
                                     const unsigned short* __restrict__ a,
                                     const unsigned short* __restrict__ b) {
  const int lane = threadIdx.x & 63;
  const int g = lane >> 4;
  const int rc = lane & 15;
  unsigned int ar[2], br[2];
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    const int k0 = 4 * g + 2 * i;
    ar[i] = (unsigned)a[rc * 16 + k0] | ((unsigned)a[rc * 16 + k0 + 1] << 16);
    br[i] = (unsigned)b[k0 * 16 + rc] | ((unsigned)b[(k0 + 1) * 16 + rc] << 16);
  }
  f32x4_v acc = {};
  acc = __builtin_amdgcn_mfma_f32_16x16x16bf16_1k(
      as_frag16(ar[0], ar[1]), as_frag16(br[0], br[1]), acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r) c[(4 * g + r) * 16 + rc] = acc[r];
}

void mfma_probe16k(torch::Tensor out, torch::Tensor a, torch::Tensor b) {
  TORCH_CHECK(a.sizes() == torch::IntArrayRef({16, 16}));
  TORCH_CHECK(b.sizes() == torch::IntArrayRef({16, 16}));
  TORCH_CHECK(out.sizes() == torch::IntArrayRef({16, 16}));
  auto stream = c10::hip::getCurrentHIPStream().stream();
  mfma_probe16k_kernel<<<1, 64, 0, stream>>>(
      out.data_ptr<float>(),
      reinterpret_cast<const unsigned short*>(a.data_ptr()),
      reinterpret_cast<const unsigned short*>(b.data_ptr()));
  HIP_CHECK_KERNEL();
}

// ===================================================================
// 16-token-tile MFMA decode attention: the residency-first variant.
//
// Same wave-autonomous split-slot scheme as paged_attn_mfma_kernel, but
// one cache block (16 tokens) per tile on 16x16 MFMAs:
//   - S^T[tok, head] via 16x16x32 (K = head_dim chunks),
//   - PV O^T[d, head] via 16x16x16bf16_1k (K = the 16 tokens) — and in
//     this swapped shape the C layout of the scores IS the B layout of
//     P (row=k=4*(l>>4)+idx, col=head), so P needs NO cross-lane
//     redistribution at all, just two v_cvt_pk_bf16_f32.
//   - 8 KiB LDS per wave (32 KiB/WG) -> ~5 workgroups/CU, matching the
//     v_dot2 kernel's residency while issuing ~5x fewer instructions.
// G <= 16 (8B: 4, 70B: 8); q-head padding is per-lane garbage, never
// written back.
template <bool FP8>
// fp8 needs headroom for the cvt_pk conversions (at 4 waves/SIMD the
// 128-reg bound put a 2-VGPR spill in the hot loop)
__global__ __launch_bounds__(256, FP8 ? 3 : 4) void paged_attn_mfma16_kernel(
    float* __restrict__ tmp_out,          // [B, Hq, S, D] f32
    float* __restrict__ tmp_ml,           // [B, Hq, S, 2]
    const unsigned short* __restrict__ q, // [B, q_stride]
    const void* __restrict__ k_cache,     // [NB, Hk, 16, D]
    const void* __restrict__ v_cache,
    const int* __restrict__ block_table,  // [B, max_blocks]
    const int* __restrict__ seq_lens,     // [B]
    long q_stride, int Hq, int Hk, int max_blocks, int S, float scale) {
  constexpr int D = 128;
  constexpr int BS = 16;
  const int b = blockIdx.x;
  const int hk = blockIdx.y;
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int g4 = lane >> 4;       // 0..3
  const int rc = lane & 15;
  const int G = Hq / Hk;
  const int slot = blockIdx.z * 4 + wid;
  const int ctx = seq_lens[b];
  if (ctx <= 0 || slot >= S) return;

  // V^T images only — K is consumed straight from HBM in fragment shape
  // (16 B/lane across 64 B-line groups; the LDS round trip was pure
  // added latency here: per-wave parked time was 67% of cycles). Two
  // V^T buffers per wave: tile bi's image is written during tile bi-1,
  // so PV never waits on fresh LDS stores.
  __shared__ __align__(16) unsigned short vtbuf[4][2][D * BS];

  const int nblocks = (ctx + BS - 1) / BS;
  const int per_slot = (nblocks + S - 1) / S;
  const int blk_begin = slot * per_slot;
  const int blk_end = min(nblocks, blk_begin + per_slot);
  const int kv_hi = min(ctx, blk_end * BS);

  const int qh = min(rc, G - 1);  // B col = q head
  // Q fragments pinned in 16 registers (L2 re-reads in the tile loop
  // measured as 4 dependent ~250-cycle stalls per tile)
  unsigned int qf[4][4];
  {
    const unsigned short* qp =
        q + (long)b * q_stride + (long)(hk * G + qh) * D;
#pragma unroll
    for (int st = 0; st < 4; ++st) {
      const uint4 v = *reinterpret_cast<const uint4*>(qp + st * 32 + g4 * 8);
      qf[st][0] = v.x; qf[st][1] = v.y; qf[st][2] = v.z; qf[st][3] = v.w;
    }
  }

  float m = -INFINITY, l_acc = 0.f;
  f32x4_v oacc[8] = {{}, {}, {}, {}, {}, {}, {}, {}};

  const unsigned short* kc16 =
      reinterpret_cast<const unsigned short*>(k_cache);
  const unsigned short* vc16 =
      reinterpret_cast<const unsigned short*>(v_cache);
  const unsigned char* kc8 = reinterpret_cast<const unsigned char*>(k_cache);
  const unsigned char* vc8 = reinterpret_cast<const unsigned char*>(v_cache);

  // K in MFMA A-fragment shape: lane (rc = token, g4 = k-group) reads
  // 16 B at tok*256B + st*64B + g4*16B — lanes sharing a token cover one
  // 64 B line, so the four per-tile loads coalesce at line granularity.
  uint4 kfr[4];
  auto kfetch = [&](int bi_f) {
    const long base =
        ((long)block_table[(long)b * max_blocks + bi_f] * Hk + hk) * BS * D;
    const long off = base + (long)rc * D + g4 * 8;
#pragma unroll
    for (int st = 0; st < 4; ++st) {
      if (FP8) {
        kfr[st] = fp8x8_to_bf16x8(
            *reinterpret_cast<const uint2*>(kc8 + off + st * 32));
      } else {
        kfr[st] = *reinterpret_cast<const uint4*>(kc16 + off + st * 32);
      }
    }
  };
  const int stg_d0 = rc * 8;
  uint4 vpre[2][2];
  auto vfetch = [&](int bi_f) {
    const long base =
        ((long)block_table[(long)b * max_blocks + bi_f] * Hk + hk) * BS * D;
#pragma unroll
    for (int pass = 0; pass < 2; ++pass) {
      const int kp = pass * 4 + g4;
      const long voff0 = base + (long)(2 * kp) * D + stg_d0;
      const long voff1 = base + (long)(2 * kp + 1) * D + stg_d0;
      if (FP8) {
        vpre[pass][0] =
            fp8x8_to_bf16x8(*reinterpret_cast<const uint2*>(vc8 + voff0));
        vpre[pass][1] =
            fp8x8_to_bf16x8(*reinterpret_cast<const uint2*>(vc8 + voff1));
      } else {
        vpre[pass][0] = *reinterpret_cast<const uint4*>(vc16 + voff0);
        vpre[pass][1] = *reinterpret_cast<const uint4*>(vc16 + voff1);
      }
    }
  };
  auto vwrite = [&](unsigned short* vt, int kvbase) {
#pragma unroll
    for (int pass = 0; pass < 2; ++pass) {
      const int kp = pass * 4 + g4;
      uint4 r0 = vpre[pass][0];
      uint4 r1 = vpre[pass][1];
      if (kvbase + 2 * kp >= kv_hi) r0 = uint4{0, 0, 0, 0};
      if (kvbase + 2 * kp + 1 >= kv_hi) r1 = uint4{0, 0, 0, 0};
      const unsigned int* a0 = reinterpret_cast<const unsigned int*>(&r0);
      const unsigned int* a1 = reinterpret_cast<const unsigned int*>(&r1);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const unsigned short e0 = (a0[j >> 1] >> ((j & 1) * 16)) & 0xffffu;
        const unsigned short e1 = (a1[j >> 1] >> ((j & 1) * 16)) & 0xffffu;
        const unsigned int packed = (unsigned)e0 | ((unsigned)e1 << 16);
        const int d = stg_d0 + j;
        const int byte = d * 32 + ((4 * kp) ^ ((d & 3) << 3));
        *reinterpret_cast<unsigned int*>(
            reinterpret_cast<char*>(vt) + byte) = packed;
      }
    }
  };

  if (blk_begin < blk_end) {
    kfetch(blk_begin);
    vfetch(blk_begin);
    vwrite(vtbuf[wid][blk_begin & 1], blk_begin * BS);
  }

  for (int bi = blk_begin; bi < blk_end; ++bi) {
    const int kvbase = bi * BS;
    const unsigned short* vt = vtbuf[wid][bi & 1];

    // ---- S^T = K . Q^T from the in-flight K fragments ----
    f32x4_v sacc = {};
#pragma unroll
    for (int st = 0; st < 4; ++st) {
      sacc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          as_frag32(kfr[st].x, kfr[st].y, kfr[st].z, kfr[st].w),
          as_frag32(qf[st][0], qf[st][1], qf[st][2], qf[st][3]),
          sacc, 0, 0, 0);
    }
    // K registers are free again: issue the next tile's fragment loads
    // (they fly under softmax + V staging + PV + the loop turnaround)
    if (bi + 1 < blk_end) kfetch(bi + 1);

    // ---- online softmax (lane holds tokens 4*g4..+3 of its head) ----
    float sv[4];
    float tmax = -INFINITY;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const bool valid = (kvbase + 4 * g4 + r) < kv_hi;
      sv[r] = valid ? sacc[r] * scale : -INFINITY;
      tmax = fmaxf(tmax, sv[r]);
    }
    tmax = fmaxf(tmax, __shfl_xor(tmax, 16, WAVE));
    tmax = fmaxf(tmax, __shfl_xor(tmax, 32, WAVE));
    const float mn = fmaxf(m, tmax);
    const float alpha = (m == -INFINITY) ? 0.f : __expf(m - mn);
    m = mn;
    float psum = 0.f;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      sv[r] = (sv[r] == -INFINITY) ? 0.f : __expf(sv[r] - mn);
      psum += sv[r];
    }
    psum += __shfl_xor(psum, 16, WAVE);
    psum += __shfl_xor(psum, 32, WAVE);
    l_acc = l_acc * alpha + psum;
#pragma unroll
    for (int dc = 0; dc < 8; ++dc)
#pragma unroll
      for (int r = 0; r < 4; ++r) oacc[dc][r] *= alpha;

    // ---- stage NEXT tile's V^T into the other buffer ----
    if (bi + 1 < blk_end) {
      vfetch(bi + 1);
      vwrite(vtbuf[wid][(bi + 1) & 1], (bi + 1) * BS);
    }

    // ---- PV: P is already a 16x16x16 B fragment ----
    const bf16x4_v pfrag = as_frag16(cvt_pk_bf16d(sv[0], sv[1]),
                                     cvt_pk_bf16d(sv[2], sv[3]));
#pragma unroll
    for (int dc = 0; dc < 8; ++dc) {
      const int d = dc * 16 + rc;  // A row = output dim
      const int byte = d * 32 + ((8 * g4) ^ ((d & 3) << 3));
      const uint2 vf = *reinterpret_cast<const uint2*>(
          reinterpret_cast<const char*>(vt) + byte);
      oacc[dc] = __builtin_amdgcn_mfma_f32_16x16x16bf16_1k(
          as_frag16(vf.x, vf.y), pfrag, oacc[dc], 0, 0, 0);
    }
  }

  // ---- epilogue ----
  if (rc >= G) return;
  const int head = hk * G + rc;
  float* top = tmp_out + (((long)b * Hq + head) * S + slot) * D;
#pragma unroll
  for (int dc = 0; dc < 8; ++dc) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      top[dc * 16 + 4 * g4 + r] = oacc[dc][r];
    }
  }
  if (g4 == 0) {
    float* ml = tmp_ml + (((long)b * Hq + head) * S + slot) * 2;
    ml[0] = m;
    ml[1] = l_acc;
  }
}

// Reduce for the MFMA path: every one of the S slots is written on every
// call (empty ranges write m = -inf), so all are read unconditionally —
// no per-split block math, no zero-ctx division hazard.
template <int D>
__global__ void paged_attn_reduce_all_kernel(
    unsigned short* __restrict__ out, const float* __restrict__ tmp_out,
    const float* __restrict__ tmp_ml, const int* __restrict__ seq_lens,
    int Hq, int S) {
  const int b = blockIdx.x;
  const int h = blockIdx.y;
  const int d = threadIdx.x;
  if (seq_lens[b] <= 0) return;
  const float* ml = tmp_ml + (((long)b * Hq + h) * S) * 2;
  float M = -INFINITY;
  for (int s = 0; s < S; ++s) M = fmaxf(M, ml[s * 2]);
  float L = 0.f;
  for (int s = 0; s < S; ++s)
    L += (ml[s * 2] == -INFINITY ? 0.f : __expf(ml[s * 2] - M)) *
         ml[s * 2 + 1];
  const float* top = tmp_out + (((long)b * Hq + h) * S) * D;
  float acc = 0.f;
  for (int s = 0; s < S; ++s) {
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
  // Opt-in (KUKEON_ATTN_MFMA=1): correct and ~5x fewer wave-instructions
  // per token, but measured 25-30% behind the v_dot2 kernel end to end —
  // its 64 KiB LDS caps residency at 2 workgroups/CU while the v_dot2
  // kernel's 8 KiB image keeps ~8x more waves in flight to hide its
  // issue stalls. Round-2 lever: 16-token tiles on 16x16x32 MFMAs
  // (quarter the LDS) or a hand-scheduled pipeline.
  // DEFAULT: the 16-token-tile MFMA kernel (fragment-direct K, double-
  // buffered V^T, zero barriers) — measured 4.9-5.4 TB/s (78-86% of the
  // HBM roofline) vs the v_dot2 kernel's 3.8-4.1 on the decode shapes.
  // KUKEON_ATTN_MFMA=0 forces v_dot2, =32 the 32-token 32x32 variant;
  // v_dot2 also remains the fallback for split counts not divisible by
  // 4 and GQA groups over 16.
  const char* mfma_env = getenv("KUKEON_ATTN_MFMA");
  const bool want32 = mfma_env && mfma_env[0] == '3' && G <= 32;
  const bool want16 = !want32 && (!mfma_env || mfma_env[0] != '0') && G <= 16;
  const bool use_mfma = (want16 || want32) &&
                        num_splits % 4 == 0 &&
                        tmp_out.numel() >= (long)B * Hq * num_splits * 128;
  if (use_mfma) {
    dim3 mgrid(B, Hk, (unsigned)(num_splits / 4));
    float* tovm = tmp_out.data_ptr<float>();
    float* tmlm = tmp_ml.data_ptr<float>();
#define PAM_LAUNCH(KERN, FP8_)                                               \
    KERN<FP8_><<<mgrid, 256, 0, stream>>>(                                   \
        tovm, tmlm, qp, k_cache.data_ptr(), v_cache.data_ptr(),              \
        block_table.data_ptr<int>(), seq_lens.data_ptr<int>(), q_stride,     \
        Hq, Hk, max_blocks, (int)num_splits, (float)scale)
    if (want16) {
      if (fp8) { PAM_LAUNCH(paged_attn_mfma16_kernel, true); }
      else { PAM_LAUNCH(paged_attn_mfma16_kernel, false); }
    } else {
      if (fp8) { PAM_LAUNCH(paged_attn_mfma_kernel, true); }
      else { PAM_LAUNCH(paged_attn_mfma_kernel, false); }
    }
#undef PAM_LAUNCH
    HIP_CHECK_KERNEL();
    dim3 rgrid(B, Hq);
    paged_attn_reduce_all_kernel<128><<<rgrid, 128, 0, stream>>>(
        op, tovm, tmlm, seq_lens.data_ptr<int>(), Hq, (int)num_splits);
    HIP_CHECK_KERNEL();
    return;
  }
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
