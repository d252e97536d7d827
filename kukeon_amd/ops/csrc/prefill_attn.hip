// Varlen causal paged prefill attention (MFMA, bf16) for gfx950 — plus the
// fragment-layout probe used by the GPU test suite to pin the
// v_mfma_f32_32x32x16_bf16 operand maps this kernel relies on.
//
// Structure (MI355X-first; guide §Appendix B attention recipe adapted to a
// paged varlen serving kernel):
//   - workgroup = 4 waves = the 4 q-heads of one GQA group (K/V staged in
//     LDS once per 32-token tile and shared by all heads: KV HBM traffic is
//     never multiplied by the group size); grid.z splits groups of 8.
//   - swapped QK^T (S^T = K·Q^T) so each lane owns the scores of one q
//     column -> online softmax is lane-local (+ one shfl_xor(32) exchange
//     between the two half-wave row sets).
//   - P -> bf16 via packed cvt (v_cvt_pk_bf16_f32) + permlane32_swap to
//     rebuild k-contiguous MFMA A.. er B-fragments without an LDS bounce.
//   - PV as O^T = V^T·P^T with V staged TRANSPOSED into LDS (ushort2
//     interleave at stage time, slot-XOR swizzle), so the V fragment read is
//     a single ds_read_b128.
//   - K tile XOR-swizzled ((row&15)<<4) -> conflict-free ds_read_b128
//     (guide G4: row-major D=128 tiles are a 16-way conflict otherwise).
//
// Assumed gfx950 fragment maps (validated by tests/test_ops_gpu.py):
//   A[32x16k]: lane l -> row = l&31, k = 8*(l>>5) + j   (j = 0..7, bf16x8)
//   B[16kx32]: lane l -> col = l&31, k = 8*(l>>5) + j
//   C/D:       lane l -> col = l&31, row = (r&3) + 8*(r>>2) + 4*(l>>5)
#include "common.h"
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

namespace kukeon {

typedef __attribute__((__vector_size__(8 * sizeof(short)))) short bf16x8_t;
typedef __attribute__((__vector_size__(16 * sizeof(float)))) float f32x16_t;

struct uint4_t { unsigned int x[4]; };
DEV_INLINE bf16x8_t as_frag(unsigned int w0, unsigned int w1, unsigned int w2,
                            unsigned int w3) {
  uint4_t v{{w0, w1, w2, w3}};
  return __builtin_bit_cast(bf16x8_t, v);
}

// ---------- layout probe: C[32,32] = A[32,16] @ B[16,32] ----------
__global__ void mfma_probe_kernel(float* __restrict__ c,
                                  const unsigned short* __restrict__ a,
                                  const unsigned short* __restrict__ b) {
  const int lane = threadIdx.x & 63;
  const int hi = lane >> 5;
  const int rc = lane & 31;
  unsigned int areg[4], breg[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int k0 = 8 * hi + 2 * i;
    areg[i] = (unsigned)a[rc * 16 + k0] | ((unsigned)a[rc * 16 + k0 + 1] << 16);
    breg[i] = (unsigned)b[k0 * 32 + rc] | ((unsigned)b[(k0 + 1) * 32 + rc] << 16);
  }
  f32x16_t acc = {};
  acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
      as_frag(areg[0], areg[1], areg[2], areg[3]),
      as_frag(breg[0], breg[1], breg[2], breg[3]), acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int row = (r & 3) + 8 * (r >> 2) + 4 * hi;
    c[row * 32 + rc] = acc[r];
  }
}

__global__ void mfma_probe16_kernel(float* __restrict__ c,
                                    const unsigned short* __restrict__ a,
                                    const unsigned short* __restrict__ b) {
  typedef __attribute__((__vector_size__(4 * sizeof(float)))) float f32x4_t;
  const int lane = threadIdx.x & 63;
  const int g = lane >> 4;     // k group
  const int rc = lane & 15;
  unsigned int areg[4], breg[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int k0 = 8 * g + 2 * i;
    areg[i] = (unsigned)a[rc * 32 + k0] | ((unsigned)a[rc * 32 + k0 + 1] << 16);
    breg[i] = (unsigned)b[k0 * 16 + rc] | ((unsigned)b[(k0 + 1) * 16 + rc] << 16);
  }
  f32x4_t acc = {};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
      as_frag(areg[0], areg[1], areg[2], areg[3]),
      as_frag(breg[0], breg[1], breg[2], breg[3]), acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r) c[(4 * g + r) * 16 + rc] = acc[r];
}

void mfma_probe16(torch::Tensor out, torch::Tensor a, torch::Tensor b) {
  TORCH_CHECK(a.sizes() == torch::IntArrayRef({16, 32}));
  TORCH_CHECK(b.sizes() == torch::IntArrayRef({32, 16}));
  TORCH_CHECK(out.sizes() == torch::IntArrayRef({16, 16}));
  auto stream = c10::hip::getCurrentHIPStream().stream();
  mfma_probe16_kernel<<<1, 64, 0, stream>>>(
      out.data_ptr<float>(),
      reinterpret_cast<const unsigned short*>(a.data_ptr()),
      reinterpret_cast<const unsigned short*>(b.data_ptr()));
  HIP_CHECK_KERNEL();
}

void mfma_probe(torch::Tensor out, torch::Tensor a, torch::Tensor b) {
  TORCH_CHECK(a.sizes() == torch::IntArrayRef({32, 16}));
  TORCH_CHECK(b.sizes() == torch::IntArrayRef({16, 32}));
  TORCH_CHECK(out.sizes() == torch::IntArrayRef({32, 32}));
  TORCH_CHECK(a.scalar_type() == torch::kBFloat16 && a.is_contiguous());
  TORCH_CHECK(out.scalar_type() == torch::kFloat32);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  mfma_probe_kernel<<<1, 64, 0, stream>>>(
      out.data_ptr<float>(),
      reinterpret_cast<const unsigned short*>(a.data_ptr()),
      reinterpret_cast<const unsigned short*>(b.data_ptr()));
  HIP_CHECK_KERNEL();
}

// ---------- helpers ----------
DEV_INLINE unsigned int cvt_pk_bf16(float lo, float hi) {
  // lowers to v_cvt_pk_bf16_f32 (don't hand-write the asm: guide T12)
  __hip_bfloat162 h = __float22bfloat162_rn(float2{lo, hi});
  unsigned int r;
  __builtin_memcpy(&r, &h, 4);
  return r;
}

// ---------- the kernel ----------
// D=128, QBLK=32 rows per workgroup, KV tile = 32 tokens (2 cache blocks).
template <int BS, bool FP8>
__global__ __launch_bounds__(256, 1) void prefill_attn_kernel(
    unsigned short* __restrict__ out,       // [T, Hq*D]
    const unsigned short* __restrict__ q,   // [T, q_stride] fused qkv rows
    const void* __restrict__ k_cache,       // [NB, Hk, BS, D] bf16|fp8
    const void* __restrict__ v_cache,
    const int* __restrict__ block_table,    // [nseq, max_blocks]
    const int* __restrict__ seq_lens,       // [nseq] total ctx after append
    const int* __restrict__ q_starts,       // [nseq, 2] = (abs_start, row0)
    const int* __restrict__ qb_seq,         // [nqb]
    const int* __restrict__ qb_start,       // [nqb] local q offset
    long q_stride, long q_offset, int Hq, int Hk, int max_blocks,
    int out_stride, float scale) {
  constexpr int D = 128;
  constexpr int KVT = 32;
  // 16-B aligned: staged via uint4 stores (guide G17)
  __shared__ __align__(16) unsigned short kbuf[KVT * D];   // XOR-swizzled K
  __shared__ __align__(16) unsigned short vtbuf[D * KVT];  // V^T tile

  const int s = qb_seq[blockIdx.x];
  const int qb0 = qb_start[blockIdx.x];
  const int hk = blockIdx.y;
  const int ctx = seq_lens[s];
  const int abs_start = q_starts[2 * s];
  const int row0 = q_starts[2 * s + 1] + qb0;
  const int qlen = ctx - abs_start;
  const int qn = min(32, qlen - qb0);
  if (qn <= 0) return;
  const int G = Hq / Hk;
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  const int hi = lane >> 5;
  const int qcol = lane & 31;
  const int h_local = blockIdx.z * 4 + wid;
  const bool active = h_local < G;
  const int head = hk * G + (active ? h_local : 0);

  // absolute position of this lane's q column (clamped for padding rows)
  const int q_abs = abs_start + qb0 + min(qcol, qn - 1);

  // ---- load Q fragments (B-operand: col=q, k=d chunks) ----
  unsigned int qfrag[8][4];
  {
    const long qrow = row0 + min(qcol, qn - 1);
    const unsigned short* qp = q + qrow * q_stride + q_offset + (long)head * D;
#pragma unroll
    for (int st = 0; st < 8; ++st) {
      const uint4 v = *reinterpret_cast<const uint4*>(qp + st * 16 + hi * 8);
      qfrag[st][0] = v.x; qfrag[st][1] = v.y;
      qfrag[st][2] = v.z; qfrag[st][3] = v.w;
    }
  }

  float m = -INFINITY, l_acc = 0.f;
  f32x16_t oacc[4] = {{}, {}, {}, {}};

  const int kv_limit = min(ctx, abs_start + qb0 + qn);
  const int ntiles = (kv_limit + KVT - 1) / KVT;
  const int nblocks = (ctx + BS - 1) / BS;

  for (int ti = 0; ti < ntiles; ++ti) {
    const int kvbase = ti * KVT;
    // ---- cooperative stage ----
    {
      __syncthreads();  // protect previous tile's readers
      // K: thread covers (tok = tid/8, d0 = (tid%8)*16), 2 x 16B
      const int tok = threadIdx.x >> 3;
      const int d0 = (threadIdx.x & 7) * 16;
      const int kvpos = kvbase + tok;
      const int bidx = min(kvpos / BS, nblocks - 1);
      const int pblk = block_table[(long)s * max_blocks + bidx];
      const long koff = (((long)pblk * Hk + hk) * BS + (kvpos % BS)) * D;
      const int swz = (tok & 15) << 4;
#pragma unroll
      for (int c = 0; c < 2; ++c) {
        uint4 v;
        if (FP8) {
          v = fp8x8_to_bf16x8(*reinterpret_cast<const uint2*>(
              reinterpret_cast<const unsigned char*>(k_cache) + koff + d0 +
              c * 8));
        } else {
          v = *reinterpret_cast<const uint4*>(
              reinterpret_cast<const unsigned short*>(k_cache) + koff + d0 +
              c * 8);
        }
        const int byte = tok * 256 + (((d0 + c * 8) * 2) ^ swz);
        *reinterpret_cast<uint4*>(
            reinterpret_cast<char*>(kbuf) + byte) = v;
      }
      // V: thread covers (kv pair = tid/16, d0 = (tid%16)*8), transpose
      const int kp = threadIdx.x >> 4;          // 0..15 -> kv = 2*kp
      const int vd0 = (threadIdx.x & 15) * 8;
      const int vpos0 = kvbase + 2 * kp;
      const int vpos1 = vpos0 + 1;
      const int vb0 = min(vpos0 / BS, nblocks - 1);
      const int vb1 = min(vpos1 / BS, nblocks - 1);
      const long voff0 =
          (((long)block_table[(long)s * max_blocks + vb0] * Hk + hk) * BS +
           (vpos0 % BS)) * D;
      const long voff1 =
          (((long)block_table[(long)s * max_blocks + vb1] * Hk + hk) * BS +
           (vpos1 % BS)) * D;
      uint4 r0, r1;
      if (FP8) {
        const unsigned char* v8 =
            reinterpret_cast<const unsigned char*>(v_cache);
        r0 = fp8x8_to_bf16x8(
            *reinterpret_cast<const uint2*>(v8 + voff0 + vd0));
        r1 = fp8x8_to_bf16x8(
            *reinterpret_cast<const uint2*>(v8 + voff1 + vd0));
      } else {
        const unsigned short* v16 =
            reinterpret_cast<const unsigned short*>(v_cache);
        r0 = *reinterpret_cast<const uint4*>(v16 + voff0 + vd0);
        r1 = *reinterpret_cast<const uint4*>(v16 + voff1 + vd0);
      }
      // zero V rows past ctx: a 0*NaN in the PV mfma would poison O
      if (vpos0 >= ctx) r0 = uint4{0, 0, 0, 0};
      if (vpos1 >= ctx) r1 = uint4{0, 0, 0, 0};
      const unsigned int* a0 = reinterpret_cast<const unsigned int*>(&r0);
      const unsigned int* a1 = reinterpret_cast<const unsigned int*>(&r1);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const unsigned short e0 = (a0[j >> 1] >> ((j & 1) * 16)) & 0xffffu;
        const unsigned short e1 = (a1[j >> 1] >> ((j & 1) * 16)) & 0xffffu;
        const unsigned int packed = (unsigned)e0 | ((unsigned)e1 << 16);
        const int d = vd0 + j;
        const int byte = d * 64 + ((4 * kp) ^ ((d & 3) << 4));
        *reinterpret_cast<unsigned int*>(
            reinterpret_cast<char*>(vtbuf) + byte) = packed;
      }
      __syncthreads();
    }
    if (!active) continue;

    // ---- S^T = K · Q^T  (A = K tile, B = Q frags) ----
    f32x16_t sacc = {};
#pragma unroll
    for (int st = 0; st < 8; ++st) {
      const int row = qcol;  // A row = kv = lane&31
      const int byte = row * 256 + (((st * 16 + hi * 8) * 2) ^ ((row & 15) << 4));
      const uint4 kf = *reinterpret_cast<const uint4*>(
          reinterpret_cast<const char*>(kbuf) + byte);
      sacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
          as_frag(kf.x, kf.y, kf.z, kf.w),
          as_frag(qfrag[st][0], qfrag[st][1], qfrag[st][2], qfrag[st][3]),
          sacc, 0, 0, 0);
    }

    // ---- online softmax over this lane's 16 kv rows ----
    float sv[16];
    float tmax = -INFINITY;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int kvrow = (r & 3) + 8 * (r >> 2) + 4 * hi;
      const int kvpos = kvbase + kvrow;
      const bool valid = (kvpos <= q_abs) && (kvpos < ctx);
      sv[r] = valid ? sacc[r] * scale : -INFINITY;
      tmax = fmaxf(tmax, sv[r]);
    }
    tmax = fmaxf(tmax, __shfl_xor(tmax, 32, WAVE));
    const float mn = fmaxf(m, tmax);
    const float alpha = (m == -INFINITY) ? 0.f : __expf(m - mn);
    m = mn;
    float psum = 0.f;
    float p[16];
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      p[r] = (sv[r] == -INFINITY) ? 0.f : __expf(sv[r] - mn);
      psum += p[r];
    }
    psum += __shfl_xor(psum, 32, WAVE);
    l_acc = l_acc * alpha + psum;
#pragma unroll
    for (int db = 0; db < 4; ++db)
#pragma unroll
      for (int r = 0; r < 16; ++r) oacc[db][r] *= alpha;

    // ---- P -> bf16 B-fragments via cvt_pk + permlane32_swap ----
    unsigned int cp[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) cp[j] = cvt_pk_bf16(p[2 * j], p[2 * j + 1]);
    unsigned int pb[2][4];  // [kv chunk][4 dwords = 8 bf16, k-contiguous]
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

    // ---- O^T += V^T · P^T ----
#pragma unroll
    for (int db = 0; db < 4; ++db) {
      const int d = db * 32 + qcol;  // A row = d
#pragma unroll
      for (int c = 0; c < 2; ++c) {
        const int byte = d * 64 + (((c * 32) + hi * 16) ^ ((d & 3) << 4));
        const uint4 vf = *reinterpret_cast<const uint4*>(
            reinterpret_cast<const char*>(vtbuf) + byte);
        oacc[db] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            as_frag(vf.x, vf.y, vf.z, vf.w),
            as_frag(pb[c][0], pb[c][1], pb[c][2], pb[c][3]),
            oacc[db], 0, 0, 0);
      }
    }
  }

  // ---- epilogue: out[row0+q][head*D + d] = O^T[d][q] / l ----
  if (!active || qcol >= qn) return;
  const float inv = l_acc > 0.f ? 1.f / l_acc : 0.f;
  unsigned short* orow = out + (long)(row0 + qcol) * out_stride +
                         (long)head * D;
#pragma unroll
  for (int db = 0; db < 4; ++db) {
#pragma unroll
    for (int quad = 0; quad < 4; ++quad) {
      const int d = db * 32 + quad * 8 + 4 * hi;
      const unsigned int w0 = cvt_pk_bf16(oacc[db][quad * 4 + 0] * inv,
                                          oacc[db][quad * 4 + 1] * inv);
      const unsigned int w1 = cvt_pk_bf16(oacc[db][quad * 4 + 2] * inv,
                                          oacc[db][quad * 4 + 3] * inv);
      *reinterpret_cast<uint2*>(orow + d) = uint2{w0, w1};
    }
  }
}

void prefill_attention(torch::Tensor out, torch::Tensor q,
                       torch::Tensor k_cache, torch::Tensor v_cache,
                       torch::Tensor block_table, torch::Tensor seq_lens,
                       torch::Tensor q_starts, torch::Tensor qb_seq,
                       torch::Tensor qb_start, int64_t q_offset, double scale) {
  const int Hk = k_cache.size(1);
  const int BS = k_cache.size(2);
  const int D = k_cache.size(3);
  const int Hq = out.size(1) / D;
  const int G = Hq / Hk;
  const int nqb = qb_seq.size(0);
  if (nqb == 0) return;
  TORCH_CHECK(D == 128, "only head_dim=128 supported");
  TORCH_CHECK(BS == 16, "kv block size must be 16");
  TORCH_CHECK(q_starts.dim() == 2 && q_starts.size(1) == 2);
  TORCH_CHECK(qb_seq.scalar_type() == torch::kInt32);
  auto stream = c10::hip::getCurrentHIPStream().stream();
  dim3 grid(nqb, Hk, (G + 3) / 4);
  const bool fp8 = k_cache.scalar_type() == torch::kUInt8;
#define PF_LAUNCH(FP8_)                                                      \
  prefill_attn_kernel<16, FP8_><<<grid, 256, 0, stream>>>(                   \
      reinterpret_cast<unsigned short*>(out.data_ptr()),                     \
      reinterpret_cast<const unsigned short*>(q.data_ptr()),                 \
      k_cache.data_ptr(), v_cache.data_ptr(), block_table.data_ptr<int>(),   \
      seq_lens.data_ptr<int>(), q_starts.data_ptr<int>(),                    \
      qb_seq.data_ptr<int>(), qb_start.data_ptr<int>(), q.stride(0),         \
      q_offset, Hq, Hk, (int)block_table.size(1), (int)out.stride(0),        \
      (float)scale)
  if (fp8) { PF_LAUNCH(true); } else { PF_LAUNCH(false); }
#undef PF_LAUNCH
  HIP_CHECK_KERNEL();
}

}  // namespace kukeon
