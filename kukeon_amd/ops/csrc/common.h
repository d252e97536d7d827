// kukeon_amd CDNA4 (gfx950) kernel common helpers.
//
// MI355X-native: wave64, 4 SIMD-32 per CU, 160 KiB LDS, HBM3E.
// No CUDA compatibility paths — HIP/gfx950 only.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <stdint.h>

#define WAVE 64

#define DEV_INLINE __device__ __forceinline__

// ---------- bf16 <-> f32 ----------
DEV_INLINE float bf2f(__hip_bfloat16 x) { return __bfloat162float(x); }
DEV_INLINE __hip_bfloat16 f2bf(float x) { return __float2bfloat16(x); }

// Reinterpret a ushort as bf16 without conversion.
DEV_INLINE float us2f(unsigned short u) {
  unsigned int w = ((unsigned int)u) << 16;
  return __builtin_bit_cast(float, w);
}
DEV_INLINE unsigned short f2us(float f) {
  // round-to-nearest-even bf16 truncation
  unsigned int w = __builtin_bit_cast(unsigned int, f);
  unsigned int lsb = (w >> 16) & 1u;
  w += 0x7fffu + lsb;
  return (unsigned short)(w >> 16);
}

// 8 bf16 = 16 bytes, the coalescing sweet spot (G13).
struct bf16x8 {
  uint4 raw;
  DEV_INLINE unsigned short us(int i) const {
    const unsigned int* p = reinterpret_cast<const unsigned int*>(&raw);
    return (unsigned short)((p[i >> 1] >> ((i & 1) * 16)) & 0xffffu);
  }
  DEV_INLINE float f(int i) const { return us2f(us(i)); }
};

DEV_INLINE bf16x8 load_bf16x8(const void* p) {
  bf16x8 v;
  v.raw = *reinterpret_cast<const uint4*>(p);
  return v;
}

DEV_INLINE uint4 pack_bf16x8(const float* f) {
  uint4 r;
  unsigned int* p = reinterpret_cast<unsigned int*>(&r);
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    p[i] = (unsigned int)f2us(f[2 * i]) | ((unsigned int)f2us(f[2 * i + 1]) << 16);
  }
  return r;
}

// ---------- fp8 (OCP e4m3fn) helpers ----------
typedef __attribute__((__vector_size__(2 * sizeof(float)))) float f32x2_cvt_t;

DEV_INLINE unsigned int cvtpk_bf16(float lo, float hi) {
  __hip_bfloat162 h = __float22bfloat162_rn(float2{lo, hi});
  unsigned int r;
  __builtin_memcpy(&r, &h, 4);
  return r;
}

// 4 fp8 bytes (one dword) -> 4 bf16 (uint2)
DEV_INLINE uint2 fp8x4_to_bf16x4(unsigned int w) {
  f32x2_cvt_t lo = __builtin_amdgcn_cvt_pk_f32_fp8(w, false);
  f32x2_cvt_t hi = __builtin_amdgcn_cvt_pk_f32_fp8(w, true);
  return uint2{cvtpk_bf16(lo[0], lo[1]), cvtpk_bf16(hi[0], hi[1])};
}

// 8 fp8 bytes (uint2) -> 8 bf16 (uint4)
DEV_INLINE uint4 fp8x8_to_bf16x8(uint2 w) {
  const uint2 a = fp8x4_to_bf16x4(w.x);
  const uint2 b = fp8x4_to_bf16x4(w.y);
  return uint4{a.x, a.y, b.x, b.y};
}

// 8 f32 -> 8 fp8 bytes (uint2)
DEV_INLINE uint2 pack_fp8x8(const float* f) {
  unsigned int w0 = __builtin_amdgcn_cvt_pk_fp8_f32(f[0], f[1], 0u, false);
  w0 = __builtin_amdgcn_cvt_pk_fp8_f32(f[2], f[3], w0, true);
  unsigned int w1 = __builtin_amdgcn_cvt_pk_fp8_f32(f[4], f[5], 0u, false);
  w1 = __builtin_amdgcn_cvt_pk_fp8_f32(f[6], f[7], w1, true);
  return uint2{w0, w1};
}

// ---------- wave reductions (wave64) ----------
DEV_INLINE float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}
DEV_INLINE float wave_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE));
  return v;
}

// Block reduce over up to 16 waves; every thread returns the result.
// `red` must point to >= nwaves floats of LDS.
template <typename Op>
DEV_INLINE float block_reduce(float v, float* red, Op op, float init) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int nwaves = (blockDim.x + WAVE - 1) / WAVE;
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = op(v, __shfl_xor(v, off, WAVE));
  if (lane == 0) red[wid] = v;
  __syncthreads();
  float r = init;
  if (threadIdx.x < nwaves) r = red[threadIdx.x];
#pragma unroll
  for (int off = 8; off > 0; off >>= 1) r = op(r, __shfl_xor(r, off, WAVE));
  // lane 0 of wave 0 has it; broadcast via LDS
  if (threadIdx.x == 0) red[0] = r;
  __syncthreads();
  r = red[0];
  __syncthreads();
  return r;
}

struct SumOp { DEV_INLINE float operator()(float a, float b) const { return a + b; } };
struct MaxOp { DEV_INLINE float operator()(float a, float b) const { return fmaxf(a, b); } };

#define HIP_CHECK_KERNEL()                                   \
  do {                                                       \
    hipError_t e_ = hipGetLastError();                       \
    TORCH_CHECK(e_ == hipSuccess, "HIP kernel launch failed: ", hipGetErrorString(e_)); \
  } while (0)
