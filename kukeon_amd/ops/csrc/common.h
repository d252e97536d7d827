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
