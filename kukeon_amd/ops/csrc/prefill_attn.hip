// Varlen causal paged prefill attention (MFMA, bf16) for gfx950 — plus the
// fragment-layout probe used by the GPU test suite to pin the
// v_mfma_f32_32x32x16_bf16 operand maps this kernel relies on.
//
// Assumed gfx950 fragment maps (validated by tests/test_ops_gpu.py::test_mfma_probe):
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

void prefill_attention(torch::Tensor out, torch::Tensor q,
                       torch::Tensor k_cache, torch::Tensor v_cache,
                       torch::Tensor block_table, torch::Tensor seq_lens,
                       torch::Tensor q_starts, torch::Tensor qb_seq,
                       torch::Tensor qb_start, int64_t q_offset, double scale) {
  TORCH_CHECK(false, "prefill_attention MFMA kernel not yet wired");
}

}  // namespace kukeon
