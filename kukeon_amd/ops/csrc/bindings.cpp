// Python bindings for the kukeon_amd gfx950 kernel library.
#include <torch/extension.h>

namespace kukeon {
void rmsnorm(torch::Tensor out, torch::Tensor input, torch::Tensor weight,
             double eps);
void fused_add_rmsnorm(torch::Tensor input, torch::Tensor residual,
                       torch::Tensor weight, double eps);
void silu_mul(torch::Tensor out, torch::Tensor gate_up);
void rope_kv_append(torch::Tensor qkv, torch::Tensor k_cache,
                    torch::Tensor v_cache, torch::Tensor cos_sin,
                    torch::Tensor positions, torch::Tensor slot_mapping,
                    int64_t num_q_heads, int64_t num_kv_heads,
                    int64_t head_dim, torch::Tensor block_table);
void decode_advance(torch::Tensor ids, torch::Tensor pos,
                    torch::Tensor seq_lens, torch::Tensor tokens,
                    torch::Tensor ring, torch::Tensor counter);
void paged_attention(torch::Tensor out, torch::Tensor q, torch::Tensor k_cache,
                     torch::Tensor v_cache, torch::Tensor block_table,
                     torch::Tensor seq_lens, int64_t q_offset,
                     int64_t num_splits, double scale, torch::Tensor tmp_out,
                     torch::Tensor tmp_ml);
void prefill_attention(torch::Tensor out, torch::Tensor q,
                       torch::Tensor k_cache, torch::Tensor v_cache,
                       torch::Tensor block_table, torch::Tensor seq_lens,
                       torch::Tensor q_starts, torch::Tensor qb_seq,
                       torch::Tensor qb_start, int64_t q_offset, double scale);
void mfma_probe(torch::Tensor out, torch::Tensor a, torch::Tensor b);
void mfma_probe16(torch::Tensor out, torch::Tensor a, torch::Tensor b);
void mfma_probe16k(torch::Tensor out, torch::Tensor a, torch::Tensor b);
void skinny_gemm2(torch::Tensor out, torch::Tensor x, torch::Tensor w,
                  torch::Tensor ws);
void skinny_gemm4(torch::Tensor out, torch::Tensor x, torch::Tensor w,
                  torch::Tensor ws);
void skinny_gemm5(torch::Tensor out, torch::Tensor x, torch::Tensor w,
                  torch::Tensor ws);
void skinny_gemm6(torch::Tensor out, torch::Tensor x, torch::Tensor w,
                  torch::Tensor ws);
void skinny_gemm6_fused_norm(torch::Tensor normed, torch::Tensor x,
                             torch::Tensor w, torch::Tensor ws,
                             torch::Tensor residual, torch::Tensor nw,
                             double eps);
void skinny_gemm5_silu_fused_norm(torch::Tensor normed, torch::Tensor gu,
                                  torch::Tensor w, torch::Tensor ws,
                                  torch::Tensor residual, torch::Tensor nw,
                                  double eps);
void skinny_gemm(torch::Tensor out, torch::Tensor x, torch::Tensor w,
                 torch::Tensor ws);
void glds_probe(torch::Tensor out, torch::Tensor src);
void skinny_gemm5_fused_norm(torch::Tensor normed, torch::Tensor x,
                             torch::Tensor w, torch::Tensor ws,
                             torch::Tensor residual, torch::Tensor nw,
                             double eps);
void skinny_gemm_fused_norm(torch::Tensor normed, torch::Tensor x,
                            torch::Tensor w, torch::Tensor ws,
                            torch::Tensor residual, torch::Tensor nw,
                            double eps);
void sample(torch::Tensor tokens, torch::Tensor logits, torch::Tensor temps,
            torch::Tensor top_k, torch::Tensor top_p, torch::Tensor seed,
            torch::Tensor workspace);
void moe_router_weights(torch::Tensor wdense, torch::Tensor logits,
                        long K);
void moe_dense_combine(torch::Tensor out, torch::Tensor y,
                       torch::Tensor wdense);
void moe_gather_tokens(torch::Tensor out, torch::Tensor input,
                       torch::Tensor row_map);
void moe_scatter_tokens(torch::Tensor out, torch::Tensor input,
                        torch::Tensor row_map, torch::Tensor weights,
                        int64_t top_k);
}  // namespace kukeon

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rmsnorm", &kukeon::rmsnorm, "RMSNorm (bf16, gfx950)");
  m.def("fused_add_rmsnorm", &kukeon::fused_add_rmsnorm,
        "residual += input; input = rmsnorm(residual)");
  m.def("silu_mul", &kukeon::silu_mul, "out = silu(gate)*up from fused [T,2I]");
  m.def("rope_kv_append", &kukeon::rope_kv_append,
        "in-place NEOX RoPE on fused qkv + paged KV append");
  m.def("paged_attention", &kukeon::paged_attention,
        "decode paged attention (GQA, split-KV)");
  m.def("prefill_attention", &kukeon::prefill_attention,
        "varlen causal paged prefill attention (MFMA)");
  m.def("mfma_probe", &kukeon::mfma_probe,
        "32x32x16 bf16 MFMA fragment-layout probe");
  m.def("mfma_probe16", &kukeon::mfma_probe16,
        "16x16x32 bf16 MFMA fragment-layout probe");
  m.def("mfma_probe16k", &kukeon::mfma_probe16k,
        "16x16x16 (K=16) bf16 MFMA fragment-layout probe");
  m.def("skinny_gemm6", &kukeon::skinny_gemm6,
        "decode GEMM v6 (barrier-free register-x pipeline)");
  m.def("skinny_gemm6_fused_norm", &kukeon::skinny_gemm6_fused_norm,
        "v6 + fused split-K reduce + residual add + RMSNorm");
  m.def("skinny_gemm5_silu_fused_norm",
        &kukeon::skinny_gemm5_silu_fused_norm,
        "silu(gate)*up + down GEMM + reduce + add + RMSNorm fused");
  m.def("skinny_gemm5", &kukeon::skinny_gemm5,
        "full-line W stream via wave-private LDS image (v5)");
  m.def("skinny_gemm4", &kukeon::skinny_gemm4,
        "G=4 deep-pipeline weight-streaming GEMM");
  m.def("skinny_gemm2", &kukeon::skinny_gemm2,
        "G-walk hand-counted weight-streaming GEMM (v2)");
  m.def("skinny_gemm", &kukeon::skinny_gemm,
        "weight-streaming decode GEMM (M<=64)");
  m.def("glds_probe", &kukeon::glds_probe,
        "asm global_lds round-trip validator");
  m.def("skinny_gemm5_fused_norm", &kukeon::skinny_gemm5_fused_norm,
        "skinny5 + fused split-K reduce + residual add + RMSNorm");
  m.def("skinny_gemm_fused_norm", &kukeon::skinny_gemm_fused_norm,
        "skinny GEMM + split-K reduce + residual add + RMSNorm");
  m.def("sample", &kukeon::sample, "top-k/top-p/temperature sampling");
  m.def("decode_advance", &kukeon::decode_advance,
        "on-device decode cursor advance (self-advancing graph)");
  m.def("moe_gather_tokens", &kukeon::moe_gather_tokens, "MoE permute");
  m.def("moe_router_weights", &kukeon::moe_router_weights,
        "softmax top-K renorm -> dense [T,E] weights, one launch");
  m.def("moe_dense_combine", &kukeon::moe_dense_combine,
        "out = sum_e w[t,e]*y[e,t,:], zero-weight experts skipped");
  m.def("moe_scatter_tokens", &kukeon::moe_scatter_tokens,
        "MoE unpermute + weighted combine");
}
