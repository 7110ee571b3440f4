// pybind11 bindings for the agentainer_amd gfx950 HIP kernels.
#include <torch/extension.h>

void rmsnorm(torch::Tensor out, torch::Tensor x, torch::Tensor w, double eps);
void fused_add_rmsnorm(torch::Tensor out, torch::Tensor x,
                       torch::Tensor residual, torch::Tensor w, double eps);
void silu_mul(torch::Tensor out, torch::Tensor gate, torch::Tensor up);
void rope_inplace(torch::Tensor q, torch::Tensor k, torch::Tensor cos_sin,
                  torch::Tensor positions);
void rope_append(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                 torch::Tensor k_cache, torch::Tensor v_cache,
                 torch::Tensor cos_sin, torch::Tensor positions,
                 torch::Tensor slot_mapping);
void kv_append(torch::Tensor k_cache, torch::Tensor v_cache, torch::Tensor k,
               torch::Tensor v, torch::Tensor slot_mapping);
void paged_decode_attention(torch::Tensor out, torch::Tensor q,
                            torch::Tensor k_cache, torch::Tensor v_cache,
                            torch::Tensor page_table, torch::Tensor seq_lens,
                            double scale);
void paged_prefill_attention(torch::Tensor out, torch::Tensor q,
                             torch::Tensor k_cache, torch::Tensor v_cache,
                             torch::Tensor page_table, torch::Tensor seq_lens,
                             torch::Tensor query_starts, torch::Tensor query_lens,
                             double scale, long max_qlen);
void greedy_sample(torch::Tensor out, torch::Tensor logits);
void topp_sample(torch::Tensor out, torch::Tensor logits, torch::Tensor temps,
                 torch::Tensor top_ps, torch::Tensor seeds);
void gather_kv_pages(torch::Tensor dst, torch::Tensor k_cache,
                     torch::Tensor v_cache, torch::Tensor page_ids);
void skinny_gemm(torch::Tensor out, torch::Tensor x, torch::Tensor w,
                 torch::Tensor ws, long split);
void skinny_gemm_packed(torch::Tensor out, torch::Tensor x,
                        torch::Tensor w_packed, long N, long K,
                        torch::Tensor ws, long split, bool nt, long kc);
void quant_fp8_rows(torch::Tensor x8, torch::Tensor sx, torch::Tensor x);
void quant_fp4_rows(torch::Tensor x4, torch::Tensor sx, torch::Tensor x);
void skinny_gemm_mxfp4(torch::Tensor out, torch::Tensor x8, torch::Tensor sx,
                       torch::Tensor w_packed, torch::Tensor w_scales, long N,
                       long K, torch::Tensor ws, long split, long combo);
void skinny_gemm_fp8(torch::Tensor out, torch::Tensor x8, torch::Tensor sx,
                     torch::Tensor w_packed, torch::Tensor sw, long N, long K,
                     torch::Tensor ws, long split);
void mfma_probe(torch::Tensor c, torch::Tensor a, torch::Tensor b);
void scatter_kv_pages(torch::Tensor k_cache, torch::Tensor v_cache,
                      torch::Tensor src, torch::Tensor page_ids);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "agentainer_amd CDNA4 (gfx950) kernels";
  m.def("rmsnorm", &rmsnorm, "RMSNorm (bf16)");
  m.def("fused_add_rmsnorm", &fused_add_rmsnorm, "residual += x; rmsnorm");
  m.def("silu_mul", &silu_mul, "silu(gate) * up");
  m.def("rope_inplace", &rope_inplace, "apply rotary embedding to q,k in place");
  m.def("rope_append", &rope_append, "fused RoPE + paged KV append");
  m.def("kv_append", &kv_append, "scatter new k/v into the paged KV cache");
  m.def("paged_decode_attention", &paged_decode_attention,
        "single-token GQA attention over the paged KV cache");
  m.def("paged_prefill_attention", &paged_prefill_attention,
        "varlen causal MFMA attention over the paged KV cache");
  m.def("greedy_sample", &greedy_sample, "argmax over vocab per row");
  m.def("topp_sample", &topp_sample, "temperature + top-p sampling per row");
  m.def("gather_kv_pages", &gather_kv_pages, "pages -> host-shaped buffer");
  m.def("skinny_gemm", &skinny_gemm, "out[M,N] = x[M,K] @ W[N,K]^T, M<=64");
  m.def("skinny_gemm_packed", &skinny_gemm_packed, "packed-weight skinny GEMM");
  m.def("skinny_gemm_mxfp4", &skinny_gemm_mxfp4, "MXFP4 block-scaled expert GEMM");
  m.def("quant_fp4_rows", &quant_fp4_rows, "per-row e2m1 activation quant");
  m.def("quant_fp8_rows", &quant_fp8_rows, "per-row bf16 -> e4m3 quant");
  m.def("skinny_gemm_fp8", &skinny_gemm_fp8, "fp8 MFMA skinny GEMM");
  m.def("mfma_probe", &mfma_probe, "single 16x16x32 bf16 MFMA on prepacked fragments");
  m.def("scatter_kv_pages", &scatter_kv_pages, "buffer -> pages");
}
