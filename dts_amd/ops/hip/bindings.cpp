// Python bindings for the dts_amd MI355X kernels.
#include <torch/extension.h>

void rmsnorm(torch::Tensor out, torch::Tensor x, torch::Tensor w, double eps);
void fused_add_rmsnorm(torch::Tensor x, torch::Tensor residual,
                       torch::Tensor w, double eps);
void silu_mul(torch::Tensor out, torch::Tensor in);
void layernorm(torch::Tensor out, torch::Tensor x, torch::Tensor w,
               torch::Tensor b, double eps);
void rope_kv_append(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                    torch::Tensor positions, torch::Tensor cos_t,
                    torch::Tensor sin_t, torch::Tensor k_cache,
                    torch::Tensor v_cache, torch::Tensor slots);
void kv_append(torch::Tensor k, torch::Tensor v, torch::Tensor k_cache,
               torch::Tensor v_cache, torch::Tensor slots);
void attn_decode_paged(torch::Tensor out, torch::Tensor q,
                       torch::Tensor kcache, torch::Tensor vcache,
                       torch::Tensor block_tables, torch::Tensor kv_lens,
                       double scale);
void attn_prefill_paged(torch::Tensor out, torch::Tensor q, torch::Tensor cu_q,
                        torch::Tensor q_pos, torch::Tensor kcache,
                        torch::Tensor vcache, torch::Tensor block_tables,
                        torch::Tensor kv_lens, double scale, int64_t swz);
void top_p_sample(torch::Tensor out, torch::Tensor logits, torch::Tensor temps,
                  torch::Tensor top_ps, torch::Tensor seeds);
void derive_seeds(torch::Tensor out, torch::Tensor bases,
                  torch::Tensor positions);
void top_p_sample_v2(torch::Tensor out, torch::Tensor logits,
                     torch::Tensor temps, torch::Tensor top_ps,
                     torch::Tensor seeds, torch::Tensor ws_max,
                     torch::Tensor ws_bins, torch::Tensor ws_slice,
                     torch::Tensor ws_tau);
void gemv_bf16(torch::Tensor out, torch::Tensor x, torch::Tensor w);
void gemm_skinny_bf16(torch::Tensor out, torch::Tensor x, torch::Tensor w);
void moe_grouped_linear(torch::Tensor out, torch::Tensor x, torch::Tensor w,
                        torch::Tensor counts, torch::Tensor offsets);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rmsnorm", &rmsnorm, "RMSNorm (bf16, CDNA4)");
  m.def("fused_add_rmsnorm", &fused_add_rmsnorm,
        "fused residual add + RMSNorm (in-place, bf16)");
  m.def("silu_mul", &silu_mul, "SwiGLU activation (bf16)");
  m.def("layernorm", &layernorm, "LayerNorm (bf16, GPT-2 path)");
  m.def("rope_kv_append", &rope_kv_append,
        "fused rotate-half RoPE + paged KV append (bf16)");
  m.def("kv_append", &kv_append, "paged KV append without RoPE (bf16)");
  m.def("attn_decode_paged", &attn_decode_paged,
        "paged GQA decode attention (bf16, wave-per-kv-head)");
  m.def("attn_prefill_paged", &attn_prefill_paged,
        "paged causal prefill attention (bf16, MFMA)", py::arg("out"),
        py::arg("q"), py::arg("cu_q"), py::arg("q_pos"), py::arg("kcache"),
        py::arg("vcache"), py::arg("block_tables"), py::arg("kv_lens"),
        py::arg("scale"), py::arg("swz") = -1);
  m.def("top_p_sample", &top_p_sample,
        "fused temperature softmax + top-p sampling (sort-free)");
  m.def("derive_seeds", &derive_seeds,
        "stateless (seed, position) mix for chained decode");
  m.def("top_p_sample_v2", &top_p_sample_v2,
        "gridded top-p sampler (vocab sliced across workgroups)");
  m.def("gemv_bf16", &gemv_bf16,
        "skinny-batch (M<=8) bf16 weight-streaming GEMV");
  m.def("gemm_skinny_bf16", &gemm_skinny_bf16,
        "skinny-M (M<=16) bf16 MFMA weight-streaming GEMM");
  m.def("moe_grouped_linear", &moe_grouped_linear,
        "grouped per-expert skinny GEMM (capture-safe MoE decode)");
}
