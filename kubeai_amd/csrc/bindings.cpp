// Python bindings for the kubeai_amd gfx950 HIP kernels.
#include <torch/extension.h>

void rmsnorm(torch::Tensor out, torch::Tensor x, torch::Tensor weight,
             double eps);
void fused_add_rmsnorm(torch::Tensor x, torch::Tensor residual,
                       torch::Tensor weight, double eps);
void rope(torch::Tensor q, torch::Tensor k, torch::Tensor positions,
          torch::Tensor cos_sin);
void reshape_and_cache_fp8(torch::Tensor k, torch::Tensor v,
                           torch::Tensor k_cache, torch::Tensor v_cache,
                           torch::Tensor slot_mapping);
void reshape_and_cache(torch::Tensor k, torch::Tensor v, torch::Tensor k_cache,
                       torch::Tensor v_cache, torch::Tensor slot_mapping);
void paged_attention_decode(torch::Tensor out, torch::Tensor q,
                            torch::Tensor k_cache, torch::Tensor v_cache,
                            torch::Tensor block_tables, torch::Tensor seq_lens,
                            double scale, int64_t window, double softcap);
void paged_attention_prefill(torch::Tensor out, torch::Tensor q,
                             torch::Tensor k_cache, torch::Tensor v_cache,
                             torch::Tensor block_tables,
                             torch::Tensor query_start_loc,
                             torch::Tensor seq_lens, double scale,
                             int64_t window, double softcap);
void silu_and_mul(torch::Tensor out, torch::Tensor x);
void gelu_and_mul(torch::Tensor out, torch::Tensor x);
void skinny_gemm(torch::Tensor y, torch::Tensor x, torch::Tensor w);
void greedy_sample(torch::Tensor out, torch::Tensor logits);
void gumbel_sample(torch::Tensor out, torch::Tensor logits,
                   torch::Tensor temperature, torch::Tensor seeds,
                   int64_t step);
void sgmv(torch::Tensor y, torch::Tensor x, torch::Tensor A, torch::Tensor B,
          torch::Tensor idx, double scale);
void register_chwbl(pybind11::module_& m);
// one-shot fused all-reduce over xGMI (allreduce.hip)
pybind11::bytes xgmi_alloc();
void xgmi_connect(int64_t rank, int64_t world,
                  const std::vector<pybind11::bytes>& handles);
void xgmi_fused_allreduce_add_rmsnorm(torch::Tensor x, torch::Tensor residual,
                                      torch::Tensor weight, double eps);
int64_t xgmi_error_count();
int64_t xgmi_max_elems();
void xgmi_shutdown();
void rmsnorm_fp8(torch::Tensor out, torch::Tensor out_scale, torch::Tensor x,
                 torch::Tensor weight, double eps);
void fused_add_rmsnorm_fp8(torch::Tensor out, torch::Tensor out_scale,
                           torch::Tensor x, torch::Tensor residual,
                           torch::Tensor weight, double eps);
void silu_and_mul_fp8(torch::Tensor out, torch::Tensor out_scale,
                      torch::Tensor x);
void quant_fp8(torch::Tensor out, torch::Tensor out_scale, torch::Tensor x);
void nucleus_stats(torch::Tensor m, torch::Tensor z, torch::Tensor logits,
                   torch::Tensor temps);
void nucleus_accept(torch::Tensor ok, torch::Tensor logits,
                    torch::Tensor cand, torch::Tensor m, torch::Tensor z,
                    torch::Tensor temps, torch::Tensor top_ps,
                    torch::Tensor top_ks);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rmsnorm", &rmsnorm, "RMSNorm (bf16, fp32 accum)");
  m.def("fused_add_rmsnorm", &fused_add_rmsnorm,
        "in-place residual add + RMSNorm");
  m.def("rope", &rope, "NeoX rotary embedding, in-place q/k");
  m.def("reshape_and_cache", &reshape_and_cache, "paged KV cache write");
  m.def("reshape_and_cache_fp8", &reshape_and_cache_fp8,
        "paged KV cache write, fp8 e5m2 cache");
  m.def("paged_attention_decode", &paged_attention_decode,
        "paged attention, one query token per seq");
  m.def("paged_attention_prefill", &paged_attention_prefill,
        "paged causal flash attention over cached KV");
  m.def("gelu_and_mul", &gelu_and_mul, "GeGLU tanh-gelu(gate)*up");
  m.def("silu_and_mul", &silu_and_mul, "SwiGLU activation");
  m.def("skinny_gemm", &skinny_gemm, "weight-streaming GEMM for M<=64");
  m.def("greedy_sample", &greedy_sample, "argmax sampling");
  m.def("gumbel_sample", &gumbel_sample, "temperature sampling (hash RNG)");
  m.def("rmsnorm_fp8", &rmsnorm_fp8, "rmsnorm with fused fp8 row quant");
  m.def("fused_add_rmsnorm_fp8", &fused_add_rmsnorm_fp8,
        "residual add + rmsnorm with fused fp8 row quant");
  m.def("silu_and_mul_fp8", &silu_and_mul_fp8, "SwiGLU with fused fp8 quant");
  m.def("quant_fp8", &quant_fp8, "bf16 -> fp8 row quant");
  m.def("nucleus_stats", &nucleus_stats, "per-row softmax max + Z");
  m.def("nucleus_accept", &nucleus_accept,
        "nucleus membership of sampled tokens (mass above < p, rank < k)");
  m.def("xgmi_alloc", &xgmi_alloc, "allocate + export the IPC comm buffer");
  m.def("xgmi_connect", &xgmi_connect, "open peer IPC buffers");
  m.def("xgmi_fused_allreduce_add_rmsnorm", &xgmi_fused_allreduce_add_rmsnorm,
        "one-shot xGMI all-reduce fused with residual add + RMSNorm");
  m.def("xgmi_error_count", &xgmi_error_count, "spin-wait timeout count");
  m.def("xgmi_max_elems", &xgmi_max_elems, "one-shot capacity (bf16 elems)");
  m.def("xgmi_shutdown", &xgmi_shutdown, "free/close comm buffers");
  m.def("sgmv", &sgmv, "segmented gather LoRA apply (one adapter segment)");
  register_chwbl(m);
}
