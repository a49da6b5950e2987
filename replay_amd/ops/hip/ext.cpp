// Python bindings for the replay_amd gfx950 HIP extension.
#include <torch/extension.h>
#include <vector>

std::vector<torch::Tensor> layer_norm_fwd(torch::Tensor x, torch::Tensor weight,
                                          torch::Tensor bias, double eps);
std::vector<torch::Tensor> layer_norm_bwd(torch::Tensor x, torch::Tensor dy,
                                          torch::Tensor weight, torch::Tensor mean,
                                          torch::Tensor rstd);
std::vector<torch::Tensor> ce_fwd(torch::Tensor logits, torch::Tensor labels,
                                  int64_t ignore_index);
torch::Tensor ce_bwd(torch::Tensor logits, torch::Tensor labels, torch::Tensor lse,
                     torch::Tensor grad_scale, torch::Tensor count, int64_t ignore_index,
                     bool inplace);
std::vector<torch::Tensor> attention_fwd(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                                         c10::optional<torch::Tensor> valid, double scale,
                                         bool causal, bool need_lse);
std::vector<torch::Tensor> attention_bwd(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                                         torch::Tensor out, torch::Tensor dout,
                                         torch::Tensor lse, c10::optional<torch::Tensor> valid,
                                         double scale, bool causal);
std::vector<torch::Tensor> threshold_compact(torch::Tensor scores, torch::Tensor thresholds,
                                             int64_t capacity,
                                             c10::optional<torch::Tensor> seen,
                                             int64_t col_offset);
std::vector<torch::Tensor> scored_topk_gemm(torch::Tensor q, torch::Tensor w,
                                            torch::Tensor thresholds, int64_t capacity);
std::vector<torch::Tensor> attention_fwd_mfma(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                                              c10::optional<torch::Tensor> valid, double scale,
                                              bool causal, bool need_lse);
std::vector<torch::Tensor> attention_bwd_mfma(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                                              torch::Tensor out, torch::Tensor dout,
                                              torch::Tensor lse,
                                              c10::optional<torch::Tensor> valid, double scale,
                                              bool causal);
std::vector<torch::Tensor> ce_linear_fwd(torch::Tensor hidden, torch::Tensor w,
                                         torch::Tensor labels);
torch::Tensor ce_linear_lse(torch::Tensor hidden, torch::Tensor w);
std::vector<torch::Tensor> ce_linear_bwd(torch::Tensor hidden, torch::Tensor w,
                                         torch::Tensor labels, torch::Tensor lse,
                                         torch::Tensor gscale, double gsign);
torch::Tensor ce_linear_wgrad(torch::Tensor hidden, torch::Tensor w, torch::Tensor labels,
                              torch::Tensor lse, torch::Tensor gscale, double gsign);
torch::Tensor ce_linear_bwd_fused_dh(torch::Tensor hidden, torch::Tensor w,
                                     torch::Tensor labels, torch::Tensor lse,
                                     torch::Tensor gscale, double gsign);
torch::Tensor metrics_reduce(torch::Tensor preds, torch::Tensor gt,
                             c10::optional<torch::Tensor> train, torch::Tensor ks);
std::vector<torch::Tensor> scored_topk_gemm_fp8(torch::Tensor q, torch::Tensor w,
                                                torch::Tensor thresholds, int64_t capacity);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("layer_norm_fwd", &layer_norm_fwd, "fused LayerNorm forward (gfx950)");
  m.def("layer_norm_bwd", &layer_norm_bwd, "fused LayerNorm backward (gfx950)");
  m.def("ce_fwd", &ce_fwd, "fused softmax-CE forward (gfx950)");
  m.def("ce_bwd", &ce_bwd, "fused softmax-CE backward (gfx950)");
  m.def("attention_fwd", &attention_fwd, "fused attention forward (gfx950)");
  m.def("attention_bwd", &attention_bwd, "fused attention backward (gfx950)");
  m.def("threshold_compact", &threshold_compact, "top-k threshold compaction (gfx950)");
  m.def("scored_topk_gemm", &scored_topk_gemm,
        "fused MFMA score-GEMM + top-k candidate selection (gfx950)");
  m.def("attention_fwd_mfma", &attention_fwd_mfma, "MFMA attention forward (gfx950)");
  m.def("attention_bwd_mfma", &attention_bwd_mfma, "MFMA attention backward (gfx950)");
  m.def("ce_linear_lse", &ce_linear_lse,
        "LSE-only fused linear forward (no label capture) (gfx950)");
  m.def("ce_linear_fwd", &ce_linear_fwd,
        "fused linear+CE forward: online LSE in the GEMM epilogue (gfx950)");
  m.def("ce_linear_bwd", &ce_linear_bwd,
        "fused linear+CE backward: recomputed dlogits + fused dhidden (gfx950)");
  m.def("ce_linear_wgrad", &ce_linear_wgrad,
        "item-owner dW accumulation without materializing dlogits (gfx950)");
  m.def("ce_linear_bwd_fused_dh", &ce_linear_bwd_fused_dh,
        "dhidden-only backward (no dlogits stores; pairs with ce_linear_wgrad)");
  m.def("scored_topk_gemm_fp8", &scored_topk_gemm_fp8,
        "fused e4m3 MFMA score-GEMM + top-k candidate selection (gfx950)");
  m.def("metrics_reduce", &metrics_reduce,
        "fused ranking-metric sums: hits + all cutoffs in one launch (gfx950)");
}
