// Torch extension bindings for the gfx950 fused-op kernels.
#include <torch/extension.h>

#include <vector>

// layernorm.hip
std::vector<torch::Tensor> layernorm_fwd(torch::Tensor x, torch::Tensor w,
                                         torch::Tensor b, double eps);
std::vector<torch::Tensor> layernorm_bwd(torch::Tensor dy, torch::Tensor x,
                                         torch::Tensor w, torch::Tensor mean,
                                         torch::Tensor rstd);
std::vector<torch::Tensor> rmsnorm_fwd(torch::Tensor x, torch::Tensor w,
                                       double eps);
std::vector<torch::Tensor> rmsnorm_bwd(torch::Tensor dy, torch::Tensor x,
                                       torch::Tensor w, torch::Tensor rstd);
std::vector<torch::Tensor> layernorm_fwd_residual(torch::Tensor x,
                                                  torch::Tensor res,
                                                  torch::Tensor w,
                                                  torch::Tensor b, double eps);
std::vector<torch::Tensor> layernorm_bwd_residual(torch::Tensor dy,
                                                  torch::Tensor x,
                                                  torch::Tensor w,
                                                  torch::Tensor mean,
                                                  torch::Tensor rstd,
                                                  torch::Tensor dsum);
// elementwise.hip
torch::Tensor bias_gelu_fwd(torch::Tensor x, torch::Tensor bias);
std::vector<torch::Tensor> bias_gelu_bwd(torch::Tensor dy, torch::Tensor x,
                                         torch::Tensor bias);
void adamw_flat(torch::Tensor master, torch::Tensor grad, torch::Tensor m,
                torch::Tensor v, torch::Tensor model, double lr, double beta1,
                double beta2, double eps, double wd, long step);
torch::Tensor rope_fwd(torch::Tensor x, torch::Tensor cos, torch::Tensor sin);
torch::Tensor colsum(torch::Tensor x);
void grad_sumsq(torch::Tensor x, torch::Tensor out, long slot);
// softmax.hip
torch::Tensor softmax_causal_fwd(torch::Tensor s, double scale);
torch::Tensor softmax_causal_bwd(torch::Tensor dy, torch::Tensor y,
                                 double scale);
torch::Tensor softmax_bias_fwd(torch::Tensor s, torch::Tensor bias,
                               long inner, long outer, double scale);
std::vector<torch::Tensor> cross_entropy_fwd(torch::Tensor logits,
                                             torch::Tensor labels,
                                             long ignore_index);
torch::Tensor cross_entropy_bwd(torch::Tensor dloss, torch::Tensor logits,
                                torch::Tensor labels, torch::Tensor lse,
                                long ignore_index);
torch::Tensor row_max(torch::Tensor x);
torch::Tensor row_sumexp(torch::Tensor x, torch::Tensor gmax);
torch::Tensor gather_label_logit(torch::Tensor x, torch::Tensor labels,
                                 long start, long ignore_index);
torch::Tensor vp_ce_bwd(torch::Tensor dloss, torch::Tensor logits,
                        torch::Tensor labels, torch::Tensor lse, long start,
                        long ignore_index);
// attention.hip
std::vector<torch::Tensor> attn_fwd(torch::Tensor q, torch::Tensor k,
                                    torch::Tensor v, bool causal,
                                    double scale, double p_drop, long seed);
std::vector<torch::Tensor> attn_bwd(torch::Tensor dout, torch::Tensor q,
                                    torch::Tensor k, torch::Tensor v,
                                    torch::Tensor o, torch::Tensor lse,
                                    bool causal, double scale,
                                    double p_drop, long seed);
std::vector<torch::Tensor> attn_fwd_packed(torch::Tensor qkv, long num_heads,
                                           double scale, double p_drop,
                                           long seed);
torch::Tensor attn_bwd_packed(torch::Tensor dout, torch::Tensor qkv,
                              torch::Tensor o, torch::Tensor lse,
                              long num_heads, double scale, double p_drop,
                              long seed);
// topp.hip
std::vector<torch::Tensor> topp_select(torch::Tensor sorted_p,
                                       torch::Tensor sorted_idx,
                                       torch::Tensor top_p, torch::Tensor u);
// moe.hip
std::vector<torch::Tensor> moe_dispatch(torch::Tensor x,
                                        torch::Tensor expert_ids,
                                        long num_experts, long top_k);
// mfma_probe.hip
torch::Tensor mfma_gemm16_probe(torch::Tensor a, torch::Tensor b);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("layernorm_fwd", &layernorm_fwd, "fused LayerNorm fwd (gfx950)");
  m.def("layernorm_bwd", &layernorm_bwd, "fused LayerNorm bwd");
  m.def("rmsnorm_fwd", &rmsnorm_fwd, "fused RMSNorm fwd");
  m.def("rmsnorm_bwd", &rmsnorm_bwd, "fused RMSNorm bwd");
  m.def("layernorm_fwd_residual", &layernorm_fwd_residual,
        "LN(a+b) fwd, also emits a+b");
  m.def("layernorm_bwd_residual", &layernorm_bwd_residual,
        "LN bwd with fused residual-grad addend");
  m.def("colsum", &colsum, "column sum [N,H] -> fp32 [H] (dbias)");
  m.def("grad_sumsq", &grad_sumsq, "fused sumsq (found_inf + grad norm)");
  m.def("bias_gelu_fwd", &bias_gelu_fwd, "fused bias+gelu fwd");
  m.def("bias_gelu_bwd", &bias_gelu_bwd, "fused bias+gelu bwd");
  m.def("adamw_flat", &adamw_flat, "fused AdamW on flat buffers");
  m.def("rope_fwd", &rope_fwd, "rotary embedding fwd");
  m.def("softmax_causal_fwd", &softmax_causal_fwd,
        "fused scale+causal-mask+softmax fwd");
  m.def("softmax_causal_bwd", &softmax_causal_bwd, "causal softmax bwd");
  m.def("softmax_bias_fwd", &softmax_bias_fwd,
        "fused scale+bias+softmax (gated attention core)");
  m.def("cross_entropy_fwd", &cross_entropy_fwd, "softmax CE fwd");
  m.def("cross_entropy_bwd", &cross_entropy_bwd, "softmax CE bwd");
  m.def("row_max", &row_max, "rowwise max (vocab-parallel CE)");
  m.def("row_sumexp", &row_sumexp, "rowwise sum(exp(x-m))");
  m.def("gather_label_logit", &gather_label_logit, "gather label logits");
  m.def("vp_ce_bwd", &vp_ce_bwd, "vocab-parallel CE bwd");
  m.def("moe_dispatch", &moe_dispatch,
        "fused MoE token dispatch (histogram+rank+scatter)");
  m.def("attn_fwd", &attn_fwd, "flash attention fwd (MFMA, causal)",
        py::arg("q"), py::arg("k"), py::arg("v"), py::arg("causal"),
        py::arg("scale"), py::arg("p_drop") = 0.0, py::arg("seed") = 0);
  m.def("attn_bwd", &attn_bwd, "flash attention bwd (MFMA)",
        py::arg("dout"), py::arg("q"), py::arg("k"), py::arg("v"),
        py::arg("o"), py::arg("lse"), py::arg("causal"), py::arg("scale"),
        py::arg("p_drop") = 0.0, py::arg("seed") = 0);
  m.def("attn_fwd_packed", &attn_fwd_packed,
        "flash attention fwd on packed QKV (no layout copies)",
        py::arg("qkv"), py::arg("num_heads"), py::arg("scale"),
        py::arg("p_drop") = 0.0, py::arg("seed") = 0);
  m.def("attn_bwd_packed", &attn_bwd_packed,
        "flash attention bwd writing packed dQKV",
        py::arg("dout"), py::arg("qkv"), py::arg("o"), py::arg("lse"),
        py::arg("num_heads"), py::arg("scale"),
        py::arg("p_drop") = 0.0, py::arg("seed") = 0);
  m.def("topp_select", &topp_select, "top-p nucleus cutoff + draw");
  m.def("mfma_gemm16_probe", &mfma_gemm16_probe,
        "debug: 16x16x32 MFMA fragment-layout probe");
}
