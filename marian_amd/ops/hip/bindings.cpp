// pybind bindings for the marian_amd gfx950 kernels.
#include <torch/extension.h>

std::vector<torch::Tensor> layernorm_fwd(torch::Tensor x, torch::Tensor gamma,
                                         torch::Tensor beta, double eps);
std::vector<torch::Tensor> layernorm_res_fwd(torch::Tensor x,
                                             torch::Tensor res,
                                             torch::Tensor gamma,
                                             torch::Tensor beta, double eps);
std::vector<torch::Tensor> layernorm_bwd(torch::Tensor x, torch::Tensor gamma,
                                         torch::Tensor mean, torch::Tensor rstd,
                                         torch::Tensor dy);
torch::Tensor softmax_fwd(torch::Tensor x);
torch::Tensor log_softmax_fwd(torch::Tensor x);
torch::Tensor softmax_bwd(torch::Tensor y, torch::Tensor dy);
torch::Tensor log_softmax_bwd(torch::Tensor y, torch::Tensor dy);
std::vector<torch::Tensor> ce_fwd(torch::Tensor logits, torch::Tensor labels,
                                  double eps, c10::optional<torch::Tensor> mask);
torch::Tensor ce_bwd(torch::Tensor logits, torch::Tensor labels,
                     torch::Tensor lse, torch::Tensor dloss, double eps,
                     c10::optional<torch::Tensor> mask);
void adam_step(torch::Tensor p, torch::Tensor g, torch::Tensor m,
               torch::Tensor v, double lr, double b1, double b2, double eps,
               double wd, int64_t t, double gscale,
               c10::optional<torch::Tensor> gscale_t,
               c10::optional<torch::Tensor> lr_t,
               c10::optional<torch::Tensor> step_t);
std::vector<torch::Tensor> attention_fwd(torch::Tensor q, torch::Tensor k,
                                         torch::Tensor v,
                                         c10::optional<torch::Tensor> kv_lens,
                                         bool causal, double scale,
                                         bool need_lse);
torch::Tensor lt_linear_act(torch::Tensor x, torch::Tensor W,
                            torch::Tensor b, bool relu);
std::vector<torch::Tensor> lt_dwt_bgrad(torch::Tensor x, torch::Tensor dy,
                                        bool want_db);
torch::Tensor lt_mm(torch::Tensor A, torch::Tensor B);
torch::Tensor wgrad_bf16(torch::Tensor x, torch::Tensor dy, int64_t splits);
std::vector<torch::Tensor> wgrad_bias_bf16(torch::Tensor x, torch::Tensor dy,
                                           int64_t splits, bool want_db);
std::vector<torch::Tensor> fp8_quantize(torch::Tensor x,
                                        torch::Tensor scale);
torch::Tensor col_sum(torch::Tensor x);
torch::Tensor sumsq_finite(torch::Tensor g);
std::vector<torch::Tensor> beam_topk(torch::Tensor logprobs,
                                     torch::Tensor scores, int64_t n_beams,
                                     int64_t k, int64_t suppress);
std::vector<torch::Tensor> attention_bwd_into(
    torch::Tensor q, torch::Tensor k, torch::Tensor v, torch::Tensor o,
    torch::Tensor d_o, torch::Tensor lse, c10::optional<torch::Tensor> kv_lens,
    bool causal, double scale,
    torch::Tensor dq, torch::Tensor dk, torch::Tensor dv);
std::vector<torch::Tensor> attention_bwd(torch::Tensor q, torch::Tensor k,
                                         torch::Tensor v, torch::Tensor o,
                                         torch::Tensor d_o, torch::Tensor lse,
                                         c10::optional<torch::Tensor> kv_lens,
                                         bool causal, double scale);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("layernorm_fwd", &layernorm_fwd, "LayerNorm forward (gfx950)");
  m.def("layernorm_res_fwd", &layernorm_res_fwd,
        "fused residual-add + LayerNorm forward");
  m.def("layernorm_bwd", &layernorm_bwd, "LayerNorm backward (gfx950)");
  m.def("softmax_fwd", &softmax_fwd, "row softmax forward");
  m.def("log_softmax_fwd", &log_softmax_fwd, "row log-softmax forward");
  m.def("softmax_bwd", &softmax_bwd, "row softmax backward");
  m.def("log_softmax_bwd", &log_softmax_bwd, "row log-softmax backward");
  m.def("ce_fwd", &ce_fwd, "fused label-smoothed CE forward");
  m.def("ce_bwd", &ce_bwd, "fused label-smoothed CE backward");
  m.def("adam_step", &adam_step, "fused Adam step",
        py::arg("p"), py::arg("g"), py::arg("m"), py::arg("v"),
        py::arg("lr"), py::arg("b1"), py::arg("b2"), py::arg("eps"),
        py::arg("wd"), py::arg("t"), py::arg("gscale"),
        py::arg("gscale_t") = c10::nullopt,
        py::arg("lr_t") = c10::nullopt,
        py::arg("step_t") = c10::nullopt);
  m.def("attention_fwd", &attention_fwd,
        "fused flash-style attention forward (bf16, Dh=64)");
  m.def("attention_bwd_into", &attention_bwd_into,
        "fused attention backward writing into given strided views");
  m.def("attention_bwd", &attention_bwd,
        "fused flash-style attention backward (bf16, Dh=64)");
  m.def("lt_mm", &lt_mm, "tuned bf16 GEMM (full algo pool, split-K)");
  m.def("wgrad_bf16", &wgrad_bf16,
        "hand-MFMA split-K weight-grad GEMM dW = x^T dy",
        py::arg("x"), py::arg("dy"), py::arg("splits") = 0);
  m.def("wgrad_bias_bf16", &wgrad_bias_bf16,
        "hand-MFMA wgrad with fused fp32 bias grad",
        py::arg("x"), py::arg("dy"), py::arg("splits") = 0,
        py::arg("want_db") = true);
  m.def("lt_linear_act", &lt_linear_act,
        "hipBLASLt GEMM with fused bias(+ReLU) epilogue");
  m.def("lt_dwt_bgrad", &lt_dwt_bgrad,
        "hipBLASLt dW^T GEMM with fused bias-gradient (BGRADB) epilogue");
  m.def("fp8_quantize", &fp8_quantize,
        "fused e4m3fn quantize + amax (delayed scaling)");
  m.def("col_sum", &col_sum, "column sum (bias gradients)");
  m.def("sumsq_finite", &sumsq_finite, "fused grad sumsq + nonfinite count");
  m.def("beam_topk", &beam_topk,
        "fused beam-search top-k (path-score add + UNK suppression)");
}
