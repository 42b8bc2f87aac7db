// Python bindings for the in-tree CDNA4 (gfx950) kernels.
// Built ahead of time via `python setup.py build_ext --inplace`
// (PYTORCH_ROCM_ARCH=gfx950); the resulting _C.so lives inside the
// package so it travels with repo snapshots to GPU boxes.

#include <torch/extension.h>

std::vector<torch::Tensor> layernorm_fwd(torch::Tensor x, torch::Tensor w,
                                         torch::Tensor b, double eps);
std::vector<torch::Tensor> layernorm_bwd(torch::Tensor dy, torch::Tensor x,
                                         torch::Tensor w, torch::Tensor mean,
                                         torch::Tensor rstd);
std::vector<torch::Tensor> layernorm_add_fwd(torch::Tensor x,
                                             torch::Tensor res,
                                             torch::Tensor w, torch::Tensor b,
                                             double eps);
std::vector<torch::Tensor> layernorm_add_bwd(torch::Tensor dy,
                                             torch::Tensor dsum,
                                             torch::Tensor sum,
                                             torch::Tensor w,
                                             torch::Tensor mean,
                                             torch::Tensor rstd);

void fused_adamw(std::vector<torch::Tensor> params,
                 std::vector<torch::Tensor> grads,
                 std::vector<torch::Tensor> exp_avgs,
                 std::vector<torch::Tensor> exp_avg_sqs,
                 std::vector<c10::optional<torch::Tensor>> mirrors,
                 double lr,
                 double beta1, double beta2, double eps, double weight_decay,
                 double bias_c1, double bias_c2,
                 c10::optional<torch::Tensor> grad_scale,
                 double grad_prescale);
torch::Tensor multi_tensor_sqnorm(std::vector<torch::Tensor> tensors);
void multi_tensor_scale(std::vector<torch::Tensor> tensors, double factor);
void multi_tensor_scale_tensor(std::vector<torch::Tensor> tensors,
                               torch::Tensor factor);

std::vector<torch::Tensor> cross_entropy_fwd(torch::Tensor logits,
                                             torch::Tensor target);
torch::Tensor cross_entropy_bwd(torch::Tensor dloss, torch::Tensor logits,
                                torch::Tensor target, torch::Tensor lse);

std::vector<torch::Tensor> fmha_fwd(torch::Tensor q, torch::Tensor k,
                                    torch::Tensor v, double scale,
                                    double p_drop, long seed);
std::vector<torch::Tensor> fmha_bwd(torch::Tensor dout, torch::Tensor q,
                                    torch::Tensor k, torch::Tensor v,
                                    torch::Tensor o, torch::Tensor lse,
                                    double scale, double p_drop, long seed);
std::vector<torch::Tensor> fmha_fwd_qkv(torch::Tensor qkv, long num_heads,
                                        double scale, double p_drop,
                                        long seed);
torch::Tensor fmha_bwd_qkv(torch::Tensor dout, torch::Tensor qkv,
                           torch::Tensor o, torch::Tensor lse,
                           long num_heads, double scale, double p_drop,
                           long seed);
torch::Tensor mfma_probe(torch::Tensor a, torch::Tensor b);
torch::Tensor mfma32_probe(torch::Tensor a, torch::Tensor b);
torch::Tensor tr16_probe(long mode);
std::vector<torch::Tensor> wgrad_gemm(torch::Tensor a, torch::Tensor b,
                                      bool with_bias);
torch::Tensor fwd_gemm(torch::Tensor x, torch::Tensor w,
                       c10::optional<torch::Tensor> bias);
std::vector<torch::Tensor> fwd_gemm_gelu(torch::Tensor x, torch::Tensor w,
                                         torch::Tensor bias);
torch::Tensor lt_gemm(torch::Tensor a, torch::Tensor b, long algo_index,
                      c10::optional<torch::Tensor> bias);
std::vector<torch::Tensor> lt_gemm_gelu(torch::Tensor a, torch::Tensor b,
                                        c10::optional<torch::Tensor> bias,
                                        long algo_index);
std::vector<torch::Tensor> lt_gemm_dgelu_bgrad(torch::Tensor dy,
                                               torch::Tensor w,
                                               torch::Tensor aux,
                                               long algo_index);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "MI355X-native CDNA4 kernels for the FSDP ViT framework";
  m.def("layernorm_fwd", &layernorm_fwd, "LayerNorm forward (bf16, fp32 stats)");
  m.def("layernorm_bwd", &layernorm_bwd, "LayerNorm backward");
  m.def("layernorm_add_fwd", &layernorm_add_fwd,
        "fused residual-add + LayerNorm forward");
  m.def("layernorm_add_bwd", &layernorm_add_bwd,
        "fused residual-add + LayerNorm backward");
  m.def("fused_adamw", &fused_adamw,
        "multi-tensor AdamW step (fp32 master, fp32/bf16 grads with "
        "folded prescale + deferred clip)",
        py::arg("params"), py::arg("grads"), py::arg("exp_avgs"),
        py::arg("exp_avg_sqs"), py::arg("mirrors"), py::arg("lr"),
        py::arg("beta1"), py::arg("beta2"), py::arg("eps"),
        py::arg("weight_decay"), py::arg("bias_c1"), py::arg("bias_c2"),
        py::arg("grad_scale") = py::none(), py::arg("grad_prescale") = 1.0);
  m.def("multi_tensor_sqnorm", &multi_tensor_sqnorm,
        "sum of squares over tensor list");
  m.def("multi_tensor_scale", &multi_tensor_scale, "in-place scalar scale");
  m.def("multi_tensor_scale_tensor", &multi_tensor_scale_tensor,
        "in-place scale by device scalar");
  m.def("cross_entropy_fwd", &cross_entropy_fwd, "fused softmax CE forward");
  m.def("cross_entropy_bwd", &cross_entropy_bwd, "fused softmax CE backward");
  m.def("fmha_fwd", &fmha_fwd,
        "flash attention forward (bf16, head_dim<=192; in-kernel "
        "attention dropout via a counter-hash mask)",
        py::arg("q"), py::arg("k"), py::arg("v"), py::arg("scale"),
        py::arg("p_drop") = 0.0, py::arg("seed") = 0);
  m.def("fmha_bwd", &fmha_bwd,
        "flash attention backward (fused FA2-style; regenerates the "
        "dropout mask from the seed)",
        py::arg("dout"), py::arg("q"), py::arg("k"), py::arg("v"),
        py::arg("o"), py::arg("lse"), py::arg("scale"),
        py::arg("p_drop") = 0.0, py::arg("seed") = 0);
  m.def("fmha_fwd_qkv", &fmha_fwd_qkv,
        "flash attention forward on the fused [B,T,3,H,D] qkv projection "
        "(zero-copy strided IO, O returned as [B,T,E])",
        py::arg("qkv"), py::arg("num_heads"), py::arg("scale"),
        py::arg("p_drop") = 0.0, py::arg("seed") = 0);
  m.def("fmha_bwd_qkv", &fmha_bwd_qkv,
        "flash attention backward producing the fused dqkv [B,T,3,H,D]",
        py::arg("dout"), py::arg("qkv"), py::arg("o"), py::arg("lse"),
        py::arg("num_heads"), py::arg("scale"), py::arg("p_drop") = 0.0,
        py::arg("seed") = 0);
  m.def("mfma_probe", &mfma_probe, "16x16x32 bf16 MFMA layout probe");
  m.def("mfma32_probe", &mfma32_probe, "32x32x16 bf16 MFMA layout probe");
  m.def("tr16_probe", &tr16_probe, "ds_read_b64_tr_b16 semantics probe");
  m.def("wgrad_gemm", &wgrad_gemm,
        "C = A^T B weight-gradient GEMM (bf16, tr16 transpose reads)");
  m.def("fwd_gemm", &fwd_gemm,
        "C = X W^T forward Linear GEMM (+fused bias), hand-written CDNA4 "
        "MFMA (csrc/fgemm.hip)",
        py::arg("x"), py::arg("w"), py::arg("bias") = py::none());
  m.def("fwd_gemm_gelu", &fwd_gemm_gelu,
        "(gelu(X W^T + bias), pre-activation) with the EXACT erf GELU "
        "fused into the hand-written forward GEMM epilogue",
        py::arg("x"), py::arg("w"), py::arg("bias"));
  m.def("lt_gemm", &lt_gemm,
        "row-major bf16 GEMM (+optional fused bias epilogue) via "
        "hipblaslt-ext with an explicit algorithm index (-1 = library "
        "heuristic); offline-search apply path, see "
        "csrc/tools/hipblaslt_search.cpp",
        py::arg("a"), py::arg("b"), py::arg("algo_index") = -1,
        py::arg("bias") = py::none());
  m.def("lt_gemm_gelu", &lt_gemm_gelu,
        "(gelu(a@b+bias), pre-activation aux) in one hipblaslt-ext GEMM "
        "(GELU_AUX epilogue; tanh-approx GELU)",
        py::arg("a"), py::arg("b"), py::arg("bias") = py::none(),
        py::arg("algo_index") = -1);
  m.def("lt_gemm_dgelu_bgrad", &lt_gemm_dgelu_bgrad,
        "(dgelu(dy@w, aux), column-sum bias grad) in one hipblaslt-ext "
        "GEMM (DGELU_BGRAD epilogue)",
        py::arg("dy"), py::arg("w"), py::arg("aux"),
        py::arg("algo_index") = -1);
}
