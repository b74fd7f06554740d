// Torch-extension bindings for the photon_amd CDNA4 HIP kernels.
// Built in-tree as photon_amd/ops/_photon_hip.so (PYTORCH_ROCM_ARCH=gfx950);
// the .so travels with the repo snapshot to GPU boxes.

#include <torch/extension.h>

#include <vector>

namespace photon_hip {

std::vector<torch::Tensor> layernorm_fwd(torch::Tensor x, torch::Tensor w,
                                         c10::optional<torch::Tensor> b,
                                         double eps,
                                         c10::optional<torch::Tensor> residual);
std::vector<torch::Tensor> layernorm_bwd(torch::Tensor dy, torch::Tensor x,
                                         torch::Tensor w, torch::Tensor mean,
                                         torch::Tensor rstd,
                                         c10::optional<torch::Tensor> dresid);
torch::Tensor ce_fwd_bwd_inplace(torch::Tensor logits, torch::Tensor targets);
void adamw_step(std::vector<torch::Tensor> ps, std::vector<torch::Tensor> gs,
                std::vector<torch::Tensor> ms, std::vector<torch::Tensor> vs,
                double lr, double b1, double b2, double eps, double wd,
                double bc1, double bc2,
                std::vector<torch::Tensor> masters);
void adopt_step(std::vector<torch::Tensor> ps, std::vector<torch::Tensor> gs,
                std::vector<torch::Tensor> ms, std::vector<torch::Tensor> vs,
                double lr, double b1, double b2, double eps, double wd,
                double clip, long step,
                std::vector<torch::Tensor> masters);
torch::Tensor multi_tensor_l2norm(std::vector<torch::Tensor> gs);
void multi_tensor_scale_clip(std::vector<torch::Tensor> gs,
                             torch::Tensor total_norm, double max_norm);
std::vector<torch::Tensor> attn_fwd_qkv(torch::Tensor qkv, long H,
                                        torch::Tensor slopes, bool causal);
torch::Tensor attn_bwd_qkv(torch::Tensor dout, torch::Tensor qkv, long H,
                           torch::Tensor slopes, torch::Tensor o,
                           torch::Tensor lse, bool causal);
std::vector<torch::Tensor> attn_fwd_launch(torch::Tensor q, torch::Tensor k,
                                           torch::Tensor v,
                                           torch::Tensor slopes, bool causal);
torch::Tensor mfma_probe(torch::Tensor A, torch::Tensor BT);
torch::Tensor pack_probe(torch::Tensor dummy);
std::vector<torch::Tensor> attn_bwd_launch(torch::Tensor dout, torch::Tensor q,
                                           torch::Tensor k, torch::Tensor v,
                                           torch::Tensor slopes,
                                           torch::Tensor o, torch::Tensor lse,
                                           bool causal);
std::vector<torch::Tensor> lt_linear_fwd(torch::Tensor x, torch::Tensor w,
                                         c10::optional<torch::Tensor> bias,
                                         bool gelu);
std::vector<torch::Tensor> lt_linear_bwd_dx(torch::Tensor dy, torch::Tensor w,
                                            c10::optional<torch::Tensor> aux_z,
                                            bool want_bgrad);
std::vector<torch::Tensor> lt_linear_bwd_dw(torch::Tensor x, torch::Tensor dy,
                                            bool want_bgrad);
torch::Tensor bias_grad(torch::Tensor dy);
torch::Tensor fused_linear(torch::Tensor x, torch::Tensor w,
                           c10::optional<torch::Tensor> b);

}  // namespace photon_hip

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  using namespace photon_hip;
  m.def("layernorm_fwd", &layernorm_fwd,
        "fused LayerNorm forward (optional fused residual add)",
        py::arg("x"), py::arg("w"), py::arg("b"), py::arg("eps"),
        py::arg("residual") = c10::nullopt);
  m.def("layernorm_bwd", &layernorm_bwd,
        "fused LayerNorm backward (optional residual-grad add-through)",
        py::arg("dy"), py::arg("x"), py::arg("w"), py::arg("mean"),
        py::arg("rstd"), py::arg("dresid") = c10::nullopt);
  m.def("ce_fwd_bwd_inplace", &ce_fwd_bwd_inplace,
        "fused CE loss + in-place softmax gradient");
  m.def("adamw_step", &adamw_step, "fused multi-tensor AdamW step");
  m.def("adopt_step", &adopt_step, "fused multi-tensor ADOPT step");
  m.def("multi_tensor_l2norm", &multi_tensor_l2norm);
  m.def("multi_tensor_scale_clip", &multi_tensor_scale_clip);
  m.def("attn_fwd", &attn_fwd_launch, "flash attention forward (ALiBi fused)");
  m.def("attn_bwd", &attn_bwd_launch, "flash attention backward");
  m.def("lt_linear_fwd", &lt_linear_fwd,
        "hipblaslt linear fwd (bias/GELU epilogue)");
  m.def("lt_linear_bwd_dx", &lt_linear_bwd_dx,
        "hipblaslt linear bwd dx (DGELU/BGRAD epilogue)");
  m.def("lt_linear_bwd_dw", &lt_linear_bwd_dw,
        "hipblaslt linear bwd dW (BGRADB epilogue)");
  m.def("bias_grad", &bias_grad, "two-stage deterministic bias-grad reduce");
  m.def("fused_linear", &fused_linear,
        "C++ autograd linear: tuned GEMMs + HIP bias-grad backward");
  m.def("attn_fwd_qkv", &attn_fwd_qkv,
        "flash attention forward on packed [B,S,3HD] qkv");
  m.def("attn_bwd_qkv", &attn_bwd_qkv,
        "flash attention backward on packed qkv; returns dqkv");
  m.def("mfma_probe", &mfma_probe);
  m.def("pack_probe", &pack_probe);
}
