// C++ autograd Functions for the linear layers: torch's tuned GEMMs plus
// the HIP bias-grad reduce in backward, with ZERO per-call Python overhead.
//
// A Python torch.autograd.Function wrapper was measured first and rejected:
// ~384 Function invocations per MPT-125M step cost ~30 ms — more than the
// bias-reduce fusion saved. This C++ version dispatches straight to
// at::linear / at::mm and photon_hip::bias_grad.
//
// Autocast parity: inputs go through at::autocast::cached_cast exactly like
// at::linear under amp, so fp32 master weights keep their per-autocast-region
// bf16 cast cache and gradients flow back to fp32 through the cast nodes.

#include <ATen/autocast_mode.h>
#include <torch/extension.h>

namespace photon_hip {

torch::Tensor bias_grad(torch::Tensor dy);

namespace {

using torch::autograd::AutogradContext;
using torch::autograd::variable_list;

class FusedLinearFn : public torch::autograd::Function<FusedLinearFn> {
 public:
  static torch::Tensor forward(AutogradContext* ctx, torch::Tensor x,
                               torch::Tensor w,
                               c10::optional<torch::Tensor> b) {
    at::AutoDispatchBelowADInplaceOrView g;
    ctx->save_for_backward({x, w});
    ctx->saved_data["has_bias"] = b.has_value();
    return b.has_value() ? at::linear(x, w, *b) : at::linear(x, w);
  }

  static variable_list backward(AutogradContext* ctx, variable_list grads) {
    auto saved = ctx->get_saved_variables();
    auto x = saved[0];
    auto w = saved[1];
    auto dy = grads[0].contiguous();
    const bool has_bias = ctx->saved_data["has_bias"].toBool();
    // flatten leading dims for the GEMMs
    auto x2 = x.reshape({-1, x.size(-1)});
    auto dy2 = dy.reshape({-1, dy.size(-1)});
    auto dx = at::matmul(dy, w);
    auto dw = at::mm(dy2.t(), x2);
    torch::Tensor db;
    if (has_bias) {
      db = (dy2.is_cuda() && dy2.scalar_type() == at::kBFloat16 &&
            dy2.size(-1) % 4 == 0)
               ? bias_grad(dy2)
               : dy2.sum(0);
    }
    return {dx, dw, db};
  }
};

}  // namespace

// y = x @ W^T + b with autocast-cached bf16 casts (drop-in for at::linear
// on the forward; backward uses the HIP bias-grad reduce).
torch::Tensor fused_linear(torch::Tensor x, torch::Tensor w,
                           c10::optional<torch::Tensor> b) {
  if (at::autocast::is_autocast_enabled(at::kCUDA)) {
    const auto dt = at::autocast::get_autocast_dtype(at::kCUDA);
    x = at::autocast::cached_cast(dt, x, c10::DeviceType::CUDA);
    w = at::autocast::cached_cast(dt, w, c10::DeviceType::CUDA);
    if (b.has_value())
      b = at::autocast::cached_cast(dt, *b, c10::DeviceType::CUDA);
  }
  return FusedLinearFn::apply(x, w, b);
}

}  // namespace photon_hip
