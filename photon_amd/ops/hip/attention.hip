// Host launchers for the CDNA4 flash-attention kernels (device code in
// attn_kernels.h — kept torch-free so scripts/attn_bench.hip can compile it
// standalone with hipcc for fast kernel iteration).

#include "host_common.h"
#include "attn_kernels.h"

namespace photon_hip {

// ---------------------------------------------------------------------------
// Host launchers
// ---------------------------------------------------------------------------

static int attn_lds_bytes(int D, int kind) {
  // kind 0: fwd (double-buffered [KBF][D] k + v row images),
  // kind 1: dq (k + v row images), 2: dkdv (q + do row images + lse/delta).
  // Transposed A-fragments are hardware tr16 reads — no transposed images.
  const int img = 64 * D;  // bytes of one [32][D] bf16 image
  // dkdv: double-buffered [QTF][D] q + do images + 2*QTF lse/delta floats
  // (QTF = 64 at D<=64, 32 at D=128 — mirrors the kernel's constexpr)
  const int qtf = D <= 64 ? 64 : 32;
  int imgs = kind == 0 ? 4 * (KBF * D * 2)
                       : (kind == 1 ? 2 * img : 4 * (qtf * D * 2) + 4 * qtf * 4);
  int bounce = WAVES * img;
  return std::max(imgs, bounce);
}

std::vector<torch::Tensor> attn_fwd_launch(torch::Tensor q, torch::Tensor k,
                                           torch::Tensor v,
                                           torch::Tensor slopes, bool causal) {
  TORCH_CHECK(q.scalar_type() == at::kBFloat16,
              "attn_fwd: bf16 only (use impl='torch' for fp32)");
  TORCH_CHECK(q.dim() == 4, "attn_fwd: expected [B,H,S,D]");
  const int B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
  TORCH_CHECK(D == 64 || D == 128, "attn_fwd: d_head must be 64 or 128");
  TORCH_CHECK(q.is_contiguous() && k.is_contiguous() && v.is_contiguous());
  auto o = torch::empty_like(q);
  auto lse = torch::empty({B, H, S}, q.options().dtype(at::kFloat));
  auto slopes_f = slopes.to(q.device(), at::kFloat).contiguous();
  dim3 grid((S + WAVES * QB - 1) / (WAVES * QB), (long)B * H);
  const int lds = attn_lds_bytes(D, 0);
  if (D == 64) {
    hipLaunchKernelGGL((attn_fwd_kernel<64>), grid, dim3(ATT_BLOCK), lds,
                       cur_stream(), (const __bf16*)q.data_ptr(),
                       (const __bf16*)k.data_ptr(),
                       (const __bf16*)v.data_ptr(), slopes_f.data_ptr<float>(),
                       (__bf16*)o.data_ptr(), lse.data_ptr<float>(), S, H,
                       (int)causal);
  } else {
    hipLaunchKernelGGL((attn_fwd_kernel<128>), grid, dim3(ATT_BLOCK), lds,
                       cur_stream(), (const __bf16*)q.data_ptr(),
                       (const __bf16*)k.data_ptr(),
                       (const __bf16*)v.data_ptr(), slopes_f.data_ptr<float>(),
                       (__bf16*)o.data_ptr(), lse.data_ptr<float>(), S, H,
                       (int)causal);
  }
  return {o, lse};
}

std::vector<torch::Tensor> attn_bwd_launch(torch::Tensor dout, torch::Tensor q,
                                           torch::Tensor k, torch::Tensor v,
                                           torch::Tensor slopes,
                                           torch::Tensor o, torch::Tensor lse,
                                           bool causal) {
  const int B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
  auto dq = torch::empty_like(q);
  auto dk = torch::empty_like(k);
  auto dv = torch::empty_like(v);
  auto slopes_f = slopes.to(q.device(), at::kFloat).contiguous();
  // delta = rowsum(dO * O), fp32
  auto delta = (dout.to(at::kFloat) * o.to(at::kFloat)).sum(-1).contiguous();
  dim3 grid((S + WAVES * QB - 1) / (WAVES * QB), (long)B * H);
#define LAUNCH_BWD(DD)                                                        \
  hipLaunchKernelGGL((attn_bwd_dq_kernel<DD>), grid, dim3(ATT_BLOCK),         \
                     attn_lds_bytes(DD, 1), cur_stream(),                     \
                     (const __bf16*)dout.data_ptr(),                          \
                     (const __bf16*)q.data_ptr(),                             \
                     (const __bf16*)k.data_ptr(),                             \
                     (const __bf16*)v.data_ptr(), slopes_f.data_ptr<float>(), \
                     lse.data_ptr<float>(), delta.data_ptr<float>(),          \
                     (__bf16*)dq.data_ptr(), S, H, (int)causal);              \
  hipLaunchKernelGGL((attn_bwd_dkdv_kernel<DD>), grid, dim3(ATT_BLOCK),       \
                     attn_lds_bytes(DD, 2), cur_stream(),                     \
                     (const __bf16*)dout.data_ptr(),                          \
                     (const __bf16*)q.data_ptr(),                             \
                     (const __bf16*)k.data_ptr(),                             \
                     (const __bf16*)v.data_ptr(), slopes_f.data_ptr<float>(), \
                     lse.data_ptr<float>(), delta.data_ptr<float>(),          \
                     (__bf16*)dk.data_ptr(), (__bf16*)dv.data_ptr(), S, H,    \
                     (int)causal)
  if (D == 64) {
    LAUNCH_BWD(64);
  } else {
    LAUNCH_BWD(128);
  }
#undef LAUNCH_BWD
  return {dq, dk, dv};
}

}  // namespace photon_hip
