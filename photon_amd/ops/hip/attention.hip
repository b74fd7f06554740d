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
  // (QTF = 64 at every D — mirrors the kernel's ATT_QTF* constexprs)
  const int qtf = 64;
  // dq: double-buffered [64][D] k + v row images
  int imgs = kind == 0 ? 4 * (KBF * D * 2)
                       : (kind == 1 ? 4 * (64 * D * 2)
                                    : 4 * (qtf * D * 2) + 4 * qtf * 4);
  int bounce = WAVES * img;
  return std::max(imgs, bounce);
}

struct Strides {
  long bs, hs, rs;
};

// [B,H,S,D] contiguous layout
static Strides bhsd_strides(int H, int S, int D) {
  return {(long)H * S * D, (long)S * D, (long)D};
}
// [B,S,3,H,D] packed qkv layout (pointers pre-offset per section)
static Strides qkv_strides(int H, int S, int D) {
  return {(long)S * 3 * H * D, (long)D, (long)3 * H * D};
}
// [B,S,H,D] packed output layout
static Strides bshd_strides(int H, int S, int D) {
  return {(long)S * H * D, (long)D, (long)H * D};
}

static std::vector<torch::Tensor> fwd_common(
    const __bf16* q, const __bf16* k, const __bf16* v, torch::Tensor slopes,
    bool causal, int B, int H, int S, int D, Strides in, Strides out_s,
    torch::Tensor o, torch::Tensor ref_opts_tensor) {
  auto lse = torch::empty({B, H, S}, ref_opts_tensor.options().dtype(at::kFloat));
  auto slopes_f = slopes.to(ref_opts_tensor.device(), at::kFloat).contiguous();
  dim3 grid((S + WAVES * QB - 1) / (WAVES * QB), (long)B * H);
  const int lds = attn_lds_bytes(D, 0);
#define LAUNCH_FWD(DD)                                                        \
  hipLaunchKernelGGL((attn_fwd_kernel<DD>), grid, dim3(ATT_BLOCK), lds,       \
                     cur_stream(), q, k, v, slopes_f.data_ptr<float>(),       \
                     (__bf16*)o.data_ptr(), lse.data_ptr<float>(), S, H,      \
                     (int)causal, in.bs, in.hs, in.rs, out_s.bs, out_s.hs,    \
                     out_s.rs)
  if (D == 64) {
    LAUNCH_FWD(64);
  } else {
    LAUNCH_FWD(128);
  }
#undef LAUNCH_FWD
  return {o, lse};
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
  auto st = bhsd_strides(H, S, D);
  return fwd_common((const __bf16*)q.data_ptr(), (const __bf16*)k.data_ptr(),
                    (const __bf16*)v.data_ptr(), slopes, causal, B, H, S, D,
                    st, st, o, q);
}

// Packed entry: qkv [B, S, 3*H*D] straight from the fused Wqkv projection —
// no .contiguous()/transpose copies; out is [B, S, H*D].
std::vector<torch::Tensor> attn_fwd_qkv(torch::Tensor qkv, long H,
                                        torch::Tensor slopes, bool causal) {
  TORCH_CHECK(qkv.scalar_type() == at::kBFloat16, "attn_fwd_qkv: bf16 only");
  TORCH_CHECK(qkv.dim() == 3 && qkv.is_contiguous(),
              "attn_fwd_qkv: expected contiguous [B,S,3*H*D]");
  const int B = qkv.size(0), S = qkv.size(1);
  const int D = (int)(qkv.size(2) / (3 * H));
  TORCH_CHECK(D == 64 || D == 128, "attn_fwd_qkv: d_head must be 64 or 128");
  auto o = torch::empty({(long)B, (long)S, H * (long)D}, qkv.options());
  auto in = qkv_strides(H, S, D);
  auto out_s = bshd_strides(H, S, D);
  const __bf16* base = (const __bf16*)qkv.data_ptr();
  return fwd_common(base, base + (long)H * D, base + 2L * H * D, slopes,
                    causal, B, (int)H, S, D, in, out_s, o, qkv);
}

static torch::Tensor delta_launch(torch::Tensor dout, torch::Tensor o,
                                  int B, int H, int S, int D, Strides st) {
  auto delta = torch::empty({(long)B, (long)H, (long)S},
                            dout.options().dtype(at::kFloat));
  const long n_rows = (long)B * S * H;
  const int rpw = 64 / (D / 4);
  long want = (n_rows + 4L * rpw - 1) / (4L * rpw);
  int G = (int)std::min<long>(std::max<long>(want, 64), 2048);
  if (D == 64) {
    hipLaunchKernelGGL((attn_delta_kernel<64>), dim3(G), dim3(256), 0,
                       cur_stream(), (const __bf16*)dout.data_ptr(),
                       (const __bf16*)o.data_ptr(), delta.data_ptr<float>(),
                       n_rows, S, H, st.bs, st.hs, st.rs);
  } else {
    hipLaunchKernelGGL((attn_delta_kernel<128>), dim3(G), dim3(256), 0,
                       cur_stream(), (const __bf16*)dout.data_ptr(),
                       (const __bf16*)o.data_ptr(), delta.data_ptr<float>(),
                       n_rows, S, H, st.bs, st.hs, st.rs);
  }
  return delta;
}

static void bwd_common(const __bf16* dout, const __bf16* q, const __bf16* k,
                       const __bf16* v, torch::Tensor slopes_f,
                       torch::Tensor lse, torch::Tensor delta, __bf16* dq,
                       __bf16* dk, __bf16* dv, bool causal, int B, int H,
                       int S, int D, Strides in, Strides out_s) {
  dim3 grid((S + WAVES * QB - 1) / (WAVES * QB), (long)B * H);
#define LAUNCH_BWD(DD)                                                        \
  hipLaunchKernelGGL((attn_bwd_dq_kernel<DD>), grid, dim3(ATT_BLOCK),         \
                     attn_lds_bytes(DD, 1), cur_stream(), dout, q, k, v,      \
                     slopes_f.data_ptr<float>(), lse.data_ptr<float>(),       \
                     delta.data_ptr<float>(), dq, S, H, (int)causal, in.bs,   \
                     in.hs, in.rs, out_s.bs, out_s.hs, out_s.rs);             \
  hipLaunchKernelGGL((attn_bwd_dkdv_kernel<DD>), grid, dim3(ATT_BLOCK),       \
                     attn_lds_bytes(DD, 2), cur_stream(), dout, q, k, v,      \
                     slopes_f.data_ptr<float>(), lse.data_ptr<float>(),       \
                     delta.data_ptr<float>(), dk, dv, S, H, (int)causal,      \
                     in.bs, in.hs, in.rs, out_s.bs, out_s.hs, out_s.rs)
  if (D == 64) {
    LAUNCH_BWD(64);
  } else {
    LAUNCH_BWD(128);
  }
#undef LAUNCH_BWD
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
  auto st = bhsd_strides(H, S, D);
  auto delta = delta_launch(dout.contiguous(), o, B, H, S, D, st);
  bwd_common((const __bf16*)dout.data_ptr(), (const __bf16*)q.data_ptr(),
             (const __bf16*)k.data_ptr(), (const __bf16*)v.data_ptr(),
             slopes_f, lse, delta, (__bf16*)dq.data_ptr(),
             (__bf16*)dk.data_ptr(), (__bf16*)dv.data_ptr(), causal, B, H, S,
             D, st, st);
  return {dq, dk, dv};
}

// Packed backward: dout/o are [B,S,H*D], qkv [B,S,3*H*D]; returns dqkv in
// the SAME packed layout (the autograd cat of dq/dk/dv disappears).
torch::Tensor attn_bwd_qkv(torch::Tensor dout, torch::Tensor qkv, long H,
                           torch::Tensor slopes, torch::Tensor o,
                           torch::Tensor lse, bool causal) {
  const int B = qkv.size(0), S = qkv.size(1);
  const int D = (int)(qkv.size(2) / (3 * H));
  auto dqkv = torch::empty_like(qkv);
  auto slopes_f = slopes.to(qkv.device(), at::kFloat).contiguous();
  auto in = qkv_strides((int)H, S, D);
  auto out_s = bshd_strides((int)H, S, D);
  // delta = rowsum(dO * O) per (b,h,s) straight off the packed layout
  auto delta = delta_launch(dout, o, B, (int)H, S, D, out_s);
  const __bf16* base = (const __bf16*)qkv.data_ptr();
  __bf16* dbase = (__bf16*)dqkv.data_ptr();
  bwd_common((const __bf16*)dout.data_ptr(), base, base + (long)H * D,
             base + 2L * H * D, slopes_f, lse, delta, dbase,
             dbase + (long)H * D, dbase + 2L * H * D, causal, B, (int)H, S,
             D, in, out_s);
  return dqkv;
}

}  // namespace photon_hip
