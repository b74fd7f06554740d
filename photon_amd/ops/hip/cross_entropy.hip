// Fused softmax cross-entropy over the vocab (50368) for CDNA4.
//
// ce_fwd_bwd_inplace: given a logits chunk [N, V] and targets [N], computes
// per-row loss = logsumexp(logits) - logits[target] (fp32) and OVERWRITES
// logits with the softmax gradient (softmax - onehot) in the input dtype.
// The surrounding GEMMs (hipBLASLt) consume that in-place gradient, so the
// full-precision logits never round-trip to HBM twice.
//
// One 256-thread workgroup per row; 8-wide vector loads; online max+sum in
// a single pass, second pass writes the gradient. V needs no multiple-of-8
// padding handled here because vocab 50368 = 8 * 6296.

#include "host_common.h"

namespace photon_hip {

template <typename T, int BLOCK>
__global__ void ce_fwd_bwd_kernel(T* __restrict__ logits,
                                  const long* __restrict__ targets,
                                  float* __restrict__ losses, long V) {
  __shared__ float scratch[BLOCK / WAVE];
  const long row = blockIdx.x;
  T* lr = logits + row * V;
  const long tgt = targets[row];

  // pass 1: online max & sum(exp(x - max))
  float m = -INFINITY, s = 0.f;
  for (long i = threadIdx.x * 8; i < V; i += BLOCK * 8) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      if (i + j < V) {
        float v = load_f32<T>(lr, i + j);
        if (v > m) {
          s *= __expf(m - v);
          m = v;
        }
        s += __expf(v - m);
      }
    }
  }
  // combine across threads: global max then rescaled sums
  float gm = block_reduce_max(m, scratch);
  s *= __expf(m - gm);
  float gs = block_reduce_sum(s, scratch);
  const float lse = gm + __logf(gs);
  if (threadIdx.x == 0) {
    losses[row] = lse - load_f32<T>(lr, tgt);
  }
  __syncthreads();

  // pass 2: write gradient (softmax - onehot)
  const float inv = 1.f / gs;
  for (long i = threadIdx.x * 8; i < V; i += BLOCK * 8) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      if (i + j < V) {
        float v = load_f32<T>(lr, i + j);
        float p = __expf(v - gm) * inv;
        if (i + j == tgt) p -= 1.f;
        store_f32<T>(lr, i + j, p);
      }
    }
  }
}

// ---------------------------------------------------------------------------
torch::Tensor ce_fwd_bwd_inplace(torch::Tensor logits, torch::Tensor targets) {
  TORCH_CHECK(logits.dim() == 2 && logits.is_contiguous(),
              "ce_fwd_bwd: logits must be contiguous 2D");
  TORCH_CHECK(targets.scalar_type() == at::kLong, "targets must be int64");
  const long N = logits.size(0), V = logits.size(1);
  auto losses = torch::empty({N}, logits.options().dtype(at::kFloat));
  constexpr int CE_BLOCK = 256;
  DISPATCH_DTYPE(logits, "ce_fwd_bwd", {
    hipLaunchKernelGGL((ce_fwd_bwd_kernel<scalar_t, CE_BLOCK>), dim3(N),
                       dim3(CE_BLOCK), 0, cur_stream(),
                       (scalar_t*)logits.data_ptr(),
                       targets.data_ptr<long>(), losses.data_ptr<float>(), V);
  });
  return losses;
}

}  // namespace photon_hip
