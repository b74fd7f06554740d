// Fused softmax cross-entropy over the vocab (50368) for CDNA4.
//
// ce_fwd_bwd_inplace: given a logits chunk [N, V] and targets [N], computes
// per-row loss = logsumexp(logits) - logits[target] (fp32) and OVERWRITES
// logits with the softmax gradient (softmax - onehot) in the input dtype.
// The surrounding GEMMs (hipBLASLt) consume that in-place gradient, so the
// full-precision logits never round-trip to HBM twice.
//
// v3: TWO passes. v2's three-pass structure assumed the 100 KB row stayed
// in the XCD L2 between passes, but at full occupancy 256 resident blocks
// per XCD hold 25 MB of rows against a 4 MB L2 — every pass went to HBM
// (measured: 4.5 ms per [65536, 50368] call = the 3R+2W HBM bound).
// Pass A reads the row ONCE with an online max+sum (rescale-by-exp on max
// growth, then a two-step block combine: max first, then sums scaled by
// exp(m_i - gm)); pass B re-reads and writes the gradient in one sweep.
// 2R + 1W = the minimum for an in-place grad without holding the row.

#include "host_common.h"

namespace photon_hip {

template <typename T, int BLOCK>
__global__ __launch_bounds__(BLOCK) void ce_fwd_bwd_kernel(
    T* __restrict__ logits, const long* __restrict__ targets,
    float* __restrict__ losses, long V) {
  __shared__ float scratch[BLOCK / WAVE];
  const long row = blockIdx.x;
  T* lr = logits + row * V;
  const long tgt = targets[row];
  const float logit_tgt = load_f32<T>(lr, tgt);

  // pass A: ONE read, online max + rescaled sum per thread
  float m = -1e30f, s = 0.f;
#pragma unroll 2
  for (long i = (long)threadIdx.x * 8; i + 7 < V; i += (long)BLOCK * 8) {
    floatx4 a = load4<T>(lr + i);
    floatx4 b = load4<T>(lr + i + 4);
    const float lm = fmaxf(
        fmaxf(fmaxf(a.x, a.y), fmaxf(a.z, a.w)),
        fmaxf(fmaxf(b.x, b.y), fmaxf(b.z, b.w)));
    if (lm > m) {
      s *= __expf(m - lm);  // exp(-huge) == 0 on the first tile
      m = lm;
    }
    s += __expf(a.x - m) + __expf(a.y - m) + __expf(a.z - m) +
         __expf(a.w - m) + __expf(b.x - m) + __expf(b.y - m) +
         __expf(b.z - m) + __expf(b.w - m);
  }
  for (long i = (V / 8) * 8 + threadIdx.x; i < V; i += BLOCK) {
    const float v = load_f32<T>(lr, i);
    if (v > m) {
      s *= __expf(m - v);
      m = v;
    }
    s += __expf(v - m);
  }
  // combine (m, s) pairs: global max first, then sums scaled into it
  const float gm = block_reduce_max(m, scratch);
  __syncthreads();  // scratch reuse between the two reductions
  const float gs = block_reduce_sum(s * __expf(m - gm), scratch);
  const float inv = 1.f / gs;
  if (threadIdx.x == 0) {
    losses[row] = gm + __logf(gs) - logit_tgt;
  }

  // pass B: re-read, write p = exp(v - gm)/gs (- onehot) in one sweep
#pragma unroll 2
  for (long i = (long)threadIdx.x * 8; i + 7 < V; i += (long)BLOCK * 8) {
    floatx4 a = load4<T>(lr + i);
    floatx4 b = load4<T>(lr + i + 4);
    a.x = __expf(a.x - gm) * inv; a.y = __expf(a.y - gm) * inv;
    a.z = __expf(a.z - gm) * inv; a.w = __expf(a.w - gm) * inv;
    b.x = __expf(b.x - gm) * inv; b.y = __expf(b.y - gm) * inv;
    b.z = __expf(b.z - gm) * inv; b.w = __expf(b.w - gm) * inv;
    store4<T>(lr + i, a);
    store4<T>(lr + i + 4, b);
  }
  for (long i = (V / 8) * 8 + threadIdx.x; i < V; i += BLOCK) {
    store_f32<T>(lr, i, __expf(load_f32<T>(lr, i) - gm) * inv);
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    store_f32<T>(lr, tgt, load_f32<T>(lr, tgt) - 1.f);
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
