// Fused LayerNorm fwd/bwd for CDNA4 (gfx950).
//
// Forward: one 256-thread workgroup per row; bf16 (or f32) input loaded as
// 8-element vectors (G13: scalar bf16 loads are 2x slower), fp32 mean/rstd
// via wave64 + LDS block reductions. Replaces llm-foundry LPLayerNorm
// (reference SURVEY.md L133 row).
//
// Backward: dx per row in one kernel (two fused row-reductions); dweight /
// dbias via deterministic two-stage column reduction (per-block partials in
// a [GRID, D] fp32 buffer, reduced by a second kernel) — no atomics, so the
// backward is bit-deterministic run to run.

#include "host_common.h"

namespace photon_hip {

// ---------------------------------------------------------------------------
// Forward: y = (x - mean) * rstd * w + b ; saves mean, rstd (fp32 per row)
// ---------------------------------------------------------------------------
// D up to 8192 (MPT-7B d_model 4096): x stays in registers (<= 4 chunks of
// 8 per thread), so mean and variance are exact two-pass fp32 without
// re-reading HBM.
template <typename T, int BLOCK>
__global__ void layernorm_fwd_kernel(
    const T* __restrict__ x, const float* __restrict__ w,
    const float* __restrict__ b, T* __restrict__ y,
    float* __restrict__ mean_out, float* __restrict__ rstd_out, int D,
    float eps) {
  __shared__ float scratch[BLOCK / WAVE];
  const long row = blockIdx.x;
  const T* xr = x + row * (long)D;
  T* yr = y + row * (long)D;

  constexpr int MAX_CHUNKS = 4;  // 4 * 8 * BLOCK elems = 8192 at BLOCK=256
  float xv[MAX_CHUNKS][8];
  const int nchunks = (D + BLOCK * 8 - 1) / (BLOCK * 8);

  float s = 0.f;
  for (int c = 0; c < nchunks; ++c) {
    const int i = (c * BLOCK + threadIdx.x) * 8;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      xv[c][j] = (i + j < D) ? load_f32<T>(xr, i + j) : 0.f;
      s += xv[c][j];
    }
  }
  s = block_reduce_sum(s, scratch);
  const float mean = s / D;
  float s2 = 0.f;
  for (int c = 0; c < nchunks; ++c) {
    const int i = (c * BLOCK + threadIdx.x) * 8;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float d = (i + j < D) ? (xv[c][j] - mean) : 0.f;
      s2 += d * d;
    }
  }
  s2 = block_reduce_sum(s2, scratch);
  const float rstd = rsqrtf(s2 / D + eps);
  if (threadIdx.x == 0) {
    mean_out[row] = mean;
    rstd_out[row] = rstd;
  }
  for (int c = 0; c < nchunks; ++c) {
    const int i = (c * BLOCK + threadIdx.x) * 8;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      if (i + j < D) {
        const float wj = w ? w[i + j] : 1.f;
        const float bj = b ? b[i + j] : 0.f;
        store_f32<T>(yr, i + j, (xv[c][j] - mean) * rstd * wj + bj);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Backward dx: dx = rstd * (dyw - mean(dyw) - xhat * mean(dyw * xhat))
//   where dyw = dy * w, xhat = (x - mean) * rstd
// ---------------------------------------------------------------------------
template <typename T, int BLOCK>
__global__ void layernorm_bwd_dx_kernel(
    const T* __restrict__ dy, const T* __restrict__ x,
    const float* __restrict__ w, const float* __restrict__ mean,
    const float* __restrict__ rstd, T* __restrict__ dx, int D) {
  __shared__ float scratch[BLOCK / WAVE];
  const long row = blockIdx.x;
  const T* dyr = dy + row * (long)D;
  const T* xr = x + row * (long)D;
  T* dxr = dx + row * (long)D;
  const float mu = mean[row], rs = rstd[row];

  float c1 = 0.f, c2 = 0.f;
  for (int i = threadIdx.x * 8; i < D; i += BLOCK * 8) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float g = load_f32<T>(dyr, i + j) * (w ? w[i + j] : 1.f);
      float xh = (load_f32<T>(xr, i + j) - mu) * rs;
      c1 += g;
      c2 += g * xh;
    }
  }
  c1 = block_reduce_sum(c1, scratch) / D;
  c2 = block_reduce_sum(c2, scratch) / D;
  for (int i = threadIdx.x * 8; i < D; i += BLOCK * 8) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float g = load_f32<T>(dyr, i + j) * (w ? w[i + j] : 1.f);
      float xh = (load_f32<T>(xr, i + j) - mu) * rs;
      store_f32<T>(dxr, i + j, rs * (g - c1 - xh * c2));
    }
  }
}

// dw/db partials: each block strides over rows, accumulating its own fp32
// partial row of length D; partials reduced by reduce_partials_kernel.
template <typename T, int BLOCK>
__global__ void layernorm_bwd_dwdb_partial_kernel(
    const T* __restrict__ dy, const T* __restrict__ x,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    float* __restrict__ dw_part, float* __restrict__ db_part, long N, int D) {
  const int nblocks = gridDim.x;
  float* dwp = dw_part + blockIdx.x * (long)D;
  float* dbp = db_part + blockIdx.x * (long)D;
  for (int i = threadIdx.x; i < D; i += BLOCK) {
    float dw = 0.f, db = 0.f;
    for (long r = blockIdx.x; r < N; r += nblocks) {
      float g = load_f32<T>(dy + r * (long)D, i);
      float xh = (load_f32<T>(x + r * (long)D, i) - mean[r]) * rstd[r];
      dw += g * xh;
      db += g;
    }
    dwp[i] = dw;
    dbp[i] = db;
  }
}

__global__ void reduce_partials_kernel(const float* __restrict__ part,
                                       float* __restrict__ out, int nparts,
                                       int D) {
  for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < D;
       i += gridDim.x * blockDim.x) {
    float s = 0.f;
    for (int p = 0; p < nparts; ++p) s += part[p * (long)D + i];
    out[i] = s;
  }
}

// ---------------------------------------------------------------------------
// Host launchers
// ---------------------------------------------------------------------------
constexpr int LN_BLOCK = 256;

std::vector<torch::Tensor> layernorm_fwd(torch::Tensor x, torch::Tensor w,
                                         c10::optional<torch::Tensor> b,
                                         double eps) {
  TORCH_CHECK(x.is_contiguous(), "layernorm_fwd: x must be contiguous");
  const int D = x.size(-1);
  TORCH_CHECK(D % 8 == 0, "layernorm_fwd: D must be a multiple of 8");
  const long N = x.numel() / D;
  auto y = torch::empty_like(x);
  auto opts = x.options().dtype(at::kFloat);
  auto mean = torch::empty({N}, opts);
  auto rstd = torch::empty({N}, opts);
  auto wf = w.contiguous().to(at::kFloat);
  torch::Tensor bf;
  const float* bptr = nullptr;
  if (b.has_value()) {
    bf = b->contiguous().to(at::kFloat);
    bptr = bf.data_ptr<float>();
  }
  DISPATCH_DTYPE(x, "layernorm_fwd", {
    hipLaunchKernelGGL((layernorm_fwd_kernel<scalar_t, LN_BLOCK>), dim3(N),
                       dim3(LN_BLOCK), 0, cur_stream(),
                       (const scalar_t*)x.data_ptr(), wf.data_ptr<float>(),
                       bptr, (scalar_t*)y.data_ptr(), mean.data_ptr<float>(),
                       rstd.data_ptr<float>(), D, (float)eps);
  });
  return {y, mean, rstd};
}

std::vector<torch::Tensor> layernorm_bwd(torch::Tensor dy, torch::Tensor x,
                                         torch::Tensor w, torch::Tensor mean,
                                         torch::Tensor rstd) {
  const int D = x.size(-1);
  const long N = x.numel() / D;
  auto dx = torch::empty_like(x);
  auto wf = w.contiguous().to(at::kFloat);
  const int NPART = 256;
  auto opts = x.options().dtype(at::kFloat);
  auto dw_part = torch::empty({NPART, (long)D}, opts);
  auto db_part = torch::empty({NPART, (long)D}, opts);
  auto dw = torch::empty({(long)D}, opts);
  auto db = torch::empty({(long)D}, opts);
  DISPATCH_DTYPE(x, "layernorm_bwd", {
    hipLaunchKernelGGL((layernorm_bwd_dx_kernel<scalar_t, LN_BLOCK>), dim3(N),
                       dim3(LN_BLOCK), 0, cur_stream(),
                       (const scalar_t*)dy.data_ptr(),
                       (const scalar_t*)x.data_ptr(), wf.data_ptr<float>(),
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),
                       (scalar_t*)dx.data_ptr(), D);
    hipLaunchKernelGGL((layernorm_bwd_dwdb_partial_kernel<scalar_t, LN_BLOCK>),
                       dim3(NPART), dim3(LN_BLOCK), 0, cur_stream(),
                       (const scalar_t*)dy.data_ptr(),
                       (const scalar_t*)x.data_ptr(), mean.data_ptr<float>(),
                       rstd.data_ptr<float>(), dw_part.data_ptr<float>(),
                       db_part.data_ptr<float>(), N, D);
  });
  const int rblocks = std::min<long>((D + LN_BLOCK - 1) / LN_BLOCK, 256);
  hipLaunchKernelGGL(reduce_partials_kernel, dim3(rblocks), dim3(LN_BLOCK), 0,
                     cur_stream(), dw_part.data_ptr<float>(),
                     dw.data_ptr<float>(), NPART, D);
  hipLaunchKernelGGL(reduce_partials_kernel, dim3(rblocks), dim3(LN_BLOCK), 0,
                     cur_stream(), db_part.data_ptr<float>(),
                     db.data_ptr<float>(), NPART, D);
  return {dx, dw.to(w.dtype()), db.to(w.dtype())};
}

}  // namespace photon_hip
