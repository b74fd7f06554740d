// Fused LayerNorm fwd/bwd for CDNA4 (gfx950).
//
// Design (MI355X-first): one 64-lane WAVE per row — no LDS, no barriers,
// pure wave64 shuffle reductions. Each lane owns the same column set
// (lane*4 + c*256) for every row, loads are 8-byte (bf16) / 16-byte (f32)
// vectors, all statistics accumulate in fp32. Rows are re-read from L1 for
// the second pass (a row is 1.5-8 KB — L1-hot), which keeps register
// pressure at occupancy 8.
//
// Backward fuses the dweight/dbias column partials into the dx pass: each
// wave accumulates its columns' dw/db in registers across all of its rows
// and writes ONE fp32 partial row; a second kernel reduces the partials.
// No atomics anywhere, so backward is bit-deterministic run to run.
//
// Replaces llm-foundry LPLayerNorm / torch LN CUDA kernels (reference
// SURVEY.md §2.3 L133 row). Requires D % 256 == 0 (all MPT presets:
// 768/1024/2048/2560/4096); other D fall back to PyTorch in ops/layernorm.py.

#include "host_common.h"

namespace photon_hip {

#define COLS_PER_WAVE 256  // 64 lanes * 4 elements

// ---------------------------------------------------------------------------
// Forward: y = (x - mean) * rstd * w + b ; saves mean, rstd (fp32 per row).
// Grid: G blocks of BLOCK threads = G*BLOCK/64 waves; wave g handles rows
// g, g+W, g+2W, ... Two-pass mean/variance (exact), row re-read from L1.
// ---------------------------------------------------------------------------
// Optional fused residual add (r != nullptr): s = x + r is rounded to T,
// written to s_out (the new residual stream) and used as the LN input —
// kills the standalone residual-add kernels (~3% of the 125M step) and
// matches the eager x+r->LN numerics (stats on the ROUNDED sum).
template <typename T>
__global__ void layernorm_fwd_kernel(const T* __restrict__ x,
                                     const float* __restrict__ w,
                                     const float* __restrict__ b,
                                     T* __restrict__ y,
                                     float* __restrict__ mean_out,
                                     float* __restrict__ rstd_out, long N,
                                     int D, float eps,
                                     const T* __restrict__ r,
                                     T* __restrict__ s_out) {
  const int lane = threadIdx.x & (WAVE - 1);
  const long wave_id =
      ((long)blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  const long n_waves = ((long)gridDim.x * blockDim.x) / WAVE;
  const int nch = D / COLS_PER_WAVE;
  const float inv_d = 1.f / (float)D;

  for (long row = wave_id; row < N; row += n_waves) {
    const T* xr = x + row * (long)D;
    T* yr = y + row * (long)D;
    float s = 0.f;
    if (r != nullptr) {
      const T* rr = r + row * (long)D;
      T* so = s_out + row * (long)D;
      for (int c = 0; c < nch; ++c) {
        const int i = c * COLS_PER_WAVE + lane * 4;
        floatx4 a = load4<T>(xr + i);
        floatx4 bv = load4<T>(rr + i);
        // round through T first (store4 does the T-rounding), then re-read
        // the rounded values: stats must see exactly the stored stream
        floatx4 o;
        o.x = a.x + bv.x;
        o.y = a.y + bv.y;
        o.z = a.z + bv.z;
        o.w = a.w + bv.w;
        store4<T>(so + i, o);
        o = load4<T>(so + i);
        s += o.x + o.y + o.z + o.w;
      }
      xr = so;  // variance + normalize passes read the sum
    } else {
      for (int c = 0; c < nch; ++c) {
        floatx4 v = load4<T>(xr + c * COLS_PER_WAVE + lane * 4);
        s += v.x + v.y + v.z + v.w;
      }
    }
    const float mean = wave_reduce_sum(s) * inv_d;
    float s2 = 0.f;
    for (int c = 0; c < nch; ++c) {
      floatx4 v = load4<T>(xr + c * COLS_PER_WAVE + lane * 4);
      float dx0 = v.x - mean, dx1 = v.y - mean, dx2 = v.z - mean,
            dx3 = v.w - mean;
      s2 += dx0 * dx0 + dx1 * dx1 + dx2 * dx2 + dx3 * dx3;
    }
    const float rstd = rsqrtf(wave_reduce_sum(s2) * inv_d + eps);
    if (lane == 0) {
      mean_out[row] = mean;
      rstd_out[row] = rstd;
    }
    for (int c = 0; c < nch; ++c) {
      const int i = c * COLS_PER_WAVE + lane * 4;
      floatx4 v = load4<T>(xr + i);
      floatx4 wv = load4<float>(w + i);
      floatx4 o;
      if (b != nullptr) {
        floatx4 bv = load4<float>(b + i);
        o.x = (v.x - mean) * rstd * wv.x + bv.x;
        o.y = (v.y - mean) * rstd * wv.y + bv.y;
        o.z = (v.z - mean) * rstd * wv.z + bv.z;
        o.w = (v.w - mean) * rstd * wv.w + bv.w;
      } else {
        o.x = (v.x - mean) * rstd * wv.x;
        o.y = (v.y - mean) * rstd * wv.y;
        o.z = (v.z - mean) * rstd * wv.z;
        o.w = (v.w - mean) * rstd * wv.w;
      }
      store4<T>(yr + i, o);
    }
  }
}

// ---------------------------------------------------------------------------
// Backward, fused: dx = rstd * (g - mean(g) - xhat * mean(g * xhat)) with
// g = dy * w, xhat = (x - mean) * rstd; simultaneously accumulates
// dw_col += g * xhat and db_col += g in registers across this wave's rows.
// MAX_CH bounds the register accumulators: 2 * MAX_CH * 4 floats.
// ---------------------------------------------------------------------------
template <typename T, int MAX_CH>
__global__ void layernorm_bwd_fused_kernel(
    const T* __restrict__ dy, const T* __restrict__ x,
    const float* __restrict__ w, const float* __restrict__ mean,
    const float* __restrict__ rstd, T* __restrict__ dx,
    float* __restrict__ dwdb_part, long N, int D,
    const T* __restrict__ dresid) {
  const int lane = threadIdx.x & (WAVE - 1);
  const long wave_id =
      ((long)blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  const long n_waves = ((long)gridDim.x * blockDim.x) / WAVE;
  const int nch = D / COLS_PER_WAVE;
  const float inv_d = 1.f / (float)D;

  floatx4 dw_acc[MAX_CH], db_acc[MAX_CH];
#pragma unroll
  for (int c = 0; c < MAX_CH; ++c) {
    dw_acc[c] = floatx4{0.f, 0.f, 0.f, 0.f};
    db_acc[c] = floatx4{0.f, 0.f, 0.f, 0.f};
  }

  for (long row = wave_id; row < N; row += n_waves) {
    const T* dyr = dy + row * (long)D;
    const T* xr = x + row * (long)D;
    T* dxr = dx + row * (long)D;
    const float mu = mean[row], rs = rstd[row];

    float c1 = 0.f, c2 = 0.f;
    for (int c = 0; c < nch; ++c) {
      const int i = c * COLS_PER_WAVE + lane * 4;
      floatx4 gy = load4<T>(dyr + i);
      floatx4 wv = load4<float>(w + i);
      floatx4 xv = load4<T>(xr + i);
      float g0 = gy.x * wv.x, g1 = gy.y * wv.y, g2 = gy.z * wv.z,
            g3 = gy.w * wv.w;
      float h0 = (xv.x - mu) * rs, h1 = (xv.y - mu) * rs,
            h2 = (xv.z - mu) * rs, h3 = (xv.w - mu) * rs;
      c1 += g0 + g1 + g2 + g3;
      c2 += g0 * h0 + g1 * h1 + g2 * h2 + g3 * h3;
    }
    c1 = wave_reduce_sum(c1) * inv_d;
    c2 = wave_reduce_sum(c2) * inv_d;
    for (int c = 0; c < nch && c < MAX_CH; ++c) {
      const int i = c * COLS_PER_WAVE + lane * 4;
      floatx4 gy = load4<T>(dyr + i);
      floatx4 wv = load4<float>(w + i);
      floatx4 xv = load4<T>(xr + i);
      float g0 = gy.x * wv.x, g1 = gy.y * wv.y, g2 = gy.z * wv.z,
            g3 = gy.w * wv.w;
      float h0 = (xv.x - mu) * rs, h1 = (xv.y - mu) * rs,
            h2 = (xv.z - mu) * rs, h3 = (xv.w - mu) * rs;
      floatx4 o;
      o.x = rs * (g0 - c1 - h0 * c2);
      o.y = rs * (g1 - c1 - h1 * c2);
      o.z = rs * (g2 - c1 - h2 * c2);
      o.w = rs * (g3 - c1 - h3 * c2);
      if (dresid != nullptr) {
        // fused residual-path gradient: dx(total) = dLN/dx + ds_out
        floatx4 dr = load4<T>(dresid + row * (long)D + i);
        o.x += dr.x;
        o.y += dr.y;
        o.z += dr.z;
        o.w += dr.w;
      }
      store4<T>(dxr + i, o);
      dw_acc[c].x += gy.x * h0;  // dw is vs raw dy (not g=dy*w)
      dw_acc[c].y += gy.y * h1;
      dw_acc[c].z += gy.z * h2;
      dw_acc[c].w += gy.w * h3;
      db_acc[c].x += gy.x;
      db_acc[c].y += gy.y;
      db_acc[c].z += gy.z;
      db_acc[c].w += gy.w;
    }
  }
  // One fp32 partial row per wave, dw and db side by side: [n_waves, 2D].
  float* dwp = dwdb_part + wave_id * (long)(2 * D);
  float* dbp = dwp + D;
  for (int c = 0; c < nch && c < MAX_CH; ++c) {
    const int i = c * COLS_PER_WAVE + lane * 4;
    store4<float>(dwp + i, dw_acc[c]);
    store4<float>(dbp + i, db_acc[c]);
  }
}

// Column-sum of a [nparts, D] fp32 matrix, parallel over BOTH dims:
// grid.y splits the partial dimension so the launch fills the chip even for
// small D (a D-only grid at D=768 is 3 workgroups on a 256-CU chip — the
// first version of this kernel spent 51% of the training step there,
// profiles/r01). Two chained launches (split=S then split=1) reduce fully.
__global__ void reduce_partials_kernel(const float* __restrict__ part,
                                       float* __restrict__ out, int nparts,
                                       int D) {
  const int col = blockIdx.x * blockDim.x + threadIdx.x;
  if (col >= D) return;
  const int per = (nparts + gridDim.y - 1) / gridDim.y;
  const int p0 = blockIdx.y * per;
  const int p1 = min(p0 + per, nparts);
  float s = 0.f;
  for (int p = p0; p < p1; ++p) s += part[(long)p * D + col];
  out[(long)blockIdx.y * D + col] = s;
}

// ---------------------------------------------------------------------------
// Host launchers
// ---------------------------------------------------------------------------
constexpr int LN_BLOCK = 256;  // 4 waves per block

static int ln_grid_blocks(long n_rows) {
  // 8 blocks/CU * 256 CUs = 2048 fills the chip; shrink for few rows.
  long want = (n_rows + 3) / 4;  // one row per wave minimum
  if (want < 64) want = 64;
  if (want > 2048) want = 2048;
  return (int)want;
}

std::vector<torch::Tensor> layernorm_fwd(torch::Tensor x, torch::Tensor w,
                                         c10::optional<torch::Tensor> b,
                                         double eps,
                                         c10::optional<torch::Tensor> residual) {
  TORCH_CHECK(x.is_contiguous(), "layernorm_fwd: x must be contiguous");
  const int D = x.size(-1);
  TORCH_CHECK(D % 256 == 0 && D <= 4096,
              "layernorm_fwd: D must be a multiple of 256 and <= 4096");
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
  torch::Tensor s_out;
  const bool has_r = residual.has_value();
  if (has_r) {
    TORCH_CHECK(residual->is_contiguous() && residual->sizes() == x.sizes() &&
                    residual->scalar_type() == x.scalar_type(),
                "layernorm_fwd: residual must match x");
    s_out = torch::empty_like(x);
  }
  const int G = ln_grid_blocks(N);
  DISPATCH_DTYPE(x, "layernorm_fwd", {
    hipLaunchKernelGGL((layernorm_fwd_kernel<scalar_t>), dim3(G),
                       dim3(LN_BLOCK), 0, cur_stream(),
                       (const scalar_t*)x.data_ptr(), wf.data_ptr<float>(),
                       bptr, (scalar_t*)y.data_ptr(), mean.data_ptr<float>(),
                       rstd.data_ptr<float>(), N, D, (float)eps,
                       has_r ? (const scalar_t*)residual->data_ptr() : nullptr,
                       has_r ? (scalar_t*)s_out.data_ptr() : nullptr);
  });
  if (has_r) return {y, mean, rstd, s_out};
  return {y, mean, rstd};
}

std::vector<torch::Tensor> layernorm_bwd(torch::Tensor dy, torch::Tensor x,
                                         torch::Tensor w, torch::Tensor mean,
                                         torch::Tensor rstd,
                                         c10::optional<torch::Tensor> dresid) {
  const int D = x.size(-1);
  TORCH_CHECK(D % 256 == 0 && D <= 4096,
              "layernorm_bwd: D must be a multiple of 256 and <= 4096");
  const long N = x.numel() / D;
  auto dx = torch::empty_like(x);
  auto wf = w.contiguous().to(at::kFloat);
  const int G = ln_grid_blocks(N);
  const int n_waves = G * (LN_BLOCK / WAVE);
  auto opts = x.options().dtype(at::kFloat);
  // dw and db partials side by side: [n_waves, 2D].
  auto dwdb_part = torch::empty({n_waves, 2L * D}, opts);
  // Register accumulators sized to D: 2 * MAX_CH * 4 VGPRs live across the
  // row loop — pick the smallest template that fits so occupancy stays high.
  DISPATCH_DTYPE(x, "layernorm_bwd", {
    const scalar_t* drp = nullptr;
    if (dresid.has_value()) {
      TORCH_CHECK(dresid->is_contiguous() && dresid->sizes() == x.sizes(),
                  "layernorm_bwd: dresid must match x");
      drp = (const scalar_t*)dresid->data_ptr();
    }
    auto launch = [&](auto kern) {
      hipLaunchKernelGGL(kern, dim3(G), dim3(LN_BLOCK), 0, cur_stream(),
                         (const scalar_t*)dy.data_ptr(),
                         (const scalar_t*)x.data_ptr(), wf.data_ptr<float>(),
                         mean.data_ptr<float>(), rstd.data_ptr<float>(),
                         (scalar_t*)dx.data_ptr(),
                         dwdb_part.data_ptr<float>(), N, D, drp);
    };
    if (D <= 1024)
      launch(layernorm_bwd_fused_kernel<scalar_t, 4>);
    else if (D <= 2048)
      launch(layernorm_bwd_fused_kernel<scalar_t, 8>);
    else
      launch(layernorm_bwd_fused_kernel<scalar_t, 16>);
  });
  // Two-stage column reduce of [n_waves, 2D]: split the partial dim so the
  // grid fills the chip, then collapse the split.
  const int D2 = 2 * D;
  const int cblocks = (D2 + LN_BLOCK - 1) / LN_BLOCK;
  const int split = std::max(1, std::min(n_waves / 8, 512 / cblocks + 1));
  auto stage = torch::empty({(long)split, (long)D2}, opts);
  auto dwdb = torch::empty({(long)D2}, opts);
  hipLaunchKernelGGL(reduce_partials_kernel, dim3(cblocks, split),
                     dim3(LN_BLOCK), 0, cur_stream(),
                     dwdb_part.data_ptr<float>(), stage.data_ptr<float>(),
                     n_waves, D2);
  hipLaunchKernelGGL(reduce_partials_kernel, dim3(cblocks, 1), dim3(LN_BLOCK),
                     0, cur_stream(), stage.data_ptr<float>(),
                     dwdb.data_ptr<float>(), split, D2);
  auto dw = dwdb.narrow(0, 0, D);
  auto db = dwdb.narrow(0, D, D);
  return {dx, dw.to(w.dtype()), db.to(w.dtype())};
}

}  // namespace photon_hip
