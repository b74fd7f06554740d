// Fused multi-tensor AdamW / ADOPT / grad-clip kernels for CDNA4.
//
// Replaces torch's _fused_adamw multi-tensor-apply + the reference "adopt"
// optimizer. Tensors are batched into a struct
// passed by kernarg (up to MT_CHUNK tensors per launch); each block
// grid-strides over one tensor's elements (block -> tensor mapping by
// cumulative block counts). All state math in fp32; params may be fp32 or
// bf16.

#include "host_common.h"

#include <vector>

namespace photon_hip {

constexpr int MT_CHUNK = 8;

template <typename T>
struct TensorBatch {
  T* p[MT_CHUNK];
  T* g[MT_CHUNK];
  float* m[MT_CHUNK];
  float* v[MT_CHUNK];
  float* w[MT_CHUNK];  // fp32 master weights (null = p IS the master)
  long n[MT_CHUNK];
  int count;
};

// blocks are distributed proportionally: block b handles tensor
// b % count, striding with gridDim.x/count-ish. Simpler: each block picks
// tensor (b % count) and strides by the number of blocks assigned to it.
template <typename T, int BLOCK>
__global__ void adamw_kernel(TensorBatch<T> batch, float lr, float b1,
                             float b2, float eps, float wd, float bc1,
                             float bc2) {
  const int t = blockIdx.x % batch.count;
  const int nb = (gridDim.x + batch.count - 1 - t) / batch.count;  // blocks on t
  const int bi = blockIdx.x / batch.count;
  T* p = batch.p[t];
  T* g = batch.g[t];
  float* m = batch.m[t];
  float* v = batch.v[t];
  float* w = batch.w[t];  // fp32 master (PURE mixed precision), or null
  const long n = batch.n[t];
  const float decay = 1.f - lr * wd;
  for (long i = (long)(bi * BLOCK + threadIdx.x); i < n; i += (long)nb * BLOCK) {
    float gf = load_f32<T>(g, i);
    float mi = m[i] = b1 * m[i] + (1.f - b1) * gf;
    float vi = v[i] = b2 * v[i] + (1.f - b2) * gf * gf;
    float pv = w ? w[i] : load_f32<T>(p, i);
    if (wd != 0.f) pv *= decay;
    pv -= lr * (mi / bc1) / (sqrtf(vi / bc2) + eps);
    if (w) w[i] = pv;
    store_f32<T>(p, i, pv);
  }
}

template <typename T, int BLOCK>
__global__ void adopt_kernel(TensorBatch<T> batch, float lr, float b1,
                             float b2, float eps, float wd, float clip,
                             int step) {
  const int t = blockIdx.x % batch.count;
  const int nb = (gridDim.x + batch.count - 1 - t) / batch.count;
  const int bi = blockIdx.x / batch.count;
  T* p = batch.p[t];
  T* g = batch.g[t];
  float* m = batch.m[t];
  float* v = batch.v[t];
  float* w = batch.w[t];
  const long n = batch.n[t];
  const float decay = 1.f - lr * wd;
  for (long i = (long)(bi * BLOCK + threadIdx.x); i < n; i += (long)nb * BLOCK) {
    float gf = load_f32<T>(g, i);
    if (step == 1) {
      v[i] = gf * gf;  // v0 init; no param update
      continue;
    }
    float c = gf / fmaxf(sqrtf(v[i]), eps);
    c = fminf(fmaxf(c, -clip), clip);
    float mi = m[i] = b1 * m[i] + (1.f - b1) * c;
    float pv = w ? w[i] : load_f32<T>(p, i);
    if (wd != 0.f) pv *= decay;
    pv -= lr * mi;
    if (w) w[i] = pv;
    store_f32<T>(p, i, pv);
    v[i] = b2 * v[i] + (1.f - b2) * gf * gf;
  }
}

// -- multi-tensor L2 norm (two-stage, deterministic) ------------------------
template <typename T, int BLOCK>
__global__ void l2norm_partial_kernel(TensorBatch<T> batch,
                                      float* __restrict__ partials) {
  __shared__ float scratch[BLOCK / WAVE];
  const int t = blockIdx.x % batch.count;
  const int nb = (gridDim.x + batch.count - 1 - t) / batch.count;
  const int bi = blockIdx.x / batch.count;
  const T* g = batch.g[t];
  const long n = batch.n[t];
  float s = 0.f;
  for (long i = (long)(bi * BLOCK + threadIdx.x); i < n; i += (long)nb * BLOCK) {
    float v = load_f32<T>(g, i);
    s += v * v;
  }
  s = block_reduce_sum(s, scratch);
  if (threadIdx.x == 0) partials[blockIdx.x] = s;
}

__global__ void l2norm_finish_kernel(const float* __restrict__ partials,
                                     int n, float* __restrict__ out,
                                     int accumulate) {
  __shared__ float scratch[256 / WAVE];
  float s = 0.f;
  for (int i = threadIdx.x; i < n; i += blockDim.x) s += partials[i];
  s = block_reduce_sum(s, scratch);
  if (threadIdx.x == 0) {
    if (accumulate)
      out[0] += s;
    else
      out[0] = s;
  }
}

// scale grads by max_norm/total_norm if above threshold; norm read on
// device (no host sync).
template <typename T, int BLOCK>
__global__ void scale_clip_kernel(TensorBatch<T> batch,
                                  const float* __restrict__ total_norm,
                                  float max_norm) {
  const float total = total_norm[0];
  const float scale = max_norm / (total + 1e-6f);
  if (scale >= 1.f) return;
  const int t = blockIdx.x % batch.count;
  const int nb = (gridDim.x + batch.count - 1 - t) / batch.count;
  const int bi = blockIdx.x / batch.count;
  T* g = batch.g[t];
  const long n = batch.n[t];
  for (long i = (long)(bi * BLOCK + threadIdx.x); i < n; i += (long)nb * BLOCK) {
    store_f32<T>(g, i, load_f32<T>(g, i) * scale);
  }
}

// ---------------------------------------------------------------------------
// Host launchers
// ---------------------------------------------------------------------------
namespace {
constexpr int OPT_BLOCK = 256;

template <typename T>
TensorBatch<T> make_batch(const std::vector<torch::Tensor>* ps,
                          const std::vector<torch::Tensor>& gs,
                          const std::vector<torch::Tensor>* ms,
                          const std::vector<torch::Tensor>* vs, size_t start,
                          size_t count,
                          const std::vector<torch::Tensor>* ws = nullptr) {
  TensorBatch<T> b{};
  b.count = (int)count;
  for (size_t i = 0; i < count; ++i) {
    b.p[i] = ps ? (T*)(*ps)[start + i].data_ptr() : nullptr;
    b.g[i] = (T*)gs[start + i].data_ptr();
    b.m[i] = ms ? (*ms)[start + i].data_ptr<float>() : nullptr;
    b.v[i] = vs ? (*vs)[start + i].data_ptr<float>() : nullptr;
    b.w[i] = (ws && !ws->empty()) ? (*ws)[start + i].data_ptr<float>()
                                  : nullptr;
    b.n[i] = gs[start + i].numel();
  }
  return b;
}
}  // namespace

void adamw_step(std::vector<torch::Tensor> ps, std::vector<torch::Tensor> gs,
                std::vector<torch::Tensor> ms, std::vector<torch::Tensor> vs,
                double lr, double b1, double b2, double eps, double wd,
                double bc1, double bc2,
                std::vector<torch::Tensor> masters = {}) {
  const int GRID = 1024;
  for (size_t s = 0; s < ps.size(); s += MT_CHUNK) {
    size_t c = std::min((size_t)MT_CHUNK, ps.size() - s);
    DISPATCH_DTYPE(ps[s], "adamw_step", {
      auto batch = make_batch<scalar_t>(&ps, gs, &ms, &vs, s, c, &masters);
      hipLaunchKernelGGL((adamw_kernel<scalar_t, OPT_BLOCK>), dim3(GRID),
                         dim3(OPT_BLOCK), 0, cur_stream(), batch, (float)lr,
                         (float)b1, (float)b2, (float)eps, (float)wd,
                         (float)bc1, (float)bc2);
    });
  }
}

void adopt_step(std::vector<torch::Tensor> ps, std::vector<torch::Tensor> gs,
                std::vector<torch::Tensor> ms, std::vector<torch::Tensor> vs,
                double lr, double b1, double b2, double eps, double wd,
                double clip, long step,
                std::vector<torch::Tensor> masters = {}) {
  const int GRID = 1024;
  for (size_t s = 0; s < ps.size(); s += MT_CHUNK) {
    size_t c = std::min((size_t)MT_CHUNK, ps.size() - s);
    DISPATCH_DTYPE(ps[s], "adopt_step", {
      auto batch = make_batch<scalar_t>(&ps, gs, &ms, &vs, s, c, &masters);
      hipLaunchKernelGGL((adopt_kernel<scalar_t, OPT_BLOCK>), dim3(GRID),
                         dim3(OPT_BLOCK), 0, cur_stream(), batch, (float)lr,
                         (float)b1, (float)b2, (float)eps, (float)wd,
                         (float)clip, (int)step);
    });
  }
}

torch::Tensor multi_tensor_l2norm(std::vector<torch::Tensor> gs) {
  TORCH_CHECK(!gs.empty(), "multi_tensor_l2norm: empty grad list");
  const int GRID = 512;
  const size_t nbatches = (gs.size() + MT_CHUNK - 1) / MT_CHUNK;
  auto partials = torch::zeros({(long)(nbatches * GRID)},
                               gs[0].options().dtype(at::kFloat));
  for (size_t b = 0; b < nbatches; ++b) {
    size_t s = b * MT_CHUNK;
    size_t c = std::min((size_t)MT_CHUNK, gs.size() - s);
    DISPATCH_DTYPE(gs[s], "multi_tensor_l2norm", {
      auto batch = make_batch<scalar_t>(nullptr, gs, nullptr, nullptr, s, c);
      hipLaunchKernelGGL((l2norm_partial_kernel<scalar_t, OPT_BLOCK>),
                         dim3(GRID), dim3(OPT_BLOCK), 0, cur_stream(), batch,
                         partials.data_ptr<float>() + b * GRID);
    });
  }
  auto total_sq = torch::empty({1}, partials.options());
  hipLaunchKernelGGL(l2norm_finish_kernel, dim3(1), dim3(256), 0, cur_stream(),
                     partials.data_ptr<float>(), (int)(nbatches * GRID),
                     total_sq.data_ptr<float>(), 0);
  return total_sq.sqrt_().squeeze(0);
}

void multi_tensor_scale_clip(std::vector<torch::Tensor> gs,
                             torch::Tensor total_norm, double max_norm) {
  const int GRID = 1024;
  auto norm = total_norm.reshape({1}).contiguous();
  for (size_t s = 0; s < gs.size(); s += MT_CHUNK) {
    size_t c = std::min((size_t)MT_CHUNK, gs.size() - s);
    DISPATCH_DTYPE(gs[s], "multi_tensor_scale_clip", {
      auto batch = make_batch<scalar_t>(nullptr, gs, nullptr, nullptr, s, c);
      hipLaunchKernelGGL((scale_clip_kernel<scalar_t, OPT_BLOCK>), dim3(GRID),
                         dim3(OPT_BLOCK), 0, cur_stream(), batch,
                         norm.data_ptr<float>(), (float)max_norm);
    });
  }
}

}  // namespace photon_hip
