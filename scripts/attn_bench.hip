// Standalone attention kernel bench — compiles with plain hipcc (no torch):
//   hipcc --offload-arch=gfx950 -O3 -I photon_amd/ops/hip scripts/attn_bench.hip -o /tmp/attn_bench
// Optional -DABENCH_NO_EXP / -DABENCH_NO_VT bisect variants (wrong results,
// perf signal only).
#include <cstdio>
#include <cstdlib>
#include <vector>
#include <cmath>
#include <hip/hip_runtime.h>
#include "attn_kernels.h"

using namespace photon_hip;

#define CHECK(x) do { hipError_t e = (x); if (e) { printf("ERR %s\n", hipGetErrorString(e)); exit(1);} } while(0)

template <int D>
void bench(int B, int H, int S, int iters) {
  size_t n = (size_t)B * H * S * D;
  __bf16 *q, *k, *v, *o;
  float *lse, *slopes;
  CHECK(hipMalloc(&q, n * 2)); CHECK(hipMalloc(&k, n * 2));
  CHECK(hipMalloc(&v, n * 2)); CHECK(hipMalloc(&o, n * 2));
  CHECK(hipMalloc(&lse, (size_t)B * H * S * 4));
  CHECK(hipMalloc(&slopes, H * 4));
  // random-ish fill (device memset pattern is fine for DVFS realism? use host rand)
  {
    std::vector<__bf16> h(n);
    for (size_t i = 0; i < n; ++i) h[i] = (__bf16)((float)(rand() % 2000 - 1000) / 500.f);
    CHECK(hipMemcpy(q, h.data(), n * 2, hipMemcpyHostToDevice));
    for (size_t i = 0; i < n; ++i) h[i] = (__bf16)((float)(rand() % 2000 - 1000) / 500.f);
    CHECK(hipMemcpy(k, h.data(), n * 2, hipMemcpyHostToDevice));
    CHECK(hipMemcpy(v, h.data(), n * 2, hipMemcpyHostToDevice));
    std::vector<float> hs(H);
    for (int i = 0; i < H; ++i) hs[i] = 0.5f / (1 << i);
    CHECK(hipMemcpy(slopes, hs.data(), H * 4, hipMemcpyHostToDevice));
  }
  dim3 grid((S + WAVES * QB - 1) / (WAVES * QB), (size_t)B * H);
  const int lds = 4 * KBF * D * 2 > WAVES * 64 * D ? 4 * KBF * D * 2 : WAVES * 64 * D;
  hipEvent_t e0, e1; hipEventCreate(&e0); hipEventCreate(&e1);
  const long bs0 = (long)H * S * D, hs0 = (long)S * D, rs0 = D;
  for (int i = 0; i < 3; ++i)
    hipLaunchKernelGGL((attn_fwd_kernel<D>), grid, dim3(ATT_BLOCK), lds, 0,
                       q, k, v, slopes, o, lse, S, H, 1, bs0, hs0, rs0,
                       bs0, hs0, rs0);
  CHECK(hipDeviceSynchronize());
  hipEventRecord(e0);
  for (int i = 0; i < iters; ++i)
    hipLaunchKernelGGL((attn_fwd_kernel<D>), grid, dim3(ATT_BLOCK), lds, 0,
                       q, k, v, slopes, o, lse, S, H, 1, bs0, hs0, rs0, bs0,
                       hs0, rs0);
  hipEventRecord(e1);
  CHECK(hipDeviceSynchronize());
  float ms; hipEventElapsedTime(&ms, e0, e1); ms /= iters;
  double eff = 4.0 * B * H * (double)S * S * D / 2 / (ms / 1e3) / 1e12;
  printf("fwd  B%d H%d S%d D%d: %8.3f ms  %7.1f TF/s eff\n", B, H, S, D, ms, eff);
  hipFree(q); hipFree(k); hipFree(v); hipFree(o); hipFree(lse); hipFree(slopes);
}

// Host fp64 reference check for small shapes (causal + ALiBi).
template <int D>
void check(int B, int H, int S) {
  size_t n = (size_t)B * H * S * D;
  std::vector<__bf16> hq(n), hk(n), hv(n);
  std::vector<float> hs(H);
  for (size_t i = 0; i < n; ++i) {
    hq[i] = (__bf16)((float)(rand() % 2000 - 1000) / 500.f);
    hk[i] = (__bf16)((float)(rand() % 2000 - 1000) / 500.f);
    hv[i] = (__bf16)((float)(rand() % 2000 - 1000) / 500.f);
  }
  for (int i = 0; i < H; ++i) hs[i] = 0.5f / (1 << i);
  __bf16 *q, *k, *v, *o;
  float *lse, *slopes;
  hipMalloc(&q, n * 2); hipMalloc(&k, n * 2); hipMalloc(&v, n * 2);
  hipMalloc(&o, n * 2); hipMalloc(&lse, (size_t)B * H * S * 4);
  hipMalloc(&slopes, H * 4);
  hipMemcpy(q, hq.data(), n * 2, hipMemcpyHostToDevice);
  hipMemcpy(k, hk.data(), n * 2, hipMemcpyHostToDevice);
  hipMemcpy(v, hv.data(), n * 2, hipMemcpyHostToDevice);
  hipMemcpy(slopes, hs.data(), H * 4, hipMemcpyHostToDevice);
  dim3 grid((S + WAVES * QB - 1) / (WAVES * QB), (size_t)B * H);
  const int lds = 4 * KBF * D * 2 > WAVES * 64 * D ? 4 * KBF * D * 2 : WAVES * 64 * D;
  const long bsS = (long)H * S * D, hsS = (long)S * D, rsS = D;
  hipLaunchKernelGGL((attn_fwd_kernel<D>), grid, dim3(ATT_BLOCK), lds, 0,
                     q, k, v, slopes, o, lse, S, H, 1, bsS, hsS, rsS, bsS,
                     hsS, rsS);
  CHECK(hipDeviceSynchronize());
  std::vector<__bf16> ho(n);
  hipMemcpy(ho.data(), o, n * 2, hipMemcpyDeviceToHost);
  double worst = 0; int wq = -1, wd = -1, wbh = -1;
  for (int bh = 0; bh < B * H; ++bh) {
    const int hidx = bh % H;
    for (int qi = 0; qi < S; ++qi) {
      double m = -1e300, l = 0;
      std::vector<double> acc(D, 0.0);
      for (int ki = 0; ki <= qi; ++ki) {
        double sdot = 0;
        for (int d = 0; d < D; ++d)
          sdot += (double)(float)hq[((size_t)bh * S + qi) * D + d] *
                  (double)(float)hk[((size_t)bh * S + ki) * D + d];
        double sv = sdot / sqrt((double)D) - (double)hs[hidx] * (qi - ki);
        double mn = m > sv ? m : sv;
        double a = exp(m - mn), pw = exp(sv - mn);
        for (int d = 0; d < D; ++d)
          acc[d] = acc[d] * a + pw * (double)(float)hv[((size_t)bh * S + ki) * D + d];
        l = l * a + pw;
        m = mn;
      }
      for (int d = 0; d < D; ++d) {
        double ref = acc[d] / l;
        double got = (double)(float)ho[((size_t)bh * S + qi) * D + d];
        double e = fabs(ref - got);
        if (e > worst) { worst = e; wq = qi; wd = d; wbh = bh; }
      }
    }
  }
  printf("check B%d H%d S%d D%d: max err %.4e at bh=%d q=%d d=%d\n", B, H, S,
         D, worst, wbh, wq, wd);
  hipFree(q); hipFree(k); hipFree(v); hipFree(o); hipFree(lse); hipFree(slopes);
}

int main(int argc, char** argv) {
  if (argc > 1 && argv[1][0] == 'c') {
    check<64>(1, 2, 192);
    check<128>(1, 1, 192);
    return 0;
  }
  bench<64>(32, 12, 2048, 20);
  bench<128>(8, 16, 4096, 10);
  return 0;
}
