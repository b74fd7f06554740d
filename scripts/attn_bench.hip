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

// Backward bench: runs fwd once for a real lse, then times dq and dkdv.
template <int D>
void bench_bwd(int B, int H, int S, int iters) {
  size_t n = (size_t)B * H * S * D;
  __bf16 *q, *k, *v, *o, *dout, *dq, *dk, *dv;
  float *lse, *delta, *slopes;
  CHECK(hipMalloc(&q, n * 2)); CHECK(hipMalloc(&k, n * 2));
  CHECK(hipMalloc(&v, n * 2)); CHECK(hipMalloc(&o, n * 2));
  CHECK(hipMalloc(&dout, n * 2)); CHECK(hipMalloc(&dq, n * 2));
  CHECK(hipMalloc(&dk, n * 2)); CHECK(hipMalloc(&dv, n * 2));
  CHECK(hipMalloc(&lse, (size_t)B * H * S * 4));
  CHECK(hipMalloc(&delta, (size_t)B * H * S * 4));
  CHECK(hipMalloc(&slopes, H * 4));
  {
    std::vector<__bf16> h(n);
    for (size_t i = 0; i < n; ++i) h[i] = (__bf16)((float)(rand() % 2000 - 1000) / 500.f);
    CHECK(hipMemcpy(q, h.data(), n * 2, hipMemcpyHostToDevice));
    for (size_t i = 0; i < n; ++i) h[i] = (__bf16)((float)(rand() % 2000 - 1000) / 500.f);
    CHECK(hipMemcpy(k, h.data(), n * 2, hipMemcpyHostToDevice));
    CHECK(hipMemcpy(v, h.data(), n * 2, hipMemcpyHostToDevice));
    CHECK(hipMemcpy(dout, h.data(), n * 2, hipMemcpyHostToDevice));
    std::vector<float> hs(H);
    for (int i = 0; i < H; ++i) hs[i] = 0.5f / (1 << i);
    CHECK(hipMemcpy(slopes, hs.data(), H * 4, hipMemcpyHostToDevice));
    std::vector<float> hd((size_t)B * H * S, 0.1f);
    CHECK(hipMemcpy(delta, hd.data(), (size_t)B * H * S * 4, hipMemcpyHostToDevice));
  }
  dim3 grid((S + WAVES * QB - 1) / (WAVES * QB), (size_t)B * H);
  const long bs0 = (long)H * S * D, hs0 = (long)S * D, rs0 = D;
  const int lds_fwd = 4 * KBF * D * 2 > WAVES * 64 * D ? 4 * KBF * D * 2 : WAVES * 64 * D;
  hipLaunchKernelGGL((attn_fwd_kernel<D>), grid, dim3(ATT_BLOCK), lds_fwd, 0,
                     q, k, v, slopes, o, lse, S, H, 1, bs0, hs0, rs0, bs0, hs0, rs0);
  CHECK(hipDeviceSynchronize());
  const int qtf = 64;  // matches ATT_QTF* kernel constexprs
  const int lds_dq = 4 * 64 * D * 2 > WAVES * 64 * D ? 4 * 64 * D * 2 : WAVES * 64 * D;
  int lds_kv = 4 * qtf * D * 2 + 4 * qtf * 4;
  if (lds_kv < WAVES * 64 * D) lds_kv = WAVES * 64 * D;
  hipEvent_t e0, e1; (void)hipEventCreate(&e0); (void)hipEventCreate(&e1);
  // dq
  for (int i = 0; i < 3; ++i)
    hipLaunchKernelGGL((attn_bwd_dq_kernel<D>), grid, dim3(ATT_BLOCK), lds_dq, 0,
                       dout, q, k, v, slopes, lse, delta, dq, S, H, 1,
                       bs0, hs0, rs0, bs0, hs0, rs0);
  CHECK(hipDeviceSynchronize());
  hipEventRecord(e0);
  for (int i = 0; i < iters; ++i)
    hipLaunchKernelGGL((attn_bwd_dq_kernel<D>), grid, dim3(ATT_BLOCK), lds_dq, 0,
                       dout, q, k, v, slopes, lse, delta, dq, S, H, 1,
                       bs0, hs0, rs0, bs0, hs0, rs0);
  hipEventRecord(e1);
  CHECK(hipDeviceSynchronize());
  float ms_dq; hipEventElapsedTime(&ms_dq, e0, e1); ms_dq /= iters;
  // dkdv
  for (int i = 0; i < 3; ++i)
    hipLaunchKernelGGL((attn_bwd_dkdv_kernel<D>), grid, dim3(ATT_BLOCK), lds_kv, 0,
                       dout, q, k, v, slopes, lse, delta, dk, dv, S, H, 1,
                       bs0, hs0, rs0, bs0, hs0, rs0);
  CHECK(hipDeviceSynchronize());
  hipEventRecord(e0);
  for (int i = 0; i < iters; ++i)
    hipLaunchKernelGGL((attn_bwd_dkdv_kernel<D>), grid, dim3(ATT_BLOCK), lds_kv, 0,
                       dout, q, k, v, slopes, lse, delta, dk, dv, S, H, 1,
                       bs0, hs0, rs0, bs0, hs0, rs0);
  hipEventRecord(e1);
  CHECK(hipDeviceSynchronize());
  float ms_kv; hipEventElapsedTime(&ms_kv, e0, e1); ms_kv /= iters;
  // dq: 2 GEMM-equivalents (S, dSK); dkdv: 2.5 (S, dP, dV, dK at half causal)
  double base = 2.0 * B * H * (double)S * S * D / 2 / 1e12;  // causal half
  printf("dq   B%d H%d S%d D%d: %8.3f ms  %7.1f TF/s eff\n", B, H, S, D, ms_dq,
         base * 2 / (ms_dq / 1e3));
  printf("dkdv B%d H%d S%d D%d: %8.3f ms  %7.1f TF/s eff\n", B, H, S, D, ms_kv,
         base * 2 / (ms_kv / 1e3));
  hipFree(q); hipFree(k); hipFree(v); hipFree(o); hipFree(dout); hipFree(dq);
  hipFree(dk); hipFree(dv); hipFree(lse); hipFree(delta); hipFree(slopes);
}

int main(int argc, char** argv) {
  if (argc > 1 && argv[1][0] == 'c') {
    check<64>(1, 2, 192);
    check<128>(1, 1, 192);
    return 0;
  }
  bench<64>(32, 12, 2048, 20);
  bench<128>(8, 16, 4096, 10);
  bench_bwd<64>(32, 12, 2048, 10);
  bench_bwd<128>(8, 16, 4096, 5);
  return 0;
}
