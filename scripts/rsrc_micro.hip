#include <hip/hip_runtime.h>
// flags=0 (default) REPRODUCES the silent-zeros bug: the V# DATA_FORMAT
// field is invalid and every load is treated as out-of-range. Build with
// -DRSRC_FLAGS=0x27FAC (dst_sel XYZW | NFMT float | DFMT 32) for correct
// loads — the constant attn_kernels.h::make_rsrc ships with.
#ifndef RSRC_FLAGS
#define RSRC_FLAGS 0
#endif
#include <cstdio>
#include <vector>
typedef __attribute__((ext_vector_type(4))) float f32x4;
// mimic the kernel: 256 threads, chunks voff = tid*16 (+ i*4096), several tiles via rebuilt rsrc
__global__ void k_rsrc(const char* src, char* dst, int extent, int tiles, int tile_bytes) {
  for (int t = 0; t < tiles; ++t) {
    const int so = t * tile_bytes;
    auto rsrc = __builtin_amdgcn_make_buffer_rsrc((void*)(src + so), (short)0, extent - so, RSRC_FLAGS);
    for (int i = 0; i < 2; ++i) {
      int voff = i * 4096 + threadIdx.x * 16;
      f32x4 v = __builtin_amdgcn_raw_buffer_load_b128(rsrc, voff, 0, 0);
      *(f32x4*)(dst + so + voff) = v;
    }
  }
}
__global__ void k_rsrc_soff(const char* src, char* dst, int extent, int tiles, int tile_bytes) {
  auto rsrc = __builtin_amdgcn_make_buffer_rsrc((void*)src, (short)0, extent, RSRC_FLAGS);
  for (int t = 0; t < tiles; ++t) {
    const int so = t * tile_bytes;
    for (int i = 0; i < 2; ++i) {
      int voff = i * 4096 + threadIdx.x * 16;
      f32x4 v = __builtin_amdgcn_raw_buffer_load_b128(rsrc, voff, so, 0);
      *(f32x4*)(dst + so + voff) = v;
    }
  }
}
int main() {
  const int tiles = 8, tile_bytes = 8192, N = tiles * tile_bytes;
  char *s, *d1, *d2; hipMalloc(&s, N); hipMalloc(&d1, N); hipMalloc(&d2, N);
  std::vector<char> h(N); for (int i = 0; i < N; ++i) h[i] = (char)(i * 1315423911u >> 13);
  hipMemcpy(s, h.data(), N, hipMemcpyHostToDevice);
  hipMemset(d1, 0, N); hipMemset(d2, 0, N);
  hipLaunchKernelGGL(k_rsrc, dim3(1), dim3(256), 0, 0, s, d1, N, tiles, tile_bytes);
  hipLaunchKernelGGL(k_rsrc_soff, dim3(1), dim3(256), 0, 0, s, d2, N, tiles, tile_bytes);
  hipDeviceSynchronize();
  std::vector<char> o1(N), o2(N);
  hipMemcpy(o1.data(), d1, N, hipMemcpyDeviceToHost);
  hipMemcpy(o2.data(), d2, N, hipMemcpyDeviceToHost);
  int bad1 = 0, bad2 = 0, first1 = -1, first2 = -1;
  for (int i = 0; i < N; ++i) {
    if (o1[i] != h[i]) { if (first1 < 0) first1 = i; ++bad1; }
    if (o2[i] != h[i]) { if (first2 < 0) first2 = i; ++bad2; }
  }
  printf("rebuilt-rsrc: %d bad (first %d); sgpr-soffset: %d bad (first %d)\n", bad1, first1, bad2, first2);
  return 0;
}
