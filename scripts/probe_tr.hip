// Probe ds_read_b64_tr_b16 lane semantics on gfx950.
// Fills LDS with u16 values = their own u16-index, has each lane issue one
// ds_read_b64_tr_b16 at a lane-chosen address, and prints which 4 source
// indices land in each lane. Run on a GPU box:
//   hipcc --offload-arch=gfx950 -O3 scripts/probe_tr.hip -o /tmp/ptr && /tmp/ptr
#include <cstdio>
#include <hip/hip_runtime.h>

typedef unsigned short u16;

__global__ void probe_kernel(u16* out, int addr_mode) {
  __shared__ u16 lds[2048];
  const int lane = threadIdx.x;
  for (int i = lane; i < 2048; i += 64) lds[i] = (u16)i;
  __syncthreads();
  // address in bytes, 8-byte aligned, lane-dependent
  unsigned byte;
  switch (addr_mode) {
    case 0: byte = lane * 8; break;                       // linear 8B/lane
    case 1: byte = (lane & 15) * 8 + (lane >> 4) * 128; break;
    case 2: byte = (lane & 3) * 8 + ((lane >> 2) & 3) * 32 + (lane >> 4) * 512; break;
    case 3: {
      // The PV A-frag pattern: row-major image [key][64 dh], row stride
      // 128 B; lane j of a 16-group points at key (j>>2), dh 4*(j&3).
      const int j = lane & 15;
      byte = (j >> 2) * 128 + (4 * (j & 3)) * 2 + (lane >> 4) * 512;
      break;
    }
    default: byte = lane * 8; break;
  }
  typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;
  typedef bf16x4 __attribute__((address_space(3)))* lds_bf16x4_ptr;
  bf16x4 rv = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (lds_bf16x4_ptr)((__attribute__((address_space(3))) char*)lds + byte));
  union { bf16x4 v; unsigned long long u; } cvt; cvt.v = rv;
  unsigned long long r = cvt.u;
  out[lane * 5 + 0] = (u16)(byte / 2);  // the index the lane POINTED at
  out[lane * 5 + 1] = (u16)(r & 0xffff);
  out[lane * 5 + 2] = (u16)((r >> 16) & 0xffff);
  out[lane * 5 + 3] = (u16)((r >> 32) & 0xffff);
  out[lane * 5 + 4] = (u16)((r >> 48) & 0xffff);
}

int main() {
  u16* d;
  hipMalloc(&d, 64 * 5 * 2);
  u16 h[64 * 5];
  for (int mode = 3; mode < 4; ++mode) {
    hipLaunchKernelGGL(probe_kernel, dim3(1), dim3(64), 0, 0, d, mode);
    hipDeviceSynchronize();
    hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
    printf("=== addr_mode %d (lane: pointed -> got[0..3])\n", mode);
    for (int l = 0; l < 64; ++l) {
      printf("l%02d:%4d ->%4d %4d %4d %4d%s", l, h[l * 5], h[l * 5 + 1],
             h[l * 5 + 2], h[l * 5 + 3], h[l * 5 + 4],
             (l % 4 == 3) ? "\n" : "   ");
    }
  }
  return 0;
}
