// Debug probes for MFMA operand lane layouts on gfx950.
// mfma_probe: loads A/B fragments under the "contiguous-8" assumption
// (lane l elem j -> k = 8*(l>>5)+j) and returns D under the documented C/D
// map (col=lane&31, row=(reg&3)+8*(reg>>2)+4*(lane>>5)). Host code compares
// against candidate layouts to pin down the real operand maps.

#include "host_common.h"

namespace photon_hip {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8_d;
typedef __attribute__((ext_vector_type(16))) float f32x16_d;

__global__ void mfma_probe_kernel(const __bf16* __restrict__ A,
                                  const __bf16* __restrict__ BT,
                                  float* __restrict__ D) {
  const int lane = threadIdx.x & 63;
  if (threadIdx.x >= 64) return;
  const int hi = lane >> 5;
  const int lq = lane & 31;
  bf16x8_d a = *(const bf16x8_d*)(A + lq * 16 + 8 * hi);
  bf16x8_d b = *(const bf16x8_d*)(BT + lq * 16 + 8 * hi);
  f32x16_d acc = {};
  acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int row = (r & 3) + 8 * (r >> 2) + 4 * hi;
    D[row * 32 + lq] = acc[r];
  }
}

// Probe pack_bfrag: each lane fills p[16] with p[r] = 1000*lane_row(r) + lq
// where lane_row(r) = (r&3)+8*(r>>2)+4*hi; pack_bfrag(rb=0) output dumped as
// 8 bf16 per lane so the host can see which (row, col) each frag slot holds.
DEV_INLINE unsigned cvt_pk_bf16_d(float lo, float hi2) {
  unsigned r;
  asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(lo), "v"(hi2));
  return r;
}

__global__ void pack_probe_kernel(float* __restrict__ out) {
  const int lane = threadIdx.x & 63;
  if (threadIdx.x >= 64) return;
  const int hi = lane >> 5;
  const int lq = lane & 31;
  float p[8];
#pragma unroll
  for (int r = 0; r < 8; ++r) {
    const int row = (r & 3) + 8 * (r >> 2) + 4 * hi;
    p[r] = (float)(row * 64 + lq);
  }
  unsigned d0 = cvt_pk_bf16_d(p[0], p[1]);
  unsigned d1 = cvt_pk_bf16_d(p[2], p[3]);
  unsigned d2 = cvt_pk_bf16_d(p[4], p[5]);
  unsigned d3 = cvt_pk_bf16_d(p[6], p[7]);
  auto r02 = __builtin_amdgcn_permlane32_swap(d0, d2, false, false);
  auto r13 = __builtin_amdgcn_permlane32_swap(d1, d3, false, false);
  unsigned u[4] = {(unsigned)r02[0], (unsigned)r13[0], (unsigned)r02[1],
                   (unsigned)r13[1]};
#pragma unroll
  for (int d = 0; d < 4; ++d) {
    out[(lane * 8 + 2 * d) ] = bf16_to_f32((unsigned short)(u[d] & 0xffff));
    out[(lane * 8 + 2 * d + 1)] = bf16_to_f32((unsigned short)(u[d] >> 16));
  }
}

torch::Tensor mfma_probe(torch::Tensor A, torch::Tensor BT) {
  auto D = torch::zeros({32, 32}, A.options().dtype(at::kFloat));
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, cur_stream(),
                     (const __bf16*)A.data_ptr(), (const __bf16*)BT.data_ptr(),
                     D.data_ptr<float>());
  return D;
}

torch::Tensor pack_probe(torch::Tensor dummy) {
  auto out = torch::zeros({64, 8}, dummy.options().dtype(at::kFloat));
  hipLaunchKernelGGL(pack_probe_kernel, dim3(1), dim3(64), 0, cur_stream(),
                     out.data_ptr<float>());
  return out;
}

}  // namespace photon_hip
