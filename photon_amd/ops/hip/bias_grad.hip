// Bias-gradient column reduce: db[n] = sum_m dy[m, n] — replaces
// at::native::reduce_kernel (measured 1.7 TB/s, ~2.7% of the 125M step)
// with a coalesced two-stage reduction.
//
// Stage 1: grid (m_chunks, ceil(N/1024)); each 256-thread block owns 1024
// consecutive columns (4 bf16 = one 8-byte load per thread per row) and a
// contiguous row range, accumulating fp32 partials — consecutive threads
// read consecutive columns, so each row access is one 2 KB contiguous
// burst per block. Stage 2 folds the m_chunks partials. No atomics:
// bit-deterministic (fixed reduction order), matching the attention
// backward's atomics-free design.
//
// hipBLASLt's BGRADB epilogue was measured as the alternative and rejected:
// fusing the bias grad into the dW GEMM disables split-K kernel selection,
// and the reduction-heavy dW shapes (k = tokens) then run ~10x slower
// (MPT-1B step 1704 -> 3515 ms). See scripts/lt_epi_probe.hip notes.

#include "host_common.h"

namespace photon_hip {

typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4g;

__global__ __launch_bounds__(256) void bias_grad_stage1(
    const __bf16* __restrict__ dy, float* __restrict__ partial, long M,
    long N, long rows_per_chunk) {
  const long col0 = (long)blockIdx.y * 1024 + (long)threadIdx.x * 4;
  if (col0 >= N) return;
  const long m0 = (long)blockIdx.x * rows_per_chunk;
  const long m1 = min(m0 + rows_per_chunk, M);
  float acc0 = 0.f, acc1 = 0.f, acc2 = 0.f, acc3 = 0.f;
  const __bf16* p = dy + m0 * N + col0;
  for (long m = m0; m < m1; ++m, p += N) {
    const bf16x4g v = *(const bf16x4g*)p;
    acc0 += (float)v[0];
    acc1 += (float)v[1];
    acc2 += (float)v[2];
    acc3 += (float)v[3];
  }
  float* out = partial + (long)blockIdx.x * N + col0;
  out[0] = acc0;
  out[1] = acc1;
  out[2] = acc2;
  out[3] = acc3;
}

template <typename OutT>
__global__ __launch_bounds__(256) void bias_grad_stage2(
    const float* __restrict__ partial, OutT* __restrict__ db, long N,
    int m_chunks) {
  const long col = (long)blockIdx.x * 256 + threadIdx.x;
  if (col >= N) return;
  float acc = 0.f;
  for (int c = 0; c < m_chunks; ++c) acc += partial[(long)c * N + col];
  db[col] = (OutT)acc;
}

// dy: [M, N] bf16 contiguous -> returns db [N] in dy's dtype.
torch::Tensor bias_grad(torch::Tensor dy) {
  TORCH_CHECK(dy.is_cuda() && dy.dim() == 2 && dy.is_contiguous() &&
                  dy.scalar_type() == at::kBFloat16,
              "bias_grad: need contiguous 2-D bf16");
  const long M = dy.size(0), N = dy.size(1);
  TORCH_CHECK(N % 4 == 0, "bias_grad: N must be a multiple of 4");
  // enough chunks to fill the chip (>=1024 blocks total), capped so the
  // partial buffer stays tiny
  const long col_blocks = (N + 1023) / 1024;
  int m_chunks = (int)std::min<long>((1024 + col_blocks - 1) / col_blocks,
                                     (M + 255) / 256);
  m_chunks = std::max(m_chunks, 1);
  const long rows_per_chunk = (M + m_chunks - 1) / m_chunks;
  auto partial = at::empty({m_chunks, N}, dy.options().dtype(at::kFloat));
  auto db = at::empty({N}, dy.options());
  dim3 g1(m_chunks, col_blocks);
  hipLaunchKernelGGL(bias_grad_stage1, g1, dim3(256), 0, cur_stream(),
                     (const __bf16*)dy.data_ptr(), partial.data_ptr<float>(),
                     M, N, rows_per_chunk);
  hipLaunchKernelGGL((bias_grad_stage2<__bf16>), dim3((N + 255) / 256),
                     dim3(256), 0, cur_stream(), partial.data_ptr<float>(),
                     (__bf16*)db.data_ptr(), N, m_chunks);
  return db;
}

}  // namespace photon_hip
