// Flash attention (causal, fused ALiBi) for CDNA4 gfx950 — MFMA bf16.
//
// Replaces flash-attn 2.6.3's CUDA kernels (reference install_env.sh:71;
// SURVEY.md §2.3 rows 1-2). MI355X-native structure (not a port):
//
//   * 256-thread workgroups = 4 wave64s; each wave owns a 32-row Q block
//     (fwd / dQ) or 32 keys (dK/dV); mfma_f32_32x32x16_bf16 tiles.
//   * "Swapped" QK^T — mfma(A=K, B=Q) gives S[key][q] with q = lane&31, so
//     the online-softmax state (m, l) is lane-local: rescales are scalar
//     per lane, row reductions are 16 regs + one shfl_xor(32).
//   * P (f32 regs) is converted to the next MFMA's B-operand fragments
//     in-register with v_cvt_pk_bf16_f32 + permlane32_swap — no LDS
//     round-trip for P.
//   * K/V tiles staged in LDS: row image [32][D] with a ((row&15)<<4) XOR
//     byte swizzle (conflict-free b128 column reads) + an explicitly
//     transposed image [D][32] for the A-operands of PV / dQ / dK / dV
//     (upgrade path: ds_read_b64_tr_b16 hardware transpose reads).
//   * Online softmax carries LSE out for the backward; backward is the
//     standard FlashAttention-2 split: one kernel for dK/dV (blocks own key
//     tiles, loop over Q) and one for dQ (blocks own Q tiles, loop over
//     KV) — atomics-free and bit-deterministic.
//
// v1 is correctness-first: single-buffered LDS, one barrier pair per tile.

#include "host_common.h"

namespace photon_hip {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(16))) float f32x16;

constexpr int ATT_BLOCK = 256;   // 4 waves
constexpr int QB = 32;           // q rows per wave
constexpr int KB = 32;           // keys per kv tile
constexpr int WAVES = 4;

DEV_INLINE unsigned swz(unsigned byte, int row) {
  return byte ^ (((unsigned)row & 15u) << 4);
}

DEV_INLINE unsigned cvt_pk_bf16(float lo, float hi) {
  unsigned r;
  asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(lo), "v"(hi));
  return r;
}

// Pack 8 consecutive f32 P-regs (reg base rb) into the MFMA B-operand
// fragment for one 16-deep k-step. Derivation: reg r holds row index
// (r&3)+8*(r>>2)+4*hi; the B fragment wants rows 8*hi+jj. cvt_pk pairs +
// permlane32_swap (upper(vdst) <-> lower(src)) rearrange exactly that.
DEV_INLINE bf16x8 pack_bfrag(const float* p, int rb) {
  unsigned d0 = cvt_pk_bf16(p[rb + 0], p[rb + 1]);
  unsigned d1 = cvt_pk_bf16(p[rb + 2], p[rb + 3]);
  unsigned d2 = cvt_pk_bf16(p[rb + 4], p[rb + 5]);
  unsigned d3 = cvt_pk_bf16(p[rb + 6], p[rb + 7]);
  auto r02 = __builtin_amdgcn_permlane32_swap(d0, d2, false, false);
  auto r13 = __builtin_amdgcn_permlane32_swap(d1, d3, false, false);
  union {
    unsigned u[4];
    bf16x8 v;
  } out;
  out.u[0] = r02[0];
  out.u[1] = r13[0];
  out.u[2] = r02[1];
  out.u[3] = r13[1];
  return out.v;
}

// Load a 16-byte (8 x bf16) fragment from an LDS row image with the XOR
// swizzle. byte = row*rowstride + coloff must be 16B-aligned pre-swizzle.
DEV_INLINE bf16x8 lds_frag(const __bf16* img, int row, int rowstride_b,
                           int coloff_b) {
  unsigned byte = swz((unsigned)(row * rowstride_b + coloff_b), row);
  return *(const bf16x8*)((const char*)img + byte);
}

DEV_INLINE void lds_store16(__bf16* img, int row, int rowstride_b,
                            int coloff_b, bf16x8 v) {
  unsigned byte = swz((unsigned)(row * rowstride_b + coloff_b), row);
  *(bf16x8*)((char*)img + byte) = v;
}

DEV_INLINE void lds_store2(__bf16* img, int row, int rowstride_b,
                           int coloff_b, __bf16 v) {
  unsigned byte = swz((unsigned)(row * rowstride_b + coloff_b), row);
  *(__bf16*)((char*)img + byte) = v;
}

// Stage a [rows=32][D] global tile into (a) the row image (swizzled,
// row stride D*2 bytes) and (b) the transposed image [D][32] (swizzled,
// row stride 64 bytes). Cooperative across the whole 256-thread block.
// Rows >= rows_valid are zero-filled.
template <int D>
DEV_INLINE void stage_tile(const __bf16* __restrict__ gsrc, long g_row0,
                           long g_rows_total, long g_row_stride,
                           __bf16* row_img, __bf16* t_img) {
  constexpr int CHUNKS = 32 * D / 8;  // 16B chunks
  for (int c = threadIdx.x; c < CHUNKS; c += ATT_BLOCK) {
    const int row = c / (D / 8);
    const int col = (c % (D / 8)) * 8;
    bf16x8 v;
    const long grow = g_row0 + row;
    if (grow < g_rows_total) {
      v = *(const bf16x8*)(gsrc + grow * g_row_stride + col);
    } else {
      v = bf16x8{};
    }
    if (row_img) lds_store16(row_img, row, D * 2, col * 2, v);
    if (t_img) {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        lds_store2(t_img, col + j, 64, row * 2, v[j]);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Forward
// ---------------------------------------------------------------------------
template <int D>
__global__ __launch_bounds__(ATT_BLOCK) void attn_fwd_kernel(
    const __bf16* __restrict__ q, const __bf16* __restrict__ k,
    const __bf16* __restrict__ v, const float* __restrict__ slopes,
    __bf16* __restrict__ out, float* __restrict__ lse_out, int S, int H,
    int causal) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  __bf16* k_img = (__bf16*)smem;                  // [32][D] swizzled
  __bf16* vt_img = (__bf16*)(smem + 64 * D);      // [D][32] swizzled
  // epilogue bounce reuses smem from offset 0: [wave][32][D]

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int hi = lane >> 5;
  const int lq = lane & 31;
  const long bh = blockIdx.y;
  const int h = bh % H;
  const float slope = slopes[h];
  const float scale = rsqrtf((float)D);

  const long base = bh * (long)S * D;
  const int q0 = blockIdx.x * (WAVES * QB) + wave * QB;
  const int my_q = q0 + lq;  // this lane's q row

  // Q fragments in registers: B-operand, frag kk covers dh [kk*16, kk*16+16)
  bf16x8 qfrag[D / 16];
  {
    const long qrow = base + (long)min(my_q, S - 1) * D;
#pragma unroll
    for (int kk = 0; kk < D / 16; ++kk) {
      if (my_q < S) {
        qfrag[kk] = *(const bf16x8*)(q + qrow + kk * 16 + 8 * hi);
      } else {
        qfrag[kk] = bf16x8{};
      }
    }
  }

  float m_run = -INFINITY;
  float l_run = 0.f;
  f32x16 o_acc[D / 32];
#pragma unroll
  for (int db = 0; db < D / 32; ++db) o_acc[db] = f32x16{};

  const int q_max_block = min(blockIdx.x * (WAVES * QB) + WAVES * QB - 1, S - 1);
  const int n_tiles = causal ? (q_max_block / KB + 1) : ((S + KB - 1) / KB);
  const int my_q_max = min(q0 + QB - 1, S - 1);

  for (int t = 0; t < n_tiles; ++t) {
    const int kv0 = t * KB;
    stage_tile<D>(k + base, kv0, S, D, k_img + 0, nullptr);
    stage_tile<D>(v + base, kv0, S, D, nullptr, vt_img);
    __syncthreads();

    const bool active = !causal || (kv0 <= my_q_max);
    if (active) {
      // S[key][q] = K Q^T : A = K row frags, B = Q regs
      f32x16 s_acc = f32x16{};
#pragma unroll
      for (int kk = 0; kk < D / 16; ++kk) {
        bf16x8 a = lds_frag(k_img, lq, D * 2, kk * 32 + hi * 16);
        s_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, qfrag[kk], s_acc,
                                                        0, 0, 0);
      }
      // scale + ALiBi + causal mask; online softmax
      float p[16];
      float tile_max = -INFINITY;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int key = kv0 + (r & 3) + 8 * (r >> 2) + 4 * hi;
        float sv = s_acc[r] * scale - slope * (float)(my_q - key);
        const bool masked = (key >= S) || (causal && key > my_q);
        sv = masked ? -INFINITY : sv;
        p[r] = sv;
        tile_max = fmaxf(tile_max, sv);
      }
      tile_max = fmaxf(tile_max, __shfl_xor(tile_max, 32, 64));
      const float m_new = fmaxf(m_run, tile_max);
      if (m_new != -INFINITY) {
        const float alpha = (m_run == -INFINITY) ? 0.f : __expf(m_run - m_new);
        float l_add = 0.f;
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          p[r] = (p[r] == -INFINITY) ? 0.f : __expf(p[r] - m_new);
          l_add += p[r];
        }
        l_add += __shfl_xor(l_add, 32, 64);
        l_run = l_run * alpha + l_add;
        m_run = m_new;
#pragma unroll
        for (int db = 0; db < D / 32; ++db) {
#pragma unroll
          for (int r = 0; r < 16; ++r) o_acc[db][r] *= alpha;
        }
        // PV: D2[dh][q] += V^T P ; A = V^T frags (transposed img), B = P
#pragma unroll
        for (int s16 = 0; s16 < 2; ++s16) {
          bf16x8 pfrag = pack_bfrag(p, 8 * s16);
#pragma unroll
          for (int db = 0; db < D / 32; ++db) {
            bf16x8 a =
                lds_frag(vt_img, db * 32 + lq, 64, s16 * 32 + hi * 16);
            o_acc[db] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                a, pfrag, o_acc[db], 0, 0, 0);
          }
        }
      }
    }
    __syncthreads();
  }

  // epilogue: bounce O through LDS for coalesced stores
  __bf16* o_img = (__bf16*)smem + wave * 32 * D;  // per-wave [32][D], linear
  const float inv_l = (l_run > 0.f) ? 1.f / l_run : 0.f;
#pragma unroll
  for (int db = 0; db < D / 32; ++db) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int dh = db * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
      o_img[lq * D + dh] = (__bf16)(o_acc[db][r] * inv_l);
    }
  }
  if (hi == 0 && my_q < S && lse_out) {
    lse_out[bh * (long)S + my_q] = m_run + __logf(l_run);
  }
  __syncthreads();
  // each wave stores its own 32 rows
  for (int c = lane; c < 32 * D / 8; c += 64) {
    const int row = c / (D / 8);
    const int col = (c % (D / 8)) * 8;
    if (q0 + row < S) {
      *(bf16x8*)(out + base + (long)(q0 + row) * D + col) =
          *(const bf16x8*)(o_img + row * D + col);
    }
  }
}

// ---------------------------------------------------------------------------
// Backward dQ: blocks own Q tiles, loop over KV tiles.
// dQ^T[dh][q] = sum_key K^T[dh][key] * dS[key][q] * scale
// ---------------------------------------------------------------------------
template <int D>
__global__ __launch_bounds__(ATT_BLOCK) void attn_bwd_dq_kernel(
    const __bf16* __restrict__ dout, const __bf16* __restrict__ q,
    const __bf16* __restrict__ k, const __bf16* __restrict__ v,
    const float* __restrict__ slopes, const float* __restrict__ lse,
    const float* __restrict__ delta, __bf16* __restrict__ dq, int S, int H,
    int causal) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  __bf16* k_img = (__bf16*)smem;                    // [32][D]
  __bf16* v_img = (__bf16*)(smem + 64 * D);         // [32][D]
  __bf16* kt_img = (__bf16*)(smem + 128 * D);       // [D][32]

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int hi = lane >> 5;
  const int lq = lane & 31;
  const long bh = blockIdx.y;
  const int h = bh % H;
  const float slope = slopes[h];
  const float scale = rsqrtf((float)D);
  const long base = bh * (long)S * D;
  const int q0 = blockIdx.x * (WAVES * QB) + wave * QB;
  const int my_q = q0 + lq;

  bf16x8 qfrag[D / 16], dofrag[D / 16];
  {
    const long row = base + (long)min(my_q, S - 1) * D;
#pragma unroll
    for (int kk = 0; kk < D / 16; ++kk) {
      if (my_q < S) {
        qfrag[kk] = *(const bf16x8*)(q + row + kk * 16 + 8 * hi);
        dofrag[kk] = *(const bf16x8*)(dout + row + kk * 16 + 8 * hi);
      } else {
        qfrag[kk] = bf16x8{};
        dofrag[kk] = bf16x8{};
      }
    }
  }
  const float my_lse = (my_q < S) ? lse[bh * (long)S + my_q] : INFINITY;
  const float my_delta = (my_q < S) ? delta[bh * (long)S + my_q] : 0.f;

  f32x16 dq_acc[D / 32];
#pragma unroll
  for (int db = 0; db < D / 32; ++db) dq_acc[db] = f32x16{};

  const int q_max_block = min(blockIdx.x * (WAVES * QB) + WAVES * QB - 1, S - 1);
  const int n_tiles = causal ? (q_max_block / KB + 1) : ((S + KB - 1) / KB);
  const int my_q_max = min(q0 + QB - 1, S - 1);

  for (int t = 0; t < n_tiles; ++t) {
    const int kv0 = t * KB;
    stage_tile<D>(k + base, kv0, S, D, k_img, kt_img);
    stage_tile<D>(v + base, kv0, S, D, v_img, nullptr);
    __syncthreads();

    const bool active = !causal || (kv0 <= my_q_max);
    if (active) {
      f32x16 s_acc = f32x16{}, dp_acc = f32x16{};
#pragma unroll
      for (int kk = 0; kk < D / 16; ++kk) {
        bf16x8 ka = lds_frag(k_img, lq, D * 2, kk * 32 + hi * 16);
        bf16x8 va = lds_frag(v_img, lq, D * 2, kk * 32 + hi * 16);
        s_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(ka, qfrag[kk], s_acc,
                                                        0, 0, 0);
        dp_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(va, dofrag[kk],
                                                         dp_acc, 0, 0, 0);
      }
      float ds[16];
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int key = kv0 + (r & 3) + 8 * (r >> 2) + 4 * hi;
        float sv = s_acc[r] * scale - slope * (float)(my_q - key);
        const bool masked = (key >= S) || (causal && key > my_q);
        const float pv = masked ? 0.f : __expf(sv - my_lse);
        ds[r] = pv * (dp_acc[r] - my_delta) * scale;
      }
#pragma unroll
      for (int s16 = 0; s16 < 2; ++s16) {
        bf16x8 dsfrag = pack_bfrag(ds, 8 * s16);
#pragma unroll
        for (int db = 0; db < D / 32; ++db) {
          bf16x8 a = lds_frag(kt_img, db * 32 + lq, 64, s16 * 32 + hi * 16);
          dq_acc[db] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              a, dsfrag, dq_acc[db], 0, 0, 0);
        }
      }
    }
    __syncthreads();
  }

  // epilogue
  __bf16* o_img = (__bf16*)smem + wave * 32 * D;
#pragma unroll
  for (int db = 0; db < D / 32; ++db) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int dh = db * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
      o_img[lq * D + dh] = (__bf16)dq_acc[db][r];
    }
  }
  __syncthreads();
  for (int c = lane; c < 32 * D / 8; c += 64) {
    const int row = c / (D / 8);
    const int col = (c % (D / 8)) * 8;
    if (q0 + row < S) {
      *(bf16x8*)(dq + base + (long)(q0 + row) * D + col) =
          *(const bf16x8*)(o_img + row * D + col);
    }
  }
}

// ---------------------------------------------------------------------------
// Backward dK/dV: blocks own key tiles, loop over Q tiles.
//   dV^T[dh][key] = sum_q dO^T[dh][q] P[q][key]
//   dK^T[dh][key] = sum_q Q^T[dh][q] dS[q][key] * scale
// ---------------------------------------------------------------------------
template <int D>
__global__ __launch_bounds__(ATT_BLOCK) void attn_bwd_dkdv_kernel(
    const __bf16* __restrict__ dout, const __bf16* __restrict__ q,
    const __bf16* __restrict__ k, const __bf16* __restrict__ v,
    const float* __restrict__ slopes, const float* __restrict__ lse,
    const float* __restrict__ delta, __bf16* __restrict__ dk,
    __bf16* __restrict__ dv, int S, int H, int causal) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  __bf16* q_img = (__bf16*)smem;                    // [32][D]
  __bf16* do_img = (__bf16*)(smem + 64 * D);        // [32][D]
  __bf16* qt_img = (__bf16*)(smem + 128 * D);       // [D][32]
  __bf16* dot_img = (__bf16*)(smem + 192 * D);      // [D][32]
  float* lse_t = (float*)(smem + 256 * D);          // [32]
  float* del_t = lse_t + 32;                        // [32]

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int hi = lane >> 5;
  const int lq = lane & 31;
  const long bh = blockIdx.y;
  const int h = bh % H;
  const float slope = slopes[h];
  const float scale = rsqrtf((float)D);
  const long base = bh * (long)S * D;
  const int k0 = blockIdx.x * (WAVES * KB) + wave * KB;
  const int my_key = k0 + lq;

  // K, V rows of this wave's keys as B-operand fragments (like Q in fwd)
  bf16x8 kfrag[D / 16], vfrag[D / 16];
  {
    const long row = base + (long)min(my_key, S - 1) * D;
#pragma unroll
    for (int kk = 0; kk < D / 16; ++kk) {
      if (my_key < S) {
        kfrag[kk] = *(const bf16x8*)(k + row + kk * 16 + 8 * hi);
        vfrag[kk] = *(const bf16x8*)(v + row + kk * 16 + 8 * hi);
      } else {
        kfrag[kk] = bf16x8{};
        vfrag[kk] = bf16x8{};
      }
    }
  }

  f32x16 dk_acc[D / 32], dv_acc[D / 32];
#pragma unroll
  for (int db = 0; db < D / 32; ++db) {
    dk_acc[db] = f32x16{};
    dv_acc[db] = f32x16{};
  }

  const int k_min_block = blockIdx.x * (WAVES * KB);
  const int t0 = causal ? (k_min_block / QB) : 0;
  const int n_tiles = (S + QB - 1) / QB;
  const int my_k_min = k0;

  for (int t = t0; t < n_tiles; ++t) {
    const int qt0 = t * QB;
    stage_tile<D>(q + base, qt0, S, D, q_img, qt_img);
    stage_tile<D>(dout + base, qt0, S, D, do_img, dot_img);
    for (int i = threadIdx.x; i < 32; i += ATT_BLOCK) {
      const int qi = qt0 + i;
      lse_t[i] = (qi < S) ? lse[bh * (long)S + qi] : INFINITY;
      del_t[i] = (qi < S) ? delta[bh * (long)S + qi] : 0.f;
    }
    __syncthreads();

    const bool active = !causal || (qt0 + QB - 1 >= my_k_min);
    if (active) {
      // S'[q][key]: A = Q row frags, B = K regs; dP'[q][key]: A = dO, B = V
      f32x16 s_acc = f32x16{}, dp_acc = f32x16{};
#pragma unroll
      for (int kk = 0; kk < D / 16; ++kk) {
        bf16x8 qa = lds_frag(q_img, lq, D * 2, kk * 32 + hi * 16);
        bf16x8 doa = lds_frag(do_img, lq, D * 2, kk * 32 + hi * 16);
        s_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qa, kfrag[kk], s_acc,
                                                        0, 0, 0);
        dp_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(doa, vfrag[kk],
                                                         dp_acc, 0, 0, 0);
      }
      float p[16], ds[16];
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int qi = qt0 + (r & 3) + 8 * (r >> 2) + 4 * hi;
        const float l = lse_t[(r & 3) + 8 * (r >> 2) + 4 * hi];
        const float dlt = del_t[(r & 3) + 8 * (r >> 2) + 4 * hi];
        float sv = s_acc[r] * scale - slope * (float)(qi - my_key);
        const bool masked = (my_key >= S) || (causal && my_key > qi) || (qi >= S);
        p[r] = masked ? 0.f : __expf(sv - l);
        ds[r] = p[r] * (dp_acc[r] - dlt) * scale;
      }
#pragma unroll
      for (int s16 = 0; s16 < 2; ++s16) {
        bf16x8 pfrag = pack_bfrag(p, 8 * s16);
        bf16x8 dsfrag = pack_bfrag(ds, 8 * s16);
#pragma unroll
        for (int db = 0; db < D / 32; ++db) {
          bf16x8 doa =
              lds_frag(dot_img, db * 32 + lq, 64, s16 * 32 + hi * 16);
          bf16x8 qa = lds_frag(qt_img, db * 32 + lq, 64, s16 * 32 + hi * 16);
          dv_acc[db] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              doa, pfrag, dv_acc[db], 0, 0, 0);
          dk_acc[db] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              qa, dsfrag, dk_acc[db], 0, 0, 0);
        }
      }
    }
    __syncthreads();
  }

  // epilogue: two bounces (dk then dv) through per-wave LDS
  __bf16* o_img = (__bf16*)smem + wave * 32 * D;
  for (int which = 0; which < 2; ++which) {
    f32x16* acc = which == 0 ? dk_acc : dv_acc;
    __bf16* dst = which == 0 ? dk : dv;
#pragma unroll
    for (int db = 0; db < D / 32; ++db) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int dh = db * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
        o_img[lq * D + dh] = (__bf16)acc[db][r];
      }
    }
    __syncthreads();
    for (int c = lane; c < 32 * D / 8; c += 64) {
      const int row = c / (D / 8);
      const int col = (c % (D / 8)) * 8;
      if (k0 + row < S) {
        *(bf16x8*)(dst + base + (long)(k0 + row) * D + col) =
            *(const bf16x8*)(o_img + row * D + col);
      }
    }
    __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// Host launchers
// ---------------------------------------------------------------------------

static int attn_lds_bytes(int D, int kind) {
  // kind 0: fwd (2 images), 1: dq (3 images), 2: dkdv (4 images + 64 floats)
  const int img = 64 * D;  // bytes of one [32][D] bf16 image
  int imgs = kind == 0 ? 2 * img : (kind == 1 ? 3 * img : 4 * img + 256);
  int bounce = WAVES * img;
  return std::max(imgs, bounce);
}

std::vector<torch::Tensor> attn_fwd_launch(torch::Tensor q, torch::Tensor k,
                                           torch::Tensor v,
                                           torch::Tensor slopes, bool causal) {
  TORCH_CHECK(q.scalar_type() == at::kBFloat16,
              "attn_fwd: bf16 only (use impl='torch' for fp32)");
  TORCH_CHECK(q.dim() == 4, "attn_fwd: expected [B,H,S,D]");
  const int B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
  TORCH_CHECK(D == 64 || D == 128, "attn_fwd: d_head must be 64 or 128");
  TORCH_CHECK(q.is_contiguous() && k.is_contiguous() && v.is_contiguous());
  auto o = torch::empty_like(q);
  auto lse = torch::empty({B, H, S}, q.options().dtype(at::kFloat));
  auto slopes_f = slopes.to(q.device(), at::kFloat).contiguous();
  dim3 grid((S + WAVES * QB - 1) / (WAVES * QB), (long)B * H);
  const int lds = attn_lds_bytes(D, 0);
  if (D == 64) {
    hipLaunchKernelGGL((attn_fwd_kernel<64>), grid, dim3(ATT_BLOCK), lds,
                       cur_stream(), (const __bf16*)q.data_ptr(),
                       (const __bf16*)k.data_ptr(),
                       (const __bf16*)v.data_ptr(), slopes_f.data_ptr<float>(),
                       (__bf16*)o.data_ptr(), lse.data_ptr<float>(), S, H,
                       (int)causal);
  } else {
    hipLaunchKernelGGL((attn_fwd_kernel<128>), grid, dim3(ATT_BLOCK), lds,
                       cur_stream(), (const __bf16*)q.data_ptr(),
                       (const __bf16*)k.data_ptr(),
                       (const __bf16*)v.data_ptr(), slopes_f.data_ptr<float>(),
                       (__bf16*)o.data_ptr(), lse.data_ptr<float>(), S, H,
                       (int)causal);
  }
  return {o, lse};
}

std::vector<torch::Tensor> attn_bwd_launch(torch::Tensor dout, torch::Tensor q,
                                           torch::Tensor k, torch::Tensor v,
                                           torch::Tensor slopes,
                                           torch::Tensor o, torch::Tensor lse,
                                           bool causal) {
  const int B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
  auto dq = torch::empty_like(q);
  auto dk = torch::empty_like(k);
  auto dv = torch::empty_like(v);
  auto slopes_f = slopes.to(q.device(), at::kFloat).contiguous();
  // delta = rowsum(dO * O), fp32
  auto delta = (dout.to(at::kFloat) * o.to(at::kFloat)).sum(-1).contiguous();
  dim3 grid((S + WAVES * QB - 1) / (WAVES * QB), (long)B * H);
#define LAUNCH_BWD(DD)                                                        \
  hipLaunchKernelGGL((attn_bwd_dq_kernel<DD>), grid, dim3(ATT_BLOCK),         \
                     attn_lds_bytes(DD, 1), cur_stream(),                     \
                     (const __bf16*)dout.data_ptr(),                          \
                     (const __bf16*)q.data_ptr(),                             \
                     (const __bf16*)k.data_ptr(),                             \
                     (const __bf16*)v.data_ptr(), slopes_f.data_ptr<float>(), \
                     lse.data_ptr<float>(), delta.data_ptr<float>(),          \
                     (__bf16*)dq.data_ptr(), S, H, (int)causal);              \
  hipLaunchKernelGGL((attn_bwd_dkdv_kernel<DD>), grid, dim3(ATT_BLOCK),       \
                     attn_lds_bytes(DD, 2), cur_stream(),                     \
                     (const __bf16*)dout.data_ptr(),                          \
                     (const __bf16*)q.data_ptr(),                             \
                     (const __bf16*)k.data_ptr(),                             \
                     (const __bf16*)v.data_ptr(), slopes_f.data_ptr<float>(), \
                     lse.data_ptr<float>(), delta.data_ptr<float>(),          \
                     (__bf16*)dk.data_ptr(), (__bf16*)dv.data_ptr(), S, H,    \
                     (int)causal)
  if (D == 64) {
    LAUNCH_BWD(64);
  } else {
    LAUNCH_BWD(128);
  }
#undef LAUNCH_BWD
  return {dq, dk, dv};
}

}  // namespace photon_hip
