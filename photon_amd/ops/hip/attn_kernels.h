// Flash attention (causal, fused ALiBi) for CDNA4 gfx950 — MFMA bf16.
//
// Replaces flash-attn 2.6.3's CUDA kernels (reference install_env.sh:71;
// SURVEY.md §2.3 rows 1-2). MI355X-native structure (not a port); round-2
// state (measured numbers in profiles/r02_final_SUMMARY.md):
//
//   * 256-thread workgroups = 4 wave64s; each wave owns a 32-row Q block
//     (fwd / dQ) or 32 keys (dK/dV); mfma_f32_32x32x16_bf16 tiles.
//   * "Swapped" QK^T — mfma(A=K, B=Q) gives S[key][q] with q = lane&31, so
//     the online-softmax state (m, l) is lane-local: rescales are scalar
//     per lane, row reductions are 16 regs + one shfl_xor(32).
//   * exp2-DOMAIN softmax: scale/slopes pre-multiplied by log2(e), every
//     exponential is one native v_exp_f32; LSE converted to natural log at
//     the epilogue (wire format unchanged). Defer-max (T13, THR=8).
//   * The KV loop is SPLIT into interior tiles (every element unmasked for
//     the whole block: straight-line softmax, zero compares) and boundary
//     tiles (diagonal/tail masking via the -1e30 sentinel).
//   * K/V tiles reach LDS by LDS-DMA (buffer_load..lds, 1 KiB per
//     wave-instruction, SRSRC descriptor with flags 0x27FAC and
//     num_records clamping rows >= S to hardware zeros): no staging
//     registers, no ds_write issue cost. The zero-bank-conflict XOR
//     swizzle (GF(2)-searched, row-preserving) is applied to the DMA
//     SOURCE offsets; double-buffered, one barrier + one counted
//     vmcnt(0) per tile. Occupancy: fwd64 4 waves/SIMD, dq64 3, others 2.
//   * P (f32 regs) is converted to the next MFMA's B-operand fragments
//     in-register with v_cvt_pk_bf16_f32 + permlane32_swap; A-operands of
//     PV / dQ / dK / dV come from ds_read_b64_tr_b16 hardware transpose
//     reads of the row images — no transposed image, no LDS round-trip
//     for P.
//   * Online softmax carries LSE out for the backward; backward is the
//     standard FlashAttention-2 split: one kernel for dK/dV (blocks own key
//     tiles, loop over Q) and one for dQ (blocks own Q tiles, loop over
//     KV) — atomics-free and bit-deterministic.

#pragma once
#include "common.h"

namespace photon_hip {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(16))) float f32x16;

#ifndef ATT_WAVES
#define ATT_WAVES 4  // waves per workgroup (8-wave variant: -DATT_WAVES=8)
#endif
constexpr int WAVES = ATT_WAVES;
constexpr int ATT_BLOCK = 64 * WAVES;
constexpr int QB = 32;           // q rows per wave
constexpr int KB = 32;           // keys per kv tile

// D=128 occupancy/spill A/B (measured): dq wins at 2 waves/SIMD despite
// ~100 B/lane spill (385 vs 244 TF/s); dkdv loses badly with its ~300 B
// spill (99 vs 177 TF/s) -> spill-free 1 wave/SIMD there.
#ifndef ATT_DQ_MINWAVES_D128
#define ATT_DQ_MINWAVES_D128 2
#endif
#ifndef ATT_DKDV_MINWAVES_D128
#define ATT_DKDV_MINWAVES_D128 1
#endif

// XOR swizzle field per row, found by exhaustive search over GF(2)-linear
// tables (scripts/swz_search.py): zero LDS bank conflicts simultaneously
// for the b128 column reads, the ds_read_b64_tr_b16 transpose reads AND
// the b128 stores, at both D=64 and D=128 (the r01 identity field left
// the tr16 reads 2-way conflicted at D64, worse at D128 — measured
// SQ_LDS_BANK_CONFLICT 10-13% of wave cycles, now 2-3%).
//
// ROW-PRESERVING constraint: the field must not flip byte bits >= the row
// stride so the swizzle is an XOR involution WITHIN each row — that makes
// the LDS-DMA source permutation trivially invertible. At D=64 (128 B
// rows) that limits the field to 3 bits; the 4-bit half-swap is fine at
// D=128 (256 B rows).
DEV_INLINE unsigned swz_field(unsigned r, bool d64) {
  if (d64) {
    // best zero-conflict 3-bit GF(2)-linear map: t = {bit1->4, bit2->1,
    // bit3->2} of row&15
    return (((r >> 1) & 1u) << 2) | ((r >> 2) & 1u) | (((r >> 3) & 1u) << 1);
  }
  return ((r & 3u) << 2) | (r >> 2);  // nibble half-swap
}

DEV_INLINE unsigned swz(unsigned byte, int row, bool d64) {
  return byte ^ (swz_field((unsigned)row & 15u, d64) << 4);
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

typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;

constexpr float LOG2E = 1.4426950408889634f;
constexpr float LN2 = 0.6931471805599453f;

// Native v_exp_f32 (GCN v_exp IS 2^x): one transcendental, no ln2 multiply
// — the whole softmax runs in the exp2 domain (scale/slopes pre-multiplied
// by log2(e); LSE converted back to natural log at the epilogue so the
// wire format is unchanged). MUST be the intrinsic, not inline asm: the
// TRANS->VALU data hazard needs the compiler's s_nop padding, which it
// cannot insert around an opaque asm block (caused max-err 3.2 corruption).
#ifdef ABENCH_EXP_VIA_LN
// bisect aid: same exp2-domain math through the old __expf lowering
DEV_INLINE float exp2_fast(float x) { return __expf(x * LN2); }
#else
DEV_INLINE float exp2_fast(float x) { return __builtin_amdgcn_exp2f(x); }
#endif

typedef __attribute__((ext_vector_type(4))) float f32x4;

// Buffer-descriptor (SRSRC, guide T8) staging: scalar 128-bit descriptor +
// thread-fixed 32-bit voffset + per-tile SGPR soffset — zero per-chunk
// address VALU, and num_records clamps rows >= S to hardware ZEROS (no
// bounds compares, no tail branches).
//
// flags MUST carry a valid DATA_FORMAT (0x27FAC = dst_sel XYZW | NFMT
// float | DFMT 32): with flags=0 the FORMAT field is invalid and EVERY
// load returns zeros — measured, scripts/rsrc_micro.hip (this silently
// corrupted a whole round of kernels while LOOKING 15% faster thanks to
// the zero-data DVFS bonus).
DEV_INLINE __amdgpu_buffer_rsrc_t make_rsrc(const __bf16* base,
                                            long num_bytes) {
  return __builtin_amdgcn_make_buffer_rsrc((void*)base, (short)0,
                                           (int)num_bytes, 0x27FAC);
}

// LDS-DMA: one wave-instruction moves 1 KiB (16 B/lane) global -> LDS with
// NO data registers and no ds_write issue cost (buffer_load_dwordx4 ..lds).
// The LDS destination is linear in lanes, so the ROW-PRESERVING XOR
// swizzle is applied to the SOURCE offsets instead (involution per row).
DEV_INLINE void lds_dma16(__amdgpu_buffer_rsrc_t rsrc, char* lds_dest,
                          int voffset, int soffset) {
  typedef __attribute__((address_space(3))) void* lds_vp;
  __builtin_amdgcn_raw_ptr_buffer_load_lds(
      rsrc, (lds_vp)lds_dest, 16, voffset, soffset, 0, 0);
}

DEV_INLINE void vm_wait0() {
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
}

DEV_INLINE float log2_fast(float x) { return __builtin_amdgcn_logf(x); }

// Hardware transpose read (gfx950 ds_read_b64_tr_b16). Probed semantics
// (scripts/probe_tr.hip, addr_mode 3): per 16-lane group,
// out[j][elem k] = element (j&3) of the 8-byte chunk loaded by lane
// (4k + (j>>2)). With lane i pointing at row (i>>2), 4-elem column chunk
// 4*(i&3) of a row-major image, output lane j receives column j of rows
// 0..3 in NATURAL order. Used to build V^T / K^T / Q^T MFMA A-fragments
// straight from ROW-major LDS images (no scalar-store transposed image —
// that image's stores were 4-way (D=64) to 32-way (D=128) bank-conflicted).
DEV_INLINE bf16x4 lds_tr16(const __bf16* img, unsigned byte) {
  typedef __attribute__((address_space(3))) bf16x4* lds_v4p;
  typedef __attribute__((address_space(3))) char* lds_cp;
  return __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (lds_v4p)((lds_cp)img + byte));
}

// ---------------------------------------------------------------------------
// Forward
//
// v2 structure (each measured on MI355X, see profiles/):
//   * KV tile = 64 keys (KBF), double-buffered LDS, ONE barrier per tile.
//   * Register-staged loads issued at the top of the compute phase for the
//     NEXT tile (T14 async-stage: HBM latency hides under this tile's
//     MFMAs), ds_writes land after the compute, into the other buffer.
//   * Defer-max online softmax (T13, THR=8): o_acc rescale only when a
//     lane's tile max outgrows the running max by more than THR; P is then
//     bounded by e^8 in fp32 accumulate (accuracy cost ~3x vs THR=0,
//     validated against fp64 + spiked-key data in tests).
//   * Mask fast-path: tiles strictly below the causal diagonal skip the
//     mask compare per element.
// ---------------------------------------------------------------------------
constexpr int KBF = 64;  // fwd kv-tile keys

#ifndef ATT_FWD_MINWAVES
#define ATT_FWD_MINWAVES 2  // min waves/SIMD: caps register alloc at 256
#endif

// Generalized addressing: element (b, h, s, d) of an input lives at
// b*bs + h*hs + s*rs + d. Contiguous [B,H,S,D]: (H*S*D, S*D, D). Packed
// [B,S,3,H,D] qkv (the Wqkv output, no .contiguous() copies): q/k/v
// pointers pre-offset by their section, strides (S*3*H*D, D, 3*H*D).
template <int D>
__global__ __launch_bounds__(ATT_BLOCK, ATT_FWD_MINWAVES) void attn_fwd_kernel(
    const __bf16* __restrict__ q, const __bf16* __restrict__ k,
    const __bf16* __restrict__ v, const float* __restrict__ slopes,
    __bf16* __restrict__ out, float* __restrict__ lse_out, int S, int H,
    int causal, long bs_i, long hs_i, long rs_i, long bs_o, long hs_o,
    long rs_o) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  // Per buffer: k row image [KBF][D] + v transposed image [D][KBF].
  constexpr int IMG = KBF * D * 2;  // bytes per image
  // Buffer b: k at smem + b*2*IMG, v^T at smem + b*2*IMG + IMG.
  auto k_img = [&](int b) { return (__bf16*)(smem + b * 2 * IMG); };
  auto v_img = [&](int b) { return (__bf16*)(smem + b * 2 * IMG + IMG); };
  // epilogue bounce reuses smem from offset 0: [wave][32][D]

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int hi = lane >> 5;
  const int lq = lane & 31;
  const long bh = blockIdx.y;
  const int h = bh % H;
  const float slope = slopes[h];
  // exp2-domain softmax: scale/slope pre-multiplied by log2(e) so every
  // exponential is ONE native v_exp_f32 (no ln->log2 v_mul per element).
  const float scale2 = rsqrtf((float)D) * LOG2E;
  const float slope2 = slope * LOG2E;

  const long ibase = (bh / H) * bs_i + (long)h * hs_i;
  const long obase = (bh / H) * bs_o + (long)h * hs_o;
  const int q0_block = blockIdx.x * (WAVES * QB);
  const int q0 = q0_block + wave * QB;
  const int my_q = q0 + lq;  // this lane's q row

  // Q fragments in registers: B-operand, frag kk covers dh [kk*16, kk*16+16)
  bf16x8 qfrag[D / 16];
  {
    const long qrow = ibase + (long)min(my_q, S - 1) * rs_i;
#pragma unroll
    for (int kk = 0; kk < D / 16; ++kk) {
      if (my_q < S) {
        qfrag[kk] = *(const bf16x8*)(q + qrow + kk * 16 + 8 * hi);
      } else {
        qfrag[kk] = bf16x8{};
      }
    }
  }

  float m_run = -1e30f;  // finite sentinel: exp2(-huge) == 0, no NaN paths
  float l_run = 0.f;
  f32x16 o_acc[D / 32];
#pragma unroll
  for (int db = 0; db < D / 32; ++db) o_acc[db] = f32x16{};

  const int q_max_block = min(q0_block + WAVES * QB - 1, S - 1);
  const int n_tiles = causal ? (q_max_block / KBF + 1) : ((S + KBF - 1) / KBF);
  const int my_q_max = min(q0 + QB - 1, S - 1);
  // Interior tiles: every (key, q) of the BLOCK unmasked — keys strictly
  // below the block's first q row (causal) resp. full 64-key tiles within
  // S (non-causal). The interior loop body carries no mask compare, no
  // active/break checks and one straight-line softmax.
  int t_int = causal ? (q0_block / KBF) : (S / KBF);
  if (t_int > n_tiles) t_int = n_tiles;

  // Hoisted per-element ALiBi constants (exp2-domain): element r of an
  // MFMA result row covers key kv0s + pat(r), pat = (r&3)+8*(r>>2)+4*hi.
  // alibi2[r] = slope2*pat; the per-subtile remainder slope2*(kv0s - my_q)
  // is a per-lane constant that folds into the exp shift (mshift).
  float alibi2[16];
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int pat = (r & 3) + 8 * (r >> 2) + 4 * hi;
    alibi2[r] = slope2 * (float)pat;
  }

  // LDS-DMA staging: NW 1-KiB windows per wave per tensor go straight
  // from HBM to LDS (no staging registers, no ds_write issue cost, rows
  // >= S hardware-zeroed by num_records). Source offsets carry the
  // swizzle (involution within each row).
  constexpr int NW = KBF * D * 2 / 1024 / WAVES;
  const long ext_kv = ((long)(S - 1) * rs_i + D) * 2;
  const auto krs = make_rsrc(k + ibase, ext_kv);
  const auto vrs = make_rsrc(v + ibase, ext_kv);
  int dma_voff[NW];
#pragma unroll
  for (int i = 0; i < NW; ++i) {
    const int dest = (wave * NW + i) * 1024 + lane * 16;  // image byte
    const int row = dest / (D * 2);
    const int src = (int)(dest ^ (swz_field(row & 15u, D == 64) << 4));
    dma_voff[i] = (int)((long)(src / (D * 2)) * rs_i * 2 + src % (D * 2));
  }
  const int tile_soff = (int)(KBF * rs_i * 2);

  auto dma_stage = [&](int t, int b) {
    const int so = t * tile_soff;
#ifndef ABENCH_NO_LOAD
#pragma unroll
    for (int i = 0; i < NW; ++i) {
      const int win = (wave * NW + i) * 1024;
      lds_dma16(krs, (char*)k_img(b) + win, dma_voff[i], so);
      lds_dma16(vrs, (char*)v_img(b) + win, dma_voff[i], so);
    }
#else
    (void)so;
    (void)b;
#endif
  };
  dma_stage(0, 0);

#if ATT_WAVES == 8 && defined(ATT_SETPRIO)
  // 8-wave workgroups put two waves on each SIMD; the second-dispatched
  // half loses issue arbitration on every segment — one static priority
  // raise for it removes the start-of-segment penalty
  // (MI355X_MICROARCH.md two-waves-per-SIMD item 4).
  if (wave >= 4) __builtin_amdgcn_s_setprio(1);
#endif

  // Hoisted LDS read addresses (tile-loop invariant; the swizzle math was
  // otherwise recomputed per read, ~60 VALU per tile). Offsets BELOW the
  // swizzle's XOR bits (4..7) cannot be folded as adds, so each distinct
  // low-offset read gets its own precomputed address; the sub (+4096 B),
  // s16 (+2048 B) and buf (+2*IMG) offsets sit above bit 7 with no carry
  // from the low fields and fold into the ds-instruction immediate.
  const int pv_tj = lane & 15;
  const int pv_tg1 = (lane >> 4) & 1;
  unsigned qk_addr[D / 16];  // per-kk QK fragment address (sub=0, buf=0)
#pragma unroll
  for (int kk = 0; kk < D / 16; ++kk) {
    qk_addr[kk] = swz((unsigned)(lq * (D * 2) + kk * 32 + hi * 16), lq, D == 64);
  }
  unsigned pv_addr[2][D / 32];  // per (rd, db) tr-read address
#pragma unroll
  for (int rd = 0; rd < 2; ++rd) {
#pragma unroll
    for (int db = 0; db < D / 32; ++db) {
      const int key0 = 8 * hi + 4 * rd + (pv_tj >> 2);
      const int dhc0 = db * 32 + 16 * pv_tg1 + 4 * (pv_tj & 3);
      pv_addr[rd][db] = swz((unsigned)(key0 * (D * 2) + dhc0 * 2), key0, D == 64);
    }
  }

  // Shared PV body: P (f32 regs) -> bf16 fragments -> V^T MFMAs via
  // hardware transpose reads from the V ROW image. A-frag rows arrive
  // permuted within each 16-row half (epilogue un-permutes).
  auto pv_accum = [&](const float* p, int buf, int sub) {
#ifndef ABENCH_NO_PV
#pragma unroll
    for (int s16 = 0; s16 < 2; ++s16) {
      bf16x8 pfrag = pack_bfrag(p, 8 * s16);
#pragma unroll
      for (int db = 0; db < D / 32; ++db) {
        union { bf16x4 h[2]; bf16x8 v8; } a;
#pragma unroll
        for (int rd = 0; rd < 2; ++rd) {
          // hoisted: buf/sub/s16 offsets are immediates (the XOR field
          // key&15 is invariant to +32/+16 key steps)
          a.h[rd] = lds_tr16(
              v_img(0), pv_addr[rd][db] + buf * 2 * IMG +
                            sub * 32 * (D * 2) + s16 * 16 * (D * 2));
        }
        o_acc[db] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a.v8, pfrag,
                                                            o_acc[db], 0, 0, 0);
      }
    }
#endif
  };
  auto qk_mfma = [&](int buf, int sub) {
    f32x16 s_acc = f32x16{};
#ifndef ABENCH_NO_QK
#pragma unroll
    for (int kk = 0; kk < D / 16; ++kk) {
      // hoisted address + immediate offsets (sub*32 rows and buf are above
      // the XOR swizzle bits; row&15 == lq&15 is sub-invariant)
      bf16x8 a = *(const bf16x8*)((const char*)k_img(0) + qk_addr[kk] +
                                  buf * 2 * IMG + sub * 32 * (D * 2));
      s_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, qfrag[kk], s_acc,
                                                      0, 0, 0);
    }
#endif
    return s_acc;
  };
  auto defer_rescale = [&](float tile_max_abs) {
    // T13 defer-max: rescale only if some lane's max grew by > THR.
    constexpr float THR = 8.f;
    if (!__all(tile_max_abs - m_run <= THR)) {
      const float m_new = fmaxf(m_run, tile_max_abs);
      const float alpha = exp2_fast(m_run - m_new);  // exp2(-huge) == 0
      l_run *= alpha;
#pragma unroll
      for (int db = 0; db < D / 32; ++db) {
#pragma unroll
        for (int r = 0; r < 16; ++r) o_acc[db][r] *= alpha;
      }
      m_run = m_new;
    }
  };

  // ---- interior tiles: no masks, no active checks, straight-line ---------
  for (int t = 0; t < t_int; ++t) {
    const int buf = t & 1;
    vm_wait0();  // this tile's DMA landed
#ifndef ABENCH_NO_BARRIER
    __syncthreads();  // LDS[buf] visible to all waves; 1-buf reads done
#endif
    if (t + 1 < n_tiles) dma_stage(t + 1, 1 - buf);  // hide HBM under t
#pragma unroll
    for (int sub = 0; sub < 2; ++sub) {
      f32x16 s_acc = qk_mfma(buf, sub);
      // shifted softmax: p' = s*scale2 + alibi2[r]; the per-subtile,
      // per-lane constant b2 = slope2*(kv0s - my_q) shifts max and exp
      // uniformly and folds into mshift — 1 fma + 1 max per element.
      float p[16];
      float tmax = -1e30f;
#ifdef ABENCH_NO_SOFTMAX
#pragma unroll
      for (int r = 0; r < 16; ++r) p[r] = s_acc[r];
      (void)tmax;
#else
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        p[r] = fmaf(s_acc[r], scale2, alibi2[r]);
        tmax = fmaxf(tmax, p[r]);
      }
      tmax = fmaxf(tmax, __shfl_xor(tmax, 32, 64));
      const float b2 = slope2 * (float)(t * KBF + sub * 32 - my_q);
      defer_rescale(tmax + b2);
      const float mshift = m_run - b2;
      float l_add = 0.f;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
#ifndef ABENCH_NO_EXP
        p[r] = exp2_fast(p[r] - mshift);
#else
        p[r] = p[r] - mshift;  // bench-only: perf bisect without v_exp
#endif
        l_add += p[r];
      }
      l_add += __shfl_xor(l_add, 32, 64);
      l_run += l_add;
#endif  // ABENCH_NO_SOFTMAX
      pv_accum(p, buf, sub);
    }
  }

  // ---- boundary tiles: diagonal/tail masking (<= 2 per block) ------------
  for (int t = t_int; t < n_tiles; ++t) {
    const int buf = t & 1;
    const int kv0 = t * KBF;
    vm_wait0();
#ifndef ABENCH_NO_BARRIER
    __syncthreads();
#endif
    if (t + 1 < n_tiles) dma_stage(t + 1, 1 - buf);
    const bool active = !causal || (kv0 <= my_q_max);
    if (active) {
#pragma unroll
      for (int sub = 0; sub < 2; ++sub) {
        const int kv0s = kv0 + sub * 32;
        if (causal && kv0s > my_q_max) break;
        f32x16 s_acc = qk_mfma(buf, sub);
        // masked lanes get -1e30 so the exp pass needs no per-element
        // compare (exp2(-huge - m) == 0)
        const float abase2 = slope2 * (float)(kv0s - my_q);
        float p[16];
        float tile_max = -1e30f;
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int pat = (r & 3) + 8 * (r >> 2) + 4 * hi;
          float sv = fmaf(s_acc[r], scale2, alibi2[r] + abase2);
          const int key = kv0s + pat;
          const bool masked = (key >= S) || (causal && key > my_q);
          sv = masked ? -1e30f : sv;
          p[r] = sv;
          tile_max = fmaxf(tile_max, sv);
        }
        tile_max = fmaxf(tile_max, __shfl_xor(tile_max, 32, 64));
        defer_rescale(tile_max);
        float l_add = 0.f;
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          p[r] = exp2_fast(p[r] - m_run);  // masked: exp2(-huge) == 0
          l_add += p[r];
        }
        l_add += __shfl_xor(l_add, 32, 64);
        l_run += l_add;
        pv_accum(p, buf, sub);
      }
    }
  }
  __syncthreads();  // protect epilogue smem reuse

  // epilogue: bounce O through LDS for coalesced stores. The tr-read
  // addressing (probed: out[j][k] = elem (j&3) of lane (4k + (j>>2)))
  // delivers A-frag rows in NATURAL order — no permutation to undo.
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
    // wire format stays natural-log: lse = ln(sum exp(sv)) with sv in
    // nats = ln2 * (m2 + log2(l)) from the exp2-domain running state
    lse_out[bh * (long)S + my_q] = LN2 * (m_run + log2_fast(l_run));
  }
  __syncthreads();
  // each wave stores its own 32 rows
  for (int c = lane; c < 32 * D / 8; c += 64) {
    const int row = c / (D / 8);
    const int col = (c % (D / 8)) * 8;
    if (q0 + row < S) {
      *(bf16x8*)(out + obase + (long)(q0 + row) * rs_o + col) =
          *(const bf16x8*)(o_img + row * D + col);
    }
  }
}

// ---------------------------------------------------------------------------
// Backward dQ: blocks own Q tiles, loop over KV tiles.
// dQ^T[dh][q] = sum_key K^T[dh][key] * dS[key][q] * scale
// ---------------------------------------------------------------------------
template <int D>
// SRSRC staging dropped D=128 register pressure (211 VGPR vs >256 with the
// old 64-bit-address register staging): 2 waves/SIMD now fits at every D.
__global__ __launch_bounds__(ATT_BLOCK, (D <= 64) ? 2 : ATT_DQ_MINWAVES_D128)
void attn_bwd_dq_kernel(
    const __bf16* __restrict__ dout, const __bf16* __restrict__ q,
    const __bf16* __restrict__ k, const __bf16* __restrict__ v,
    const float* __restrict__ slopes, const float* __restrict__ lse,
    const float* __restrict__ delta, __bf16* __restrict__ dq, int S, int H,
    int causal, long bs_i, long hs_i, long rs_i, long bs_o, long hs_o,
    long rs_o) {
  // v2: 64-key double-buffered K/V tiles, ONE barrier per tile, register
  // staging (the fwd pipeline); K^T A-fragments via hardware tr16 reads.
  constexpr int KBQ = 64;           // kv keys per staged tile (2 subtiles)
  constexpr int IMG2 = KBQ * D * 2;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  auto k_img = [&](int b) { return (__bf16*)(smem + b * 2 * IMG2); };
  auto v_img = [&](int b) { return (__bf16*)(smem + b * 2 * IMG2 + IMG2); };

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int hi = lane >> 5;
  const int lq = lane & 31;
  const long bh = blockIdx.y;
  const int h = bh % H;
  const float slope = slopes[h];
  // exp2 domain (cf. fwd); log2(scale) = -log2(D)/2 is exact for D=2^k
  const float scale2 = rsqrtf((float)D) * LOG2E;
  const float slope2 = slope * LOG2E;
  constexpr float LOG2_SCALE = (D == 64) ? -3.0f : (D == 128) ? -3.5f : 0.f;
  static_assert(D == 64 || D == 128, "log2(scale) fold assumes D in {64,128}");
  const long ibase = (bh / H) * bs_i + (long)h * hs_i;
  const long obase = (bh / H) * bs_o + (long)h * hs_o;
  const int q0_block = blockIdx.x * (WAVES * QB);
  const int q0 = q0_block + wave * QB;
  const int my_q = q0 + lq;

  bf16x8 qfrag[D / 16], dofrag[D / 16];
  {
    const long qrow = ibase + (long)min(my_q, S - 1) * rs_i;
    const long orow = obase + (long)min(my_q, S - 1) * rs_o;
#pragma unroll
    for (int kk = 0; kk < D / 16; ++kk) {
      if (my_q < S) {
        qfrag[kk] = *(const bf16x8*)(q + qrow + kk * 16 + 8 * hi);
        dofrag[kk] = *(const bf16x8*)(dout + orow + kk * 16 + 8 * hi);
      } else {
        qfrag[kk] = bf16x8{};
        dofrag[kk] = bf16x8{};
      }
    }
  }
  // lse pre-multiplied into the exp2 domain
  const float my_lse2 =
      (my_q < S) ? lse[bh * (long)S + my_q] * LOG2E : INFINITY;
  const float my_delta = (my_q < S) ? delta[bh * (long)S + my_q] : 0.f;

  f32x16 dq_acc[D / 32];
#pragma unroll
  for (int db = 0; db < D / 32; ++db) dq_acc[db] = f32x16{};

  const int q_max_block = min(q0_block + WAVES * QB - 1, S - 1);
  const int n_tiles = causal ? (q_max_block / KBQ + 1) : ((S + KBQ - 1) / KBQ);
  const int my_q_max = min(q0 + QB - 1, S - 1);
  int t_int = causal ? (q0_block / KBQ) : (S / KBQ);
  if (t_int > n_tiles) t_int = n_tiles;

  // LDS-DMA staging (cf. fwd): swizzled SOURCE offsets, no staging regs
  constexpr int NW = KBQ * D * 2 / 1024 / WAVES;
  const long ext_kv = ((long)(S - 1) * rs_i + D) * 2;
  const auto krs = make_rsrc(k + ibase, ext_kv);
  const auto vrs = make_rsrc(v + ibase, ext_kv);
  int dma_voff[NW];
#pragma unroll
  for (int i = 0; i < NW; ++i) {
    const int dest = (wave * NW + i) * 1024 + lane * 16;
    const int row = dest / (D * 2);
    const int src = (int)(dest ^ (swz_field(row & 15u, D == 64) << 4));
    dma_voff[i] = (int)((long)(src / (D * 2)) * rs_i * 2 + src % (D * 2));
  }
  const int tile_soff = (int)(KBQ * rs_i * 2);
  auto dma_stage = [&](int t, int b) {
    const int so = t * tile_soff;
#pragma unroll
    for (int i = 0; i < NW; ++i) {
      const int win = (wave * NW + i) * 1024;
      lds_dma16(krs, (char*)k_img(b) + win, dma_voff[i], so);
      lds_dma16(vrs, (char*)v_img(b) + win, dma_voff[i], so);
    }
  };
  dma_stage(0, 0);

  // Hoisted LDS read addresses (cf. fwd): same k/v row-image layout, so
  // one address set serves both images (v_img = k_img + IMG2 immediate).
  const int pv_tj = lane & 15;
  const int pv_tg1 = (lane >> 4) & 1;
  unsigned qk_addr[D / 16];
#pragma unroll
  for (int kk = 0; kk < D / 16; ++kk) {
    qk_addr[kk] = swz((unsigned)(lq * (D * 2) + kk * 32 + hi * 16), lq, D == 64);
  }
  unsigned tr_addr[2][D / 32];
#pragma unroll
  for (int rd = 0; rd < 2; ++rd) {
#pragma unroll
    for (int db = 0; db < D / 32; ++db) {
      const int key0 = 8 * hi + 4 * rd + (pv_tj >> 2);
      const int dhc0 = db * 32 + 16 * pv_tg1 + 4 * (pv_tj & 3);
      tr_addr[rd][db] = swz((unsigned)(key0 * (D * 2) + dhc0 * 2), key0, D == 64);
    }
  }

  auto qkdp_mfma = [&](int buf, int sub, f32x16& s_acc, f32x16& dp_acc) {
#pragma unroll
    for (int kk = 0; kk < D / 16; ++kk) {
      const char* base = (const char*)k_img(0) + qk_addr[kk] +
                         buf * 2 * IMG2 + sub * 32 * (D * 2);
      bf16x8 ka = *(const bf16x8*)base;
      bf16x8 va = *(const bf16x8*)(base + IMG2);
      s_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(ka, qfrag[kk], s_acc,
                                                      0, 0, 0);
      dp_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(va, dofrag[kk],
                                                       dp_acc, 0, 0, 0);
    }
  };
  auto dsk_mfma = [&](const float* ds, int buf, int sub) {
#pragma unroll
    for (int s16 = 0; s16 < 2; ++s16) {
      bf16x8 dsfrag = pack_bfrag(ds, 8 * s16);
#pragma unroll
      for (int db = 0; db < D / 32; ++db) {
        union { bf16x4 h[2]; bf16x8 v8; } a;
#pragma unroll
        for (int rd = 0; rd < 2; ++rd) {
          a.h[rd] = lds_tr16(
              k_img(0), tr_addr[rd][db] + buf * 2 * IMG2 +
                            sub * 32 * (D * 2) + s16 * 16 * (D * 2));
        }
        dq_acc[db] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            a.v8, dsfrag, dq_acc[db], 0, 0, 0);
      }
    }
  };

  // ---- interior tiles (no masks): p*scale = exp2(sv2 - lse2) ------------
  for (int t = 0; t < t_int; ++t) {
    const int buf = t & 1;
    vm_wait0();
    __syncthreads();
    if (t + 1 < n_tiles) dma_stage(t + 1, 1 - buf);
#pragma unroll
    for (int sub = 0; sub < 2; ++sub) {
      f32x16 s_acc = f32x16{}, dp_acc = f32x16{};
      qkdp_mfma(buf, sub, s_acc, dp_acc);
      // abase2 folds log2(scale) AND the lse: exp2 yields p*scale, so ds
      // skips the per-element multiply
      const float ab2 = fmaf(slope2, (float)(t * KBQ + sub * 32 - my_q),
                             LOG2_SCALE) - my_lse2;
      float ds[16];
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int pat = (r & 3) + 8 * (r >> 2) + 4 * hi;
        const float sv = fmaf(s_acc[r], scale2, fmaf(slope2, (float)pat, ab2));
        const float pv = exp2_fast(sv);
        ds[r] = pv * (dp_acc[r] - my_delta);
      }
      dsk_mfma(ds, buf, sub);
    }
  }

  // ---- boundary tiles (diagonal/tail) ------------------------------------
  for (int t = t_int; t < n_tiles; ++t) {
    const int buf = t & 1;
    const int kv0 = t * KBQ;
    vm_wait0();
    __syncthreads();
    if (t + 1 < n_tiles) dma_stage(t + 1, 1 - buf);
#pragma unroll
    for (int sub = 0; sub < 2; ++sub) {
      const int kv0s = kv0 + sub * 32;
      if (causal && kv0s > my_q_max) break;
      f32x16 s_acc = f32x16{}, dp_acc = f32x16{};
      qkdp_mfma(buf, sub, s_acc, dp_acc);
      const float ab2 =
          fmaf(slope2, (float)(kv0s - my_q), LOG2_SCALE) - my_lse2;
      float ds[16];
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int pat = (r & 3) + 8 * (r >> 2) + 4 * hi;
        float sv = fmaf(s_acc[r], scale2, fmaf(slope2, (float)pat, ab2));
        const int key = kv0s + pat;
        const bool masked = (key >= S) || (causal && key > my_q);
        sv = masked ? -1e30f : sv;
        const float pv = exp2_fast(sv);  // masked: exp2(-huge) == 0
        ds[r] = pv * (dp_acc[r] - my_delta);
      }
      dsk_mfma(ds, buf, sub);
    }
  }
  __syncthreads();  // protect epilogue smem reuse

  // epilogue (tr-read rows are natural order; see fwd epilogue note)
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
      *(bf16x8*)(dq + ibase + (long)(q0 + row) * rs_i + col) =
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
// SRSRC staging dropped D=128 register pressure (204 VGPR vs ~420 with the
// old register staging): 2 waves/SIMD now fits at every D.
__global__ __launch_bounds__(ATT_BLOCK, (D <= 64) ? 2 : ATT_DKDV_MINWAVES_D128)
void attn_bwd_dkdv_kernel(
    const __bf16* __restrict__ dout, const __bf16* __restrict__ q,
    const __bf16* __restrict__ k, const __bf16* __restrict__ v,
    const float* __restrict__ slopes, const float* __restrict__ lse,
    const float* __restrict__ delta, __bf16* __restrict__ dk,
    __bf16* __restrict__ dv, int S, int H, int causal, long bs_i, long hs_i,
    long rs_i, long bs_o, long hs_o, long rs_o) {
  // v2: double-buffered Q/dO tiles, ONE barrier per tile, register-staged
  // loads (same pipeline as attn_fwd_kernel), Q^T/dO^T A-fragments via
  // hardware tr16 reads from the row images. Tile height shrinks at D=128
  // so LDS (4 images) stays at 32 KB and registers below the spill line —
  // QTF=64 at D=128 measured 2x SLOWER (18.8 vs 9.0 ms) from occupancy.
#ifndef ATT_QTF64
#define ATT_QTF64 64  // q rows per staged dkdv tile at D<=64 (A/B knob)
#endif
#ifndef ATT_QTF128
// 64 q-rows per staged dkdv tile at D=128: with LDS-DMA staging the
// registers no longer bind, and halving the barrier count measured
// 176 -> 207 TF/s (r01's QTF=32 choice was a register-pressure artifact).
#define ATT_QTF128 64
#endif
  constexpr int QTF = (D <= 64) ? ATT_QTF64 : ATT_QTF128;
  constexpr int NSUB = QTF / 32;            // 32-row compute subtiles
  constexpr int IMG2 = QTF * D * 2; // bytes per [QTF][D] image
  extern __shared__ __attribute__((aligned(16))) char smem[];
  auto q_img = [&](int b) { return (__bf16*)(smem + b * 2 * IMG2); };
  auto do_img = [&](int b) { return (__bf16*)(smem + b * 2 * IMG2 + IMG2); };
  float* lse_buf = (float*)(smem + 4 * IMG2);        // [2][QTF]
  float* del_buf = (float*)(smem + 4 * IMG2 + 2 * QTF * 4);

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int hi = lane >> 5;
  const int lq = lane & 31;
  const long bh = blockIdx.y;
  const int h = bh % H;
  const float slope = slopes[h];
  // exp2 domain (cf. fwd/dq)
  const float scale2 = rsqrtf((float)D) * LOG2E;
  const float slope2 = slope * LOG2E;
  constexpr float LOG2_SCALE = (D == 64) ? -3.0f : (D == 128) ? -3.5f : 0.f;
  static_assert(D == 64 || D == 128, "log2(scale) fold assumes D in {64,128}");
  const long ibase = (bh / H) * bs_i + (long)h * hs_i;
  const long obase = (bh / H) * bs_o + (long)h * hs_o;
  const int k0 = blockIdx.x * (WAVES * KB) + wave * KB;
  const int my_key = k0 + lq;

  // K, V rows of this wave's keys as B-operand fragments (like Q in fwd)
  bf16x8 kfrag[D / 16], vfrag[D / 16];
  {
    const long row = ibase + (long)min(my_key, S - 1) * rs_i;
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
  const int block_k_max = (int)(blockIdx.x + 1) * (WAVES * KB) - 1;
  const int t0 = causal ? (k_min_block / QTF) : 0;
  const int n_tiles = (S + QTF - 1) / QTF;
  const int my_k_min = k0;

  float lse_r = 0.f, del_r = 0.f;

  // LDS-DMA staging (cf. fwd). q and dout have separate strides (packed
  // QKV input vs contiguous dout) -> separate descriptors/voffsets. The
  // tiny lse/delta side-channel stays register-staged (loads at issue
  // time, LDS writes after compute).
  constexpr int NW = QTF * D * 2 / 1024 / WAVES;
  const auto qrs = make_rsrc(q + ibase, ((long)(S - 1) * rs_i + D) * 2);
  const auto ors = make_rsrc(dout + obase, ((long)(S - 1) * rs_o + D) * 2);
  int dvoff_q[NW], dvoff_o[NW];
#pragma unroll
  for (int i = 0; i < NW; ++i) {
    const int dest = (wave * NW + i) * 1024 + lane * 16;
    const int row = dest / (D * 2);
    const int src = (int)(dest ^ (swz_field(row & 15u, D == 64) << 4));
    const int srow = src / (D * 2), scol = src % (D * 2);
    dvoff_q[i] = (int)((long)srow * rs_i * 2 + scol);
    dvoff_o[i] = (int)((long)srow * rs_o * 2 + scol);
  }
  const int tsoff_q = (int)(QTF * rs_i * 2);
  const int tsoff_o = (int)(QTF * rs_o * 2);

  auto dma_stage = [&](int t, int b) {
#pragma unroll
    for (int i = 0; i < NW; ++i) {
      const int win = (wave * NW + i) * 1024;
      lds_dma16(qrs, (char*)q_img(b) + win, dvoff_q[i], t * tsoff_q);
      lds_dma16(ors, (char*)do_img(b) + win, dvoff_o[i], t * tsoff_o);
    }
    if (threadIdx.x < QTF) {
      const long qi = (long)t * QTF + threadIdx.x;
      // lse pre-multiplied into the exp2 domain at stage time
      lse_r = (qi < S) ? lse[bh * (long)S + qi] * LOG2E : INFINITY;
      del_r = (qi < S) ? delta[bh * (long)S + qi] : 0.f;
    }
  };
  auto misc_write = [&](int b) {
    if (threadIdx.x < QTF) {
      lse_buf[b * QTF + threadIdx.x] = lse_r;
      del_buf[b * QTF + threadIdx.x] = del_r;
    }
  };

  dma_stage(t0, t0 & 1);
  misc_write(t0 & 1);

  // Hoisted LDS read addresses (cf. fwd); q and do share the layout.
  const int pv_tj = lane & 15;
  const int pv_tg1 = (lane >> 4) & 1;
  unsigned qk_addr[D / 16];
#pragma unroll
  for (int kk = 0; kk < D / 16; ++kk) {
    qk_addr[kk] = swz((unsigned)(lq * (D * 2) + kk * 32 + hi * 16), lq, D == 64);
  }
  unsigned tr_addr[2][D / 32];
#pragma unroll
  for (int rd = 0; rd < 2; ++rd) {
#pragma unroll
    for (int db = 0; db < D / 32; ++db) {
      const int key0 = 8 * hi + 4 * rd + (pv_tj >> 2);
      const int dhc0 = db * 32 + 16 * pv_tg1 + 4 * (pv_tj & 3);
      tr_addr[rd][db] = swz((unsigned)(key0 * (D * 2) + dhc0 * 2), key0, D == 64);
    }
  }

  // S'[q][key]: A = Q row frags, B = K regs; dP'[q][key]: A = dO, B = V
  auto qkdp_mfma = [&](int buf, int sub, f32x16& s_acc, f32x16& dp_acc) {
#pragma unroll
    for (int kk = 0; kk < D / 16; ++kk) {
      const char* base = (const char*)q_img(0) + qk_addr[kk] +
                         buf * 2 * IMG2 + sub * 32 * (D * 2);
      bf16x8 qa = *(const bf16x8*)base;
      bf16x8 doa = *(const bf16x8*)(base + IMG2);
      s_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qa, kfrag[kk], s_acc,
                                                      0, 0, 0);
      dp_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(doa, vfrag[kk],
                                                       dp_acc, 0, 0, 0);
    }
  };
  auto dkdv_mfma = [&](const float* p, const float* ds, int buf, int sub) {
#pragma unroll
    for (int s16 = 0; s16 < 2; ++s16) {
      bf16x8 pfrag = pack_bfrag(p, 8 * s16);
      bf16x8 dsfrag = pack_bfrag(ds, 8 * s16);
#pragma unroll
      for (int db = 0; db < D / 32; ++db) {
        union { bf16x4 h[2]; bf16x8 v8; } doa, qa;
#pragma unroll
        for (int rd = 0; rd < 2; ++rd) {
          const unsigned byte = tr_addr[rd][db] + buf * 2 * IMG2 +
                                sub * 32 * (D * 2) + s16 * 16 * (D * 2);
          qa.h[rd] = lds_tr16(q_img(0), byte);
          doa.h[rd] = lds_tr16((const __bf16*)((const char*)q_img(0) + IMG2),
                               byte);
        }
        dv_acc[db] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            doa.v8, pfrag, dv_acc[db], 0, 0, 0);
        dk_acc[db] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            qa.v8, dsfrag, dk_acc[db], 0, 0, 0);
      }
    }
  };
  // masked (diagonal/tail) tile body — abase2 folds the ALiBi row constant
  // and log2(scale): exp2 yields p*scale (dv un-scales once in epilogue)
  auto masked_tile = [&](int t) {
    const int buf = t & 1;
    const int qt0 = t * QTF;
#pragma unroll
    for (int sub = 0; sub < NSUB; ++sub) {
      const int qt0s = qt0 + sub * 32;
      if (causal && qt0s + 31 < my_k_min) continue;
      f32x16 s_acc = f32x16{}, dp_acc = f32x16{};
      qkdp_mfma(buf, sub, s_acc, dp_acc);
      const float ab2 = fmaf(slope2, (float)(my_key - qt0s), LOG2_SCALE);
      float p[16], ds[16];
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int pat = (r & 3) + 8 * (r >> 2) + 4 * hi;
        const float l2 = lse_buf[buf * QTF + sub * 32 + pat];
        const float dlt = del_buf[buf * QTF + sub * 32 + pat];
        float sv = fmaf(s_acc[r], scale2, fmaf(slope2, -(float)pat, ab2));
        const int qi = qt0s + pat;
        const bool masked =
            (my_key >= S) || (causal && my_key > qi) || (qi >= S);
        sv = masked ? -1e30f : sv;
        p[r] = exp2_fast(sv - l2);  // == p_raw * scale
        ds[r] = p[r] * (dp_acc[r] - dlt);
      }
      dkdv_mfma(p, ds, buf, sub);
    }
  };
  // interior (unmasked) tile body
  auto interior_tile = [&](int t) {
    const int buf = t & 1;
    const int qt0 = t * QTF;
#pragma unroll
    for (int sub = 0; sub < NSUB; ++sub) {
      const int qt0s = qt0 + sub * 32;
      f32x16 s_acc = f32x16{}, dp_acc = f32x16{};
      qkdp_mfma(buf, sub, s_acc, dp_acc);
      const float ab2 = fmaf(slope2, (float)(my_key - qt0s), LOG2_SCALE);
      float p[16], ds[16];
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int pat = (r & 3) + 8 * (r >> 2) + 4 * hi;
        const float l2 = lse_buf[buf * QTF + sub * 32 + pat];
        const float dlt = del_buf[buf * QTF + sub * 32 + pat];
        const float sv =
            fmaf(s_acc[r], scale2, fmaf(slope2, -(float)pat, ab2));
        p[r] = exp2_fast(sv - l2);  // == p_raw * scale
        ds[r] = p[r] * (dp_acc[r] - dlt);
      }
      dkdv_mfma(p, ds, buf, sub);
    }
  };

  // three phases over Q tiles: diagonal head (masked), interior, tail.
  // Interior tiles have every q strictly above the block's keys (causal)
  // and full rows: no compares, no continue, one straight-line softmax.
  const int t_diag_end =
      causal ? min(block_k_max / QTF, n_tiles - 1) : (t0 - 1);
  int t_full_end = S / QTF;
  if (t_full_end > n_tiles) t_full_end = n_tiles;
  for (int t = t0; t <= t_diag_end; ++t) {
    vm_wait0();
    __syncthreads();
    if (t + 1 < n_tiles) dma_stage(t + 1, 1 - (t & 1));
    masked_tile(t);
    if (t + 1 < n_tiles) misc_write(1 - (t & 1));
  }
  for (int t = t_diag_end + 1; t < t_full_end; ++t) {
    vm_wait0();
    __syncthreads();
    if (t + 1 < n_tiles) dma_stage(t + 1, 1 - (t & 1));
    interior_tile(t);
    if (t + 1 < n_tiles) misc_write(1 - (t & 1));
  }
  for (int t = max(t_full_end, t_diag_end + 1); t < n_tiles; ++t) {
    vm_wait0();
    __syncthreads();
    if (t + 1 < n_tiles) dma_stage(t + 1, 1 - (t & 1));
    masked_tile(t);
    if (t + 1 < n_tiles) misc_write(1 - (t & 1));
  }
  __syncthreads();  // protect epilogue smem reuse

  // dv accumulated p*scale (see abase fold): undo once here.
  {
    const float inv_scale = sqrtf((float)D);
#pragma unroll
    for (int db = 0; db < D / 32; ++db) {
#pragma unroll
      for (int r = 0; r < 16; ++r) dv_acc[db][r] *= inv_scale;
    }
  }

  // epilogue: two bounces (dk then dv) through per-wave LDS
  // (tr-read rows are natural order; see fwd epilogue note)
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
        *(bf16x8*)(dst + ibase + (long)(k0 + row) * rs_i + col) =
            *(const bf16x8*)(o_img + row * D + col);
      }
    }
    __syncthreads();
  }
}


// ---------------------------------------------------------------------------
// delta[b,h,s] = sum_d dO[b,s,h,d] * O[b,s,h,d]  (fp32 out) — replaces the
// torch chain (two fp32 casts + a reduce kernel) for the FlashAttention-2
// delta term. 256/D rows per wave: each row is covered by a D/4-lane
// subgroup loading 4 bf16 per lane; subgroup reduction via shfl_xor.
// Strides (bs, hs, rs) address both tensors (packed [B,S,H,D] or [B,H,S,D]).
// ---------------------------------------------------------------------------
template <int D>
__global__ __launch_bounds__(256) void attn_delta_kernel(
    const __bf16* __restrict__ dout, const __bf16* __restrict__ o,
    float* __restrict__ delta, long n_rows /* B*S*H */, int S, int H,
    long bs, long hs, long rs) {
  constexpr int LPR = D / 4;          // lanes per row (16 or 32)
  constexpr int RPW = 64 / LPR;       // rows per wave (4 or 2)
  const int lane = threadIdx.x & 63;
  const int sub = lane / LPR;         // row slot within the wave
  const int sl = lane % LPR;          // lane within the row subgroup
  const long wave_id = ((long)blockIdx.x * blockDim.x + threadIdx.x) / 64;
  const long n_waves = ((long)gridDim.x * blockDim.x) / 64;

  for (long r0 = wave_id * RPW; r0 < n_rows; r0 += n_waves * RPW) {
    const long idx = r0 + sub;  // this subgroup's (b, s, h) row
    float acc = 0.f;
    long out_pos = 0;
    if (idx < n_rows) {
      const long b = idx / ((long)S * H);
      const long rem = idx % ((long)S * H);
      const long s_ = rem / H;
      const long h_ = rem % H;
      const long off = b * bs + h_ * hs + s_ * rs + sl * 4;
      floatx4 a = load4<unsigned short>((const unsigned short*)(dout + off));
      floatx4 bb = load4<unsigned short>((const unsigned short*)(o + off));
      acc = a.x * bb.x + a.y * bb.y + a.z * bb.z + a.w * bb.w;
      out_pos = (b * H + h_) * (long)S + s_;
    }
#pragma unroll
    for (int offx = LPR / 2; offx > 0; offx >>= 1)
      acc += __shfl_xor(acc, offx, 64);
    if (sl == 0 && idx < n_rows) delta[out_pos] = acc;
  }
}

}  // namespace photon_hip
