// Common device helpers for photon_amd CDNA4 (gfx950) kernels.
// Wave width is 64 on CDNA4 — hard-coded per the platform guide.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64
#define DEV_INLINE __device__ __forceinline__

typedef __attribute__((ext_vector_type(2))) float floatx2;
typedef __attribute__((ext_vector_type(4))) float floatx4;
typedef __attribute__((ext_vector_type(4))) short shortx4;
typedef __attribute__((ext_vector_type(8))) short shortx8;

DEV_INLINE float bf16_to_f32(unsigned short u) {
  union { float f; unsigned int i; } w;
  w.i = ((unsigned int)u) << 16;
  return w.f;
}

DEV_INLINE unsigned short f32_to_bf16(float f) {
  union { float f; unsigned int i; } w;
  w.f = f;
  // round-to-nearest-even
  unsigned int lsb = (w.i >> 16) & 1u;
  w.i += 0x7fffu + lsb;
  return (unsigned short)(w.i >> 16);
}

// -- dtype-templated scalar access ------------------------------------------
template <typename T>
DEV_INLINE float load_f32(const T* p, long i);
template <>
DEV_INLINE float load_f32<unsigned short>(const unsigned short* p, long i) {
  return bf16_to_f32(p[i]);
}
template <>
DEV_INLINE float load_f32<float>(const float* p, long i) { return p[i]; }

template <typename T>
DEV_INLINE void store_f32(T* p, long i, float v);
template <>
DEV_INLINE void store_f32<unsigned short>(unsigned short* p, long i, float v) {
  p[i] = f32_to_bf16(v);
}
template <>
DEV_INLINE void store_f32<float>(float* p, long i, float v) { p[i] = v; }

// -- wave64 reductions ------------------------------------------------------
DEV_INLINE float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

DEV_INLINE float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE));
  return v;
}

// Block reduction over T threads (T multiple of 64, <= 1024), using LDS.
// `scratch` needs T/WAVE floats. Result broadcast to all threads.
DEV_INLINE float block_reduce_sum(float v, float* scratch) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int nw = blockDim.x / WAVE;
  v = wave_reduce_sum(v);
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  float r = (lane < nw) ? scratch[lane] : 0.f;
  r = wave_reduce_sum(r);  // only first wave's lanes matter, but all compute
  if (wid != 0) r = 0.f;
  // broadcast via LDS
  if (threadIdx.x == 0) scratch[0] = r;
  __syncthreads();
  r = scratch[0];
  __syncthreads();
  return r;
}

DEV_INLINE float block_reduce_max(float v, float* scratch) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int nw = blockDim.x / WAVE;
  v = wave_reduce_max(v);
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  float r = (lane < nw) ? scratch[lane] : -INFINITY;
  r = wave_reduce_max(r);
  if (threadIdx.x == 0) scratch[0] = r;
  __syncthreads();
  r = scratch[0];
  __syncthreads();
  return r;
}

// ---------------------------------------------------------------------------
// Vector access helpers: 4 elements per lane per chunk.
// bf16: 8-byte load (shortx4); f32: 16-byte load (floatx4).
// ---------------------------------------------------------------------------
template <typename T>
DEV_INLINE floatx4 load4(const T* p);
template <>
DEV_INLINE floatx4 load4<unsigned short>(const unsigned short* p) {
  shortx4 r = *reinterpret_cast<const shortx4*>(p);
  floatx4 f;
  f.x = bf16_to_f32((unsigned short)r.x);
  f.y = bf16_to_f32((unsigned short)r.y);
  f.z = bf16_to_f32((unsigned short)r.z);
  f.w = bf16_to_f32((unsigned short)r.w);
  return f;
}
template <>
DEV_INLINE floatx4 load4<float>(const float* p) {
  return *reinterpret_cast<const floatx4*>(p);
}

template <typename T>
DEV_INLINE void store4(T* p, floatx4 v);
template <>
DEV_INLINE void store4<unsigned short>(unsigned short* p, floatx4 v) {
  shortx4 r;
  r.x = (short)f32_to_bf16(v.x);
  r.y = (short)f32_to_bf16(v.y);
  r.z = (short)f32_to_bf16(v.z);
  r.w = (short)f32_to_bf16(v.w);
  *reinterpret_cast<shortx4*>(p) = r;
}
template <>
DEV_INLINE void store4<float>(float* p, floatx4 v) {
  *reinterpret_cast<floatx4*>(p) = v;
}


#define HIP_CHECK(expr)                                                     \
  do {                                                                      \
    hipError_t _e = (expr);                                                 \
    if (_e != hipSuccess) {                                                 \
      TORCH_CHECK(false, "HIP error: ", hipGetErrorString(_e));             \
    }                                                                       \
  } while (0)
