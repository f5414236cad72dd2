// Shared helpers for the deeprest_amd CDNA4 (gfx950) kernels.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp8.h>
#include <stdint.h>
#include <stdio.h>
#include <algorithm>

#define DR_WAVE 64  // CDNA wavefront width (gfx950)

#define DR_HIP_CHECK(expr)                                                     \
  do {                                                                         \
    hipError_t _e = (expr);                                                    \
    if (_e != hipSuccess) {                                                    \
      printf("HIP error %s at %s:%d\n", hipGetErrorString(_e), __FILE__,       \
             __LINE__);                                                        \
    }                                                                          \
  } while (0)

namespace dr {

// ---- bf16 <-> f32 (bit ops; RNE rounding on store) ----
__device__ __forceinline__ float bf2f(uint16_t x) {
  union { uint32_t u; float f; } v;
  v.u = static_cast<uint32_t>(x) << 16;
  return v.f;
}

__device__ __forceinline__ uint16_t f2bf(float f) {
  // single v_cvt_pk_bf16_f32 (RNE) — the manual add/shift form costs 3 VALU
  __hip_bfloat16 b = __float2bfloat16(f);
  return *reinterpret_cast<uint16_t*>(&b);
}

// two f32 -> packed 2x bf16 in one v_cvt_pk_bf16_f32
__device__ __forceinline__ uint32_t f2bf2(float lo, float hi) {
  __hip_bfloat162 b2 = __float22bfloat162_rn(make_float2(lo, hi));
  return *reinterpret_cast<uint32_t*>(&b2);
}

// generic load/store as float for T in {float, bf16-as-ushort}
template <typename T> __device__ __forceinline__ float ldf(const T* p);
template <> __device__ __forceinline__ float ldf<float>(const float* p) { return *p; }
template <> __device__ __forceinline__ float ldf<uint16_t>(const uint16_t* p) { return bf2f(*p); }

template <typename T> __device__ __forceinline__ void stf(T* p, float v);
template <> __device__ __forceinline__ void stf<float>(float* p, float v) { *p = v; }
template <> __device__ __forceinline__ void stf<uint16_t>(uint16_t* p, float v) { *p = f2bf(v); }

// ---- wave reductions (64-wide) ----
__device__ __forceinline__ float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, DR_WAVE);
  return v;
}

__device__ __forceinline__ float wave_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, DR_WAVE));
  return v;
}

// reduce within contiguous 16-lane groups (MFMA C-layout row groups)
__device__ __forceinline__ float group16_sum(float v) {
#pragma unroll
  for (int off = 8; off > 0; off >>= 1) v += __shfl_xor(v, off, DR_WAVE);
  return v;
}

__device__ __forceinline__ float group16_max(float v) {
#pragma unroll
  for (int off = 8; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, DR_WAVE));
  return v;
}

__device__ __forceinline__ float sigmoidf_(float x) {
  // fast path: v_exp + fast divide; exp(-x) -> inf for x << 0 gives 1/inf = 0
  return __fdividef(1.0f, 1.0f + __expf(-x));
}

__device__ __forceinline__ float tanhf_(float x) {
  // clamp so exp never overflows into inf/inf = NaN; tanh(+-15) == +-1 in f32
  float xc = fminf(fmaxf(x, -15.f), 15.f);
  float e = __expf(2.f * xc);
  return __fdividef(e - 1.f, e + 1.f);
}

// ---- MFMA fragment types (gfx950) ----
using f32x4 = __attribute__((__vector_size__(4 * sizeof(float)))) float;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;   // 4 VGPRs: A/B frag of 16x16x32
using s16x8 = __attribute__((ext_vector_type(8))) short;

}  // namespace dr
