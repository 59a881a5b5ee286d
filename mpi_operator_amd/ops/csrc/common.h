// Common helpers for the MI355X (gfx950/CDNA4) kernels.
// Pure HIP — no torch headers in kernel TUs (keeps hipcc compile fast).
#pragma once
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <stdint.h>

#define DEV_INLINE __device__ __forceinline__

// CDNA wavefront is 64 lanes (not 32) — hard-coded per the gfx950 contract.
constexpr int WAVE = 64;

using bf16 = __hip_bfloat16;
// 8 bf16 = 16 B: one fully-coalesced lane access (1 KiB per wave instruction).
typedef __attribute__((ext_vector_type(8))) uint16_t ushort8;
typedef __attribute__((ext_vector_type(4))) uint16_t ushort4v;
typedef __attribute__((ext_vector_type(4))) float float4v;
typedef __attribute__((ext_vector_type(8))) float float8v;
typedef __attribute__((ext_vector_type(2))) float float2v;

DEV_INLINE float bf2f(uint16_t u) {
  union { uint32_t i; float f; } v;
  v.i = uint32_t(u) << 16;
  return v.f;
}

DEV_INLINE uint16_t f2bf(float f) {
  union { float f; uint32_t i; } v;
  v.f = f;
  // round-to-nearest-even
  uint32_t lsb = (v.i >> 16) & 1u;
  v.i += 0x7fffu + lsb;
  return uint16_t(v.i >> 16);
}

// unpack 8 bf16 (as ushort8) to 8 floats
DEV_INLINE void bf8_to_f8(const ushort8 &u, float *f) {
#pragma unroll
  for (int i = 0; i < 8; ++i) f[i] = bf2f(u[i]);
}

DEV_INLINE ushort8 f8_to_bf8(const float *f) {
  ushort8 u;
#pragma unroll
  for (int i = 0; i < 8; ++i) u[i] = f2bf(f[i]);
  return u;
}

// wave-wide sum over all 64 lanes (result in every lane)
DEV_INLINE float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

DEV_INLINE float wave_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE));
  return v;
}

#define HIP_KERNEL_CHECK()                                                    \
  do {                                                                        \
    hipError_t e = hipGetLastError();                                         \
    if (e != hipSuccess) return e;                                            \
  } while (0)

// ceil-div
DEV_INLINE constexpr int cdiv(int a, int b) { return (a + b - 1) / b; }
static inline int cdiv_h(long a, long b) { return (int)((a + b - 1) / b); }
