// Shared MFMA-GEMM tile constants, bf16 vector types, and the plain
// k-contiguous (NT) loader. The tile kernel itself lives in mix_gemm.h —
// one 32x32x16-MFMA structure serves every staging combination (NT/TN/
// gather); launch_nt_gemm below is the plain NT×NT instantiation.
#pragma once
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;

DEV_INLINE bf16x8 us8_to_bf8v(ushort8 u) {
  union { ushort8 u; bf16x8 b; } v;
  v.u = u;
  return v.b;
}

constexpr int BM = 128, BN = 128, BK = 64;
constexpr int NT_THREADS = 256; // 4 waves, 2x2 wave grid

// A/B loader concept:
//   struct Loader { DEV_INLINE ushort8 load(int row, int k) const; };
// row: global output row (A) / col (B); k: global reduce index (multiple of 8)
// Loader handles bounds & padding by returning zeros.

struct GemmLoader {
  const uint16_t *p;
  int rows;
  long ld; // in elements
  int kdim;
  DEV_INLINE ushort8 load(int row, int k) const {
    if (row < rows && k < kdim) return *(const ushort8 *)(p + (long)row * ld + k);
    return ushort8{0, 0, 0, 0, 0, 0, 0, 0};
  }
};
