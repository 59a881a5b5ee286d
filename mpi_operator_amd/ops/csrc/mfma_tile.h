// NT-GEMM MFMA tile template for gfx950: C[M][N] = sum_k A[m][k]*B[n][k].
// Both operands are loaded reduce-dim-contiguous (the natural MFMA fragment
// direction), which is what NHWC activations / channels-last weights give
// for conv fwd and dgrad (implicit GEMM via pluggable address functors).
//
// Structure (v1, correctness-first — the "step-2/3" rung of the guide's
// optimization ladder): 128x128 tile, BK=64, 4 waves (2x2), double-buffered
// LDS with XOR-swizzled 16B slots, register staging with predicated loads
// (zero-fill handles conv padding and edge tiles), issue-early/write-late.
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
// LDS tile: [128 rows][8 slots of 16B]; slot is XOR-swizzled by (row&7).
constexpr int SLOTS = BK / 8; // 8 slots/row

DEV_INLINE int lds_slot(int row, int slot) { return row * SLOTS + (slot ^ (row & 7)); }

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

template <class LA, class LB, bool C_F32>
__global__ __launch_bounds__(NT_THREADS) void nt_gemm_k(
    LA la, LB lb, void *__restrict__ cptr, int M, int N, int K, long ldc,
    int tiles_n, int kt_per_split, long split_stride) {
  int tile = blockIdx.x;
  // split-K (wgrad: reduce dim = all output pixels, few C tiles): split s
  // writes its fp32 partial at cptr + s*split_stride; a reduce kernel sums.
  int split = blockIdx.y;
  int tm = tile / tiles_n, tn = tile % tiles_n;
  int row0 = tm * BM, col0 = tn * BN;
  int tid = threadIdx.x;
  int lane = tid & 63, wave = tid >> 6;
  int wr = wave >> 1, wc = wave & 1; // wave's 64x64 quadrant

  __shared__ ushort8 ldsA[2][BM * SLOTS];
  __shared__ ushort8 ldsB[2][BN * SLOTS];

  // staging geometry: thread loads rows (tid/8 + 32*i), logical slot tid%8
  const int s_row = tid >> 3, s_slot = tid & 7;

  float4v acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = float4v{0.f, 0.f, 0.f, 0.f};

  int nk_total = (K + BK - 1) / BK;
  int t0 = split * kt_per_split;
  int nk = min(kt_per_split, nk_total - t0);
  if (nk <= 0) nk = 0; // still participate in epilogue (writes zeros)
  ushort8 ra[4], rb[4];

  // load tile t into staging registers
  auto load_regs = [&](int t) {
    int kb = (t0 + t) * BK + s_slot * 8;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      ra[i] = la.load(row0 + s_row + 32 * i, kb);
      rb[i] = lb.load(col0 + s_row + 32 * i, kb);
    }
  };
  auto write_lds = [&](int buf) {
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      ldsA[buf][lds_slot(s_row + 32 * i, s_slot)] = ra[i];
      ldsB[buf][lds_slot(s_row + 32 * i, s_slot)] = rb[i];
    }
  };

  if (nk > 0) {
    load_regs(0);
    write_lds(0);
  }
  __syncthreads();

  for (int t = 0; t < nk; ++t) {
    if (t + 1 < nk) load_regs(t + 1); // issue early: HBM latency hides under MFMA
    int buf = t & 1;
    // compute on buf
#pragma unroll
    for (int kk = 0; kk < BK / 32; ++kk) { // two 16x16x32 K-steps
      bf16x8 af[4], bf_[4];
      int slot = kk * 4 + (lane >> 4); // 16B slot holding this lane's 8 k
#pragma unroll
      for (int mi = 0; mi < 4; ++mi) {
        int arow = wr * 64 + mi * 16 + (lane & 15);
        af[mi] = us8_to_bf8v(ldsA[buf][lds_slot(arow, slot)]);
      }
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        int brow = wc * 64 + ni * 16 + (lane & 15);
        bf_[ni] = us8_to_bf8v(ldsB[buf][lds_slot(brow, slot)]);
      }
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mi], bf_[ni], acc[mi][ni], 0, 0, 0);
    }
    if (t + 1 < nk) write_lds((t + 1) & 1); // write-late, after compute
    __syncthreads();
  }

  // epilogue: C/D fragment map (16x16x32): col = lane&15, row = (lane>>4)*4 + r
  cptr = (void *)((char *)cptr + split * split_stride * (C_F32 ? 4 : 2));
#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      int col = col0 + wc * 64 + ni * 16 + (lane & 15);
      if (col >= N) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = row0 + wr * 64 + mi * 16 + (lane >> 4) * 4 + r;
        if (row >= M) continue;
        if (C_F32)
          ((float *)cptr)[(long)row * ldc + col] = acc[mi][ni][r];
        else
          ((uint16_t *)cptr)[(long)row * ldc + col] = f2bf(acc[mi][ni][r]);
      }
    }
  }
}

template <class LA, class LB>
static hipError_t launch_nt_gemm(const LA &la, const LB &lb, void *c, int M,
                                 int N, int K, long ldc, bool c_f32,
                                 hipStream_t s, int splits = 1) {
  int tiles_m = (M + BM - 1) / BM, tiles_n = (N + BN - 1) / BN;
  int nk = (K + BK - 1) / BK;
  if (splits > nk) splits = nk > 0 ? nk : 1;
  int kts = (nk + splits - 1) / splits;
  long split_stride = (long)M * ldc; // partial slab stride (elements)
  dim3 grid(tiles_m * tiles_n, splits);
  if (c_f32)
    nt_gemm_k<LA, LB, true><<<grid, NT_THREADS, 0, s>>>(la, lb, c, M, N, K, ldc,
                                                        tiles_n, kts, split_stride);
  else
    nt_gemm_k<LA, LB, false><<<grid, NT_THREADS, 0, s>>>(la, lb, c, M, N, K, ldc,
                                                         tiles_n, kts, split_stride);
  return hipGetLastError();
}
