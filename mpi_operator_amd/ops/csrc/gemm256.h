// 256x256-tile counted-vmcnt NT GEMM (guide's pipelined-glds structure).
//
// The mix_gemm GldsNt path runs the "2-buffer + vmcnt(0) + barrier" loop:
// the DMA for tile t+1 overlaps tile t's MFMAs, but the barrier at the end
// of every K-step drains the whole VM pipeline. This kernel keeps the DMA
// pipeline TWO K-tiles deep across raw barriers:
//
//   - 512 threads (8 waves, 2Mx4N), 256x256 C-tile, 128x64 per wave
//   - BK=32, 3 LDS tile-buffers (96 KB), glds staging (2 per thread/side)
//   - counted `s_waitcnt vmcnt(4)` at each K-tile boundary (tile t+1's 4
//     glds stay in flight while tile t computes; never vmcnt(0) mid-loop)
//   - raw `s_barrier` + explicit lgkmcnt(0) — __syncthreads() would emit
//     vmcnt(0) while a glds is outstanding and drain the pipeline
//   - s_setprio(1) around the MFMA block of each K-step
//
// FULL TILES ONLY: callers route here only when M%256==0, N%256==0 and
// K%32==0 (and splits==1). Edge handling would break the counted-vmcnt
// contract: a wave that skips an edge glds issues fewer VM ops, so a
// counted wait no longer proves the OLDER tile landed.
//
// Covers the all-NT GEMMs: 1x1 conv forward (stages 1-3 of ResNet) and
// every BERT linear forward (x·w^T with both operands k-contiguous).
// Reference equivalent: cuDNN/cuBLAS GEMMs inside the tf_cnn_benchmarks
// image (reference README.md:127-130) — rebuilt as CDNA4-native code.
#pragma once

// included from mix_gemm.h AFTER GemmLoader/LinearWriter/f2bf/us8_to_bf8v
// are declared; do not include standalone.

template <bool C_F32, bool BIAS>
__global__ __launch_bounds__(512) void nt256_gemm_k(
    const uint16_t *__restrict__ a, long lda, const uint16_t *__restrict__ b,
    long ldb, void *__restrict__ cptr, int M, int N, int K, long ldc,
    int tiles_n, int xcd_cpx, const float *__restrict__ bias) {
  int tile = blockIdx.x;
  if (xcd_cpx) // T1: contiguous tile chunk per XCD
    tile = (tile & 7) * xcd_cpx + (tile >> 3);
  int tm = tile / tiles_n, tn = tile % tiles_n;
  int row0 = tm * 256, col0 = tn * 256;
  int tid = threadIdx.x;
  int lane = tid & 63, wave = tid >> 6;
  int wr = wave >> 2, wc = wave & 3; // 2x4 wave grid, 128x64 out per wave

  // one tile = 256 rows x 4 octets (BK=32, 64 B rows); ONE shared array
  // (guide §5 trap 4a), buffers indexed arithmetically
  constexpr int TSZ = 256 * 4;
  __shared__ ushort8 lds[3 * 2 * TSZ]; // 3 bufs x (A,B) = 96 KB
#define G2_A(t) (lds + ((t) % 3) * 2 * TSZ)
#define G2_B(t) (lds + ((t) % 3) * 2 * TSZ + TSZ)

  float16v acc[4][2] = {};
  int nk = K / 32; // contract: K % 32 == 0

  // glds stager: 2 per thread per side; source octet pre-swizzled with
  // slot^(row&3) so the lane-linear LDS image reads conflict-free
  int s_row = tid >> 2, s_slot = tid & 3;
  auto stage = [&](int t) {
    int kb = t * 32;
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      int r = s_row + 128 * i;
      int k = kb + (s_slot ^ (r & 3)) * 8;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void *)(a +
              (long)(row0 + r) * lda + k),
          (__attribute__((address_space(3))) void *)(G2_A(t) + i * 512 + tid),
          16, 0, 0);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void *)(b +
              (long)(col0 + r) * ldb + k),
          (__attribute__((address_space(3))) void *)(G2_B(t) + i * 512 + tid),
          16, 0, 0);
    }
  };

  if (nk > 0) stage(0);
  if (nk > 1) stage(1);

  for (int t = 0; t < nk; ++t) {
    // tile t landed when at most tile t+1's 4 glds remain outstanding
    if (t + 1 < nk)
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier(); // raw: no implied vmcnt drain
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    if (t + 2 < nk) stage(t + 2); // DMA lands under this tile's MFMAs
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8 af[4], bf_[2];
      int slot = kk * 2 + (lane >> 5);
#pragma unroll
      for (int mi = 0; mi < 4; ++mi) {
        int ar = wr * 128 + mi * 32 + (lane & 31);
        af[mi] = us8_to_bf8v(G2_A(t)[ar * 4 + (slot ^ (ar & 3))]);
      }
#pragma unroll
      for (int ni = 0; ni < 2; ++ni) {
        int br = wc * 64 + ni * 32 + (lane & 31);
        bf_[ni] = us8_to_bf8v(G2_B(t)[br * 4 + (slot ^ (br & 3))]);
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 2; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              af[mi], bf_[ni], acc[mi][ni], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
    // no trailing barrier: the next iteration's top barrier separates these
    // reads (lgkm-waited before the MFMAs) from the glds that will reuse
    // this buffer two tiles from now
  }

  // 32x32 C/D map (col = lane&31, row = (reg&3)+8*(reg>>2)+4*(lane>>5));
  // full tiles: no row/col guards needed
#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      long row = row0 + wr * 128 + mi * 32 + (r & 3) + 8 * (r >> 2) +
                 4 * (lane >> 5);
      long base = row * ldc;
#pragma unroll
      for (int ni = 0; ni < 2; ++ni) {
        int col = col0 + wc * 64 + ni * 32 + (lane & 31);
        float v = BIAS ? acc[mi][ni][r] + bias[col] : acc[mi][ni][r];
        if (C_F32)
          ((float *)cptr)[base + col] = v;
        else
          ((uint16_t *)cptr)[base + col] = f2bf(v);
      }
    }
  }
}
#undef G2_A
#undef G2_B

template <class LA, class LB>
static hipError_t launch_nt256(const LA &la, const LB &lb, void *c, int M,
                               int N, int K, long ldc, bool c_f32,
                               hipStream_t s, const float *bias = nullptr) {
  int tiles_m = M / 256, tiles_n = N / 256;
  int nwg = tiles_m * tiles_n;
  int cpx = (nwg % 8 == 0 && nwg >= 32) ? nwg / 8 : 0;
  // compile-time bias split: a runtime `bias ? bias[col] : 0` select in
  // the epilogue measured 20.3 -> 37.6 us mean on the ResNet 1x1 shapes
  if (c_f32) {
    if (bias)
      nt256_gemm_k<true, true><<<nwg, 512, 0, s>>>(
          la.p, la.ld, lb.p, lb.ld, c, M, N, K, ldc, tiles_n, cpx, bias);
    else
      nt256_gemm_k<true, false><<<nwg, 512, 0, s>>>(
          la.p, la.ld, lb.p, lb.ld, c, M, N, K, ldc, tiles_n, cpx, bias);
  } else {
    if (bias)
      nt256_gemm_k<false, true><<<nwg, 512, 0, s>>>(
          la.p, la.ld, lb.p, lb.ld, c, M, N, K, ldc, tiles_n, cpx, bias);
    else
      nt256_gemm_k<false, false><<<nwg, 512, 0, s>>>(
          la.p, la.ld, lb.p, lb.ld, c, M, N, K, ldc, tiles_n, cpx, bias);
  }
  return hipGetLastError();
}
