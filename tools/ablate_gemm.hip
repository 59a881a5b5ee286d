// Standalone ablation of the mix-GEMM tile at a ResNet wgrad shape:
// which phase binds — staging loads, LDS round trip, MFMA, or epilogue?
// (guide §5 mistake 8: ablate empirically before optimizing.)
//
// Build: hipcc --offload-arch=gfx950 -O3 tools/ablate_gemm.hip -o tools/ablate_gemm
// Run (GPU box): ./tools/ablate_gemm
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
#include "../mpi_operator_amd/ops/csrc/mfma_tile.h"
#include "../mpi_operator_amd/ops/csrc/mix_gemm.h"

// variant bits: 1 = skip global loads, 2 = skip MFMA, 4 = skip epilogue,
// 8 = skip LDS writes (with 1: pure-LDS-read+MFMA loop)
template <int SKIP>
__global__ __launch_bounds__(NT_THREADS) void ablate_k(
    TnStage<TnRowMajor> sa, TnStage<TnRowMajor> sb, float *cptr, int M, int N,
    int K, int tiles_n, int kt, long sstride) {
  int tile = blockIdx.x;
  int split = blockIdx.y;
  int tm = tile / tiles_n, tn = tile % tiles_n;
  int row0 = tm * BM, col0 = tn * BN;
  int tid = threadIdx.x;
  int lane = tid & 63, wave = tid >> 6;
  int wr = wave >> 1, wc = wave & 1;
  constexpr int ASZ = BM * MXP;
  __shared__ ushort8 lds[2 * 2 * ASZ];
  float16v acc[2][2] = {};
  int nk_total = (K + BK - 1) / BK;
  int t0 = split * kt;
  int nk = min(kt, nk_total - t0);
  if (nk < 0) nk = 0;
  if (nk > 0 && !(SKIP & 1)) {
    sa.load(tid, row0, t0 * BK, nullptr);
    sb.load(tid, col0, t0 * BK, nullptr);
    if (!(SKIP & 8)) {
      sa.write(tid, lds);
      sb.write(tid, lds + ASZ);
    }
  }
  __syncthreads();
  for (int t = 0; t < nk; ++t) {
    int buf = t & 1;
    if (t + 1 < nk && !(SKIP & 1)) {
      sa.load(tid, row0, (t0 + t + 1) * BK, nullptr);
      sb.load(tid, col0, (t0 + t + 1) * BK, nullptr);
    }
    if (!(SKIP & 2)) {
#pragma unroll
      for (int kk = 0; kk < BK / 16; ++kk) {
        bf16x8 af[2], bf_[2];
        int slot = kk * 2 + (lane >> 5);
#pragma unroll
        for (int mi = 0; mi < 2; ++mi)
          af[mi] = us8_to_bf8v(lds[buf * 2 * ASZ + (wr * 64 + mi * 32 + (lane & 31)) * MXP + slot]);
#pragma unroll
        for (int ni = 0; ni < 2; ++ni)
          bf_[ni] = us8_to_bf8v(lds[buf * 2 * ASZ + ASZ + (wc * 64 + ni * 32 + (lane & 31)) * MXP + slot]);
#pragma unroll
        for (int mi = 0; mi < 2; ++mi)
#pragma unroll
          for (int ni = 0; ni < 2; ++ni)
            acc[mi][ni] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                af[mi], bf_[ni], acc[mi][ni], 0, 0, 0);
      }
    }
    if (t + 1 < nk && !(SKIP & 1) && !(SKIP & 8)) {
      sa.write(tid, lds + (buf ^ 1) * 2 * ASZ);
      sb.write(tid, lds + (buf ^ 1) * 2 * ASZ + ASZ);
    }
    __syncthreads();
  }
  if (SKIP & 4) { // keep acc alive
    if (acc[0][0][0] == 1234.5f) cptr[0] = 1.f;
    return;
  }
  cptr += split * sstride;
#pragma unroll
  for (int mi = 0; mi < 2; ++mi)
#pragma unroll
    for (int ni = 0; ni < 2; ++ni) {
      int col = col0 + wc * 64 + ni * 32 + (lane & 31);
      if (col >= N) continue;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int row = row0 + wr * 64 + mi * 32 + (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
        if (row >= M) continue;
        cptr[(long)row * N + col] = acc[mi][ni][r];
      }
    }
}

template <int SKIP>
static float run(const uint16_t *a, const uint16_t *b, float *c, int M, int N,
                 long Kpix, int splits, int iters) {
  TnRowMajor la{a, (long)M, (int)Kpix, M};
  TnRowMajor lb{b, (long)N, (int)Kpix, N};
  int tiles_m = (M + 127) / 128, tiles_n = (N + 127) / 128;
  int nk = (int)((Kpix + 63) / 64);
  if (splits > nk) splits = nk;
  int kt = (nk + splits - 1) / splits;
  dim3 grid(tiles_m * tiles_n, splits);
  hipEvent_t e0, e1;
  hipEventCreate(&e0);
  hipEventCreate(&e1);
  // warmup
  ablate_k<SKIP><<<grid, NT_THREADS>>>(TnStage<TnRowMajor>{la},
                                       TnStage<TnRowMajor>{lb}, c, M, N,
                                       (int)Kpix, tiles_n, kt, (long)M * N);
  hipDeviceSynchronize();
  hipEventRecord(e0);
  for (int i = 0; i < iters; ++i)
    ablate_k<SKIP><<<grid, NT_THREADS>>>(TnStage<TnRowMajor>{la},
                                         TnStage<TnRowMajor>{lb}, c, M, N,
                                         (int)Kpix, tiles_n, kt, (long)M * N);
  hipEventRecord(e1);
  hipEventSynchronize(e1);
  float ms;
  hipEventElapsedTime(&ms, e0, e1);
  return ms / iters;
}

int main() {
  // b2-1x1 wgrad shape: dw[128][512] over M=50176 pixels
  int M = 128, N = 512;
  long Kpix = 50176;
  int splits = 64, iters = 50;
  uint16_t *a, *b;
  float *c;
  hipMalloc(&a, Kpix * M * 2);
  hipMalloc(&b, Kpix * N * 2);
  hipMalloc(&c, (long)splits * M * N * 4);
  hipMemset(a, 0x3c, Kpix * M * 2);
  hipMemset(b, 0x3c, Kpix * N * 2);
  double gf = 2.0 * M * N * Kpix / 1e12;
  float full = run<0>(a, b, c, M, N, Kpix, splits, iters);
  float noload = run<1>(a, b, c, M, N, Kpix, splits, iters);
  float nomfma = run<2>(a, b, c, M, N, Kpix, splits, iters);
  float noepi = run<4>(a, b, c, M, N, Kpix, splits, iters);
  float nolw = run<9>(a, b, c, M, N, Kpix, splits, iters); // no loads+no lds writes
  printf("wgrad-ish TnTn M=%d N=%d K=%ld splits=%d\n", M, N, Kpix, splits);
  printf("full      %.1f us  %.1f TF\n", full * 1e3, gf / (full / 1e3));
  printf("no-load   %.1f us  (staging global loads cost %.1f us)\n", noload * 1e3, (full - noload) * 1e3);
  printf("no-mfma   %.1f us  (mfma+lds-read cost %.1f us)\n", nomfma * 1e3, (full - nomfma) * 1e3);
  printf("no-epilog %.1f us  (epilogue cost %.1f us)\n", noepi * 1e3, (full - noepi) * 1e3);
  printf("lds-only  %.1f us  (pure mfma+ds_read floor)\n", nolw * 1e3);
  return 0;
}
