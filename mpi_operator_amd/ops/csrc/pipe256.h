// 256²-tile 4-phase deep-pipelined NT GEMM (guide §5 "8-phase template"
// structure, re-derived for the 32x32x16 MFMA and a k-half staging grain).
//
// Why this exists: the 3-buffer BK=32 counted-vmcnt kernel (gemm256.h)
// stages whole K-tiles and groups all ds_reads per K-step — the guide's
// ladder shows the per-phase ds_read ∥ glds ∥ MFMA interleave is the lever
// (+28-41% over 1-phase; counted-vs-drain +38%), reaching ~1320 TF @4k²
// uniform-random vs ~750 for the 2-barrier 128² structure.
//
// Geometry:
//   - 512 threads, 8 waves as 2(M)×4(N); per-wave C = 128×64 as
//     4 m-frags × 2 n-frags of 32×32 (mfma_f32_32x32x16_bf16 — 2382 TF
//     µbench vs 2075 for 16x16x32)
//   - K-tile BK=64 split into two 32-deep k-halves; LDS image per half per
//     operand = [256 rows][32 k] bf16 (16 KiB) staged by 2 glds/thread
//   - 2 tile-buffers × 2 halves × 2 operands = 128 KiB LDS, 1 block/CU
//   - 4 phases per K-tile: (k-half, m-half). Each phase:
//       [vmcnt check at ph1/ph3] → 4-8 ds_read_b128 → stage one half-tile
//       of tile t+1 (2 glds) → s_barrier → lgkmcnt(0) → setprio(1) →
//       8 MFMA → setprio(0) → s_barrier
//   - counted vmcnt(2), never 0 in the loop. Proof of safety: glds retire
//     in issue order per wave; every thread stages a slice of EVERY half,
//     so each wave's own vmcnt plus the phase barrier collectivizes the
//     guarantee; each half is proven one full phase (two barriers) before
//     its first ds_read.
//       stage order: t.ph0→A(t+1,kh0), ph1→B(t+1,kh0), ph2→A(t+1,kh1),
//       ph3→B(t+1,kh1). Check at ph1 proves staged-through t-1.ph3 =
//       B(t,kh1) (read at t.ph2); check at ph3 proves through t.ph1 =
//       B(t+1,kh0) (read at t+1.ph0).
//
// LDS swizzle: glds is lane-linear (dst = base + tid·16) so the conflict
// fix lives in the SOURCE octet permutation q ^= SWZ(row) with the same
// XOR on the read side (guide T2, glds form). SWZ is a template parameter
// — ablated on hardware, see tools/ablate_gemm.
//
// FULL TILES ONLY: M%256==0, N%256==0, K%64==0 (edge shapes keep the
// mix_gemm path). Reference role: the cuDNN/cuBLAS GEMMs inside the
// tf_cnn_benchmarks image (reference README.md:127-130), rebuilt CDNA4-
// native.
#pragma once

// included from mix_gemm.h AFTER mfma_tile.h declarations.

// source-octet swizzles (see tools/ablate_gemm for the measured A/B):
//   0: none; 1: q^=(row>>2)&3 (16-slot-distinct under both consecutive-16
//   and stride-4 lane groupings — the derivation in the file header);
//   2: q^=row&3 (gemm256's round-1 choice)
template <int SWZ> DEV_INLINE int p256_swz(int q, int row) {
  if (SWZ == 1) return q ^ ((row >> 2) & 3);
  if (SWZ == 2) return q ^ (row & 3);
  return q;
}

// SP: 0 = per-phase setprio(1) around the MFMA cluster (guide T5); 1 =
// static form — the younger dispatch half (tid>=256) gets priority 1 once
// before the loop, nothing per-phase (T5 "static form": the condition must
// be wave-uniform via readfirstlane or s_setprio runs unconditionally).
template <bool C_F32, bool BIAS, int SWZ = 1, int NPB = 2, int SP = 0,
          class WR = LinearWriter, bool USE_WR = false>
__global__ __launch_bounds__(512) void pipe256_gemm_k(
    const uint16_t *__restrict__ a, long lda, const uint16_t *__restrict__ b,
    long ldb, void *__restrict__ cptr, int M, int N, int K, long ldc,
    int tiles_n, int xcd_cpx, const float *__restrict__ bias,
    WR wrt = WR{}) {
  int tile = blockIdx.x;
  if (xcd_cpx) tile = (tile & 7) * xcd_cpx + (tile >> 3);
  int tm = tile / tiles_n, tn = tile % tiles_n;
  int row0 = tm * 256, col0 = tn * 256;
  int tid = threadIdx.x;
  int lane = tid & 63, wave = tid >> 6;
  int wr = wave >> 2, wc = wave & 3; // 2x4 wave grid, 128x64 C per wave

  // one shared array, arithmetic bases (guide §5 trap 4a)
  constexpr int HSZ = 256 * 4; // one k-half: 256 rows x 4 slots of 16 B
  __shared__ ushort8 lds[8 * HSZ]; // [buf][op][kh] = 128 KiB
#define P256_IMG(buf, op, kh) (lds + (((buf) * 2 + (op)) * 2 + (kh)) * HSZ)

  if (USE_WR) {
    if constexpr (WR::STATS) wrt.reset();
  }
  float16v acc[4][2] = {};
  int nk = K / 64; // contract: K % 64 == 0

  // stage one k-half of one operand of tile t: 2 glds per thread.
  // thread-linear LDS slot (idx) ↔ (row=idx>>2, slot=idx&3); source octet
  // q = slot ^ SWZ(row) so the read side XORs the same.
  const uint16_t *ap = a + (long)row0 * lda;
  const uint16_t *bp = b + (long)col0 * ldb;
  auto stage = [&](int t, int op, int kh) {
    const uint16_t *p = op ? bp : ap;
    long ld = op ? ldb : lda;
    ushort8 *img = P256_IMG(t & 1, op, kh);
    int kb = t * 64 + kh * 32;
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      int idx = i * 512 + tid;
      int r = idx >> 2;
      int k = kb + p256_swz<SWZ>(idx & 3, r) * 8;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void *)(p + (long)r * ld + k),
          (__attribute__((address_space(3))) void *)(img + idx), 16, 0, 0);
    }
  };

  // frag readers: af[2] for m-frags (mh*2+{0,1}), bf[2] for the wave's 2
  // n-frags; kk ∈ {0,1} = k-step of 16 within the half.
  auto read_a = [&](bf16x8 af[2][2], int kh, int mh, int buf) {
#pragma unroll
    for (int mi = 0; mi < 2; ++mi)
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
        int r = wr * 128 + (mh * 2 + mi) * 32 + (lane & 31);
        int q = kk * 2 + (lane >> 5);
        af[mi][kk] =
            us8_to_bf8v(P256_IMG(buf, 0, kh)[r * 4 + p256_swz<SWZ>(q, r)]);
      }
  };
  auto read_b = [&](bf16x8 bf_[2][2], int kh, int buf) {
#pragma unroll
    for (int ni = 0; ni < 2; ++ni)
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
        int r = wc * 64 + ni * 32 + (lane & 31);
        int q = kk * 2 + (lane >> 5);
        bf_[ni][kk] =
            us8_to_bf8v(P256_IMG(buf, 1, kh)[r * 4 + p256_swz<SWZ>(q, r)]);
      }
  };

  // prologue: tile 0 fully staged; kh0 proven (vmcnt(4) leaves kh1's 4 in
  // flight), kh1 proven by the ph1 check inside the loop.
  if (nk > 0) {
    stage(0, 0, 0);
    stage(0, 1, 0);
    stage(0, 0, 1);
    stage(0, 1, 1);
    asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }

  if (SP == 1 && __builtin_amdgcn_readfirstlane(threadIdx.x) >= 256)
    __builtin_amdgcn_s_setprio(1);

  bf16x8 af[2][2], bf_[2][2];
  for (int t = 0; t < nk; ++t) {
    int buf = t & 1;
    bool pre = t + 1 < nk;
#pragma unroll
    for (int ph = 0; ph < 4; ++ph) {
      int kh = ph >> 1, mh = ph & 1;
      if (ph == 1 || ph == 3) { // counted: proves the half read 1 phase later
        if (pre) // steady state: 2 newest (this tile's fresh stages) allowed
          asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
        else // last tile stages nothing newer — vmcnt(2) would leave its
             // own kh1 half unproven: drain (guide: epilogue 4→2→0)
          asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      }
      if (mh == 0) read_b(bf_, kh, buf);
      read_a(af, kh, mh, buf);
      if (pre) stage(t + 1, ph & 1, ph >> 1); // ph0:A-kh0 ph1:B-kh0 ...
      __builtin_amdgcn_s_barrier();
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      if (SP == 0) __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
#pragma unroll
        for (int mi = 0; mi < 2; ++mi)
#pragma unroll
          for (int ni = 0; ni < 2; ++ni)
            acc[mh * 2 + mi][ni] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                af[mi][kk], bf_[ni][kk], acc[mh * 2 + mi][ni], 0, 0, 0);
      if (SP == 0) __builtin_amdgcn_s_setprio(0);
      if (NPB == 2) __builtin_amdgcn_s_barrier();
    }
  }
  if (SP == 1) __builtin_amdgcn_s_setprio(0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory"); // epilogue drain

  // 32x32 C/D map: col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      long row = row0 + wr * 128 + mi * 32 + (r & 3) + 8 * (r >> 2) +
                 4 * (lane >> 5);
      if (USE_WR) {
        typename WR::RowCtx rc = wrt.row_ctx((int)row);
#pragma unroll
        for (int ni = 0; ni < 2; ++ni) {
          int col = col0 + wc * 64 + ni * 32 + (lane & 31);
          if (C_F32)
            wrt.store_f32((float *)cptr, rc, col, acc[mi][ni][r]);
          else
            wrt.store_bf16((uint16_t *)cptr, rc, col, acc[mi][ni][r]);
        }
        continue;
      }
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
  if (USE_WR) {
    if constexpr (WR::STATS) wrt.flush(lane);
  }
}
#undef P256_IMG

// ======================================================================
// 16-wave (1024-thread) variant of the 256² pipeline: identical tile,
// LDS images, 4-phase schedule and swizzle, but 4(M)×4(N) waves of
// 64×64 each — 16 waves/CU instead of 8 at the same 128 KiB footprint
// (the 512-thread form leaves half the wave slots empty at 1 block/CU).
// Reference role: the forward cuBLAS GEMMs of the tf_cnn_benchmarks /
// Horovod images (reference README.md:127-130), MI355X-native.
// Per-thread glds counts HALVE (1 granule per stage): every counted
// vmcnt is half the 8-wave kernel's. A/B gate: MPIAMD_P256X16.
// ======================================================================
template <bool C_F32, bool BIAS, int SWZ = 1, int SP = 1,
          class WR = LinearWriter, bool USE_WR = false>
__global__ __launch_bounds__(1024) void pipe256x16_gemm_k(
    const uint16_t *__restrict__ a, long lda, const uint16_t *__restrict__ b,
    long ldb, void *__restrict__ cptr, int M, int N, int K, long ldc,
    int tiles_n, int xcd_cpx, const float *__restrict__ bias,
    WR wrt = WR{}) {
  int tile = blockIdx.x;
  if (xcd_cpx) tile = (tile & 7) * xcd_cpx + (tile >> 3);
  int tm = tile / tiles_n, tn = tile % tiles_n;
  int row0 = tm * 256, col0 = tn * 256;
  int tid = threadIdx.x;
  int lane = tid & 63, wave = tid >> 6;
  int wr = wave >> 2, wc = wave & 3; // 4x4 wave grid, 64x64 C per wave

  constexpr int HSZ = 256 * 4;
  __shared__ ushort8 lds[8 * HSZ]; // [buf][op][kh] = 128 KiB
#define P256X_IMG(buf, op, kh) (lds + (((buf) * 2 + (op)) * 2 + (kh)) * HSZ)

  if (USE_WR) {
    if constexpr (WR::STATS) wrt.reset();
  }
  float16v acc[2][2] = {};
  int nk = K / 64;

  const uint16_t *ap = a + (long)row0 * lda;
  const uint16_t *bp = b + (long)col0 * ldb;
  auto stage = [&](int t, int op, int kh) { // 1 glds per thread
    const uint16_t *p = op ? bp : ap;
    long ld = op ? ldb : lda;
    ushort8 *img = P256X_IMG(t & 1, op, kh);
    int kb = t * 64 + kh * 32;
    int r = tid >> 2;
    int k = kb + p256_swz<SWZ>(tid & 3, r) * 8;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void *)(p + (long)r * ld + k),
        (__attribute__((address_space(3))) void *)(img + tid), 16, 0, 0);
  };

  auto read_a = [&](bf16x8 af[2], int kh, int mh, int buf) {
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      int r = wr * 64 + mh * 32 + (lane & 31);
      int q = kk * 2 + (lane >> 5);
      af[kk] = us8_to_bf8v(P256X_IMG(buf, 0, kh)[r * 4 + p256_swz<SWZ>(q, r)]);
    }
  };
  auto read_b = [&](bf16x8 bf_[2][2], int kh, int buf) {
#pragma unroll
    for (int ni = 0; ni < 2; ++ni)
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
        int r = wc * 64 + ni * 32 + (lane & 31);
        int q = kk * 2 + (lane >> 5);
        bf_[ni][kk] =
            us8_to_bf8v(P256X_IMG(buf, 1, kh)[r * 4 + p256_swz<SWZ>(q, r)]);
      }
  };

  if (nk > 0) {
    stage(0, 0, 0);
    stage(0, 1, 0);
    stage(0, 0, 1);
    stage(0, 1, 1);
    asm volatile("s_waitcnt vmcnt(2)" ::: "memory"); // kh0 proven (2 ops)
    __builtin_amdgcn_s_barrier();
  }

  if (SP == 1 && __builtin_amdgcn_readfirstlane(threadIdx.x) >= 512)
    __builtin_amdgcn_s_setprio(1);

  bf16x8 af[2], bf_[2][2];
  for (int t = 0; t < nk; ++t) {
    int buf = t & 1;
    bool pre = t + 1 < nk;
#pragma unroll
    for (int ph = 0; ph < 4; ++ph) {
      int kh = ph >> 1, mh = ph & 1;
      if (ph == 1 || ph == 3) {
        if (pre) // 1 op per stage now: one fresh stage allowed in flight
          asm volatile("s_waitcnt vmcnt(1)" ::: "memory");
        else
          asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      }
      if (mh == 0) read_b(bf_, kh, buf);
      read_a(af, kh, mh, buf);
      if (pre) stage(t + 1, ph & 1, ph >> 1);
      __builtin_amdgcn_s_barrier();
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
#pragma unroll
        for (int ni = 0; ni < 2; ++ni)
          acc[mh][ni] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              af[kk], bf_[ni][kk], acc[mh][ni], 0, 0, 0);
    }
  }
  if (SP == 1) __builtin_amdgcn_s_setprio(0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");

#pragma unroll
  for (int mh = 0; mh < 2; ++mh) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      long row = row0 + wr * 64 + mh * 32 + (r & 3) + 8 * (r >> 2) +
                 4 * (lane >> 5);
      if (USE_WR) {
        typename WR::RowCtx rc = wrt.row_ctx((int)row);
#pragma unroll
        for (int ni = 0; ni < 2; ++ni) {
          int col = col0 + wc * 64 + ni * 32 + (lane & 31);
          if (C_F32)
            wrt.store_f32((float *)cptr, rc, col, acc[mh][ni][r]);
          else
            wrt.store_bf16((uint16_t *)cptr, rc, col, acc[mh][ni][r]);
        }
        continue;
      }
      long base = row * ldc;
#pragma unroll
      for (int ni = 0; ni < 2; ++ni) {
        int col = col0 + wc * 64 + ni * 32 + (lane & 31);
        float v = BIAS ? acc[mh][ni][r] + bias[col] : acc[mh][ni][r];
        if (C_F32)
          ((float *)cptr)[base + col] = v;
        else
          ((uint16_t *)cptr)[base + col] = f2bf(v);
      }
    }
  }
  if (USE_WR) {
    if constexpr (WR::STATS) wrt.flush(lane);
  }
}
#undef P256X_IMG

// Default ON: +0.4% same-box on BOTH models (BERT 1513->1520 seq/s,
// ResNet101 3812->3829 img/s). MPIAMD_P256X16=0 reverts to 8 waves.
static inline bool use_p256x16() {
  static const bool on = [] {
    const char *e = getenv("MPIAMD_P256X16");
    return !(e && e[0] == '0');
  }();
  return on;
}

// Writer-parameterized launch (fused epilogues: GELU+bias, BN stats, ...)
template <class WR, class LA, class LB>
static hipError_t launch_pipe256_wr(const LA &la, const LB &lb, void *c,
                                    int M, int N, int K, long ldc, bool c_f32,
                                    const WR &wrt, hipStream_t s) {
  int tiles_m = M / 256, tiles_n = N / 256;
  int nwg = tiles_m * tiles_n;
  int cpx = (nwg % 8 == 0 && nwg >= 32) ? nwg / 8 : 0;
  constexpr int SWZ = 1, NPB = 1, SP = 1;
  if (use_p256x16()) {
    if (c_f32)
      pipe256x16_gemm_k<true, false, SWZ, SP, WR, true><<<nwg, 1024, 0, s>>>(
          la.p, la.ld, lb.p, lb.ld, c, M, N, K, ldc, tiles_n, cpx, nullptr,
          wrt);
    else
      pipe256x16_gemm_k<false, false, SWZ, SP, WR, true><<<nwg, 1024, 0, s>>>(
          la.p, la.ld, lb.p, lb.ld, c, M, N, K, ldc, tiles_n, cpx, nullptr,
          wrt);
    return hipGetLastError();
  }
  if (c_f32)
    pipe256_gemm_k<true, false, SWZ, NPB, SP, WR, true><<<nwg, 512, 0, s>>>(
        la.p, la.ld, lb.p, lb.ld, c, M, N, K, ldc, tiles_n, cpx, nullptr, wrt);
  else
    pipe256_gemm_k<false, false, SWZ, NPB, SP, WR, true><<<nwg, 512, 0, s>>>(
        la.p, la.ld, lb.p, lb.ld, c, M, N, K, ldc, tiles_n, cpx, nullptr, wrt);
  return hipGetLastError();
}

template <class LA, class LB>
static hipError_t launch_pipe256(const LA &la, const LB &lb, void *c, int M,
                                 int N, int K, long ldc, bool c_f32,
                                 hipStream_t s, const float *bias = nullptr) {
  int tiles_m = M / 256, tiles_n = N / 256;
  int nwg = tiles_m * tiles_n;
  int cpx = (nwg % 8 == 0 && nwg >= 32) ? nwg / 8 : 0;
  // production config = measured best (tools/pipe_bench, same-box sweep):
  // swz1 + 1 barrier/phase + static setprio: 1146 TF @4k³ vs 789 for the
  // round-1 gemm256 (+44%); +17-42% on the real model shapes
  constexpr int SWZ = 1, NPB = 1, SP = 1;
  if (use_p256x16()) {
    if (c_f32) {
      if (bias)
        pipe256x16_gemm_k<true, true, SWZ, SP><<<nwg, 1024, 0, s>>>(
            la.p, la.ld, lb.p, lb.ld, c, M, N, K, ldc, tiles_n, cpx, bias);
      else
        pipe256x16_gemm_k<true, false, SWZ, SP><<<nwg, 1024, 0, s>>>(
            la.p, la.ld, lb.p, lb.ld, c, M, N, K, ldc, tiles_n, cpx, bias);
    } else {
      if (bias)
        pipe256x16_gemm_k<false, true, SWZ, SP><<<nwg, 1024, 0, s>>>(
            la.p, la.ld, lb.p, lb.ld, c, M, N, K, ldc, tiles_n, cpx, bias);
      else
        pipe256x16_gemm_k<false, false, SWZ, SP><<<nwg, 1024, 0, s>>>(
            la.p, la.ld, lb.p, lb.ld, c, M, N, K, ldc, tiles_n, cpx, bias);
    }
    return hipGetLastError();
  }
  if (c_f32) {
    if (bias)
      pipe256_gemm_k<true, true, SWZ, NPB, SP><<<nwg, 512, 0, s>>>(
          la.p, la.ld, lb.p, lb.ld, c, M, N, K, ldc, tiles_n, cpx, bias);
    else
      pipe256_gemm_k<true, false, SWZ, NPB, SP><<<nwg, 512, 0, s>>>(
          la.p, la.ld, lb.p, lb.ld, c, M, N, K, ldc, tiles_n, cpx, bias);
  } else {
    if (bias)
      pipe256_gemm_k<false, true, SWZ, NPB, SP><<<nwg, 512, 0, s>>>(
          la.p, la.ld, lb.p, lb.ld, c, M, N, K, ldc, tiles_n, cpx, bias);
    else
      pipe256_gemm_k<false, false, SWZ, NPB, SP><<<nwg, 512, 0, s>>>(
          la.p, la.ld, lb.p, lb.ld, c, M, N, K, ldc, tiles_n, cpx, bias);
  }
  return hipGetLastError();
}
