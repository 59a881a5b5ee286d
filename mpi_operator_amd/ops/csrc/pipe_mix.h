// Deep-pipelined 128²-tile GEMM with NT *and* TN (k-strided) operand
// staging — the round-2 replacement for mix_gemm's 2-barrier register-
// staged tile (measured there: staging loads ~60% of the TN kernel).
//
// Everything is glds (HBM→LDS DMA, no staging registers, no ds_write
// pass), kept in flight across raw barriers by counted vmcnt:
//
//   - 256 threads, 4 waves as 2×2; per-wave C = 64×64 as 2×2 frags of
//     32×32 (mfma_f32_32x32x16_bf16)
//   - K-tile BK=64 as two 32-deep k-halves; 2 phases per tile (one per
//     k-half); per phase: [vmcnt(4)] → frag ds_reads → stage one k-half
//     of A+B for a FUTURE tile (4 glds) → s_barrier → lgkmcnt(0) →
//     8 MFMA → [optional s_barrier]
//   - LDS: 2 tile-buffers × 2 halves × 2 operands × 8 KiB = 64 KiB →
//     2 blocks/CU (8 waves/CU)
//   - stage schedule (slot freed by the k-half-split read pattern):
//       t.ph0 stages (t+1, k1);  t.ph1 stages (t+2, k0)
//     every consumed half was staged 3 phases earlier and is proven by a
//     vmcnt(4) one full phase (≥1 barrier) before its first read; the
//     LAST tile's checks drain (vmcnt 0) because the skipped stages would
//     otherwise leave the final halves uncounted.
//
// Operand layouts:
//   NT (k-contiguous rows): LDS image [128 rows][4 k-octets] lane-linear,
//     source octet pre-swizzled q ^= (row>>2)&3 (pipe256's measured-best),
//     consumed by ds_read_b128.
//   TN (k-strided, memory rows ARE k): LDS image = [4k][16col] subtiles,
//     cq-fastest (64 subtiles per half), filled by glds of 8 columns at
//     one k (coalesced 32-B runs that the coalescer merges into full
//     lines), consumed by ds_read_b64_tr_b16 PAIRS — the gfx950 LDS
//     transpose read. Probed semantics (tools/pipe_bench --probe): lane l
//     receives column (l&15) of the 128-B-aligned block at its address,
//     elems j = block[(l&15) + 16j], j=0..3 — so two tr reads at k-subtiles
//     (4k apart) assemble the 8-deep k fragment with zero VALU repacking,
//     and a 32-lane service group touching two ADJACENT blocks covers all
//     64 banks conflict-free.
//
// Sources are SrcMap structs: ptr16(k, col8) → global address of the
// 16-B granule, or nullptr for out-of-range/padding — the launcher passes
// a 16-B zeros page those lanes load instead, so EVERY lane issues the
// same number of VM ops (the counted-vmcnt contract) and ragged M/N/K
// need no edge kernels.
//
// Replaces (reference role): the cuDNN/cuBLAS backward GEMMs of the
// tf_cnn_benchmarks image (reference README.md:127-130) — conv wgrad /
// dgrad and linear dw/dx shapes — as MI355X-native code.
#pragma once

#include <mutex>

// included from mix_gemm.h after mfma_tile.h; needs LinearWriter etc.

typedef __attribute__((ext_vector_type(2))) unsigned int uint2v_pm;

// MPIAMD_PIPEMIX=0 reverts every pipe_mix route to the round-1 register-
// staged mix_gemm tile (A/B lever).
static inline bool use_pipemix() {
  static const bool on = [] {
    const char *e = getenv("MPIAMD_PIPEMIX");
    return !(e && e[0] == '0');
  }();
  return on;
}

// The conv GATHER routes measured net-negative on the 4-WAVE pipeline
// (3296 vs 3390 same-box), but the verdict REVERSED under the 8-wave
// kernel: 3742 -> 3820 img/s ResNet101 / 6508 ResNet50 same-box — the
// extra VALU of the gather ptr16 hides behind the doubled wave count.
// Default ON; MPIAMD_PIPEGATHER=0 reverts to the register-staged mix
// gathers.
static inline bool use_pipegather() {
  static const bool on = [] {
    const char *e = getenv("MPIAMD_PIPEGATHER");
    return !(e && e[0] == '0');
  }();
  return use_pipemix() && on;
}

// wgrad-only override: the -2.9% PIPEGATHER measurement bundled fwd/dgrad
// and wgrad gathers; A/B'd alone the Xcol wgrad pipe WINS (3566 -> 3573
// img/s ResNet101, 6129 ResNet50 same-box) — default ON.
// MPIAMD_PIPEGATHER_WGRAD=0 reverts to the register-staged mix gather.
static inline bool use_pipegather_wgrad() {
  static const bool on = [] {
    const char *e = getenv("MPIAMD_PIPEGATHER_WGRAD");
    return !(e && e[0] == '0');
  }();
  return (use_pipemix() && on) || use_pipegather();
}

// ---- SrcMaps ----------------------------------------------------------
// concept (stateful: each thread stages the same 2 (row/col, k-offset)
// granules every k-tile, so the expensive decomposition hoists into init):
//   void init(int i, int rc, int koff)       — granule i: output-dim index
//                                              rc, fixed k offset koff
//   const uint16_t *ptr16(int i, int kb)     — address of granule i at
//                                              reduce base kb (the 8 elems
//                                              at k = kb + koff for NT /
//                                              the 8 cols at k = kb + koff
//                                              for TN), nullptr = zeros.

struct PlainNtSrc { // row-major [rows][kdim]: rows along the output dim
  const uint16_t *p;
  long ld;
  int rows, kdim;
  const uint16_t *base_[2];
  int koff_[2];
  DEV_INLINE void init(int i, int rc, int koff) {
    base_[i] = rc < rows ? p + (long)rc * ld + koff : nullptr;
    koff_[i] = koff;
  }
  DEV_INLINE const uint16_t *ptr16(int i, int kb) const {
    return (base_[i] && kb + koff_[i] < kdim) ? base_[i] + kb : nullptr;
  }
};

struct PlainTnSrc { // row-major [kdim][cols]: rows along k
  const uint16_t *p;
  long ld;
  int kdim, cols;
  const uint16_t *base_[2];
  int koff_[2];
  DEV_INLINE void init(int i, int rc, int koff) {
    base_[i] = rc < cols ? p + (long)koff * ld + rc : nullptr;
    koff_[i] = koff;
  }
  DEV_INLINE const uint16_t *ptr16(int i, int kb) const {
    return (base_[i] && kb + koff_[i] < kdim) ? base_[i] + (long)kb * ld
                                              : nullptr;
  }
};

// small-divisor helpers for the per-stage k decompositions: ResNet's C/Q
// are powers of two (shift+mask) and rs = k/C stays < R·S ≤ 49, so a
// 16-bit reciprocal multiply is exact — no integer divides in ptr16.
DEV_INLINE int ilog2_if_pow2(int v) {
  return (v & (v - 1)) == 0 ? (31 - __builtin_clz(v)) : -1;
}
DEV_INLINE int div_small(int num, int recip16) { // exact for num ≤ ~1000
  return (num * recip16) >> 16;
}

// conv forward x-patch gather (NT: rows are output pixels, k = (r,s,c)).
// Row decomposition (n, ho·stride-pad, wo·stride-pad) hoisted; per stage
// only k's (r,s,c) split + 2 bounds checks remain.
struct ConvFwdSrc {
  const uint16_t *x;
  int H, W, C, HO, WO, S, stride, pad, K;
  long M;
  long nbase_[2];
  int hb_[2], wb_[2], koff_[2];
  int csh_, srecip_;
  bool ok_[2];
  DEV_INLINE void init(int i, int rc, int koff) {
    long m = rc;
    ok_[i] = m < M;
    int wo = (int)(m % WO);
    long t = m / WO;
    int ho = (int)(t % HO);
    int n = (int)(t / HO);
    nbase_[i] = (long)n * H * W;
    hb_[i] = ho * stride - pad;
    wb_[i] = wo * stride - pad;
    koff_[i] = koff;
    csh_ = ilog2_if_pow2(C);
    srecip_ = (1 << 16) / S + 1;
  }
  DEV_INLINE const uint16_t *ptr16(int i, int kb) const {
    int k = kb + koff_[i];
    int c, rs;
    if (csh_ >= 0) {
      c = k & (C - 1);
      rs = k >> csh_;
    } else {
      c = k % C;
      rs = k / C;
    }
    int rr = div_small(rs, srecip_);
    int s_ = rs - rr * S;
    int h = hb_[i] + rr, w = wb_[i] + s_;
    if (k < K && ok_[i] && (unsigned)h < (unsigned)H && (unsigned)w < (unsigned)W)
      return x + (nbase_[i] + (long)h * W + w) * C + c;
    return nullptr;
  }
};

// conv dgrad dy-gather (NT: rows are input pixels, k = (r,s,q)); STRIDE
// a template param so the inner %/÷ are shifts.
template <int STRIDE> struct ConvDgradSrc {
  const uint16_t *dy;
  int H, W, Q, HO, WO, S, pad, K;
  long M;
  long nbase_[2];
  int hb_[2], wb_[2], koff_[2];
  int qsh_, srecip_;
  bool ok_[2];
  DEV_INLINE void init(int i, int rc, int koff) {
    long m = rc;
    ok_[i] = m < M;
    int w_ = (int)(m % W);
    long t = m / W;
    int h_ = (int)(t % H);
    int n = (int)(t / H);
    nbase_[i] = (long)n * HO * WO;
    hb_[i] = h_ + pad;
    wb_[i] = w_ + pad;
    koff_[i] = koff;
    qsh_ = ilog2_if_pow2(Q);
    srecip_ = (1 << 16) / S + 1;
  }
  DEV_INLINE const uint16_t *ptr16(int i, int kb) const {
    int k = kb + koff_[i];
    int q, rs;
    if (qsh_ >= 0) {
      q = k & (Q - 1);
      rs = k >> qsh_;
    } else {
      q = k % Q;
      rs = k / Q;
    }
    int rr = div_small(rs, srecip_);
    int s_ = rs - rr * S;
    int hn = hb_[i] - rr, wn = wb_[i] - s_;
    int ho = hn / STRIDE, wo = wn / STRIDE;
    if (k < K && ok_[i] && hn >= 0 && wn >= 0 &&
        (STRIDE == 1 || (hn % STRIDE == 0 && wn % STRIDE == 0)) && ho < HO &&
        wo < WO)
      return dy + (nbase_[i] + (long)ho * WO + wo) * Q + q;
    return nullptr;
  }
};

// dgrad weight view (TN: k = (r,s,q) with q fastest; cols are input
// channels): element (c, k) = w[q][(r·S+s)·C + c].
struct DgradWTnSrc {
  const uint16_t *w;
  int C, Q, K, RSC;
  int col_[2], koff_[2];
  DEV_INLINE void init(int i, int rc, int koff) {
    col_[i] = rc;
    koff_[i] = koff;
  }
  DEV_INLINE const uint16_t *ptr16(int i, int kb) const {
    int k = kb + koff_[i];
    if (k >= K || col_[i] >= C) return nullptr;
    int q = k % Q, rs = k / Q;
    return w + (long)q * RSC + rs * C + col_[i];
  }
};

// stride-2 dgrad parity gather (NT: rows = parity-subgrid pixels, k =
// (tap, q) with q fastest; tap shifts affine in the tap index). Ported
// from conv.hip's DgradS2Stage to the SrcMap contract; Q is pow2 on every
// ResNet shape (shift/mask), ntw <= 4 (16-bit reciprocal).
struct DgradS2Src {
  const uint16_t *dy;
  int HO, WO, Q, W2, H2, K;
  long M;
  int dh0, dw0, nth, ntw;
  long nbase_[2];
  int hb_[2], wb_[2], koff_[2];
  bool ok_[2];
  int qsh_, ntwrecip_;
  DEV_INLINE void init(int i, int rc, int koff) {
    long m = rc;
    ok_[i] = m < M;
    int w_ = (int)(m % W2);
    long t = m / W2;
    int h_ = (int)(t % H2);
    int n = (int)(t / H2);
    nbase_[i] = (long)n * HO * WO;
    hb_[i] = h_;
    wb_[i] = w_;
    koff_[i] = koff;
    qsh_ = ilog2_if_pow2(Q);
    ntwrecip_ = (1 << 16) / ntw + 1;
  }
  DEV_INLINE const uint16_t *ptr16(int i, int kb) const {
    int k = kb + koff_[i];
    int q, ti;
    if (qsh_ >= 0) {
      q = k & (Q - 1);
      ti = k >> qsh_;
    } else {
      q = k % Q;
      ti = k / Q;
    }
    int ih = div_small(ti, ntwrecip_);
    int iw = ti - ih * ntw;
    int ho = hb_[i] + dh0 - ih, wo = wb_[i] + dw0 - iw;
    if (k < K && ok_[i] && (unsigned)ho < (unsigned)HO &&
        (unsigned)wo < (unsigned)WO)
      return dy + (nbase_[i] + (long)ho * WO + wo) * Q + q;
    return nullptr;
  }
};

// stride-2 dgrad weight view (TN: cols = C, k = (tap, q)):
// element (c, k) = w[q][((r0+2*ih)*S + s0+2*iw)*C + c]
struct DgradWS2TnSrc {
  const uint16_t *w;
  int C, Q, K, RSC, S, r0, s0, nth, ntw;
  int col_[2], koff_[2];
  int qsh_, ntwrecip_;
  DEV_INLINE void init(int i, int rc, int koff) {
    col_[i] = rc;
    koff_[i] = koff;
    qsh_ = ilog2_if_pow2(Q);
    ntwrecip_ = (1 << 16) / ntw + 1;
  }
  DEV_INLINE const uint16_t *ptr16(int i, int kb) const {
    int k = kb + koff_[i];
    if (k >= K || col_[i] >= C) return nullptr;
    int q, ti;
    if (qsh_ >= 0) {
      q = k & (Q - 1);
      ti = k >> qsh_;
    } else {
      q = k % Q;
      ti = k / Q;
    }
    int ih = div_small(ti, ntwrecip_);
    int iw = ti - ih * ntw;
    return w + (long)q * RSC + ((r0 + 2 * ih) * S + s0 + 2 * iw) * C + col_[i];
  }
};

// wgrad implicit-im2col gather (TN: k = output pixel m, cols = (r,s,c)).
// Column decomposition (rr, ss, c) hoisted. The pixel decomposition rides
// a CARRY CHAIN: pipe_mix stages k-halves in strictly increasing kb order
// (prologue 0,32,64,96 then +32 per phase), so each granule's (n, ho, wo)
// advances by a precomputed (Δho = 32/WO, Δwo = 32%WO) instead of two
// int64 divmods per stage — the cost that made the first gather-pipeline
// cut 2.9% slower end-to-end.
struct XcolSrc {
  const uint16_t *x;
  int H, W, C, HO, WO, S, stride, pad, RSC;
  long M;
  int rr_[2], ss_[2], coff_[2], koff_[2];
  mutable long mcur_[2];
  mutable int n_[2], ho_[2], wo_[2];
  int dho_, dwo_;
  bool cok_[2];
  DEV_INLINE void init(int i, int rc, int koff) {
    cok_[i] = rc < RSC;
    int c = rc % C, rs = rc / C;
    ss_[i] = rs % S;
    rr_[i] = rs / S;
    coff_[i] = c;
    koff_[i] = koff;
    mcur_[i] = -1;
    dho_ = 32 / WO;
    dwo_ = 32 % WO;
  }
  DEV_INLINE const uint16_t *ptr16(int i, int kb) const {
    long m = kb + koff_[i];
    if (mcur_[i] != m) { // first call of this split: full decomposition
      wo_[i] = (int)(m % WO);
      long t = m / WO;
      ho_[i] = (int)(t % HO);
      n_[i] = (int)(t / HO);
    }
    const uint16_t *out = nullptr;
    if (m < M && cok_[i]) {
      int h = ho_[i] * stride + rr_[i] - pad, w = wo_[i] * stride + ss_[i] - pad;
      if ((unsigned)h < (unsigned)H && (unsigned)w < (unsigned)W)
        out = x + ((long)(n_[i] * H + h) * W + w) * C + coff_[i];
    }
    // advance to the next stage's pixel (m + 32), carrying wo→ho→n
    int wo = wo_[i] + dwo_, ho = ho_[i] + dho_;
    if (wo >= WO) {
      wo -= WO;
      ++ho;
    }
    while (ho >= HO) { // bounded by 32/(WO*HO)+1 — ≥7×7 outputs: once
      ho -= HO;
      ++n_[i];
    }
    wo_[i] = wo;
    ho_[i] = ho;
    mcur_[i] = m + 32;
    return out;
  }
};

// ---- stagers ----------------------------------------------------------
constexpr int PM_BM = 128, PM_BK = 64; // tile 128x128, k-halves of 32
constexpr int PM_THREADS = 256;
constexpr int PM_HSZ = 512; // ushort8 slots per (operand, k-half) = 8 KiB

DEV_INLINE int pm_swz(int q, int row) { return q ^ ((row >> 2) & 3); }

// NT k-half: [128 rows][4 octets], lane-linear glds, source-swizzled.
// init() binds each thread's 2 fixed (row, k-octet-offset) granules into
// the SrcMap so per-stage address work is the SrcMap's ptr16 only.
template <class SRC> struct NtPipe {
  static constexpr bool TR = false;
  SRC s;
  DEV_INLINE void init(int tid, int base) {
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      int idx = i * PM_THREADS + tid; // [0,512)
      int row = idx >> 2;
      s.init(i, base + row, pm_swz(idx & 3, row) * 8);
    }
  }
  DEV_INLINE void stage(int tid, int kb, ushort8 *img,
                        const uint16_t *zeros) const {
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      int idx = i * PM_THREADS + tid;
      const uint16_t *src = s.ptr16(i, kb);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void *)(src ? src : zeros),
          (__attribute__((address_space(3))) void *)(img + idx), 16, 0, 0);
    }
  }
  // frag read: 32 rows starting at frag0, k-step ks (16 deep) of the half
  DEV_INLINE bf16x8 read(const ushort8 *img, int lane, int frag0, int ks) const {
    int row = frag0 + (lane & 31);
    int q = ks * 2 + (lane >> 5);
    return us8_to_bf8v(img[row * 4 + pm_swz(q, row)]);
  }
};

// TN k-half: 64 subtiles of [4k][16col], cq-fastest; tr_b16 consumption.
template <class SRC> struct TnPipe {
  static constexpr bool TR = true;
  SRC s;
  DEV_INLINE void init(int tid, int base) {
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      int idx = i * PM_THREADS + tid;     // [0,512) 16-B slots
      int st = idx >> 3;                  // subtile: kq*8 + cq
      int kq = st >> 3, cq = st & 7;
      int kl = (idx & 7) >> 1, ch = idx & 1;
      s.init(i, base + cq * 16 + ch * 8, kq * 4 + kl);
    }
  }
  DEV_INLINE void stage(int tid, int kb, ushort8 *img,
                        const uint16_t *zeros) const {
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      int idx = i * PM_THREADS + tid;
      const uint16_t *src = s.ptr16(i, kb);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void *)(src ? src : zeros),
          (__attribute__((address_space(3))) void *)(img + idx), 16, 0, 0);
    }
  }
  DEV_INLINE bf16x8 read(const ushort8 *img, int lane, int frag0, int ks) const {
    // two tr reads at k-subtiles (ks*4 + (lane>>5)*2 + {0,1}) deliver
    // column (lane&15) of block (kq, cq) = k 0..3 / 4..7 of this lane's
    // operand row — no repacking
    unsigned base = (unsigned)(unsigned long)(
        __attribute__((address_space(3))) const void *)img;
    int kq0 = ks * 4 + ((lane >> 5) & 1) * 2;
    int cq = (frag0 >> 4) + ((lane >> 4) & 1);
    unsigned a0 = base + (unsigned)((kq0 * 8 + cq) * 128 + (lane & 15) * 8);
    uint2v_pm lo, hi;
    // "=&v" early-clobber is load-bearing: with plain "=v" the allocator
    // may overlap lo with a0, and the first read clobbers the address the
    // second read consumes (manifested only under register pressure)
    asm volatile("ds_read_b64_tr_b16 %0, %2\n\t"
                 "ds_read_b64_tr_b16 %1, %2 offset:1024\n\t"
                 "s_waitcnt lgkmcnt(0)"
                 : "=&v"(lo), "=&v"(hi)
                 : "v"(a0)
                 : "memory");
    union { unsigned u[4]; bf16x8 v; } r;
    r.u[0] = lo.x; r.u[1] = lo.y; r.u[2] = hi.x; r.u[3] = hi.y;
    return r.v;
  }
};

// ---- kernel -----------------------------------------------------------
template <class SA, class SB, bool C_F32, class WR, int NPB = 1>
__global__ __launch_bounds__(PM_THREADS) void pipe_mix_k(
    SA sa, SB sb, void *__restrict__ cptr, int M, int N, int K, WR wrt,
    int tiles_n, int kt_per_split, long split_stride, int xcd_cpx, int swap,
    const uint16_t *__restrict__ zeros) {
  int tile = swap ? blockIdx.y : blockIdx.x;
  int split = swap ? blockIdx.x : blockIdx.y;
  if (xcd_cpx) tile = (tile & 7) * xcd_cpx + (tile >> 3);
  int tm = tile / tiles_n, tn = tile % tiles_n;
  int row0 = tm * PM_BM, col0 = tn * PM_BM;
  int tid = threadIdx.x;
  int lane = tid & 63, wave = tid >> 6;
  int wr = wave >> 1, wc = wave & 1;

  // 128-B aligned: the tr_b16 read resolves its block as addr & ~127
  __shared__ __align__(128) ushort8 lds[8 * PM_HSZ]; // [buf][op][kh] = 64 KiB
  if constexpr (WR::STATS) wrt.reset();
#define PM_IMG(buf, op, kh) (lds + (((buf) * 2 + (op)) * 2 + (kh)) * PM_HSZ)

  float16v acc[2][2] = {};

  int nk_total = (K + PM_BK - 1) / PM_BK;
  int t0 = split * kt_per_split;
  int nk = min(kt_per_split, nk_total - t0);
  if (nk < 0) nk = 0;

  sa.init(tid, row0);
  sb.init(tid, col0);
  auto stage_half = [&](int t, int kh) {
    int kb = (t0 + t) * PM_BK + kh * 32;
    sa.stage(tid, kb, PM_IMG(t & 1, 0, kh), zeros);
    sb.stage(tid, kb, PM_IMG(t & 1, 1, kh), zeros);
  };

  if (nk > 0) {
    // Prove BOTH halves of tile 0 at this barrier. (0,k1) is read at
    // t0.ph1, and the ph1-top check runs in the SAME phase as that read:
    // a per-wave vmcnt without an intervening barrier cannot collectivize
    // other waves' stages, so a fast wave could read (0,k1) while a slow
    // wave's prologue glds was still in flight — a timing race that
    // single-kernel microbenches never hit but a loaded full-model step
    // did (rare corruption → diverging loss). Staging ALL of tile 1 here
    // keeps 8 glds in flight across the wait (prologue latency hiding).
    stage_half(0, 0);
    stage_half(0, 1);
    if (nk > 1) {
      stage_half(1, 0);
      stage_half(1, 1); // t0.ph0 skips its (1,k1) stage in exchange
      asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();
  }

  for (int t = 0; t < nk; ++t) {
    int buf = t & 1;
#pragma unroll
    for (int kh = 0; kh < 2; ++kh) {
      // counted check: proves the half read THIS phase's successor needs.
      // ph0 proves (t,k1) [staged 3 phases ago]; ph1 proves (t+1,k0).
      // When the pipeline stops staging (last tiles), drain instead —
      // vmcnt(4) would leave the final halves unproven.
      if (t > 0 || kh > 0) { // tile0.ph0 was proven by the prologue wait
        bool steady = (kh == 0) ? (t + 1 < nk) : (t + 2 < nk);
        if (steady)
          asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
        else
          asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      }
      bf16x8 af[2][2], bf_[2][2];
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
#pragma unroll
        for (int mi = 0; mi < 2; ++mi)
          af[mi][ks] = sa.read(PM_IMG(buf, 0, kh), lane, wr * 64 + mi * 32, ks);
#pragma unroll
        for (int ni = 0; ni < 2; ++ni)
          bf_[ni][ks] = sb.read(PM_IMG(buf, 1, kh), lane, wc * 64 + ni * 32, ks);
      }
      if (kh == 0) {
        // t==0: (1,k1) was already staged by the prologue
        if (t > 0 && t + 1 < nk) stage_half(t + 1, 1);
      } else {
        if (t + 2 < nk) stage_half(t + 2, 0);
      }
      __builtin_amdgcn_s_barrier();
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
#pragma unroll
        for (int mi = 0; mi < 2; ++mi)
#pragma unroll
          for (int ni = 0; ni < 2; ++ni)
            acc[mi][ni] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                af[mi][ks], bf_[ni][ks], acc[mi][ni], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      if (NPB == 2) __builtin_amdgcn_s_barrier();
    }
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");

  // 32x32 C/D map (shared with mix_gemm): row-major loop, RowCtx hoisted
  cptr = (void *)((char *)cptr + split * split_stride * (C_F32 ? 4 : 2));
#pragma unroll
  for (int mi = 0; mi < 2; ++mi) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int row = row0 + wr * 64 + mi * 32 + (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
      if (row >= M) continue;
      typename WR::RowCtx rc = wrt.row_ctx(row);
#pragma unroll
      for (int ni = 0; ni < 2; ++ni) {
        int col = col0 + wc * 64 + ni * 32 + (lane & 31);
        if (col >= N) continue;
        if (C_F32)
          wrt.store_f32((float *)cptr, rc, col, acc[mi][ni][r]);
        else
          wrt.store_bf16((uint16_t *)cptr, rc, col, acc[mi][ni][r]);
      }
    }
  }
  if constexpr (WR::STATS) wrt.flush(lane);
}
#undef PM_IMG

// 16-B zeros page for padding/ragged lanes (per-device, lazily allocated;
// glds must read SOMETHING for the counted-vmcnt contract to hold).
inline const uint16_t *pm_zeros_page() {
  // process-lifetime allocation (256 B/device, never freed by design);
  // guarded: concurrent first calls from two host threads must not leak
  static std::mutex mu;
  static uint16_t *pages[64] = {};
  int dev = 0;
  hipGetDevice(&dev);
  if (!pages[dev]) {
    std::lock_guard<std::mutex> lk(mu);
    if (!pages[dev]) {
      void *p = nullptr;
      if (hipMalloc(&p, 256) != hipSuccess) return nullptr;
      hipMemset(p, 0, 256);
      pages[dev] = (uint16_t *)p;
    }
  }
  return pages[dev];
}

// ---- 8-wave variant plumbing (definitions at end of file) ----
template <class SRC> struct Nt8Pipe;
template <class SRC> struct Tn8Pipe;
template <class P> struct Pipe8Of;
template <class S> struct Pipe8Of<NtPipe<S>> { using type = Nt8Pipe<S>; };
template <class S> struct Pipe8Of<TnPipe<S>> { using type = Tn8Pipe<S>; };
static inline bool use_pipe8();
template <class SA8, class SB8, class WR>
static hipError_t launch_pipe_mix8_wr(const SA8 &sa, const SB8 &sb, void *c,
                                      int M, int N, int K, const WR &wrt,
                                      long ldc, bool c_f32, hipStream_t s,
                                      int splits);

template <class SA, class SB, class WR>
static hipError_t launch_pipe_mix_wr(const SA &sa, const SB &sb, void *c,
                                     int M, int N, int K, const WR &wrt,
                                     long ldc, bool c_f32, hipStream_t s,
                                     int splits = 1) {
  // MPIAMD_PIPE8: route every non-STATS writer through the 8-wave kernel
  // (the band-slab BnStatsWriter assumes the 2x2 wave structure)
  if constexpr (!WR::STATS) {
    if (use_pipe8())
      return launch_pipe_mix8_wr(typename Pipe8Of<SA>::type{sa.s},
                                 typename Pipe8Of<SB>::type{sb.s}, c, M, N,
                                 K, wrt, ldc, c_f32, s, splits);
  }
  const uint16_t *zeros = pm_zeros_page();
  if (!zeros) return hipErrorOutOfMemory;
  int tiles_m = (M + PM_BM - 1) / PM_BM, tiles_n = (N + PM_BM - 1) / PM_BM;
  int nk = (K + PM_BK - 1) / PM_BK;
  if (splits > nk) splits = nk > 0 ? nk : 1;
  int kts = (nk + splits - 1) / splits;
  long split_stride = (long)M * ldc;
  int nwg = tiles_m * tiles_n;
  // XCD fold also applies under split-K: the 2-D grid dispatches x-fastest,
  // so within each split slice consecutive tile ids still round-robin the
  // XCDs and the fold restores contiguous tile bands per XCD L2.
  // MPIAMD_SK_CPX=0 restores the old splits==1-only gate for A/B.
  static const bool sk_cpx = [] {
    const char *e = getenv("MPIAMD_SK_CPX");
    return !(e && e[0] == '0');
  }();
  int cpx = (nwg % 8 == 0 && nwg >= 32 && (splits == 1 || sk_cpx))
                ? nwg / 8 : 0;
  dim3 grid(nwg, splits);
  if (c_f32)
    pipe_mix_k<SA, SB, true, WR><<<grid, PM_THREADS, 0, s>>>(
        sa, sb, c, M, N, K, wrt, tiles_n, kts, split_stride, cpx, 0, zeros);
  else
    pipe_mix_k<SA, SB, false, WR><<<grid, PM_THREADS, 0, s>>>(
        sa, sb, c, M, N, K, wrt, tiles_n, kts, split_stride, cpx, 0, zeros);
  return hipGetLastError();
}

// ======================================================================
// 8-wave (512-thread) variant: same 128² tile, same LDS images and stage
// schedule, but 2×4 waves of 64M×32N each — 2 blocks/CU become 16
// waves/CU instead of 8 (the 4-wave form leaves half the wave slots
// empty at its 64 KiB LDS footprint; PMC: 43-48% parked, 20-30% stall on
// the dw/dx pool). Cost: B-frags are read by 2 waves each (LDS
// read:MFMA 1.5 vs 1.0). A/B gated at the launchers (MPIAMD_PIPE8).
// Reference role: same as pipe_mix_k — the cuDNN/cuBLAS backward GEMMs
// of the tf_cnn_benchmarks image (reference README.md:127-130), measured
// +3-4% end-to-end over the 4-wave form on both models.
// Per-thread VM-op counts HALVE (1 granule per stage instead of 2), so
// every counted vmcnt is half the 4-wave kernel's.
// ======================================================================

template <class SRC> struct Nt8Pipe {
  static constexpr bool TR = false;
  SRC s;
  DEV_INLINE void init(int tid, int base) {
    int row = tid >> 2;
    s.init(0, base + row, pm_swz(tid & 3, row) * 8);
  }
  DEV_INLINE void stage(int tid, int kb, ushort8 *img,
                        const uint16_t *zeros) const {
    const uint16_t *src = s.ptr16(0, kb);
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void *)(src ? src : zeros),
        (__attribute__((address_space(3))) void *)(img + tid), 16, 0, 0);
  }
  DEV_INLINE bf16x8 read(const ushort8 *img, int lane, int frag0, int ks) const {
    int row = frag0 + (lane & 31);
    int q = ks * 2 + (lane >> 5);
    return us8_to_bf8v(img[row * 4 + pm_swz(q, row)]);
  }
};

template <class SRC> struct Tn8Pipe {
  static constexpr bool TR = true;
  SRC s;
  DEV_INLINE void init(int tid, int base) {
    int st = tid >> 3;
    int kq = st >> 3, cq = st & 7;
    int kl = (tid & 7) >> 1, ch = tid & 1;
    s.init(0, base + cq * 16 + ch * 8, kq * 4 + kl);
  }
  DEV_INLINE void stage(int tid, int kb, ushort8 *img,
                        const uint16_t *zeros) const {
    const uint16_t *src = s.ptr16(0, kb);
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void *)(src ? src : zeros),
        (__attribute__((address_space(3))) void *)(img + tid), 16, 0, 0);
  }
  DEV_INLINE bf16x8 read(const ushort8 *img, int lane, int frag0, int ks) const {
    unsigned base = (unsigned)(unsigned long)(
        __attribute__((address_space(3))) const void *)img;
    int kq0 = ks * 4 + ((lane >> 5) & 1) * 2;
    int cq = (frag0 >> 4) + ((lane >> 4) & 1);
    unsigned a0 = base + (unsigned)((kq0 * 8 + cq) * 128 + (lane & 15) * 8);
    uint2v_pm lo, hi;
    asm volatile("ds_read_b64_tr_b16 %0, %2\n\t"
                 "ds_read_b64_tr_b16 %1, %2 offset:1024\n\t"
                 "s_waitcnt lgkmcnt(0)"
                 : "=&v"(lo), "=&v"(hi)
                 : "v"(a0)
                 : "memory");
    union { unsigned u[4]; bf16x8 v; } r;
    r.u[0] = lo.x; r.u[1] = lo.y; r.u[2] = hi.x; r.u[3] = hi.y;
    return r.v;
  }
};

template <class SA, class SB, bool C_F32, class WR>
__global__ __launch_bounds__(512) void pipe_mix8_k(
    SA sa, SB sb, void *__restrict__ cptr, int M, int N, int K, WR wrt,
    int tiles_n, int kt_per_split, long split_stride, int xcd_cpx, int swap,
    const uint16_t *__restrict__ zeros) {
  int tile = swap ? blockIdx.y : blockIdx.x;
  int split = swap ? blockIdx.x : blockIdx.y;
  if (xcd_cpx) tile = (tile & 7) * xcd_cpx + (tile >> 3);
  int tm = tile / tiles_n, tn = tile % tiles_n;
  int row0 = tm * PM_BM, col0 = tn * PM_BM;
  int tid = threadIdx.x;
  int lane = tid & 63, wave = tid >> 6;
  int wr = wave >> 2, wc = wave & 3; // 2×4: wave C = 64M × 32N

  __shared__ __align__(128) ushort8 lds[8 * PM_HSZ];
  if constexpr (WR::STATS) wrt.reset();
#define PM8_IMG(buf, op, kh) (lds + (((buf) * 2 + (op)) * 2 + (kh)) * PM_HSZ)

  float16v acc[2] = {};

  int nk_total = (K + PM_BK - 1) / PM_BK;
  int t0 = split * kt_per_split;
  int nk = min(kt_per_split, nk_total - t0);
  if (nk < 0) nk = 0;

  sa.init(tid, row0);
  sb.init(tid, col0);
  auto stage_half = [&](int t, int kh) {
    int kb = (t0 + t) * PM_BK + kh * 32;
    sa.stage(tid, kb, PM8_IMG(t & 1, 0, kh), zeros);
    sb.stage(tid, kb, PM8_IMG(t & 1, 1, kh), zeros);
  };

  if (nk > 0) {
    // prologue proves BOTH tile-0 halves at the barrier (same race rule
    // as the 4-wave kernel); per-thread op counts are HALVED here
    stage_half(0, 0);
    stage_half(0, 1);
    if (nk > 1) {
      stage_half(1, 0);
      stage_half(1, 1);
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();
  }

  for (int t = 0; t < nk; ++t) {
    int buf = t & 1;
#pragma unroll
    for (int kh = 0; kh < 2; ++kh) {
      if (t > 0 || kh > 0) {
        bool steady = (kh == 0) ? (t + 1 < nk) : (t + 2 < nk);
        if (steady)
          asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
        else
          asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      }
      bf16x8 af[2][2], bf_[2];
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
#pragma unroll
        for (int mi = 0; mi < 2; ++mi)
          af[mi][ks] = sa.read(PM8_IMG(buf, 0, kh), lane, wr * 64 + mi * 32, ks);
        bf_[ks] = sb.read(PM8_IMG(buf, 1, kh), lane, wc * 32, ks);
      }
      if (kh == 0) {
        if (t > 0 && t + 1 < nk) stage_half(t + 1, 1);
      } else {
        if (t + 2 < nk) stage_half(t + 2, 0);
      }
      __builtin_amdgcn_s_barrier();
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
#pragma unroll
        for (int mi = 0; mi < 2; ++mi)
          acc[mi] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              af[mi][ks], bf_[ks], acc[mi], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");

  cptr = (void *)((char *)cptr + split * split_stride * (C_F32 ? 4 : 2));
#pragma unroll
  for (int mi = 0; mi < 2; ++mi) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int row = row0 + wr * 64 + mi * 32 + (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
      if (row >= M) continue;
      typename WR::RowCtx rc = wrt.row_ctx(row);
      int col = col0 + wc * 32 + (lane & 31);
      if (col >= N) continue;
      if (C_F32)
        wrt.store_f32((float *)cptr, rc, col, acc[mi][r]);
      else
        wrt.store_bf16((uint16_t *)cptr, rc, col, acc[mi][r]);
    }
  }
}
#undef PM8_IMG

// Default ON: same-box A/B with the 8-wave kernel on every pipe route —
// BERT-Large bs32 1421 -> 1468 seq/s, bs8 607 -> 684, ResNet101
// 3560 -> 3697 img/s. The extra waves (16/CU vs 8 at the same 64 KiB LDS
// footprint) buy more than the duplicated B-frag LDS reads cost.
// MPIAMD_PIPE8=0 reverts to the 4-wave kernel.
static inline bool use_pipe8() {
  static const bool on = [] {
    const char *e = getenv("MPIAMD_PIPE8");
    return !(e && e[0] == '0');
  }();
  return on;
}

template <class SA8, class SB8, class WR>
static hipError_t launch_pipe_mix8_wr(const SA8 &sa, const SB8 &sb, void *c,
                                      int M, int N, int K, const WR &wrt,
                                      long ldc, bool c_f32, hipStream_t s,
                                      int splits) {
  const uint16_t *zeros = pm_zeros_page();
  if (!zeros) return hipErrorOutOfMemory;
  int tiles_m = (M + PM_BM - 1) / PM_BM, tiles_n = (N + PM_BM - 1) / PM_BM;
  int nk = (K + PM_BK - 1) / PM_BK;
  if (splits > nk) splits = nk > 0 ? nk : 1;
  int kts = (nk + splits - 1) / splits;
  long split_stride = (long)M * ldc;
  int nwg = tiles_m * tiles_n;
  int cpx = (nwg % 8 == 0 && nwg >= 32) ? nwg / 8 : 0;
  dim3 grid(nwg, splits);
  if (c_f32)
    pipe_mix8_k<SA8, SB8, true, WR><<<grid, 512, 0, s>>>(
        sa, sb, c, M, N, K, wrt, tiles_n, kts, split_stride, cpx, 0, zeros);
  else
    pipe_mix8_k<SA8, SB8, false, WR><<<grid, 512, 0, s>>>(
        sa, sb, c, M, N, K, wrt, tiles_n, kts, split_stride, cpx, 0, zeros);
  return hipGetLastError();
}
