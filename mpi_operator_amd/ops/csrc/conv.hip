// Implicit-GEMM convolution on MFMA (gfx950) for NHWC bf16 activations and
// channels-last ([K][R][S][C]) bf16 weights — the reference workload's
// cuDNN conv replaced by hand-written CDNA4 kernels.
//
//   fwd:   y[m=(n,ho,wo)][kout]  = sum_{k=(r,s,c)}   x[patch(m,k)] * w[kout][k]
//   dgrad: dx[m=(n,h,w)][c]      = sum_{k=(r,s,q)}  dy[opatch(m,k)] * wT[k][c]
//   wgrad: dw[kout][(r,s,c)]     = sum_{m}          dyT[kout][m] * PT[(r,s,c)][m]
//
// fwd/dgrad/wgrad all run on the mixed-staging MFMA GEMM (mix_gemm.h):
// gather loaders turn padding/stride validity into zero-fill, and k-strided
// operands (wgrad's dy and implicit im2col, dgrad's weight) are transposed
// in the LDS write pass — nothing is ever materialized or pre-transposed.
// wgrad is split-K with exact fp32 partial slabs + a reduce kernel.
#include "mfma_tile.h"
#include "mix_gemm.h"

extern "C" hipError_t splitk_reduce(const float *, int, long, void *, int,
                                    hipStream_t);

// Custom NT stager for conv fwd: the 4 staged rows (output pixels) are
// FIXED for the whole kernel, so their (n, ho, wo) decomposition is hoisted
// into init(); per k-step only the (r,s,c) split of one k remains — the
// innermost gather is 2 adds + 2 bounds checks + 1 load.
struct ConvFwdStage {
  static constexpr int PITCH = MXP;
  static constexpr bool GLDS = false;
  static DEV_INLINE int rslot(int slot, int) { return slot; }
  const uint16_t *x;
  int H, W, C, HO, WO, S, stride, pad, K;
  long M;
  ushort8 r[4];
  long nbase_[4]; // n*H*W (row-invariant part of the address)
  int hb_[4], wb_[4];
  bool ok_[4];
  DEV_INLINE void init(int tid, int base) {
    int s_row = tid >> 3;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      long m = base + s_row + 32 * i;
      ok_[i] = m < M;
      int wo = (int)(m % WO);
      long t = m / WO;
      int ho = (int)(t % HO);
      int n = (int)(t / HO);
      nbase_[i] = (long)n * H * W;
      hb_[i] = ho * stride - pad;
      wb_[i] = wo * stride - pad;
    }
  }
  DEV_INLINE void load(int tid, int, int kb, ushort8 *) {
    int k = kb + (tid & 7) * 8;
    int c = k % C, rs = k / C;
    int s_ = rs % S, rr = rs / S;
    bool kok = k < K;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int h = hb_[i] + rr, w = wb_[i] + s_;
      if (kok && ok_[i] && (unsigned)h < (unsigned)H && (unsigned)w < (unsigned)W)
        r[i] = *(const ushort8 *)(x + (nbase_[i] + (long)h * W + w) * C + c);
      else
        r[i] = ushort8{0, 0, 0, 0, 0, 0, 0, 0};
    }
  }
  DEV_INLINE void write(int tid, ushort8 *img) const {
    int s_row = tid >> 3, s_slot = tid & 7;
#pragma unroll
    for (int i = 0; i < 4; ++i) img[(s_row + 32 * i) * MXP + s_slot] = r[i];
  }
};

// NT stager for dgrad: row (input pixel) decomposition hoisted to init();
// STRIDE is a template parameter so the innermost %/÷ are shifts, not the
// full integer division a runtime stride emits.
template <int STRIDE> struct ConvDgradStage {
  static constexpr int PITCH = MXP;
  static constexpr bool GLDS = false;
  static DEV_INLINE int rslot(int slot, int) { return slot; }
  const uint16_t *dy;
  int H, W, Q /*Kout*/, HO, WO, S, pad, K;
  long M;
  ushort8 r[4];
  long nbase_[4]; // n*HO*WO
  int hb_[4], wb_[4];
  bool ok_[4];
  DEV_INLINE void init(int tid, int base) {
    int s_row = tid >> 3;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      long m = base + s_row + 32 * i;
      ok_[i] = m < M;
      int w_ = (int)(m % W);
      long t = m / W;
      int h_ = (int)(t % H);
      int n = (int)(t / H);
      nbase_[i] = (long)n * HO * WO;
      hb_[i] = h_ + pad;
      wb_[i] = w_ + pad;
    }
  }
  DEV_INLINE void load(int tid, int, int kb, ushort8 *) {
    int k = kb + (tid & 7) * 8;
    int q = k % Q, rs = k / Q;
    int s_ = rs % S, rr = rs / S;
    bool kok = k < K;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int hn = hb_[i] - rr, wn = wb_[i] - s_;
      int ho = hn / STRIDE, wo = wn / STRIDE;
      bool ok = kok && ok_[i] && hn >= 0 && wn >= 0 &&
                (STRIDE == 1 || (hn % STRIDE == 0 && wn % STRIDE == 0)) &&
                ho < HO && wo < WO;
      r[i] = ok ? *(const ushort8 *)(dy + (nbase_[i] + (long)ho * WO + wo) * Q + q)
                : ushort8{0, 0, 0, 0, 0, 0, 0, 0};
    }
  }
  DEV_INLINE void write(int tid, ushort8 *img) const {
    int s_row = tid >> 3, s_slot = tid & 7;
#pragma unroll
    for (int i = 0; i < 4; ++i) img[(s_row + 32 * i) * MXP + s_slot] = r[i];
  }
};

// TN stager for wgrad's implicit-im2col operand: each thread's two column
// octets (r,s,c) are fixed for the whole kernel — decomposed once in
// init(); per k-step only the even output-pixel m is decomposed (odd m is
// derived by carry), so the inner gather is adds + bounds + load.
struct XcolStage {
  static constexpr int PITCH = MXP;
  static constexpr bool GLDS = false;
  static DEV_INLINE int rslot(int slot, int row) {
    return slot ^ ((row >> 3) & 7); // matches the TN write's slot-XOR
  }
  const uint16_t *x;
  int H, W, C, HO, WO, S, stride, pad, RSC;
  long M;
  ushort8 r[4];
  int k0_[2], coff_[2], rr_[2], ss_[2];
  bool cok_[2];
  DEV_INLINE void init(int tid, int base) {
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      int p = tid + it * 256;
      k0_[it] = (p >> 4) * 2;
      int rsc0 = base + (p & 15) * 8; // 16 lanes: 16 contiguous col octets
      cok_[it] = rsc0 < RSC;
      int c = rsc0 % C, rs = rsc0 / C;
      ss_[it] = rs % S;
      rr_[it] = rs / S;
      coff_[it] = c;
    }
  }
  DEV_INLINE ushort8 gather(int it, int n, int ho, int wo, bool mok) const {
    int h = ho * stride + rr_[it] - pad, w = wo * stride + ss_[it] - pad;
    if (mok && cok_[it] && (unsigned)h < (unsigned)H && (unsigned)w < (unsigned)W)
      return *(const ushort8 *)(x + ((long)(n * H + h) * W + w) * C + coff_[it]);
    return ushort8{0, 0, 0, 0, 0, 0, 0, 0};
  }
  DEV_INLINE void load(int tid, int, int kb, ushort8 *) {
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      long m = kb + k0_[it]; // even
      int wo = (int)(m % WO);
      long t = m / WO;
      int ho = (int)(t % HO);
      int n = (int)(t / HO);
      r[it * 2] = gather(it, n, ho, wo, m < M);
      // odd m = even + 1: carry-propagate instead of re-dividing
      int wo2 = wo + 1, ho2 = ho, n2 = n;
      if (wo2 == WO) {
        wo2 = 0;
        if (++ho2 == HO) {
          ho2 = 0;
          ++n2;
        }
      }
      r[it * 2 + 1] = gather(it, n2, ho2, wo2, m + 1 < M);
    }
  }
  DEV_INLINE void write(int tid, ushort8 *img) const {
    uint32_t *img32 = (uint32_t *)img;
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      int p = tid + it * 256;
      int k0 = (p >> 4) * 2;
      int col0 = (p & 15) * 8;
      int slot = k0 >> 3, within = (k0 & 7) >> 1;
      const ushort8 &va = r[it * 2], &vb = r[it * 2 + 1];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int row = col0 + j;
        img32[row * (MXP * 4) + rslot(slot, row) * 4 + within] =
            (uint32_t)va[j] | ((uint32_t)vb[j] << 16);
      }
    }
  }
};

// Late-stage convs have few 128x128 output tiles (7x7 spatial: ~100 blocks
// on 256 CUs): when `splits` > 1 the GEMM runs split-K into the caller's
// fp32 slab and splitk_reduce emits bf16 — same scheme as wgrad.
extern "C" hipError_t conv_fwd(const void *x, const void *w, void *y, int N,
                               int H, int W, int C, int Kout, int R, int S,
                               int stride, int pad, int HO, int WO, int splits,
                               float *partial, hipStream_t strm) {
  long M = (long)N * HO * WO;
  int K = R * S * C;
  int nk = (K + BK - 1) / BK;
  if (splits > nk) splits = nk > 0 ? nk : 1;
  void *out = splits > 1 ? (void *)partial : y;
  bool f32 = splits > 1;
  hipError_t e;
  if (R == 1 && S == 1 && stride == 1 && pad == 0) {
    // 1x1 conv IS a GEMM on the NHWC image viewed [M][C] — skip the
    // gather arithmetic entirely (≈47% of ResNet bottleneck FLOPs)
    GemmLoader la{(const uint16_t *)x, (int)M, (long)C, C};
    GemmLoader lb{(const uint16_t *)w, Kout, (long)C, C};
    e = launch_nt_gemm(la, lb, out, (int)M, Kout, C, Kout, f32, strm, splits);
  } else if (use_pipegather()) {
    // gather fwd on the deep pipeline: x patches via per-lane glds source
    // addresses (zeros page for padding), weights plain NT
    NtPipe<ConvFwdSrc> sa{
        {(const uint16_t *)x, H, W, C, HO, WO, S, stride, pad, K, M}};
    NtPipe<PlainNtSrc> sb{{(const uint16_t *)w, (long)K, Kout, K}};
    e = launch_pipe_mix_wr(sa, sb, out, (int)M, Kout, K,
                           LinearWriter{(long)Kout}, Kout, f32, strm, splits);
  } else {
    ConvFwdStage sa{(const uint16_t *)x, H, W, C, HO, WO, S, stride, pad, K, M};
    GemmLoader lb{(const uint16_t *)w, Kout, (long)K, K};
    e = launch_mix_gemm(sa, NtStage<GemmLoader>{lb}, out, (int)M, Kout, K,
                        Kout, f32, strm, splits);
  }
  if (e != hipSuccess || splits <= 1) return e;
  return splitk_reduce(partial, splits, M * Kout, y, 1, strm);
}

// conv forward that ALSO emits BatchNorm partial statistics from the GEMM
// epilogue (BnStatsWriter slab [ceil(M/64)][2][Kout]). splits==1 only —
// the caller (bindings conv2d_fwd_bn) checks and falls back otherwise.
extern "C" hipError_t conv_fwd_bn(const void *x, const void *w, void *y,
                                  int N, int H, int W, int C, int Kout, int R,
                                  int S, int stride, int pad, int HO, int WO,
                                  float *bn_slab, hipStream_t strm) {
  long M = (long)N * HO * WO;
  int K = R * S * C;
  BnStatsWriter wrt{(long)Kout, bn_slab, Kout};
  if (R == 1 && S == 1 && stride == 1 && pad == 0) {
    GemmLoader la{(const uint16_t *)x, (int)M, (long)C, C};
    GemmLoader lb{(const uint16_t *)w, Kout, (long)C, C};
    if (M % 256 == 0 && Kout % 256 == 0 && C % 64 == 0 &&
        (long)(M / 256) * (Kout / 256) >= 128)
      return launch_pipe256_wr(la, lb, y, (int)M, Kout, C, Kout, false, wrt,
                               strm);
    if (use_pipemix()) {
      NtPipe<PlainNtSrc> sa{{(const uint16_t *)x, (long)C, (int)M, C}};
      NtPipe<PlainNtSrc> sb{{(const uint16_t *)w, (long)C, Kout, C}};
      return launch_pipe_mix_wr(sa, sb, y, (int)M, Kout, C, wrt, Kout, false,
                                strm);
    }
    GldsNt ga{la.p, la.rows, la.ld, la.kdim};
    GldsNt gb{lb.p, lb.rows, lb.ld, lb.kdim};
    return launch_mix_gemm_wr(ga, gb, y, (int)M, Kout, C, wrt, Kout, false,
                              strm);
  }
  if (use_pipegather()) {
    NtPipe<ConvFwdSrc> sa{
        {(const uint16_t *)x, H, W, C, HO, WO, S, stride, pad, K, M}};
    NtPipe<PlainNtSrc> sb{{(const uint16_t *)w, (long)K, Kout, K}};
    return launch_pipe_mix_wr(sa, sb, y, (int)M, Kout, K, wrt, Kout, false,
                              strm);
  }
  ConvFwdStage sa{(const uint16_t *)x, H, W, C, HO, WO, S, stride, pad, K, M};
  GemmLoader lb{(const uint16_t *)w, Kout, (long)K, K};
  return launch_mix_gemm_wr(sa, NtStage<GemmLoader>{lb}, y, (int)M, Kout, K,
                            wrt, Kout, false, strm);
}

// 1x1 stride-1 dgrad that ACCUMULATES into dx (bottleneck backward: the
// conv1 input-grad lands directly on the skip-connection gradient, so the
// residual join needs no separate elementwise add pass).
extern "C" hipError_t conv_dgrad_1x1_acc(const void *dy, const void *w,
                                         void *dx_acc, long M, int C, int Kout,
                                         hipStream_t strm) {
  if (use_pipemix()) {
    NtPipe<PlainNtSrc> sa{{(const uint16_t *)dy, (long)Kout, (int)M, Kout}};
    TnPipe<PlainTnSrc> sb{{(const uint16_t *)w, (long)C, Kout, C}};
    return launch_pipe_mix_wr(sa, sb, dx_acc, (int)M, C, Kout,
                              LinearAccWriter{(long)C}, C, false, strm);
  }
  GemmLoader la{(const uint16_t *)dy, (int)M, (long)Kout, Kout};
  TnRowMajor lb{(const uint16_t *)w, (long)C, Kout, C};
  return launch_mix_gemm_wr(NtStage<GemmLoader>{la}, TnStage<TnRowMajor>{lb},
                            dx_acc, (int)M, C, Kout, LinearAccWriter{(long)C},
                            C, false, strm);
}

struct DgradWTn {
  const uint16_t *w;
  int C, Q, K, RSC;
  DEV_INLINE ushort8 load(int k, int c0) const {
    if (k >= K || c0 >= C) return ushort8{0, 0, 0, 0, 0, 0, 0, 0};
    int q = k % Q, rs = k / Q;
    return *(const ushort8 *)(w + (long)q * RSC + rs * C + c0);
  }
};

// ---- stride-2 dgrad, parity-decomposed --------------------------------
// A stride-2 conv's dgrad only receives from taps whose parity matches the
// input pixel: the naive gather zero-fills 3/4 of every fragment (measured
// 50 TF). Decompose dx into its 4 (h%2, w%2) sub-images; each is a DENSE
// GEMM over the parity's taps, scattered back by Stride2Writer.

// NT stager over sub-image rows: k = (tap, q) with q fastest.
struct DgradS2Stage {
  static constexpr int PITCH = MXP;
  static constexpr bool GLDS = false;
  static DEV_INLINE int rslot(int slot, int) { return slot; }
  const uint16_t *dy;
  int HO, WO, Q, W2, H2, K;
  long M;
  // Tap shifts are AFFINE in the tap index (taps step by 2): dh_i = dh0-i,
  // dw_j = dw0-j. (A runtime-indexed table here allocated the whole stager
  // in scratch memory — 384 B/lane, every inner load a scratch round-trip.)
  int dh0, dw0, nth, ntw;
  ushort8 r[4];
  long nbase_[4];
  int hb_[4], wb_[4];
  bool ok_[4];
  DEV_INLINE void init(int tid, int base) {
    int s_row = tid >> 3;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      long m = base + s_row + 32 * i;
      ok_[i] = m < M;
      int w_ = (int)(m % W2);
      long t = m / W2;
      int h_ = (int)(t % H2);
      int n = (int)(t / H2);
      nbase_[i] = (long)n * HO * WO;
      hb_[i] = h_;
      wb_[i] = w_;
    }
  }
  DEV_INLINE void load(int tid, int, int kb, ushort8 *) {
    int k = kb + (tid & 7) * 8;
    int q = k % Q, ti = k / Q;
    int ih = ti / ntw, iw = ti - ih * ntw;
    int dho = dh0 - ih, dwo = dw0 - iw;
    bool kok = k < K;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int ho = hb_[i] + dho, wo = wb_[i] + dwo;
      bool ok = kok && ok_[i] && (unsigned)ho < (unsigned)HO &&
                (unsigned)wo < (unsigned)WO;
      r[i] = ok ? *(const ushort8 *)(dy + (nbase_[i] + (long)ho * WO + wo) * Q + q)
                : ushort8{0, 0, 0, 0, 0, 0, 0, 0};
    }
  }
  DEV_INLINE void write(int tid, ushort8 *img) const {
    int s_row = tid >> 3, s_slot = tid & 7;
#pragma unroll
    for (int i = 0; i < 4; ++i) img[(s_row + 32 * i) * MXP + s_slot] = r[i];
  }
};

// TN weight view over the parity's taps: element (c, k=(ti,q)) =
// w[q][(r(ti)*S + s(ti))*C + c]
struct DgradWS2Tn {
  const uint16_t *w;
  int C, Q, K, RSC, S;
  int r0, s0, nth, ntw; // parity taps: r = r0 + 2·ih, s = s0 + 2·iw
  DEV_INLINE ushort8 load(int k, int c0) const {
    if (k >= K || c0 >= C) return ushort8{0, 0, 0, 0, 0, 0, 0, 0};
    int q = k % Q, ti = k / Q;
    int ih = ti / ntw, iw = ti - ih * ntw;
    return *(const ushort8 *)(w + (long)q * RSC +
                              ((r0 + 2 * ih) * S + s0 + 2 * iw) * C + c0);
  }
};

static hipError_t conv_dgrad_s2(const void *dy, const void *w, void *dx,
                                int N, int H, int W, int C, int Kout, int R,
                                int S, int pad, int HO, int WO,
                                hipStream_t strm) {
  int n_live = 0;
  for (int ph = 0; ph < 2; ++ph)
    for (int pw = 0; pw < 2; ++pw) {
      int nth = 0, ntw = 0;
      for (int r = (ph + pad) & 1; r < R; r += 2) nth++;
      for (int s = (pw + pad) & 1; s < S; s += 2) ntw++;
      if (nth > 0 && ntw > 0) n_live++;
    }
  // n_live==1 (1x1 s2): the single parity's epilogue zero-fills its three
  // sibling pixels (Stride2ZeroWriter) — no full-tensor memset pass.
  if (n_live != 1 && n_live != 4) // exotic shapes: zero the dead parities
    (void)hipMemsetAsync(dx, 0, (long)N * H * W * C * 2, strm);

  for (int ph = 0; ph < 2; ++ph) {
    int H2 = (H - ph + 1) / 2;
    for (int pw = 0; pw < 2; ++pw) {
      int W2 = (W - pw + 1) / 2;
      DgradS2Stage sa{};
      DgradWS2Tn lb{};
      int r_first = (ph + pad) & 1, s_first = (pw + pad) & 1;
      sa.nth = (R - r_first + 1) / 2;
      sa.ntw = (S - s_first + 1) / 2;
      sa.dh0 = (ph + pad - r_first) / 2;
      sa.dw0 = (pw + pad - s_first) / 2;
      lb.r0 = r_first;
      lb.s0 = s_first;
      if (sa.nth == 0 || sa.ntw == 0 || H2 == 0 || W2 == 0) continue;
      long M = (long)N * H2 * W2;
      int K = sa.nth * sa.ntw * Kout;
      sa.dy = (const uint16_t *)dy;
      sa.HO = HO; sa.WO = WO; sa.Q = Kout; sa.W2 = W2; sa.H2 = H2;
      sa.K = K; sa.M = M;
      lb.w = (const uint16_t *)w;
      lb.C = C; lb.Q = Kout; lb.K = K; lb.RSC = R * S * C; lb.S = S;
      lb.nth = sa.nth; lb.ntw = sa.ntw;
      static const bool pipes2 = [] { // MPIAMD_PIPES2=0: A/B lever
        const char *env = getenv("MPIAMD_PIPES2");
        return !(env && env[0] == '0');
      }();
      hipError_t e;
      if (use_pipegather() && pipes2) {
        // pipe route (8-wave under MPIAMD_PIPE8): same parity GEMM with
        // the glds/tr staging; Stride2*Writer epilogues are pipe-generic
        NtPipe<DgradS2Src> sa2{{(const uint16_t *)dy, HO, WO, Kout, W2, H2,
                                K, M, sa.dh0, sa.dw0, sa.nth, sa.ntw}};
        TnPipe<DgradWS2TnSrc> sb2{{(const uint16_t *)w, C, Kout, K,
                                   R * S * C, S, r_first, s_first, sa.nth,
                                   sa.ntw}};
        if (n_live == 1) {
          Stride2ZeroWriter wrt{W2, H2, ph, pw, W, H, C};
          e = launch_pipe_mix_wr(sa2, sb2, dx, (int)M, C, K, wrt, C, false,
                                 strm);
        } else {
          Stride2Writer wrt{W2, H2, ph, pw, W, H, C};
          e = launch_pipe_mix_wr(sa2, sb2, dx, (int)M, C, K, wrt, C, false,
                                 strm);
        }
      } else if (n_live == 1) {
        Stride2ZeroWriter wrt{W2, H2, ph, pw, W, H, C};
        e = launch_mix_gemm_wr(sa, TnStage<DgradWS2Tn>{lb}, dx, (int)M, C, K,
                               wrt, C, false, strm);
      } else {
        Stride2Writer wrt{W2, H2, ph, pw, W, H, C};
        e = launch_mix_gemm_wr(sa, TnStage<DgradWS2Tn>{lb}, dx, (int)M, C, K,
                               wrt, C, false, strm);
      }
      if (e != hipSuccess) return e;
    }
  }
  return hipSuccess;
}

// dgrad: w used directly (channels_last [Kout][RSC]); the k-strided B view
// w[q][rs·C + c0..7] is transposed in the LDS write pass (TN staging).
extern "C" hipError_t conv_dgrad(const void *dy, const void *w, void *dx,
                                 int N, int H, int W, int C, int Kout, int R,
                                 int S, int stride, int pad, int HO, int WO,
                                 int splits, float *partial,
                                 hipStream_t strm) {
  long M = (long)N * H * W;
  int K = R * S * Kout;
  if (stride != 1)
    return conv_dgrad_s2(dy, w, dx, N, H, W, C, Kout, R, S, pad, HO, WO, strm);
  int nk = (K + BK - 1) / BK;
  if (splits > nk) splits = nk > 0 ? nk : 1;
  void *out = splits > 1 ? (void *)partial : dx;
  bool f32 = splits > 1;
  hipError_t e;
  if (R == 1 && S == 1 && pad == 0) {
    // 1x1 dgrad: dx[M][C] = dy[M][Q] · w[Q][C] (w TN via tr_b16 pipeline)
    if (use_pipemix()) {
      NtPipe<PlainNtSrc> sa{{(const uint16_t *)dy, (long)Kout, (int)M, Kout}};
      TnPipe<PlainTnSrc> sb{{(const uint16_t *)w, (long)C, Kout, C}};
      e = launch_pipe_mix_wr(sa, sb, out, (int)M, C, Kout, LinearWriter{(long)C},
                             C, f32, strm, splits);
    } else {
      GemmLoader la{(const uint16_t *)dy, (int)M, (long)Kout, Kout};
      TnRowMajor lb{(const uint16_t *)w, (long)C, Kout, C};
      e = launch_mix_gemm(NtStage<GemmLoader>{la}, TnStage<TnRowMajor>{lb}, out,
                          (int)M, C, Kout, C, f32, strm, splits);
    }
  } else if (use_pipegather()) {
    NtPipe<ConvDgradSrc<1>> sa{
        {(const uint16_t *)dy, H, W, Kout, HO, WO, S, pad, K, M}};
    TnPipe<DgradWTnSrc> sb{{(const uint16_t *)w, C, Kout, K, R * S * C}};
    e = launch_pipe_mix_wr(sa, sb, out, (int)M, C, K, LinearWriter{(long)C},
                           C, f32, strm, splits);
  } else {
    // TN B: k=(r,s,q) with q fastest; element (c, k) = w[q][(r*S+s)*C + c]
    DgradWTn lb{(const uint16_t *)w, C, Kout, K, R * S * C};
    ConvDgradStage<1> sa{(const uint16_t *)dy, H, W, Kout, HO, WO, S, pad, K, M};
    e = launch_mix_gemm(sa, TnStage<DgradWTn>{lb}, out, (int)M, C, K, C, f32,
                        strm, splits);
  }
  if (e != hipSuccess || splits <= 1) return e;
  return splitk_reduce(partial, splits, M * C, dx, 1, strm);
}

// float4 lanes; SPLITS is a template parameter for the common counts so
// the accumulation loop FULLY UNROLLS — the runtime-splits version issued
// its slab loads as a 2-chain serial loop and sat 73% wave-parked (PMC),
// 4.7x off bandwidth at the BERT dw shapes.
template <bool OUT_BF16, int SPLITS = 0>
__global__ void splitk_reduce_k(const float4v *__restrict__ partial,
                                int splits_rt, long len4,
                                void *__restrict__ out) {
  const int splits = SPLITS ? SPLITS : splits_rt;
  for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < len4;
       i += (long)gridDim.x * blockDim.x) {
    float4v a = partial[i];
    float4v b = {0.f, 0.f, 0.f, 0.f};
    int s = 1;
    if (SPLITS == 0) {
      // runtime split count (wgrad runs 24..256): fixed 8-deep inner
      // chunks keep 8 independent loads in flight — the plain runtime
      // loop issued a 2-chain serial walk and sat 73% wave-parked
      for (; s + 8 <= splits; s += 8) {
        float4v c0 = {0.f, 0.f, 0.f, 0.f};
        float4v c1 = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int j = 0; j < 8; j += 2) {
          c0 += partial[(long)(s + j) * len4 + i];
          c1 += partial[(long)(s + j + 1) * len4 + i];
        }
        a += c0;
        b += c1;
      }
    }
#pragma unroll
    for (; s + 1 < splits; s += 2) {
      a += partial[(long)s * len4 + i];
      b += partial[(long)(s + 1) * len4 + i];
    }
    if (s < splits) b += partial[(long)s * len4 + i];
    a += b;
    if (OUT_BF16) {
      uint16_t *o = (uint16_t *)out + i * 4;
#pragma unroll
      for (int j = 0; j < 4; ++j) o[j] = f2bf(a[j]);
    } else {
      ((float4v *)out)[i] = a;
    }
  }
}

// Split-K reduce + bias epilogue for FORWARD activations (bf16 out): the
// fc2-class shapes (M=4096, N=1024) give 256 pipe-mix workgroups = 1 per
// CU; split-K x2 fills the second block slot, and the bias lands here
// instead of in the (now fp32-slab) GEMM epilogue. col = flat % N.
template <int SPLITS>
__global__ void splitk_bias_reduce_k(const float4v *__restrict__ partial,
                                     int splits_rt, long len4, int N4,
                                     const float4v *__restrict__ bias,
                                     uint16_t *__restrict__ out) {
  const int splits = SPLITS ? SPLITS : splits_rt;
  for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < len4;
       i += (long)gridDim.x * blockDim.x) {
    float4v a = partial[i];
    float4v b = {0.f, 0.f, 0.f, 0.f};
    int s = 1;
#pragma unroll
    for (; s + 1 < splits; s += 2) {
      a += partial[(long)s * len4 + i];
      b += partial[(long)(s + 1) * len4 + i];
    }
    if (s < splits) b += partial[(long)s * len4 + i];
    a += b + bias[(int)(i % N4)];
    uint16_t *o = out + i * 4;
#pragma unroll
    for (int j = 0; j < 4; ++j) o[j] = f2bf(a[j]);
  }
}

extern "C" hipError_t splitk_bias_reduce(const float *partial, int splits,
                                         long len, int N, const float *bias,
                                         void *out, hipStream_t s) {
  long len4 = len / 4;
  int grid = (int)((len4 + 255) / 256);
  if (grid > 4096) grid = 4096;
  switch (splits) {
  case 2:
    splitk_bias_reduce_k<2><<<grid, 256, 0, s>>>(
        (const float4v *)partial, splits, len4, N / 4, (const float4v *)bias,
        (uint16_t *)out);
    break;
  case 4:
    splitk_bias_reduce_k<4><<<grid, 256, 0, s>>>(
        (const float4v *)partial, splits, len4, N / 4, (const float4v *)bias,
        (uint16_t *)out);
    break;
  default:
    splitk_bias_reduce_k<0><<<grid, 256, 0, s>>>(
        (const float4v *)partial, splits, len4, N / 4, (const float4v *)bias,
        (uint16_t *)out);
  }
  HIP_KERNEL_CHECK();
  return hipSuccess;
}

// Per-column slab reduce for SHORT outputs with MANY chunks (colsum db:
// len = N ≤ 4k, chunks up to 256 — the float4 kernel collapsed to one
// block there): one 256-thread block per column, threads stride chunks.
__global__ void slab_colreduce_k(const float *__restrict__ partial,
                                 float *__restrict__ out, int chunks,
                                 long N) {
  long c = blockIdx.x;
  if (c >= N) return;
  float a = 0.f;
  for (int g = threadIdx.x; g < chunks; g += 256)
    a += partial[(long)g * N + c];
  __shared__ float red[256 / WAVE];
  a = wave_sum(a);
  if ((threadIdx.x & (WAVE - 1)) == 0) red[threadIdx.x / WAVE] = a;
  __syncthreads();
  if (threadIdx.x == 0) out[c] = red[0] + red[1] + red[2] + red[3];
}

extern "C" hipError_t slab_colreduce(const float *partial, float *out,
                                     int chunks, long N, hipStream_t s) {
  slab_colreduce_k<<<(int)N, 256, 0, s>>>(partial, out, chunks, N);
  HIP_KERNEL_CHECK();
  return hipSuccess;
}

extern "C" hipError_t splitk_reduce(const float *partial, int splits, long len,
                                    void *out, int out_bf16, hipStream_t s) {
  long len4 = len / 4; // len = Kout*RSC, both %8 ⇒ %4
  long blocks = (len4 + 255) / 256;
  if (blocks > 2048) blocks = 2048;
  const float4v *p4 = (const float4v *)partial;
  int nb = (int)blocks;
#define SKR_DISPATCH(BF)                                                        do {                                                                            switch (splits) {                                                             case 2: splitk_reduce_k<BF, 2><<<nb, 256, 0, s>>>(p4, splits, len4, out); break;     case 3: splitk_reduce_k<BF, 3><<<nb, 256, 0, s>>>(p4, splits, len4, out); break;     case 4: splitk_reduce_k<BF, 4><<<nb, 256, 0, s>>>(p4, splits, len4, out); break;     case 5: splitk_reduce_k<BF, 5><<<nb, 256, 0, s>>>(p4, splits, len4, out); break;     case 6: splitk_reduce_k<BF, 6><<<nb, 256, 0, s>>>(p4, splits, len4, out); break;     case 8: splitk_reduce_k<BF, 8><<<nb, 256, 0, s>>>(p4, splits, len4, out); break;     case 16: splitk_reduce_k<BF, 16><<<nb, 256, 0, s>>>(p4, splits, len4, out); break;     default: splitk_reduce_k<BF, 0><<<nb, 256, 0, s>>>(p4, splits, len4, out); break;     }                                                                           } while (0)
  if (out_bf16)
    SKR_DISPATCH(true);
  else
    SKR_DISPATCH(false);
#undef SKR_DISPATCH
  return hipGetLastError();
}

// Implicit wgrad: dw[q][rsc] = Σ_m dy[m][q]·xcol[m][rsc], both operands
// k-strided (k = output pixel m) and TN-staged straight from dy / x —
// no transposes, no im2col buffer. fp32 partial slabs → splitk_reduce.
extern "C" hipError_t conv_wgrad_implicit(const void *dy, const void *x,
                                          float *partial, void *dw, int N,
                                          int H, int W, int C, int Kout, int R,
                                          int S, int stride, int pad, int HO,
                                          int WO, int splits, int dw_bf16,
                                          hipStream_t strm) {
  long M = (long)N * HO * WO;
  int RSC = R * S * C;
  int nk = (int)((M + BK - 1) / BK);
  if (splits > nk) splits = nk > 0 ? nk : 1; // must match the launch's clamp:
  // the reduce below must sum exactly the slabs the GEMM wrote.
  TnRowMajor la{(const uint16_t *)dy, Kout, (int)M, Kout};
  hipError_t e;
  if (R == 1 && S == 1 && stride == 1 && pad == 0) {
    // 1x1 wgrad: the im2col column of x IS x itself — plain TN view
    if (use_pipemix()) {
      TnPipe<PlainTnSrc> sa{{(const uint16_t *)dy, (long)Kout, (int)M, Kout}};
      TnPipe<PlainTnSrc> sb{{(const uint16_t *)x, (long)C, (int)M, C}};
      e = launch_pipe_mix_wr(sa, sb, partial, Kout, RSC, (int)M,
                             LinearWriter{(long)RSC}, RSC, true, strm, splits);
    } else {
      TnRowMajor lb{(const uint16_t *)x, (long)C, (int)M, C};
      e = launch_mix_gemm(TnStage<TnRowMajor>{la}, TnStage<TnRowMajor>{lb},
                          partial, Kout, RSC, (int)M, RSC, true, strm, splits);
    }
  } else if (use_pipegather_wgrad()) {
    TnPipe<PlainTnSrc> sa{{(const uint16_t *)dy, (long)Kout, (int)M, Kout}};
    TnPipe<XcolSrc> sb{
        {(const uint16_t *)x, H, W, C, HO, WO, S, stride, pad, RSC, M}};
    e = launch_pipe_mix_wr(sa, sb, partial, Kout, RSC, (int)M,
                           LinearWriter{(long)RSC}, RSC, true, strm, splits);
  } else {
    XcolStage sb{(const uint16_t *)x, H, W, C, HO, WO, S, stride, pad, RSC, M};
    e = launch_mix_gemm(TnStage<TnRowMajor>{la}, sb, partial, Kout, RSC,
                        (int)M, RSC, true, strm, splits);
  }
  if (e != hipSuccess) return e;
  return splitk_reduce(partial, splits, (long)Kout * RSC, dw, dw_bf16, strm);
}
