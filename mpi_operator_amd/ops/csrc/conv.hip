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

struct ConvFwdALoader {
  const uint16_t *x;
  int H, W, C, HO, WO, S, stride, pad;
  long M;
  int K, SC;
  DEV_INLINE ushort8 load(int m, int k) const {
    if (m >= M || k >= K) return ushort8{0, 0, 0, 0, 0, 0, 0, 0};
    int c = k % C, rs = k / C;
    int s_ = rs % S, r = rs / S;
    int wo = m % WO;
    long t = m / WO;
    int ho = t % HO;
    int n = t / HO;
    int h = ho * stride + r - pad, w = wo * stride + s_ - pad;
    if ((unsigned)h >= (unsigned)H || (unsigned)w >= (unsigned)W)
      return ushort8{0, 0, 0, 0, 0, 0, 0, 0};
    return *(const ushort8 *)(x + ((long)(n * H + h) * W + w) * C + c);
  }
};

struct ConvDgradALoader {
  const uint16_t *dy;
  int H, W, Q /*Kout*/, HO, WO, S, stride, pad;
  long M;
  int K;
  DEV_INLINE ushort8 load(int m, int k) const {
    if (m >= M || k >= K) return ushort8{0, 0, 0, 0, 0, 0, 0, 0};
    int q = k % Q, rs = k / Q;
    int s_ = rs % S, r = rs / S;
    int w_ = m % W;
    long t = m / W;
    int h_ = t % H;
    int n = t / H;
    int hn = h_ + pad - r, wn = w_ + pad - s_;
    if (hn < 0 || wn < 0 || hn % stride || wn % stride)
      return ushort8{0, 0, 0, 0, 0, 0, 0, 0};
    int ho = hn / stride, wo = wn / stride;
    if (ho >= HO || wo >= WO) return ushort8{0, 0, 0, 0, 0, 0, 0, 0};
    return *(const ushort8 *)(dy + ((long)(n * HO + ho) * WO + wo) * Q + q);
  }
};

extern "C" hipError_t conv_fwd(const void *x, const void *w, void *y, int N,
                               int H, int W, int C, int Kout, int R, int S,
                               int stride, int pad, int HO, int WO,
                               hipStream_t strm) {
  long M = (long)N * HO * WO;
  int K = R * S * C;
  if (R == 1 && S == 1 && stride == 1 && pad == 0) {
    // 1x1 conv IS a GEMM on the NHWC image viewed [M][C] — skip the
    // gather arithmetic entirely (≈47% of ResNet bottleneck FLOPs)
    GemmLoader la{(const uint16_t *)x, (int)M, (long)C, C};
    GemmLoader lb{(const uint16_t *)w, Kout, (long)C, C};
    return launch_nt_gemm(la, lb, y, (int)M, Kout, C, Kout, false, strm);
  }
  ConvFwdALoader la{(const uint16_t *)x, H, W, C, HO, WO, S, stride, pad, M, K, S * C};
  GemmLoader lb{(const uint16_t *)w, Kout, (long)K, K};
  return launch_nt_gemm(la, lb, y, (int)M, Kout, K, Kout, false, strm);
}

// dgrad: w used directly (channels_last [Kout][RSC]); the k-strided B view
// w[q][rs·C + c0..7] is transposed in the LDS write pass (TN staging).
extern "C" hipError_t conv_dgrad(const void *dy, const void *w, void *dx,
                                 int N, int H, int W, int C, int Kout, int R,
                                 int S, int stride, int pad, int HO, int WO,
                                 hipStream_t strm) {
  long M = (long)N * H * W;
  int K = R * S * Kout;
  if (R == 1 && S == 1 && stride == 1 && pad == 0) {
    // 1x1 dgrad: dx[M][C] = dy[M][Q] · w[Q][C] (w TN-staged, no gather)
    GemmLoader la{(const uint16_t *)dy, (int)M, (long)Kout, Kout};
    TnRowMajor lb{(const uint16_t *)w, (long)C, Kout, C};
    return launch_mix_gemm(NtStage<GemmLoader>{la}, TnStage<TnRowMajor>{lb},
                           dx, (int)M, C, Kout, C, false, strm);
  }
  ConvDgradALoader la{(const uint16_t *)dy, H, W, Kout, HO, WO, S, stride, pad, M, K};
  // TN B: k=(r,s,q) with q fastest; element (c, k) = w[q][(r*S+s)*C + c]
  struct DgradWTn {
    const uint16_t *w;
    int C, Q, K, RSC;
    DEV_INLINE ushort8 load(int k, int c0) const {
      if (k >= K || c0 >= C) return ushort8{0, 0, 0, 0, 0, 0, 0, 0};
      int q = k % Q, rs = k / Q;
      return *(const ushort8 *)(w + (long)q * RSC + rs * C + c0);
    }
  } lb{(const uint16_t *)w, C, Kout, K, R * S * C};
  return launch_mix_gemm(NtStage<ConvDgradALoader>{la}, TnStage<DgradWTn>{lb},
                         dx, (int)M, C, K, C, false, strm);
}

template <bool OUT_BF16>
__global__ void splitk_reduce_k(const float *__restrict__ partial, int splits,
                                long len, void *__restrict__ out) {
  for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < len;
       i += (long)gridDim.x * blockDim.x) {
    float a = 0;
    for (int s = 0; s < splits; ++s) a += partial[(long)s * len + i];
    if (OUT_BF16)
      ((uint16_t *)out)[i] = f2bf(a);
    else
      ((float *)out)[i] = a;
  }
}

extern "C" hipError_t splitk_reduce(const float *partial, int splits, long len,
                                    void *out, int out_bf16, hipStream_t s) {
  long blocks = (len + 255) / 256;
  if (blocks > 2048) blocks = 2048;
  if (out_bf16)
    splitk_reduce_k<true><<<(int)blocks, 256, 0, s>>>(partial, splits, len, out);
  else
    splitk_reduce_k<false><<<(int)blocks, 256, 0, s>>>(partial, splits, len, out);
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
    TnRowMajor lb{(const uint16_t *)x, (long)C, (int)M, C};
    e = launch_mix_gemm(TnStage<TnRowMajor>{la}, TnStage<TnRowMajor>{lb},
                        partial, Kout, RSC, (int)M, RSC, true, strm, splits);
  } else {
    TnXcol lb{(const uint16_t *)x, H, W, C, HO, WO, S, stride, pad, M, RSC};
    e = launch_mix_gemm(TnStage<TnRowMajor>{la}, TnStage<TnXcol>{lb},
                        partial, Kout, RSC, (int)M, RSC, true, strm, splits);
  }
  if (e != hipSuccess) return e;
  return splitk_reduce(partial, splits, (long)Kout * RSC, dw, dw_bf16, strm);
}
