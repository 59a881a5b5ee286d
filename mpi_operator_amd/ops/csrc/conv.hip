// Implicit-GEMM convolution on MFMA (gfx950) for NHWC bf16 activations and
// channels-last ([K][R][S][C]) bf16 weights — the reference workload's
// cuDNN conv replaced by hand-written CDNA4 kernels.
//
//   fwd:   y[m=(n,ho,wo)][kout]  = sum_{k=(r,s,c)}   x[patch(m,k)] * w[kout][k]
//   dgrad: dx[m=(n,h,w)][c]      = sum_{k=(r,s,q)}  dy[opatch(m,k)] * wT[k][c]
//   wgrad: dw[kout][(r,s,c)]     = sum_{m}          dyT[kout][m] * PT[(r,s,c)][m]
//
// fwd/dgrad instantiate the NT-GEMM tile template with gather loaders
// (padding/stride validity → zero-fill, so edges and stride-2 dgrad need no
// special cases); wgrad runs split-K NT-GEMM over transposed dy and a
// transposed im2col buffer, accumulating exactly in fp32.
#include "mfma_tile.h"

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
  ConvFwdALoader la{(const uint16_t *)x, H, W, C, HO, WO, S, stride, pad, M, K, S * C};
  GemmLoader lb{(const uint16_t *)w, Kout, (long)K, K};
  return launch_nt_gemm(la, lb, y, (int)M, Kout, K, Kout, false, strm);
}

// dgrad: wT is the [RSC][Kout] transpose of w (transpose2d_bf16 of w[K][RSC])
extern "C" hipError_t conv_dgrad(const void *dy, const void *wT, void *dx,
                                 int N, int H, int W, int C, int Kout, int R,
                                 int S, int stride, int pad, int HO, int WO,
                                 hipStream_t strm) {
  long M = (long)N * H * W;
  int K = R * S * Kout;
  ConvDgradALoader la{(const uint16_t *)dy, H, W, Kout, HO, WO, S, stride, pad, M, K};
  // B row = c; k=(r,s,q): wT[(r*S+s)*C + c][q] → addr = ((r*S+s)*C + c)*Q + q
  struct DgradBLoader {
    const uint16_t *wT;
    int C, Q, K;
    DEV_INLINE ushort8 load(int c, int k) const {
      if (c >= C || k >= K) return ushort8{0, 0, 0, 0, 0, 0, 0, 0};
      int q = k % Q, rs = k / Q;
      return *(const ushort8 *)(wT + ((long)rs * C + c) * Q + q);
    }
  } lb{(const uint16_t *)wT, C, Kout, K};
  return launch_nt_gemm(la, lb, dx, (int)M, C, K, C, false, strm);
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

// wgrad GEMM over transposed operands (dyT [Kout][M], PT [RSC][M8*8]):
// partial fp32 slabs [splits][Kout][RSC] → splitk_reduce → dw fp32.
extern "C" hipError_t conv_wgrad_gemm(const void *dyT, const void *PT,
                                      float *partial, void *dw, int Kout,
                                      int RSC, long M, long ldPT, int splits,
                                      int dw_bf16, hipStream_t strm) {
  int nk = (int)((M + BK - 1) / BK);
  if (splits > nk) splits = nk > 0 ? nk : 1; // must match the launch's clamp:
  // the reduce below must sum exactly the slabs the GEMM wrote.
  GemmLoader la{(const uint16_t *)dyT, Kout, M, (int)M};
  GemmLoader lb{(const uint16_t *)PT, RSC, ldPT, (int)M};
  hipError_t e = launch_nt_gemm(la, lb, partial, Kout, RSC, (int)M, RSC, true,
                                strm, splits);
  if (e != hipSuccess) return e;
  return splitk_reduce(partial, splits, (long)Kout * RSC, dw, dw_bf16, strm);
}
