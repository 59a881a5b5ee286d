// GEMM entry points over the MFMA tile templates: NT (both operands
// k-contiguous), NT×TN and TN×TN (k-strided operands transposed in the LDS
// write pass — linear backward runs with zero materialized transposes).
#include "mfma_tile.h"
#include "mix_gemm.h"

extern "C" hipError_t splitk_reduce(const float *, int, long, void *, int,
                                    hipStream_t); // conv.hip
extern "C" hipError_t splitk_bias_reduce(const float *, int, long, int,
                                         const float *, void *,
                                         hipStream_t); // conv.hip

// A TN operand's 16-B column granules may overread up to 7 elements past
// its `dim` columns; safe when the dim is %8, or when the ROW STRIDE is %8
// and covers the rounded dim (the padded-allocation contract linear_bwd
// uses for the ragged vocab head).
static inline bool tn_cols_ok(int dim, long ld) {
  return dim % 8 == 0 || (ld % 8 == 0 && ld >= ((dim + 7) & ~7));
}

extern "C" hipError_t gemm_nt(const void *a, const void *b, void *c, int M,
                              int N, int K, long lda, long ldb, long ldc,
                              int c_f32, hipStream_t s) {
  GemmLoader la{(const uint16_t *)a, M, lda, K};
  GemmLoader lb{(const uint16_t *)b, N, ldb, K};
  return launch_nt_gemm(la, lb, c, M, N, K, ldc, c_f32 != 0, s);
}

// linear forward: bias (fp32 [N]) folded into the GEMM epilogue — the
// separate bias_add pass was a full activation read+write.
extern "C" hipError_t gemm_nt_bias(const void *a, const void *b,
                                   const float *bias, void *c, int M, int N,
                                   int K, long lda, long ldb, long ldc,
                                   hipStream_t s) {
  GemmLoader la{(const uint16_t *)a, M, lda, K};
  GemmLoader lb{(const uint16_t *)b, N, ldb, K};
  return launch_nt_gemm(la, lb, c, M, N, K, ldc, false, s, 1, bias);
}

// Forward split-K (fc2-class M=4096/N=1024 shapes give 256 pipe-mix
// workgroups = exactly 1/CU, half the block slots idle; deep K amortizes
// the fp32 slab + bias-reduce pass). Measured same-box +0.7% on BERT-Large
// bs32 (1372 -> 1381): default ON, MPIAMD_FWD_SK=0 disables. Heuristic
// shared with the dw path.
extern "C" int gemm_tn_tn_splits(int M, int N, int K); // below
extern "C" int gemm_fwd_splits(int M, int N, int K) {
  static const bool on = [] {
    const char *e = getenv("MPIAMD_FWD_SK");
    return !(e && e[0] == '0');
  }();
  if (!on || N % 8) return 1;
  int sp = gemm_tn_tn_splits(M, N, K);
  return sp > 4 ? 4 : sp;
}

extern "C" hipError_t gemm_nt_bias_sk(const void *a, const void *b,
                                      const float *bias, float *partial,
                                      void *c, int M, int N, int K, long lda,
                                      long ldb, long ldc, int splits,
                                      hipStream_t s) {
  if (splits <= 1 || !use_pipemix() || ldc != N)
    return gemm_nt_bias(a, b, bias, c, M, N, K, lda, ldb, ldc, s);
  NtPipe<PlainNtSrc> sa{{(const uint16_t *)a, lda, M, K}};
  NtPipe<PlainNtSrc> sb{{(const uint16_t *)b, ldb, N, K}};
  hipError_t e = launch_pipe_mix_wr(sa, sb, partial, M, N, K,
                                    LinearWriter{ldc}, ldc, true, s, splits);
  if (e != hipSuccess) return e;
  return splitk_bias_reduce(partial, splits, (long)M * N, N, bias, c, s);
}

// FFN fc1 fused forward: g = gelu(x·w1ᵀ + b); the epilogue also saves
// gelu'(h) (bf16, layout of C) so the backward epilogue is exp-free
// (GeluBiasWriter). Both operands NT k-contiguous.
// MPIAMD_GELU_DERIV (default 1): fwd saves gelu'(h) so bwd is exp-free;
// =0 restores the save-pre/recompute-tanh pair for A/B.
static bool gelu_deriv_mode() {
  static const bool on = [] {
    const char *e = getenv("MPIAMD_GELU_DERIV");
    return !(e && e[0] == '0');
  }();
  return on;
}

template <class WR>
static hipError_t gelu_fwd_route(WR wrt, const void *x, const void *w, void *g,
                                 int M, int N, int K, long lda, long ldb,
                                 long ldc, hipStream_t s) {
  GemmLoader la{(const uint16_t *)x, M, lda, K};
  GemmLoader lb{(const uint16_t *)w, N, ldb, K};
  if (M % 256 == 0 && N % 256 == 0 && K % 64 == 0 && ldc == N &&
      (long)(M / 256) * (N / 256) >= 128)
    return launch_pipe256_wr(la, lb, g, M, N, K, ldc, false, wrt, s);
  if (use_pipemix()) {
    NtPipe<PlainNtSrc> sa{{(const uint16_t *)x, lda, M, K}};
    NtPipe<PlainNtSrc> sb{{(const uint16_t *)w, ldb, N, K}};
    return launch_pipe_mix_wr(sa, sb, g, M, N, K, wrt, ldc, false, s);
  }
  GldsNt ga{la.p, la.rows, la.ld, la.kdim};
  GldsNt gb{lb.p, lb.rows, lb.ld, lb.kdim};
  return launch_mix_gemm_wr(ga, gb, g, M, N, K, wrt, ldc, false, s);
}

extern "C" hipError_t gemm_nt_gelu_bias(const void *x, const void *w,
                                        const float *bias, void *deriv, void *g,
                                        int M, int N, int K, long lda,
                                        long ldb, long ldc, hipStream_t s) {
  if (gelu_deriv_mode())
    return gelu_fwd_route(GeluBiasWriter{ldc, bias, (uint16_t *)deriv}, x, w,
                          g, M, N, K, lda, ldb, ldc, s);
  return gelu_fwd_route(GeluBiasPreWriter{ldc, bias, (uint16_t *)deriv}, x, w,
                        g, M, N, K, lda, ldb, ldc, s);
}

extern "C" hipError_t transpose2d_bf16(const void *, void *, int, int, long,
                                       hipStream_t); // defined below

// FFN backward: dh = (dy·w2) ⊙ gelu'(h) — the fc2-dx GEMM multiplying by
// the SAVED derivative in the epilogue (GeluBwdWriter); A NT, B (w2) TN.
template <class WR>
static hipError_t gelubwd_route(WR wrt, const void *dy, const void *w,
                                void *dh, int M, int N, int K, long lda,
                                long ldb, long ldc, hipStream_t s,
                                void *wt_buf = nullptr) {
  // Full-256-tile shapes (fc2-dx: 4096x4096x1024) run 355 TF on the
  // TN-staged pipe_mix but 550+ TF on the NT pipe256 — worth materializing
  // w2^T once per call (~8 us for 33 MB) to take the NT route. wt_buf is
  // caller-allocated [N][K] bf16.
  if (wt_buf && M % 256 == 0 && N % 256 == 0 && K % 64 == 0 && ldc == N &&
      ldb == K && (long)(M / 256) * (N / 256) >= 128) {
    hipError_t e = transpose2d_bf16(w, wt_buf, K, N, K, s);
    if (e != hipSuccess) return e;
    GemmLoader la{(const uint16_t *)dy, M, lda, K};
    GemmLoader lb{(const uint16_t *)wt_buf, N, K, K};
    return launch_pipe256_wr(la, lb, dh, M, N, K, ldc, false, wrt, s);
  }
  if (use_pipemix() && N % 8 == 0) {
    NtPipe<PlainNtSrc> sa{{(const uint16_t *)dy, lda, M, K}};
    TnPipe<PlainTnSrc> sb{{(const uint16_t *)w, ldb, K, N}};
    return launch_pipe_mix_wr(sa, sb, dh, M, N, K, wrt, ldc, false, s);
  }
  GemmLoader la{(const uint16_t *)dy, M, lda, K};
  TnRowMajor lb{(const uint16_t *)w, ldb, K, N};
  return launch_mix_gemm_wr(NtStage<GemmLoader>{la}, TnStage<TnRowMajor>{lb},
                            dh, M, N, K, wrt, ldc, false, s);
}

// wt_buf: optional [N][K] bf16 scratch enabling the transpose+pipe256
// route. Measured a WASH same-box (1275 vs 1274 avg at bs32) despite the
// per-kernel mix-vs-256 gap in prof10 — pipe256 with the deriv-reading
// epilogue loses its edge. Default OFF (MPIAMD_GELUBWD_WT=1 enables).
extern "C" int gemm_gelubwd_wants_wt(int M, int N, int K) {
  static const bool on = [] {
    const char *e = getenv("MPIAMD_GELUBWD_WT");
    return e && e[0] == '1';
  }();
  return on && M % 256 == 0 && N % 256 == 0 && K % 64 == 0 &&
         (long)(M / 256) * (N / 256) >= 128;
}

extern "C" hipError_t gemm_nt_tn_gelubwd(const void *dy, const void *w,
                                         const void *deriv, void *dh, int M,
                                         int N, int K, long lda, long ldb,
                                         long ldc, void *wt_buf,
                                         hipStream_t s) {
  if (gelu_deriv_mode())
    return gelubwd_route(GeluBwdWriter{ldc, (const uint16_t *)deriv}, dy, w,
                         dh, M, N, K, lda, ldb, ldc, s, wt_buf);
  return gelubwd_route(GeluBwdPreWriter{ldc, (const uint16_t *)deriv}, dy, w,
                       dh, M, N, K, lda, ldb, ldc, s, wt_buf);
}

// dx ACCUMULATE form: C += A·B (LinearAccWriter) — the transformer
// residual-join backward folds the dgrad into the LN-dx buffer instead of
// a separate full-activation add pass (autograd's CUDAFunctor_add join).
extern "C" hipError_t gemm_nt_tn_acc(const void *a, const void *b, void *c,
                                     int M, int N, int K, long lda, long ldb,
                                     long ldc, hipStream_t s) {
  if (use_pipemix() && tn_cols_ok(N, ldb)) {
    NtPipe<PlainNtSrc> sa{{(const uint16_t *)a, lda, M, K}};
    TnPipe<PlainTnSrc> sb{{(const uint16_t *)b, ldb, K, N}};
    return launch_pipe_mix_wr(sa, sb, c, M, N, K, LinearAccWriter{ldc}, ldc,
                              false, s);
  }
  GemmLoader la{(const uint16_t *)a, M, lda, K};
  TnRowMajor lb{(const uint16_t *)b, ldb, K, N};
  return launch_mix_gemm_wr(NtStage<GemmLoader>{la}, TnStage<TnRowMajor>{lb},
                            c, M, N, K, LinearAccWriter{ldc}, ldc, false, s);
}

// C[M][N] = A[M][K-contig] · B(k-strided [K rows][N cols])  — linear dx
extern "C" hipError_t gemm_nt_tn(const void *a, const void *b, void *c, int M,
                                 int N, int K, long lda, long ldb, long ldc,
                                 int c_f32, hipStream_t s) {
  // pipe TN staging loads 16-B column granules: a cols%8!=0 operand would
  // read past the allocation on its last k-row (BERT vocab 30522 class) —
  // those shapes keep the elementwise-tail mix path
  if (use_pipemix() && tn_cols_ok(N, ldb)) {
    NtPipe<PlainNtSrc> sa{{(const uint16_t *)a, lda, M, K}};
    TnPipe<PlainTnSrc> sb{{(const uint16_t *)b, ldb, K, N}};
    return launch_pipe_mix_wr(sa, sb, c, M, N, K, LinearWriter{ldc}, ldc,
                              c_f32 != 0, s);
  }
  GemmLoader la{(const uint16_t *)a, M, lda, K};
  TnRowMajor lb{(const uint16_t *)b, ldb, K, N};
  return launch_mix_gemm(NtStage<GemmLoader>{la}, TnStage<TnRowMajor>{lb}, c,
                         M, N, K, ldc, c_f32 != 0, s);
}

// split-K form (small-batch dx underfills the chip, e.g. BERT bs=8:
// dx[1024][4096] = 64 workgroups with K_reduce up to 4096): fp32 slabs in
// `partial` reduced into bf16 C. Same splits heuristic as the dw path.
extern "C" int gemm_tn_tn_splits(int M, int N, int K); // defined below

// dx split-K, SHAPE-aware after three verdict flips: under the 8-wave
// pipeline a 256-workgroup dx grid fills the chip and the slab+reduce tax
// loses (bs32: 1542 with splits vs 1553 without), but SMALL grids still
// need it badly (bs8's 64-wg dx: 690 with splits vs 614 without; so does
// bert-base's 192-wg grid). Split only when tiles < 256.
// MPIAMD_DX_SK: 0 = never split, 1 = always use the dw heuristic.
extern "C" int gemm_nt_tn_splits(int M, int N, int K) {
  static const int mode = [] {
    const char *e = getenv("MPIAMD_DX_SK");
    return e ? (e[0] == '1' ? 1 : 0) : -1;
  }();
  if (mode == 0) return 1;
  long tiles = ((M + 127) / 128) * ((N + 127) / 128);
  if (mode != 1 && tiles >= 256) return 1;
  return gemm_tn_tn_splits(M, N, K);
}

extern "C" hipError_t gemm_nt_tn_sk(const void *a, const void *b,
                                    float *partial, void *c, int M, int N,
                                    int K, long lda, long ldb, long ldc,
                                    int splits, hipStream_t s) {
  if (use_pipemix() && tn_cols_ok(N, ldb)) {
    NtPipe<PlainNtSrc> sa{{(const uint16_t *)a, lda, M, K}};
    TnPipe<PlainTnSrc> sb{{(const uint16_t *)b, ldb, K, N}};
    if (splits <= 1)
      return launch_pipe_mix_wr(sa, sb, c, M, N, K, LinearWriter{ldc}, ldc,
                                false, s);
    hipError_t e = launch_pipe_mix_wr(sa, sb, partial, M, N, K,
                                      LinearWriter{ldc}, ldc, true, s, splits);
    if (e != hipSuccess) return e;
    return splitk_reduce(partial, splits, (long)M * ldc, c, 1, s);
  }
  GemmLoader la{(const uint16_t *)a, M, lda, K};
  TnRowMajor lb{(const uint16_t *)b, ldb, K, N};
  if (splits <= 1)
    return launch_mix_gemm(NtStage<GemmLoader>{la}, TnStage<TnRowMajor>{lb},
                           c, M, N, K, ldc, false, s);
  hipError_t e =
      launch_mix_gemm(NtStage<GemmLoader>{la}, TnStage<TnRowMajor>{lb},
                      partial, M, N, K, ldc, true, s, splits);
  if (e != hipSuccess) return e;
  return splitk_reduce(partial, splits, (long)M * ldc, c, 1, s);
}

// C[M][N] = Σ_k A(k-strided rows, M cols) · B(k-strided rows, N cols)
// — linear dw (A = dy [Mbatch][N_out] viewed k-strided, B = x likewise)
extern "C" hipError_t gemm_tn_tn(const void *a, const void *b, void *c, int M,
                                 int N, int K, long lda, long ldb, long ldc,
                                 int c_f32, hipStream_t s) {
  if (use_pipemix() && tn_cols_ok(M, lda) && tn_cols_ok(N, ldb)) {
    TnPipe<PlainTnSrc> sa{{(const uint16_t *)a, lda, K, M}};
    TnPipe<PlainTnSrc> sb{{(const uint16_t *)b, ldb, K, N}};
    return launch_pipe_mix_wr(sa, sb, c, M, N, K, LinearWriter{ldc}, ldc,
                              c_f32 != 0, s);
  }
  TnRowMajor la{(const uint16_t *)a, lda, K, M};
  TnRowMajor lb{(const uint16_t *)b, ldb, K, N};
  return launch_mix_gemm(TnStage<TnRowMajor>{la}, TnStage<TnRowMajor>{lb}, c,
                         M, N, K, ldc, c_f32 != 0, s);
}

// split-K form of gemm_tn_tn for weight-gradient shapes whose tile grid
// underfills the chip (BERT attn-out dw = 64 workgroups on 256 CUs with a
// K of batch*seq): fp32 slabs in `partial` [splits][M*ldc] reduced into a
// dense fp32 C. Caller sizes `partial` with `splits` from gemm_tn_tn_splits.
extern "C" int gemm_tn_tn_splits(int M, int N, int K) {
  static const bool off = [] {
    const char *e = getenv("MPIAMD_LINEAR_SK");
    return e && e[0] == '0';
  }();
  long tiles = ((M + 127) / 128) * ((N + 127) / 128);
  int nk = (K + 63) / 64;
  // nk>=32: at bs=8-class shapes (nk=16) the slab+reduce overhead beats
  // the fill gain (same-box 312/290 vs 325 seq/s); the bs=32 attn-out dw
  // (nk=64) is the stable +5% winner this path exists for
  if (off || nk < 32) return 1;
  // fc2-dw-class (tiles 256..511, deep K): 2-way split measured 672 vs
  // 545-617 TF at sk4/sk1 on (1024,4096,4096) — 512 blocks = 2/CU
  if (tiles >= 512) return 1;
  if (tiles >= 256) return 2;
  long s = 512 / tiles;
  if (s > nk / 8) s = nk / 8;
  if (s > 64) s = 64;
  if (s >= 8) s &= ~7; // split-major-capable granularity
  return s < 1 ? 1 : (int)s;
}

extern "C" hipError_t gemm_tn_tn_sk(const void *a, const void *b,
                                    float *partial, void *c, int M, int N,
                                    int K, long lda, long ldb, long ldc,
                                    int splits, int out_bf16, hipStream_t s) {
  // out_bf16: write C in bf16 directly (weight-grad path — the fp32 C +
  // torch .to(bf16) cast afterwards was ~200 cast kernels / 1.3 ms per
  // BERT-Large step, prof8)
  if (use_pipemix() && tn_cols_ok(M, lda) && tn_cols_ok(N, ldb)) {
    TnPipe<PlainTnSrc> sa{{(const uint16_t *)a, lda, K, M}};
    TnPipe<PlainTnSrc> sb{{(const uint16_t *)b, ldb, K, N}};
    if (splits <= 1)
      return launch_pipe_mix_wr(sa, sb, c, M, N, K, LinearWriter{ldc}, ldc,
                                out_bf16 == 0, s);
    hipError_t e = launch_pipe_mix_wr(sa, sb, partial, M, N, K,
                                      LinearWriter{ldc}, ldc, true, s, splits);
    if (e != hipSuccess) return e;
    return splitk_reduce(partial, splits, (long)M * ldc, c, out_bf16, s);
  }
  TnRowMajor la{(const uint16_t *)a, lda, K, M};
  TnRowMajor lb{(const uint16_t *)b, ldb, K, N};
  if (splits <= 1)
    return launch_mix_gemm(TnStage<TnRowMajor>{la}, TnStage<TnRowMajor>{lb},
                           c, M, N, K, ldc, out_bf16 == 0, s);
  hipError_t e = launch_mix_gemm(TnStage<TnRowMajor>{la},
                                 TnStage<TnRowMajor>{lb}, partial, M, N, K,
                                 ldc, true, s, splits);
  if (e != hipSuccess) return e;
  return splitk_reduce(partial, splits, (long)M * ldc, c, out_bf16, s);
}

// bf16 2-D transpose: out[j][i] = in[i][j], output leading dim ldo >= R
// (callers pad ldo to a multiple of 8 — the NT-GEMM's reduce-dim granule —
// and pre-zero the output so pad columns contribute 0). LDS 64x64 tile.
__global__ void transpose2d_k(const uint16_t *__restrict__ in,
                              uint16_t *__restrict__ out, int R, int C,
                              long ldo) {
  __shared__ uint16_t t[64][65];
  int bi = blockIdx.x * 64, bj = blockIdx.y * 64;
  // 256 threads: load 64x64 (each thread 16 elems, 8-wide rows)
  int lr = threadIdx.x / 8, lc = (threadIdx.x % 8) * 8;
  for (int rr = lr; rr < 64; rr += 32) {
    int gi = bi + rr;
    if (gi < R) {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int gj = bj + lc + j;
        t[rr][lc + j] = (gj < C) ? in[(long)gi * C + gj] : 0;
      }
    }
  }
  __syncthreads();
  for (int rr = lr; rr < 64; rr += 32) {
    int go = bj + rr; // output row = input col
    if (go < C) {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int gi = bi + lc + j;
        if (gi < R) out[(long)go * ldo + gi] = t[lc + j][rr];
      }
    }
  }
}

extern "C" hipError_t transpose2d_bf16(const void *in, void *out, int R, int C,
                                       long ldo, hipStream_t s) {
  dim3 grid((R + 63) / 64, (C + 63) / 64);
  transpose2d_k<<<grid, 256, 0, s>>>((const uint16_t *)in, (uint16_t *)out, R,
                                     C, ldo);
  return hipGetLastError();
}

// Single-wave MFMA layout probe: loads A[16][32], B[16][32] (row-major, the
// assumed per-lane fragment mapping) and writes D[16][16] via the assumed
// C-map. Feeding basis/unique-valued operands from the test reveals the true
// hardware mapping if the assumption is wrong.
__global__ void mfma_probe_k(const uint16_t *__restrict__ a,
                             const uint16_t *__restrict__ b,
                             float *__restrict__ d) {
  int lane = threadIdx.x & 63;
  bf16x8 af = us8_to_bf8v(*(const ushort8 *)(a + (lane & 15) * 32 + (lane >> 4) * 8));
  bf16x8 bf_ = us8_to_bf8v(*(const ushort8 *)(b + (lane & 15) * 32 + (lane >> 4) * 8));
  float4v acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf_, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r)
    d[((lane >> 4) * 4 + r) * 16 + (lane & 15)] = acc[r];
}

extern "C" hipError_t mfma_probe(const void *a, const void *b, float *d,
                                 hipStream_t s) {
  mfma_probe_k<<<1, 64, 0, s>>>((const uint16_t *)a, (const uint16_t *)b, d);
  return hipGetLastError();
}
