// Mixed-staging MFMA GEMM for gfx950: C[M][N] = sum_k A(m,k)·B(n,k) where
// EITHER operand may be stored k-contiguous (NT: rows along the output dim,
// 16 B loads along k) or k-strided (TN: memory rows ARE k, 16 B loads along
// the output dim, transposed during the LDS write pass).
//
// This removes every materialized transpose / im2col in the framework:
//   conv wgrad   : dw[q][rsc] = Σ_m dy[m][q]·xcol[m][rsc]   (TN × TN-gather)
//   conv dgrad   : dx[m][c]   = Σ_k dy-gather · w[q][rs·C+c] (NT-gather × TN)
//   linear dx    : dx[m][k]   = Σ_n dy[m][n]·w[n][k]         (NT × TN)
//   linear dw    : dw[n][k]   = Σ_m dy[m][n]·x[m][k]         (TN × TN)
// (profiled before: im2col_t_k 15% + transpose2d_k 7.4% of step time —
//  both are pure data movement this file deletes.)
//
// LDS image: [128 rows][9 slots of 16 B] (pitch 144 B). The 9-slot pitch
// makes ds_read_b128 conflict-free WITHOUT an XOR swizzle: a 16-lane read
// group covers rows r..r+15 at one slot, and banks (36·row + 4·slot) mod 64
// are distinct because 36·row mod 64 has period 16. The TN write pass packs
// two k's per lane into one ds_write_b32 (its ≤2-way write conflicts are
// free: LDS-array cycles stay under the instruction's own 4).
//
// Schedule: register staging, issue-early / write-late (load tile t+1's
// global data during tile t's MFMAs, write it to LDS after them) — the
// measured-best register-staging form for this 2-barrier structure.
#pragma once
#include "mfma_tile.h"

constexpr int MXP = 9; // slots per LDS image row (pitch 144 B)

// ---- staging policies ----
// NT: loader.load(row, k0) -> 8 bf16 along k (row-major k-contiguous).
// TN: loader.load(k, col0) -> 8 bf16 along the output dim (k-strided).

template <class L> struct NtStage {
  static constexpr int PITCH = MXP;
  static constexpr bool GLDS = false;
  static DEV_INLINE int rslot(int slot, int) { return slot; }
  L l;
  ushort8 r[4];
  DEV_INLINE void init(int, int) {}
  DEV_INLINE void load(int tid, int base, int kb, ushort8 *) {
    int s_row = tid >> 3, s_slot = tid & 7;
#pragma unroll
    for (int i = 0; i < 4; ++i)
      r[i] = l.load(base + s_row + 32 * i, kb + s_slot * 8);
  }
  DEV_INLINE void write(int tid, ushort8 *img) const {
    int s_row = tid >> 3, s_slot = tid & 7;
#pragma unroll
    for (int i = 0; i < 4; ++i) img[(s_row + 32 * i) * MXP + s_slot] = r[i];
  }
};

// TN pair assignment: 16 consecutive lanes cover 16 consecutive column
// octets of ONE k-pair, so each wave instruction reads 4 CONTIGUOUS 256 B
// spans instead of 64 scattered 16 B lines (ablation: staging loads were
// 62% of the TN kernel). The transposed LDS write then lands 16 lanes on
// one slot column — the (row>>3)&7 slot-XOR spreads them across banks
// (matching XOR on the read side via rslot()).
template <class L> struct TnStage {
  static constexpr int PITCH = MXP;
  static constexpr bool GLDS = false;
  static DEV_INLINE int rslot(int slot, int row) {
    return slot ^ ((row >> 3) & 7);
  }
  L l;
  ushort8 r[4]; // [it][k-parity]: 2 pair-loads of 2 adjacent k each
  DEV_INLINE void init(int, int) {}
  DEV_INLINE void load(int tid, int base, int kb, ushort8 *) {
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      int p = tid + it * 256;  // pair index in [0,512)
      int k0 = (p >> 4) * 2;   // even k within the 64-deep tile
      int col0 = (p & 15) * 8; // 8 output-dim columns, contiguous per wave
      r[it * 2] = l.load(kb + k0, base + col0);
      r[it * 2 + 1] = l.load(kb + k0 + 1, base + col0);
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

// Plain k-contiguous operand staged by global_load_lds (direct HBM→LDS DMA,
// no staging registers, no ds_write pass — the measured +69% rung of the
// guide's ladder at this tile). glds writes wave-uniform-base + lane*16, so
// the LDS image is lane-linear (pitch 8, no pad); bank conflicts are dodged
// by pre-swizzling the SOURCE k-octet with (row&7) and XOR-ing the same on
// the read side.
struct GldsNt {
  static constexpr int PITCH = 8;
  static constexpr bool GLDS = true;
  static DEV_INLINE int rslot(int slot, int row) { return slot ^ (row & 7); }
  const uint16_t *p;
  int rows;
  long ld;
  int kdim;
  DEV_INLINE void init(int, int) {}
  DEV_INLINE void load(int tid, int base, int kb, ushort8 *img) {
    int s_row = tid >> 3, s_slot = tid & 7;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int row = base + s_row + 32 * i;
      int k = kb + (s_slot ^ (row & 7)) * 8; // pre-swizzled source octet
      if (row < rows && k < kdim) {
        auto *dst = (__attribute__((address_space(3))) ushort8 *)(img + i * 256 + tid);
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) void *)(p + (long)row * ld + k),
            (__attribute__((address_space(3))) void *)dst, 16, 0, 0);
      } else {
        img[i * 256 + tid] = ushort8{0, 0, 0, 0, 0, 0, 0, 0}; // edge zero-fill
      }
    }
  }
  DEV_INLINE void write(int, ushort8 *) const {} // loads land in LDS directly
};

typedef __attribute__((ext_vector_type(16))) float float16v;

// Epilogue address map: row/col → element offset in the output. The
// default is a plain row-major matrix; Stride2Writer scatters a parity
// sub-image back into an NHWC tensor (stride-2 dgrad decomposition).
// Writer concept: RowCtx row_ctx(row) hoists the per-row address work out
// of the column loop; store_*(ptr, ctx, col, v) places one value.
struct LinearWriter {
  static constexpr bool ACC = false;
  static constexpr bool STATS = false;
  long ldc;
  typedef long RowCtx;
  DEV_INLINE RowCtx row_ctx(int row) const { return (long)row * ldc; }
  DEV_INLINE void store_f32(float *p, RowCtx b, int col, float v) const {
    p[b + col] = v;
  }
  DEV_INLINE void store_bf16(uint16_t *p, RowCtx b, int col, float v) const {
    p[b + col] = f2bf(v);
  }
};

// linear forward with the bias folded into the epilogue (fp32 bias, the
// separate bias_add pass was a full activation read+write — 2.4% of a
// BERT-Large step).
struct BiasWriter {
  static constexpr bool ACC = false;
  static constexpr bool STATS = false;
  long ldc;
  const float *bias;
  typedef long RowCtx;
  DEV_INLINE RowCtx row_ctx(int row) const { return (long)row * ldc; }
  DEV_INLINE void store_f32(float *p, RowCtx b, int col, float v) const {
    p[b + col] = v + bias[col];
  }
  DEV_INLINE void store_bf16(uint16_t *p, RowCtx b, int col, float v) const {
    p[b + col] = f2bf(v + bias[col]);
  }
};

// tanh-approx GELU (BERT's activation) and its derivative, fp32.
// tanh via one hardware exp (v_exp_f32): libm tanhf costs ~10× as many
// VALU ops and dominated the fused-GELU GEMM epilogues (113 vs 58 µs on
// the fc2-dx kernel). |rel err| ~1e-7 — far below bf16 output rounding.
DEV_INLINE float tanh_fast(float u) {
  float a = __expf(-2.f * fabsf(u));
  float t = (1.f - a) / (1.f + a);
  return u < 0.f ? -t : t;
}
DEV_INLINE float gelu_f(float x) {
  float u = 0.7978845608028654f * (x + 0.044715f * x * x * x);
  return 0.5f * x * (1.f + tanh_fast(u));
}
DEV_INLINE float dgelu_f(float x) {
  float u = 0.7978845608028654f * (x + 0.044715f * x * x * x);
  float t = tanh_fast(u);
  return 0.5f * (1.f + t) +
         0.5f * x * (1.f - t * t) * 0.7978845608028654f *
             (1.f + 3.f * 0.044715f * x * x);
}
// gelu and its derivative off ONE tanh evaluation (forward epilogue saves
// the derivative so the backward epilogue is exp-free — recomputing
// gelu'(h) there cost ~27 us of v_exp per fc2-dx GEMM, 16.7M elements at
// the transcendental issue rate, serialized after the MFMA work)
DEV_INLINE float gelu_pair_f(float x, float &dg) {
  float u = 0.7978845608028654f * (x + 0.044715f * x * x * x);
  float t = tanh_fast(u);
  dg = 0.5f * (1.f + t) +
       0.5f * x * (1.f - t * t) * 0.7978845608028654f *
           (1.f + 3.f * 0.044715f * x * x);
  return 0.5f * x * (1.f + t);
}

// FFN fc1 epilogue: writes BOTH the pre-activation h = xW₁ᵀ+b (saved for
// backward) and g = gelu(h) — the separate torch GELU pass re-read the
// whole intermediate activation (BERT-Large: 32 MB/layer at 8 TB/s).
struct GeluBiasWriter {
  static constexpr bool ACC = false;
  static constexpr bool STATS = false;
  long ldc;
  const float *bias;
  uint16_t *deriv; // gelu'(h), bf16, same layout as C (saved for backward)
  typedef long RowCtx;
  DEV_INLINE RowCtx row_ctx(int row) const { return (long)row * ldc; }
  DEV_INLINE void store_f32(float *p, RowCtx b, int col, float v) const {
    float h = v + bias[col], dg;
    p[b + col] = gelu_pair_f(h, dg);
    deriv[b + col] = f2bf(dg);
  }
  DEV_INLINE void store_bf16(uint16_t *p, RowCtx b, int col, float v) const {
    float h = v + bias[col], dg;
    p[b + col] = f2bf(gelu_pair_f(h, dg));
    deriv[b + col] = f2bf(dg);
  }
};

// FFN backward epilogue: the fc2-dx GEMM produces dg; multiplying by
// gelu'(h) here yields dh directly — kills the separate dgelu pass
// (2 reads + 1 write over the intermediate activation per layer).
struct GeluBwdWriter {
  static constexpr bool ACC = false;
  static constexpr bool STATS = false;
  long ldc;
  const uint16_t *deriv; // gelu'(h) saved by GeluBiasWriter — exp-free
  typedef long RowCtx;
  DEV_INLINE RowCtx row_ctx(int row) const { return (long)row * ldc; }
  DEV_INLINE void store_f32(float *p, RowCtx b, int col, float v) const {
    p[b + col] = v * bf2f(deriv[b + col]);
  }
  DEV_INLINE void store_bf16(uint16_t *p, RowCtx b, int col, float v) const {
    p[b + col] = f2bf(v * bf2f(deriv[b + col]));
  }
};

// legacy pair (MPIAMD_GELU_DERIV=0): forward saves the PRE-ACTIVATION and
// backward recomputes gelu' (one tanh per element in the bwd epilogue)
struct GeluBiasPreWriter {
  static constexpr bool ACC = false;
  static constexpr bool STATS = false;
  long ldc;
  const float *bias;
  uint16_t *pre; // h (pre-activation), bf16
  typedef long RowCtx;
  DEV_INLINE RowCtx row_ctx(int row) const { return (long)row * ldc; }
  DEV_INLINE void store_f32(float *p, RowCtx b, int col, float v) const {
    float h = v + bias[col];
    pre[b + col] = f2bf(h);
    p[b + col] = gelu_f(h);
  }
  DEV_INLINE void store_bf16(uint16_t *p, RowCtx b, int col, float v) const {
    float h = v + bias[col];
    pre[b + col] = f2bf(h);
    p[b + col] = f2bf(gelu_f(h));
  }
};

struct GeluBwdPreWriter {
  static constexpr bool ACC = false;
  static constexpr bool STATS = false;
  long ldc;
  const uint16_t *pre;
  typedef long RowCtx;
  DEV_INLINE RowCtx row_ctx(int row) const { return (long)row * ldc; }
  DEV_INLINE void store_f32(float *p, RowCtx b, int col, float v) const {
    p[b + col] = v * dgelu_f(bf2f(pre[b + col]));
  }
  DEV_INLINE void store_bf16(uint16_t *p, RowCtx b, int col, float v) const {
    p[b + col] = f2bf(v * dgelu_f(bf2f(pre[b + col])));
  }
};

// Conv-forward epilogue that ALSO produces the BatchNorm per-channel
// partial statistics (Σy, Σy²) — removes bn_partials' full re-read of the
// activation it just wrote (~9% of a ResNet step at 8 TB/s).
//
// Deterministic, atomic-free: the slab has one entry per 64 OUTPUT ROWS
// ([ceil(M/64)][2][C], pre-zeroed by the host). In every tile kernel here
// (mix/pipe_mix 2×2 waves, pipe256 2×4) a lane's epilogue rows all fall in
// ONE 64-row band per m-frag half, and a (band, channel) pair is written
// by exactly one lane pair (l, l^32) — combined with one shfl, stored
// plain. Stats use the ROUNDED bf16 value (what bn_partials would read).
struct BnStatsWriter {
  static constexpr bool ACC = false;
  static constexpr bool STATS = true;
  long ldc;
  float *slab; // [ceil(M/64)][2][C]
  int C;
  // mutable per-lane accumulators: [row-band half][col slot (bit5 of col)]
  mutable float s0[2][2], s1[2][2];
  mutable int tm[2], colc[2];
  mutable int half_;
  typedef long RowCtx;
  DEV_INLINE void reset() const {
    s0[0][0] = s0[0][1] = s0[1][0] = s0[1][1] = 0.f;
    s1[0][0] = s1[0][1] = s1[1][0] = s1[1][1] = 0.f;
    tm[0] = tm[1] = -1;
    colc[0] = colc[1] = -1;
    half_ = 0;
  }
  DEV_INLINE RowCtx row_ctx(int row) const {
    // pipe256's 4 m-frags span two 64-row bands per lane; track which
    half_ = (row >> 6) & 1;
    tm[half_] = row >> 6;
    return (long)row * ldc;
  }
  DEV_INLINE void store_f32(float *p, RowCtx b, int col, float v) const {
    p[b + col] = v; // split-K callers never attach stats (caller contract)
  }
  DEV_INLINE void store_bf16(uint16_t *p, RowCtx b, int col, float v) const {
    uint16_t r = f2bf(v);
    p[b + col] = r;
    float vr = bf2f(r);
    int ci = (col >> 5) & 1;
    colc[ci] = col;
    s0[half_][ci] += vr;
    s1[half_][ci] += vr * vr;
  }
  DEV_INLINE void flush(int lane) const {
#pragma unroll
    for (int h = 0; h < 2; ++h) {
      // bands are per-(lane-half) identical; tm/col agree across l, l^32
#pragma unroll
      for (int ci = 0; ci < 2; ++ci) {
        float a = s0[h][ci] + __shfl_xor(s0[h][ci], 32, 64);
        float b = s1[h][ci] + __shfl_xor(s1[h][ci], 32, 64);
        if (lane < 32 && tm[h] >= 0 && colc[ci] >= 0) {
          float *e = slab + (long)tm[h] * 2 * C;
          e[colc[ci]] = a;
          e[C + colc[ci]] = b;
        }
      }
    }
  }
};

// += into an existing bf16 tensor (bottleneck backward: conv1's dgrad
// accumulates onto the skip-connection gradient — no separate add pass).
struct LinearAccWriter {
  static constexpr bool ACC = true;
  static constexpr bool STATS = false;
  long ldc;
  typedef long RowCtx;
  DEV_INLINE RowCtx row_ctx(int row) const { return (long)row * ldc; }
  DEV_INLINE void store_f32(float *p, RowCtx b, int col, float v) const {
    p[b + col] += v;
  }
  DEV_INLINE void store_bf16(uint16_t *p, RowCtx b, int col, float v) const {
    p[b + col] = f2bf(v + bf2f(p[b + col]));
  }
};

struct Stride2Writer {
  static constexpr bool ACC = false;
  static constexpr bool STATS = false;
  int W2, H2, ph, pw, W, H, C;
  typedef long RowCtx;
  DEV_INLINE RowCtx row_ctx(int row) const {
    int w_ = row % W2;
    long t = row / W2;
    int h_ = (int)(t % H2);
    int n = (int)(t / H2);
    return (((long)n * H + 2 * h_ + ph) * W + 2 * w_ + pw) * (long)C;
  }
  DEV_INLINE void store_f32(float *p, RowCtx b, int col, float v) const {
    p[b + col] = v;
  }
  DEV_INLINE void store_bf16(uint16_t *p, RowCtx b, int col, float v) const {
    p[b + col] = f2bf(v);
  }
};

// 1x1 stride-2 dgrad: parity (pad%2, pad%2) is the ONLY contributor, so the
// epilogue zeroes the three sibling pixels itself — no 100 MB memset pass.
struct Stride2ZeroWriter {
  static constexpr bool ACC = false;
  static constexpr bool STATS = false;
  int W2, H2, ph, pw, W, H, C;
  struct RowCtx {
    long base;
    int sib; // bit0: w+1 valid, bit1: h+1 valid
  };
  DEV_INLINE RowCtx row_ctx(int row) const {
    int w_ = row % W2;
    long t = row / W2;
    int h_ = (int)(t % H2);
    int n = (int)(t / H2);
    int hh = 2 * h_ + ph, ww = 2 * w_ + pw;
    RowCtx c;
    c.base = (((long)n * H + hh) * W + ww) * (long)C;
    c.sib = (ww + 1 < W ? 1 : 0) | (hh + 1 < H ? 2 : 0);
    return c;
  }
  DEV_INLINE void store_f32(float *p, RowCtx c, int col, float v) const {
    p[c.base + col] = v; // (f32 split-K path never uses ZSIB)
  }
  DEV_INLINE void store_bf16(uint16_t *p, RowCtx c, int col, float v) const {
    p[c.base + col] = f2bf(v);
    if (c.sib & 1) p[c.base + C + col] = 0;
    if (c.sib & 2) {
      p[c.base + (long)W * C + col] = 0;
      if (c.sib & 1) p[c.base + (long)W * C + C + col] = 0;
    }
  }
};

// ONEBUF: single LDS buffer (36 KB vs 72 KB) with a second barrier per
// k-step — 4 workgroups/CU instead of 2 (16 waves/CU): trades one barrier
// for 2x the latency-hiding wave pool. A/B via MPIAMD_GEMM_ONEBUF=0/1.
template <class SA, class SB, bool C_F32, class WR = LinearWriter,
          bool ONEBUF = false>
__global__ __launch_bounds__(NT_THREADS) void mix_gemm_k(
    SA sa, SB sb, void *__restrict__ cptr, int M, int N, int K, WR wrt,
    int tiles_n, int kt_per_split, long split_stride, int xcd_cpx, int swap) {
  // SPLIT-MAJOR grid (x = split, y = tile) when swap: the N-tiles sharing
  // an A panel are y-adjacent, which lands them on ONE XCD's L2 when
  // gridDim.x%8==0. Tile-major dispatch spread a panel's sharers over 4+
  // XCDs and the panel re-reads all went to HBM (ablation: staging at full
  // HBM BW; +16-28% on split-K wgrad shapes in the standalone A/B).
  int tile = swap ? blockIdx.y : blockIdx.x;
  int split = swap ? blockIdx.x : blockIdx.y;
  if (xcd_cpx) // T1: contiguous tile chunk per XCD (L2 reuse of panels)
    tile = (tile & 7) * xcd_cpx + (tile >> 3);
  int tm = tile / tiles_n, tn = tile % tiles_n;
  int row0 = tm * BM, col0 = tn * BN;
  int tid = threadIdx.x;
  int lane = tid & 63, wave = tid >> 6;
  int wr = wave >> 1, wc = wave & 1;

  // ONE shared object (a second __shared__ array makes hipcc drain vmcnt(0)
  // before every ds_read while a glds is in flight — guide §5 trap 4a)
  constexpr int ASZ = BM * SA::PITCH, BSZ = BM * SB::PITCH;
  constexpr int NBUF = ONEBUF ? 1 : 2;
  __shared__ ushort8 lds[NBUF * (ASZ + BSZ)];
  // NOTE: compute image bases arithmetically at each use — a runtime-indexed
  // private array of LDS pointers de-optimizes every ds_read (measured -15%
  // across ALL kernels when these were ushort8* imgA[2] arrays)
#define MXG_A(b) (lds + ((b) % NBUF) * (ASZ + BSZ))
#define MXG_B(b) (lds + ((b) % NBUF) * (ASZ + BSZ) + ASZ)

  // 32x32x16 MFMA (higher ceiling than 16x16x32): each wave computes a
  // 64x64 quadrant as 2x2 fragments of 32x32, 16 fp32 accumulators each.
  float16v acc[2][2] = {};

  int nk_total = (K + BK - 1) / BK;
  int t0 = split * kt_per_split;
  int nk = min(kt_per_split, nk_total - t0);
  if (nk < 0) nk = 0;

  // Register staging, issue-early / write-late (distance-1). A distance-2
  // two-register-set variant was measured SLOWER on the gather stagers
  // (address-compute duplication outweighed the extra latency cover) and
  // neutral on plain ones — don't re-add it.
  sa.init(tid, row0);
  sb.init(tid, col0);
  if (nk > 0) {
    sa.load(tid, row0, t0 * BK, MXG_A(0));
    sb.load(tid, col0, t0 * BK, MXG_B(0));
    sa.write(tid, MXG_A(0));
    sb.write(tid, MXG_B(0));
  }
  __syncthreads();

  for (int t = 0; t < nk; ++t) {
    int buf = ONEBUF ? 0 : (t & 1);
    if (t + 1 < nk) { // issue-early: loads (or glds DMA) land under MFMAs
      sa.load(tid, row0, (t0 + t + 1) * BK, MXG_A(buf ^ 1));
      sb.load(tid, col0, (t0 + t + 1) * BK, MXG_B(buf ^ 1));
    }
#pragma unroll
    for (int kk = 0; kk < BK / 16; ++kk) { // 4 k-steps of 16
      bf16x8 af[2], bf_[2];
      // lane l: row = l&31, k = kk*16 + (l>>5)*8 .. +7 → slot kk*2+(l>>5)
      int slot = kk * 2 + (lane >> 5);
#pragma unroll
      for (int mi = 0; mi < 2; ++mi) {
        int arow = wr * 64 + mi * 32 + (lane & 31);
        af[mi] = us8_to_bf8v(MXG_A(buf)[arow * SA::PITCH + SA::rslot(slot, arow)]);
      }
#pragma unroll
      for (int ni = 0; ni < 2; ++ni) {
        int brow = wc * 64 + ni * 32 + (lane & 31);
        bf_[ni] = us8_to_bf8v(MXG_B(buf)[brow * SB::PITCH + SB::rslot(slot, brow)]);
      }
#pragma unroll
      for (int mi = 0; mi < 2; ++mi)
#pragma unroll
        for (int ni = 0; ni < 2; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              af[mi], bf_[ni], acc[mi][ni], 0, 0, 0);
    }
    if (ONEBUF) {
      if (t + 1 < nk) { // reuse the single buffer: drain-readers barrier,
        __syncthreads(); // then overwrite with tile t+1
        sa.write(tid, MXG_A(0));
        sb.write(tid, MXG_B(0));
      }
    } else if (t + 1 < nk) { // write-late into the other buffer
      sa.write(tid, MXG_A(buf ^ 1));
      sb.write(tid, MXG_B(buf ^ 1));
    }
    __syncthreads();
  }

  // 32x32 C/D map: col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
  // Row-major loop order so each row's address context is computed once.
  if constexpr (WR::STATS) wrt.reset();
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

#undef MXG_A
#undef MXG_B

template <class SA, class SB, class WR>
static hipError_t launch_mix_gemm_wr(const SA &sa, const SB &sb, void *c,
                                     int M, int N, int K, const WR &wrt,
                                     long ldc, bool c_f32, hipStream_t s,
                                     int splits = 1) {
  int tiles_m = (M + BM - 1) / BM, tiles_n = (N + BN - 1) / BN;
  int nk = (K + BK - 1) / BK;
  if (splits > nk) splits = nk > 0 ? nk : 1;
  int kts = (nk + splits - 1) / splits;
  long split_stride = (long)M * ldc;
  int nwg = tiles_m * tiles_n;
  // T1 XCD swizzle: give each XCD a contiguous chunk of tiles so neighbor
  // tiles (sharing operand panels) hit the same per-XCD L2. Needs nwg%8==0
  // and enough tiles to matter.
  // default OFF: the standalone-harness win (+16-28% on synthetic TnTn
  // wgrad shapes) did not transfer to the full model (-0.4% same-box on the
  // bench) — real wgrads gather B via XcolStage and use smaller splits
  static const int swap_env = [] {
    const char *e = getenv("MPIAMD_SPLITMAJOR");
    return e && e[0] == '1' ? 1 : 0;
  }();
  int swap = (splits > 1 && splits % 8 == 0) ? swap_env : 0;
  int cpx = (nwg % 8 == 0 && nwg >= 32 && (splits == 1 || swap)) ? nwg / 8 : 0;
  dim3 grid = swap ? dim3(splits, nwg) : dim3(nwg, splits);
  static const bool onebuf_env = [] {
    const char *e = getenv("MPIAMD_GEMM_ONEBUF");
    return e && e[0] == '1';
  }();
  // glds targets the next buffer while the current one is being read — a
  // single buffer would race
  const bool onebuf = onebuf_env && !SA::GLDS && !SB::GLDS;
  if (onebuf) {
    if (c_f32)
      mix_gemm_k<SA, SB, true, WR, true><<<grid, NT_THREADS, 0, s>>>(
          sa, sb, c, M, N, K, wrt, tiles_n, kts, split_stride, cpx, swap);
    else
      mix_gemm_k<SA, SB, false, WR, true><<<grid, NT_THREADS, 0, s>>>(
          sa, sb, c, M, N, K, wrt, tiles_n, kts, split_stride, cpx, swap);
    return hipGetLastError();
  }
  if (c_f32)
    mix_gemm_k<SA, SB, true, WR><<<grid, NT_THREADS, 0, s>>>(
        sa, sb, c, M, N, K, wrt, tiles_n, kts, split_stride, cpx, swap);
  else
    mix_gemm_k<SA, SB, false, WR><<<grid, NT_THREADS, 0, s>>>(
        sa, sb, c, M, N, K, wrt, tiles_n, kts, split_stride, cpx, swap);
  return hipGetLastError();
}

template <class SA, class SB>
static hipError_t launch_mix_gemm(const SA &sa, const SB &sb, void *c, int M,
                                  int N, int K, long ldc, bool c_f32,
                                  hipStream_t s, int splits = 1) {
  return launch_mix_gemm_wr(sa, sb, c, M, N, K, LinearWriter{ldc}, ldc, c_f32,
                            s, splits);
}

#include "gemm256.h"
#include "pipe256.h"
#include "pipe_mix.h"

// Plain NT×NT entry (both operands k-contiguous row-major). glds staging
// vs register staging selectable for same-box A/B (MPIAMD_GLDS=0 reverts).
// Full-tile shapes take the 256²-tile counted-vmcnt pipeline (gemm256.h);
// MPIAMD_GEMM256=0 reverts those to the mix_gemm path.
template <class LA, class LB>
static hipError_t launch_nt_gemm(const LA &la, const LB &lb, void *c, int M,
                                 int N, int K, long ldc, bool c_f32,
                                 hipStream_t s, int splits = 1,
                                 const float *bias = nullptr) {
  static const bool use_256 = [] {
    const char *e = getenv("MPIAMD_GEMM256");
    return !(e && e[0] == '0');
  }();
  // occupancy floor: 256² tiles run 1 block/CU (96 KB LDS), so a grid
  // under ~half the 256 CUs loses more to idle CUs than the pipeline wins
  // (measured: M=12544,N=512 → 98 wgs: 338 vs 506 TF on the 128² path;
  // M=50176,N=256 → 196 wgs: 505 vs 417 — the crossover is between)
  // 4-phase deep pipeline (pipe256.h) for K%64 full-tile shapes;
  // MPIAMD_PIPE=0 reverts to the round-1 BK=32 3-buffer kernel (gemm256.h)
  static const bool use_pipe = [] {
    const char *e = getenv("MPIAMD_PIPE");
    return !(e && e[0] == '0');
  }();
  if (use_256 && splits <= 1 && M % 256 == 0 && N % 256 == 0 && K % 32 == 0 &&
      K > 0 && la.kdim == K && lb.kdim == K && ldc == N &&
      (long)(M / 256) * (N / 256) >= 128) {
    if (use_pipe && K % 64 == 0)
      return launch_pipe256(la, lb, c, M, N, K, ldc, c_f32, s, bias);
    return launch_nt256(la, lb, c, M, N, K, ldc, c_f32, s, bias);
  }
  // non-full-tile / small-grid shapes: the 128²-tile deep pipeline
  // (pipe_mix.h) — ragged edges handled by its zeros-page staging.
  // MPIAMD_PIPENT=0 reverts just this NT fallback (bisect lever).
  static const bool use_pipent = [] {
    const char *e = getenv("MPIAMD_PIPENT");
    return !(e && e[0] == '0');
  }();
  if (use_pipemix() && use_pipent) {
    NtPipe<PlainNtSrc> ga{{la.p, la.ld, la.rows, la.kdim}};
    NtPipe<PlainNtSrc> gb{{lb.p, lb.ld, lb.rows, lb.kdim}};
    if (bias)
      return launch_pipe_mix_wr(ga, gb, c, M, N, K, BiasWriter{ldc, bias},
                                ldc, c_f32, s, splits);
    return launch_pipe_mix_wr(ga, gb, c, M, N, K, LinearWriter{ldc}, ldc,
                              c_f32, s, splits);
  }
  static const bool use_glds = [] {
    const char *e = getenv("MPIAMD_GLDS");
    return !(e && e[0] == '0');
  }();
  if (use_glds) {
    GldsNt ga{la.p, la.rows, la.ld, la.kdim};
    GldsNt gb{lb.p, lb.rows, lb.ld, lb.kdim};
    if (bias)
      return launch_mix_gemm_wr(ga, gb, c, M, N, K, BiasWriter{ldc, bias},
                                ldc, c_f32, s, splits);
    return launch_mix_gemm(ga, gb, c, M, N, K, ldc, c_f32, s, splits);
  }
  if (bias)
    return launch_mix_gemm_wr(NtStage<LA>{la}, NtStage<LB>{lb}, c, M, N, K,
                              BiasWriter{ldc, bias}, ldc, c_f32, s, splits);
  return launch_mix_gemm(NtStage<LA>{la}, NtStage<LB>{lb}, c, M, N, K, ldc,
                         c_f32, s, splits);
}

// ---- TN loaders ----

// k-strided view of a row-major [K][cols] matrix (dy, x, w as TN operand).
struct TnRowMajor {
  const uint16_t *p;
  long ld;  // elements per k-row
  int kdim; // rows (= reduce extent)
  int cols;
  DEV_INLINE ushort8 load(int k, int c0) const {
    if (k >= kdim || c0 >= cols) return ushort8{0, 0, 0, 0, 0, 0, 0, 0};
    if (c0 + 8 <= cols) return *(const ushort8 *)(p + (long)k * ld + c0);
    ushort8 v{0, 0, 0, 0, 0, 0, 0, 0}; // ragged tail (cols not %8)
#pragma unroll
    for (int j = 0; j < 8; ++j)
      if (c0 + j < cols) v[j] = p[(long)k * ld + c0 + j];
    return v;
  }
};
