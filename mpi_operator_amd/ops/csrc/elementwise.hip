// Elementwise / multi-tensor kernels: fused residual add+ReLU (fwd/bwd),
// global average pool (fwd/bwd), fused multi-tensor SGD-momentum.
// All memory-bound: vectorized 16 B/lane bf16 accesses (guide G13),
// grid-stride with capped grids (guide G11).
#include "common.h"

static inline int ew_grid(long lane_tasks, int block = 256) {
  long blocks = (lane_tasks + block - 1) / block;
  if (blocks > 2048) blocks = 2048;
  if (blocks < 1) blocks = 1;
  return (int)blocks;
}

// ---------------- plain add (c = a + b) ----------------
// bottleneck backward's residual join: skip grad + conv1 dgrad. Separate
// tensors, 16 B vectorized, no RMW (an accumulate epilogue serialized on
// load/store aliasing within one tensor — measured 2x slower than dgrad).
__global__ void add_bf16_k(const ushort8 *__restrict__ a,
                           const ushort8 *__restrict__ b,
                           ushort8 *__restrict__ c, long n8) {
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < n8; i += 2 * stride) {
    long i2 = i + stride;
    ushort8 va = a[i], vb = b[i];
    ushort8 va2 = i2 < n8 ? a[i2] : ushort8{0, 0, 0, 0, 0, 0, 0, 0};
    ushort8 vb2 = i2 < n8 ? b[i2] : ushort8{0, 0, 0, 0, 0, 0, 0, 0};
    float fa[8], fb[8], fa2[8], fb2[8];
    bf8_to_f8(va, fa);
    bf8_to_f8(vb, fb);
    bf8_to_f8(va2, fa2);
    bf8_to_f8(vb2, fb2);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      fa[j] += fb[j];
      fa2[j] += fb2[j];
    }
    c[i] = f8_to_bf8(fa);
    if (i2 < n8) c[i2] = f8_to_bf8(fa2);
  }
}

extern "C" hipError_t add_bf16(const void *a, const void *b, void *c, long n,
                               hipStream_t s) {
  long n8 = n / 8;
  add_bf16_k<<<ew_grid(n8), 256, 0, s>>>((const ushort8 *)a,
                                         (const ushort8 *)b, (ushort8 *)c, n8);
  HIP_KERNEL_CHECK();
  return hipSuccess;
}

// ---------------- add + relu ----------------
__global__ void add_relu_fwd_k(const ushort8 *__restrict__ a,
                               const ushort8 *__restrict__ b,
                               ushort8 *__restrict__ y, long n8) {
  for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += (long)gridDim.x * blockDim.x) {
    ushort8 va = a[i], vb = b[i];
    float fa[8], fb[8];
    bf8_to_f8(va, fa);
    bf8_to_f8(vb, fb);
#pragma unroll
    for (int j = 0; j < 8; ++j) fa[j] = fmaxf(fa[j] + fb[j], 0.f);
    y[i] = f8_to_bf8(fa);
  }
}

extern "C" hipError_t add_relu_fwd(const void *a, const void *b, void *y,
                                   long n, hipStream_t s) {
  long n8 = n / 8;
  add_relu_fwd_k<<<ew_grid(n8), 256, 0, s>>>((const ushort8 *)a,
                                             (const ushort8 *)b, (ushort8 *)y, n8);
  HIP_KERNEL_CHECK();
  return hipSuccess;
}

__global__ void add_relu_bwd_k(const ushort8 *__restrict__ dy,
                               const ushort8 *__restrict__ y,
                               ushort8 *__restrict__ dx, long n8) {
  for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += (long)gridDim.x * blockDim.x) {
    ushort8 vd = dy[i], vy = y[i];
    ushort8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = bf2f(vy[j]) > 0.f ? vd[j] : uint16_t(0);
    dx[i] = o;
  }
}

// residual-join backward in ONE pass: dxt = dx0 + dout·(y > 0).
// Replaces the materialized g = add_relu_bwd(dout, y) followed by
// add_bf16(g, dx0): bn_bwd applies the ReLU mask itself (relu=1 reads y
// anyway), so g was pure data movement — 2 passes (4 reads 2 writes) of
// the block activation become 1 (3 reads 1 write).
__global__ void add_relu_bwd_add_k(const ushort8 *__restrict__ dy,
                                   const ushort8 *__restrict__ y,
                                   const ushort8 *__restrict__ dx0,
                                   ushort8 *__restrict__ dxt, long n8) {
  for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += (long)gridDim.x * blockDim.x) {
    ushort8 vd = dy[i], vy = y[i], v0 = dx0[i];
    ushort8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float g = bf2f(vy[j]) > 0.f ? bf2f(vd[j]) : 0.f;
      o[j] = f2bf(g + bf2f(v0[j]));
    }
    dxt[i] = o;
  }
}

// mask form: the 1-byte-per-octet relu mask (bn_apply_k) replaces the
// 16-byte y re-read — the join backward drops from 3 reads to 2 + 1/16.
__global__ void add_relu_bwd_add_mask_k(const ushort8 *__restrict__ dy,
                                        const uint8_t *__restrict__ mask,
                                        const ushort8 *__restrict__ dx0,
                                        ushort8 *__restrict__ dxt, long n8) {
  for (long i = blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += (long)gridDim.x * blockDim.x) {
    ushort8 vd = dy[i], v0 = dx0[i];
    int m = mask[i];
    ushort8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float g = (m & (1 << j)) ? bf2f(vd[j]) : 0.f;
      o[j] = f2bf(g + bf2f(v0[j]));
    }
    dxt[i] = o;
  }
}

extern "C" hipError_t add_relu_bwd_add_mask(const void *dy, const void *mask,
                                            const void *dx0, void *dxt,
                                            long n, hipStream_t s) {
  long n8 = n / 8;
  add_relu_bwd_add_mask_k<<<ew_grid(n8), 256, 0, s>>>(
      (const ushort8 *)dy, (const uint8_t *)mask, (const ushort8 *)dx0,
      (ushort8 *)dxt, n8);
  HIP_KERNEL_CHECK();
  return hipSuccess;
}

extern "C" hipError_t add_relu_bwd_add(const void *dy, const void *y,
                                       const void *dx0, void *dxt, long n,
                                       hipStream_t s) {
  long n8 = n / 8;
  add_relu_bwd_add_k<<<ew_grid(n8), 256, 0, s>>>(
      (const ushort8 *)dy, (const ushort8 *)y, (const ushort8 *)dx0,
      (ushort8 *)dxt, n8);
  HIP_KERNEL_CHECK();
  return hipSuccess;
}

extern "C" hipError_t add_relu_bwd(const void *dy, const void *y, void *dx,
                                   long n, hipStream_t s) {
  long n8 = n / 8;
  add_relu_bwd_k<<<ew_grid(n8), 256, 0, s>>>((const ushort8 *)dy,
                                             (const ushort8 *)y, (ushort8 *)dx, n8);
  HIP_KERNEL_CHECK();
  return hipSuccess;
}

// ---------------- global average pool (NHWC) ----------------
// y[n, c] = mean over HW of x[n, hw, c]; consecutive lanes take consecutive
// 8-channel groups → coalesced 16 B/lane reads at every hw step.
__global__ void gap_fwd_k(const ushort8 *__restrict__ x, ushort8 *__restrict__ y,
                          int N, int HW, int C8) {
  for (long t = blockIdx.x * blockDim.x + threadIdx.x; t < (long)N * C8;
       t += (long)gridDim.x * blockDim.x) {
    int n = t / C8, cb = t % C8;
    const ushort8 *row = x + (long)n * HW * C8 + cb;
    float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    for (int hw = 0; hw < HW; ++hw) {
      ushort8 v = row[(long)hw * C8];
      float f[8];
      bf8_to_f8(v, f);
#pragma unroll
      for (int j = 0; j < 8; ++j) acc[j] += f[j];
    }
    float inv = 1.f / HW;
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[j] *= inv;
    y[t] = f8_to_bf8(acc);
  }
}

extern "C" hipError_t gap_fwd(const void *x, void *y, int N, int HW, int C,
                              hipStream_t s) {
  int C8 = C / 8;
  gap_fwd_k<<<ew_grid((long)N * C8), 256, 0, s>>>((const ushort8 *)x,
                                                  (ushort8 *)y, N, HW, C8);
  HIP_KERNEL_CHECK();
  return hipSuccess;
}

__global__ void gap_bwd_k(const ushort8 *__restrict__ dy, ushort8 *__restrict__ dx,
                          int N, int HW, int C8, float inv_hw) {
  for (long t = blockIdx.x * blockDim.x + threadIdx.x; t < (long)N * HW * C8;
       t += (long)gridDim.x * blockDim.x) {
    int cb = t % C8;
    int n = t / ((long)HW * C8);
    ushort8 v = dy[(long)n * C8 + cb];
    float f[8];
    bf8_to_f8(v, f);
#pragma unroll
    for (int j = 0; j < 8; ++j) f[j] *= inv_hw;
    dx[t] = f8_to_bf8(f);
  }
}

extern "C" hipError_t gap_bwd(const void *dy, void *dx, int N, int HW, int C,
                              hipStream_t s) {
  int C8 = C / 8;
  gap_bwd_k<<<ew_grid((long)N * HW * C8), 256, 0, s>>>(
      (const ushort8 *)dy, (ushort8 *)dx, N, HW, C8, 1.f / HW);
  HIP_KERNEL_CHECK();
  return hipSuccess;
}

// ---------------- fused multi-tensor SGD-momentum ----------------
// One launch covers up to SGD_MAX_T tensors (descriptors in kernarg space —
// apex multi_tensor_apply-style, no per-step device metadata uploads).
// Updates, per element: g' = grad(+wd*master); mom = mu*mom + g';
// step = nesterov ? g' + mu*mom : mom; master -= lr*step; out_bf16 = master.
constexpr int SGD_MAX_T = 40;
constexpr int SGD_BLOCK = 256;
constexpr int SGD_EPB = SGD_BLOCK * 32; // elements per block (8192)

struct SgdDesc {
  const void *grad;
  float *master;
  float *mom;
  uint16_t *out; // nullptr if the param itself is fp32 (master IS the param)
  long numel;
};

struct SgdArgs {
  SgdDesc d[SGD_MAX_T];
  int first_block[SGD_MAX_T + 1]; // prefix of per-tensor block counts
  int nt;
};

template <bool GRAD_BF16>
DEV_INLINE void sgd_octet(const SgdDesc &D, long i, float lr, float mu,
                          float wd, int nesterov) {
  float g[8], mst[8], mm[8];
  if (GRAD_BF16) {
    ushort8 v = *(const ushort8 *)((const uint16_t *)D.grad + i);
    bf8_to_f8(v, g);
  } else {
    const float *gp = (const float *)D.grad + i;
    float4v a = *(const float4v *)gp, b = *(const float4v *)(gp + 4);
    g[0] = a[0]; g[1] = a[1]; g[2] = a[2]; g[3] = a[3];
    g[4] = b[0]; g[5] = b[1]; g[6] = b[2]; g[7] = b[3];
  }
  float4v ma = *(const float4v *)(D.master + i);
  float4v mb = *(const float4v *)(D.master + i + 4);
  float4v va = *(const float4v *)(D.mom + i);
  float4v vb = *(const float4v *)(D.mom + i + 4);
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    mst[j] = j < 4 ? ma[j] : mb[j - 4];
    mm[j] = j < 4 ? va[j] : vb[j - 4];
    float gj = g[j] + wd * mst[j];
    mm[j] = mu * mm[j] + gj;
    float step = nesterov ? gj + mu * mm[j] : mm[j];
    mst[j] -= lr * step;
  }
  ma = float4v{mst[0], mst[1], mst[2], mst[3]};
  mb = float4v{mst[4], mst[5], mst[6], mst[7]};
  va = float4v{mm[0], mm[1], mm[2], mm[3]};
  vb = float4v{mm[4], mm[5], mm[6], mm[7]};
  *(float4v *)(D.master + i) = ma;
  *(float4v *)(D.master + i + 4) = mb;
  *(float4v *)(D.mom + i) = va;
  *(float4v *)(D.mom + i + 4) = vb;
  if (D.out) *(ushort8 *)(D.out + i) = f8_to_bf8(mst);
}

template <bool GRAD_BF16>
__global__ void sgd_step_k(SgdArgs args, float lr, float mu, float wd,
                           int nesterov) {
  // find this block's tensor (nt <= 40: linear scan, wave-uniform)
  int bid = blockIdx.x;
  int t = 0;
  while (t + 1 <= args.nt - 1 && bid >= args.first_block[t + 1]) ++t;
  const SgdDesc &D = args.d[t];
  long base = (long)(bid - args.first_block[t]) * SGD_EPB;
  if (base + SGD_EPB <= D.numel) {
    // interior block: compile-time trip count so the compiler fully unrolls
    // the rounds and schedules loads ahead — the runtime-bounded branchy
    // loop below issue-stalled 66% / ran ~1.8x off the HBM roofline on the
    // BERT-Large optimizer step (PMC prof7).
#pragma unroll
    for (int r = 0; r < SGD_EPB / (SGD_BLOCK * 8); ++r)
      sgd_octet<GRAD_BF16>(D, base + r * (SGD_BLOCK * 8L) + threadIdx.x * 8L,
                           lr, mu, wd, nesterov);
    return;
  }
  for (long i = base + threadIdx.x * 8L; i < min(base + SGD_EPB, D.numel);
       i += SGD_BLOCK * 8L) {
    bool full = (i + 8 <= D.numel);
    int m = full ? 8 : (int)(D.numel - i);
    float g[8], mst[8], mm[8];
    if (GRAD_BF16) {
      const uint16_t *gp = (const uint16_t *)D.grad + i;
      if (full) {
        ushort8 v = *(const ushort8 *)gp;
        bf8_to_f8(v, g);
      } else
        for (int j = 0; j < m; ++j) g[j] = bf2f(gp[j]);
    } else {
      const float *gp = (const float *)D.grad + i;
      if (full) {
        float4v a = *(const float4v *)gp, b = *(const float4v *)(gp + 4);
        g[0] = a[0]; g[1] = a[1]; g[2] = a[2]; g[3] = a[3];
        g[4] = b[0]; g[5] = b[1]; g[6] = b[2]; g[7] = b[3];
      } else
        for (int j = 0; j < m; ++j) g[j] = gp[j];
    }
    // full octets take explicit 16 B vector loads/stores for master and
    // momentum — the runtime-bound j<m loop alone left them as per-element
    // dword traffic (measured 1.5 TB/s on BERT-Large's optimizer step)
    if (full) {
      float4v ma = *(const float4v *)(D.master + i);
      float4v mb = *(const float4v *)(D.master + i + 4);
      float4v va = *(const float4v *)(D.mom + i);
      float4v vb = *(const float4v *)(D.mom + i + 4);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        mst[j] = j < 4 ? ma[j] : mb[j - 4];
        mm[j] = j < 4 ? va[j] : vb[j - 4];
        float gj = g[j] + wd * mst[j];
        mm[j] = mu * mm[j] + gj;
        float step = nesterov ? gj + mu * mm[j] : mm[j];
        mst[j] -= lr * step;
      }
      ma = float4v{mst[0], mst[1], mst[2], mst[3]};
      mb = float4v{mst[4], mst[5], mst[6], mst[7]};
      va = float4v{mm[0], mm[1], mm[2], mm[3]};
      vb = float4v{mm[4], mm[5], mm[6], mm[7]};
      *(float4v *)(D.master + i) = ma;
      *(float4v *)(D.master + i + 4) = mb;
      *(float4v *)(D.mom + i) = va;
      *(float4v *)(D.mom + i + 4) = vb;
      if (D.out) *(ushort8 *)(D.out + i) = f8_to_bf8(mst);
    } else {
      for (int j = 0; j < m; ++j) {
        mst[j] = D.master[i + j];
        mm[j] = D.mom[i + j];
        float gj = g[j] + wd * mst[j];
        mm[j] = mu * mm[j] + gj;
        float step = nesterov ? gj + mu * mm[j] : mm[j];
        mst[j] -= lr * step;
        D.master[i + j] = mst[j];
        D.mom[i + j] = mm[j];
      }
      if (D.out)
        for (int j = 0; j < m; ++j) D.out[i + j] = f2bf(mst[j]);
    }
  }
}

extern "C" hipError_t sgd_step_launch(const SgdDesc *descs, int nt,
                                      int grad_is_bf16, float lr, float mu,
                                      float wd, int nesterov, hipStream_t s) {
  for (int start = 0; start < nt; start += SGD_MAX_T) {
    SgdArgs a;
    a.nt = (nt - start < SGD_MAX_T) ? nt - start : SGD_MAX_T;
    int blocks = 0;
    for (int i = 0; i < a.nt; ++i) {
      a.d[i] = descs[start + i];
      a.first_block[i] = blocks;
      blocks += (int)((a.d[i].numel + SGD_EPB - 1) / SGD_EPB);
    }
    a.first_block[a.nt] = blocks;
    if (grad_is_bf16)
      sgd_step_k<true><<<blocks, SGD_BLOCK, 0, s>>>(a, lr, mu, wd, nesterov);
    else
      sgd_step_k<false><<<blocks, SGD_BLOCK, 0, s>>>(a, lr, mu, wd, nesterov);
    HIP_KERNEL_CHECK();
  }
  return hipSuccess;
}

// ---------------- linear-layer helpers ----------------
__global__ void bias_add_k(uint16_t *__restrict__ y, const float *__restrict__ b,
                           long M, int N) {
  for (long t = blockIdx.x * blockDim.x + threadIdx.x; t < M * N;
       t += (long)gridDim.x * blockDim.x)
    y[t] = f2bf(bf2f(y[t]) + b[t % N]);
}

extern "C" hipError_t bias_add(void *y, const float *b, long M, int N,
                               hipStream_t s) {
  bias_add_k<<<ew_grid(M * N), 256, 0, s>>>((uint16_t *)y, b, M, N);
  HIP_KERNEL_CHECK();
  return hipSuccess;
}

// bias-grad column sum. colsum8_k: each thread owns one 8-column octet
// (coalesced 16 B loads), blocks tile (row-chunks × col-octet groups) and
// write per-block partial rows into a [gridDim.x][N] fp32 slab that
// splitk_reduce folds into db — the atomicAdd ending this replaces put
// gridDim.x-way RMW contention on every db word (40 µs for a 33 MB read
// that should take ~6; the atomics were the tail), and the one-thread-per
// -column serial version before THAT ran 16 blocks and was 48% of a
// BERT-Large step.
__global__ void colsum8_k(const ushort8 *__restrict__ dy,
                          float *__restrict__ partial, long M, int C8,
                          long ld8) {
  int cb = blockIdx.y * 32 + (threadIdx.x & 31);
  int rl = threadIdx.x >> 5; // 8 row lanes per block
  float a[8] = {0};
  if (cb < C8) {
    for (long m = (long)blockIdx.x * 8 + rl; m < M; m += (long)gridDim.x * 8) {
      float f[8];
      bf8_to_f8(dy[m * ld8 + cb], f);
#pragma unroll
      for (int j = 0; j < 8; ++j) a[j] += f[j];
    }
  }
  __shared__ float lds[256 * 8];
#pragma unroll
  for (int j = 0; j < 8; ++j) lds[threadIdx.x * 8 + j] = a[j];
  __syncthreads();
  if (rl == 0 && cb < C8) {
    for (int r = 1; r < 8; ++r)
#pragma unroll
      for (int j = 0; j < 8; ++j)
        a[j] += lds[((r << 5) | (threadIdx.x & 31)) * 8 + j];
    float *row = partial + (long)blockIdx.x * C8 * 8;
#pragma unroll
    for (int j = 0; j < 8; ++j) row[cb * 8 + j] = a[j];
  }
}

// fallback for ragged N (e.g. the 2-way NSP head)
__global__ void colsum_k(const uint16_t *__restrict__ dy, float *__restrict__ db,
                         long M, int N, long ld) {
  for (int n = blockIdx.x * blockDim.x + threadIdx.x; n < N;
       n += gridDim.x * blockDim.x) {
    float a = 0;
    for (long m = 0; m < M; ++m) a += bf2f(dy[m * ld + n]);
    db[n] = a;
  }
}

// M-parallel ragged colsum (the single-thread-per-column fallback ran the
// MLM-head db — [4096][30522] — at 765 µs, 25× off bandwidth): 2-D grid of
// row-chunks × col-chunks into fp32 partials, then a per-column reduce.
__global__ void colsum_ragged_k(const uint16_t *__restrict__ dy,
                                float *__restrict__ partial, long M, int N,
                                long ld, long rchunk) {
  int n = blockIdx.y * 256 + threadIdx.x;
  if (n >= N) return;
  long r0 = blockIdx.x * rchunk, r1 = min(M, r0 + rchunk);
  float a = 0;
  for (long m = r0; m < r1; ++m) a += bf2f(dy[m * ld + n]);
  partial[(long)blockIdx.x * N + n] = a;
}

__global__ void colsum_ragged_reduce_k(const float *__restrict__ partial,
                                       float *__restrict__ db, int chunks,
                                       int N) {
  int n = blockIdx.x * 256 + threadIdx.x;
  if (n >= N) return;
  float a = 0;
  for (int g = 0; g < chunks; ++g) a += partial[(long)g * N + n];
  db[n] = a;
}

// row-chunk count of the fast path — the caller sizes the fp32 partial
// slab as [colsum_chunks(M,N)][N]
extern "C" int colsum_chunks(long M, int N) {
  if (N % 8 != 0) {
    if (N < 256 || M < 512) return 0; // tiny (NSP head): serial fallback
    int gy = (N + 255) / 256;
    long gx = 2048 / gy; // ~512 blocks in flight
    long maxgx = M;
    if (gx > maxgx) gx = maxgx;
    if (gx < 1) gx = 1;
    return (int)gx;
  }
  int gy = (N / 8 + 31) / 32;
  long gx = 1024 / gy;
  long maxgx = (M + 7) / 8;
  if (gx > maxgx) gx = maxgx;
  if (gx < 1) gx = 1;
  return (int)gx;
}

extern "C" hipError_t splitk_reduce(const float *partial, int splits, long len,
                                    void *out, int out_bf16,
                                    hipStream_t s); // conv.hip
extern "C" hipError_t slab_colreduce(const float *, float *, int, long,
                                     hipStream_t); // conv.hip

extern "C" hipError_t colsum_bf16(const void *dy, float *partial, float *db,
                                  long M, int N, long ld, hipStream_t s) {
  if (N % 8 == 0 && ld % 8 == 0) {
    int C8 = N / 8;
    int gy = (C8 + 31) / 32;
    int gx = colsum_chunks(M, N);
    colsum8_k<<<dim3(gx, gy), 256, 0, s>>>((const ushort8 *)dy, partial, M, C8,
                                           ld / 8);
    HIP_KERNEL_CHECK();
    // short output + many chunks: per-column reduce (the float4 kernel
    // collapsed to one block at len=N)
    return slab_colreduce(partial, db, gx, (long)N, s);
  }
  int gx = colsum_chunks(M, N);
  if (gx > 0) { // M-parallel ragged path (vocab-scale db)
    long rchunk = (M + gx - 1) / gx;
    dim3 grid(gx, (N + 255) / 256);
    colsum_ragged_k<<<grid, 256, 0, s>>>((const uint16_t *)dy, partial, M, N,
                                         ld, rchunk);
    HIP_KERNEL_CHECK();
    return slab_colreduce(partial, db, gx, (long)N, s);
  }
  colsum_k<<<cdiv_h(N, 256), 256, 0, s>>>((const uint16_t *)dy, db, M, N, ld);
  HIP_KERNEL_CHECK();
  return hipSuccess;
}
