// Fused BatchNorm(+ReLU) for NHWC bf16 activations, fp32 statistics.
// Training fwd = {partial per-channel sum/sumsq (deterministic, no atomics)
// → finalize (mean/invstd/scale/shift) → normalize+ReLU apply}.
// Backward = {partial sum(dy·mask), sum(dy·mask·xhat) → finalize coeffs →
// elementwise dx}. ReLU is folded into both directions (mask from y>0).
// Reference op being replaced: the external image's cuDNN BN+ReLU
// (SURVEY.md §2.3 N7).
#include "common.h"

// Per-thread fixed channel-group ownership: thread t owns channel block
// cb = t % C8 and row-lane t / C8 — consecutive lanes read consecutive
// 16 B groups (fully coalesced); accumulation stays in registers.
// Requires C8 = C/8 <= 256 (C <= 2048 — every ResNet/BERT channel width).
template <int WHAT> // 0: fwd stats (sum, sumsq); 1: bwd stats (dy*m, dy*m*xhat)
__global__ void bn_partials_k(const ushort8 *__restrict__ x,
                              const ushort8 *__restrict__ dy,
                              const ushort8 *__restrict__ y,
                              const uint8_t *__restrict__ mask,
                              const float *__restrict__ mean,
                              const float *__restrict__ invstd,
                              float *__restrict__ partial, // [grid][2][C]
                              long M, int C8, int relu) {
  int C = C8 * 8;
  int cb = threadIdx.x % C8;
  int row_lane = threadIdx.x / C8;
  int rows_per_block = blockDim.x / C8;
  float a0[8] = {0}, a1[8] = {0};
  float mn[8], is[8];
  if (WHAT == 1 && row_lane < rows_per_block) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      mn[j] = mean[cb * 8 + j];
      is[j] = invstd[cb * 8 + j];
    }
  }
  if (row_lane < rows_per_block) {
    long stride = (long)gridDim.x * rows_per_block;
    // two rows per iteration: two independent load chains in flight
    // (single-chain version sat ~80% wave-parked on HBM latency)
    for (long row = (long)blockIdx.x * rows_per_block + row_lane; row < M;
         row += 2 * stride) {
      long r2 = row + stride;
      bool has2 = r2 < M;
      long off = row * C8 + cb, off2 = r2 * C8 + cb;
      float fx[8], fx2[8];
      ushort8 vx = x[off];
      ushort8 vx2 = has2 ? x[off2] : ushort8{0, 0, 0, 0, 0, 0, 0, 0};
      bf8_to_f8(vx, fx);
      bf8_to_f8(vx2, fx2);
      if (WHAT == 0) {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          a0[j] += fx[j] + fx2[j];
          a1[j] += fx[j] * fx[j] + fx2[j] * fx2[j];
        }
      } else {
        ushort8 vdy = dy[off];
        ushort8 vdy2 = has2 ? dy[off2] : ushort8{0, 0, 0, 0, 0, 0, 0, 0};
        float fdy[8], fdy2[8];
        bf8_to_f8(vdy, fdy);
        bf8_to_f8(vdy2, fdy2);
        if (relu) {
          if (mask) {
            int m1 = mask[off], m2 = has2 ? mask[off2] : 0;
#pragma unroll
            for (int j = 0; j < 8; ++j) {
              if (!(m1 & (1 << j))) fdy[j] = 0.f;
              if (!(m2 & (1 << j))) fdy2[j] = 0.f;
            }
          } else {
            ushort8 vy = y[off];
            ushort8 vy2 = has2 ? y[off2] : ushort8{0, 0, 0, 0, 0, 0, 0, 0};
#pragma unroll
            for (int j = 0; j < 8; ++j) {
              if (!(bf2f(vy[j]) > 0.f)) fdy[j] = 0.f;
              if (!(bf2f(vy2[j]) > 0.f)) fdy2[j] = 0.f;
            }
          }
        }
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          a0[j] += fdy[j] + fdy2[j];
          a1[j] += fdy[j] * ((fx[j] - mn[j]) * is[j]) +
                   fdy2[j] * ((fx2[j] - mn[j]) * is[j]);
        }
      }
    }
  }
  // block reduce across row lanes via LDS
  __shared__ float lds[2][256 * 8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    lds[0][threadIdx.x * 8 + j] = a0[j];
    lds[1][threadIdx.x * 8 + j] = a1[j];
  }
  __syncthreads();
  if (row_lane == 0) {
    for (int rl = 1; rl < rows_per_block; ++rl) {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        a0[j] += lds[0][(rl * C8 + cb) * 8 + j];
        a1[j] += lds[1][(rl * C8 + cb) * 8 + j];
      }
    }
    float *p0 = partial + (long)blockIdx.x * 2 * C + cb * 8;
    float *p1 = p0 + C;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      p0[j] = a0[j];
      p1[j] = a1[j];
    }
  }
}

// finalize: 8 lanes cooperate per channel (grid entries split across lanes,
// shfl-reduced) — the serial-per-channel version was one-wave latency-bound
// at 124 µs/call and 31% of the whole step.
DEV_INLINE void lane8_sums(const float *__restrict__ partial, int grid, int C,
                           int c, int lane8, float &s0, float &s1) {
  s0 = 0.f;
  s1 = 0.f;
  for (int g = lane8; g < grid; g += 32) {
    s0 += partial[(long)g * 2 * C + c];
    s1 += partial[(long)g * 2 * C + C + c];
  }
#pragma unroll
  for (int off = 16; off > 0; off >>= 1) {
    s0 += __shfl_down(s0, off, 32);
    s1 += __shfl_down(s1, off, 32);
  }
}

// fwd: mean/invstd + scale/shift + fused running-stats update (saves ~5 eager
// tensor ops per BN layer per step on the torch side).
__global__ void bn_finalize_fwd_k(const float *__restrict__ partial, int grid,
                                  int C, const float *__restrict__ gamma,
                                  const float *__restrict__ beta, float inv_m,
                                  float eps, float *__restrict__ mean,
                                  float *__restrict__ invstd,
                                  float *__restrict__ scale,
                                  float *__restrict__ shift,
                                  float *__restrict__ running_mean,
                                  float *__restrict__ running_var,
                                  float momentum, float unbias) {
  int t = blockIdx.x * blockDim.x + threadIdx.x;
  int c = t / 32, lane8 = t % 32;
  if (c >= C) return;
  float s, sq;
  lane8_sums(partial, grid, C, c, lane8, s, sq);
  if (lane8 != 0) return;
  float mu = s * inv_m;
  float var = fmaxf(sq * inv_m - mu * mu, 0.f);
  float is = rsqrtf(var + eps);
  mean[c] = mu;
  invstd[c] = is;
  float sc = gamma[c] * is;
  scale[c] = sc;
  shift[c] = beta[c] - mu * sc;
  if (running_mean) {
    running_mean[c] = running_mean[c] * (1.f - momentum) + mu * momentum;
    running_var[c] = running_var[c] * (1.f - momentum) + var * unbias * momentum;
  }
}

// bwd: dbeta/dgamma + the three per-channel dx coefficients
__global__ void bn_finalize_bwd_k(const float *__restrict__ partial, int grid,
                                  int C, const float *__restrict__ gamma,
                                  const float *__restrict__ invstd, float inv_m,
                                  float *__restrict__ dbeta,
                                  float *__restrict__ dgamma,
                                  float *__restrict__ k1, // gamma*invstd
                                  float *__restrict__ k2, // k1*dbeta/m
                                  float *__restrict__ k3) { // k1*dgamma/m (×xhat in apply)
  int t = blockIdx.x * blockDim.x + threadIdx.x;
  int c = t / 32, lane8 = t % 32;
  if (c >= C) return;
  float s0, s1;
  lane8_sums(partial, grid, C, c, lane8, s0, s1);
  if (lane8 != 0) return;
  dbeta[c] = s0;
  dgamma[c] = s1;
  float g_is = gamma[c] * invstd[c];
  k1[c] = g_is;
  k2[c] = g_is * s0 * inv_m;
  k3[c] = g_is * s1 * inv_m;
}

__global__ void bn_finalize_fwd_bands_k(
    const float *__restrict__ partial, int grid, int C,
    const float *__restrict__ gamma, const float *__restrict__ beta,
    float inv_m, float eps, float *__restrict__ mean,
    float *__restrict__ invstd, float *__restrict__ scale,
    float *__restrict__ shift, float *__restrict__ running_mean,
    float *__restrict__ running_var, float momentum,
    float unbias); // defined below (shared with the conv-epilogue pre path)

// bwd finalize, bands layout: one 256-thread block per channel striding
// the slab rows — the lane8 version gave each channel only 32 lanes and
// just C/8 blocks of parallelism (8 blocks at C=64), and its strided
// cross-XCD slab reads ran ~8 us/call x 104 BN layers x both directions.
__global__ void bn_finalize_bwd_bands_k(
    const float *__restrict__ partial, int grid, int C,
    const float *__restrict__ gamma, const float *__restrict__ invstd,
    float inv_m, float *__restrict__ dbeta, float *__restrict__ dgamma,
    float *__restrict__ k1, float *__restrict__ k2, float *__restrict__ k3) {
  int c = blockIdx.x;
  if (c >= C) return;
  float s0 = 0.f, s1 = 0.f;
  for (int g = threadIdx.x; g < grid; g += 256) {
    s0 += partial[(long)g * 2 * C + c];
    s1 += partial[(long)g * 2 * C + C + c];
  }
  __shared__ float red[2][256 / WAVE];
  s0 = wave_sum(s0);
  s1 = wave_sum(s1);
  if ((threadIdx.x & (WAVE - 1)) == 0) {
    red[0][threadIdx.x / WAVE] = s0;
    red[1][threadIdx.x / WAVE] = s1;
  }
  __syncthreads();
  if (threadIdx.x != 0) return;
  s0 = red[0][0] + red[0][1] + red[0][2] + red[0][3];
  s1 = red[1][0] + red[1][1] + red[1][2] + red[1][3];
  dbeta[c] = s0;
  dgamma[c] = s1;
  float g_is = gamma[c] * invstd[c];
  k1[c] = g_is;
  k2[c] = g_is * s0 * inv_m;
  k3[c] = g_is * s1 * inv_m;
}

// apply scale/shift (+optional residual add) (+ReLU): fwd-train, fwd-eval
// and the bottleneck-join fusion (res != nullptr folds the skip connection
// into this pass — one fewer full activation read+write per block).
__global__ void bn_apply_k(const ushort8 *__restrict__ x,
                           const ushort8 *__restrict__ res,
                           const float *__restrict__ scale,
                           const float *__restrict__ shift,
                           ushort8 *__restrict__ y,
                           uint8_t *__restrict__ mask, // y>0 bits, [M][C8]
                           long M, int C8, int relu) {
  int cb = threadIdx.x % C8;
  int row_lane = threadIdx.x / C8;
  int rows_per_block = blockDim.x / C8;
  if (row_lane >= rows_per_block) return;
  float sc[8], sh[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    sc[j] = scale[cb * 8 + j];
    sh[j] = shift[cb * 8 + j];
  }
  long stride = (long)gridDim.x * rows_per_block;
  for (long row = (long)blockIdx.x * rows_per_block + row_lane; row < M;
       row += 2 * stride) {
    long r2 = row + stride;
    long off = row * C8 + cb, off2 = r2 * C8 + cb;
    ushort8 v = x[off];
    ushort8 v2 = r2 < M ? x[off2] : ushort8{0, 0, 0, 0, 0, 0, 0, 0};
    float f[8], f2[8], g[8] = {0}, g2[8] = {0};
    bf8_to_f8(v, f);
    bf8_to_f8(v2, f2);
    if (res) {
      ushort8 w = res[off];
      ushort8 w2 = r2 < M ? res[off2] : ushort8{0, 0, 0, 0, 0, 0, 0, 0};
      bf8_to_f8(w, g);
      bf8_to_f8(w2, g2);
    }
    int m1 = 0, m2 = 0;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      f[j] = f[j] * sc[j] + sh[j] + g[j];
      f2[j] = f2[j] * sc[j] + sh[j] + g2[j];
      if (relu) {
        // mask records the PRE-clamp sign so backward's dy gating matches
        // the y>0 test it replaces (bf16-rounded y>0 iff f>0 here: rounding
        // keeps sign and relu output is never negative)
        if (f[j] > 0.f) m1 |= 1 << j;
        if (f2[j] > 0.f) m2 |= 1 << j;
        f[j] = fmaxf(f[j], 0.f);
        f2[j] = fmaxf(f2[j], 0.f);
      }
    }
    y[off] = f8_to_bf8(f);
    if (r2 < M) y[off2] = f8_to_bf8(f2);
    if (relu && mask) {
      mask[off] = (uint8_t)m1;
      if (r2 < M) mask[off2] = (uint8_t)m2;
    }
  }
}

// dx = k1*dy·mask - k2 - k3*xhat, xhat = (x-mean)*invstd
__global__ void bn_bwd_apply_k(const ushort8 *__restrict__ dy,
                               const ushort8 *__restrict__ x,
                               const ushort8 *__restrict__ y,
                               const uint8_t *__restrict__ mask,
                               const float *__restrict__ mean,
                               const float *__restrict__ invstd,
                               const float *__restrict__ k1,
                               const float *__restrict__ k2,
                               const float *__restrict__ k3,
                               ushort8 *__restrict__ dx, long M, int C8,
                               int relu) {
  int cb = threadIdx.x % C8;
  int row_lane = threadIdx.x / C8;
  int rows_per_block = blockDim.x / C8;
  if (row_lane >= rows_per_block) return;
  float mn[8], is[8], a[8], b[8], c3[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    int c = cb * 8 + j;
    mn[j] = mean[c];
    is[j] = invstd[c];
    a[j] = k1[c];
    b[j] = k2[c];
    c3[j] = k3[c];
  }
  long stride = (long)gridDim.x * rows_per_block;
  for (long row = (long)blockIdx.x * rows_per_block + row_lane; row < M;
       row += 2 * stride) {
    long r2 = row + stride;
    long off = row * C8 + cb, off2 = r2 * C8 + cb;
    ushort8 vdy = dy[off], vx = x[off];
    ushort8 vdy2 = r2 < M ? dy[off2] : ushort8{0, 0, 0, 0, 0, 0, 0, 0};
    ushort8 vx2 = r2 < M ? x[off2] : ushort8{0, 0, 0, 0, 0, 0, 0, 0};
    float fdy[8], fx[8], fdy2[8], fx2[8];
    bf8_to_f8(vdy, fdy);
    bf8_to_f8(vx, fx);
    bf8_to_f8(vdy2, fdy2);
    bf8_to_f8(vx2, fx2);
    if (relu) {
      if (mask) { // 1-byte relu mask replaces the 16-byte y re-read
        int m1 = mask[off], m2 = r2 < M ? mask[off2] : 0;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          if (!(m1 & (1 << j))) fdy[j] = 0.f;
          if (!(m2 & (1 << j))) fdy2[j] = 0.f;
        }
      } else {
        ushort8 vy = y[off];
        ushort8 vy2 = r2 < M ? y[off2] : ushort8{0, 0, 0, 0, 0, 0, 0, 0};
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          if (!(bf2f(vy[j]) > 0.f)) fdy[j] = 0.f;
          if (!(bf2f(vy2[j]) > 0.f)) fdy2[j] = 0.f;
        }
      }
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      fdy[j] = a[j] * fdy[j] - b[j] - c3[j] * ((fx[j] - mn[j]) * is[j]);
      fdy2[j] = a[j] * fdy2[j] - b[j] - c3[j] * ((fx2[j] - mn[j]) * is[j]);
    }
    dx[off] = f8_to_bf8(fdy);
    if (r2 < M) dx[off2] = f8_to_bf8(fdy2);
  }
}

// Partials grid cap: 512 measured best THREE times now — flat 1024 lost
// 6.6%, and the channel-aware cap (up to 4096 bands for narrow layers)
// lost 4.4% (3363 vs 3517 same-box) even with the bands finalize: the
// deeper slab costs more in finalize reads + L2 pressure than the extra
// partials parallelism buys. MPIAMD_BN_GRID overrides for A/Bs; bindings
// size the slab with bn_grid_cap — keep them in sync.
extern "C" int bn_grid_cap(int C8) {
  static const long ovr = [] {
    const char *e = getenv("MPIAMD_BN_GRID");
    long v = e ? atol(e) : 0;
    return (v >= 1 && v <= 8192) ? v : 0;
  }();
  (void)C8;
  return ovr ? (int)ovr : 512;
}

static void bn_geom(long M, int C8, int &grid, int &rows_per_block) {
  rows_per_block = 256 / C8;
  long g = (M + rows_per_block - 1) / rows_per_block;
  long cap = bn_grid_cap(C8);
  grid = (int)(g > cap ? cap : (g < 1 ? 1 : g));
}

static int bn_apply_grid(long M, int C8) {
  // cap 2048 measured best (3846-3854 vs 3833-3834 img/s at 4096, 3800
  // at 8192, four-sample same-box sweep); MPIAMD_BN_APPLY_GRID overrides
  static const long cap = [] {
    const char *e = getenv("MPIAMD_BN_APPLY_GRID");
    long v = e ? atol(e) : 0;
    return (v >= 64 && v <= 16384) ? v : 2048L;
  }();
  int rpb = 256 / C8;
  long g = (M + rpb - 1) / rpb;
  return (int)(g > cap ? cap : (g < 1 ? 1 : g));
}

extern "C" hipError_t bn_fwd_train_launch(
    const void *x, const void *res, const float *gamma, const float *beta,
    float eps, int relu, void *y, uint8_t *relu_mask, float *mean,
    float *invstd, float *scale, float *shift, float *partial,
    float *running_mean, float *running_var, float momentum, long M, int C,
    hipStream_t s) {
  int C8 = C / 8;
  if (C8 < 1 || C8 > 256) return hipErrorInvalidValue;
  int grid, rpb;
  bn_geom(M, C8, grid, rpb);
  bn_partials_k<0><<<grid, 256, 0, s>>>((const ushort8 *)x, nullptr, nullptr,
                                        nullptr, nullptr, nullptr, partial, M,
                                        C8, 0);
  HIP_KERNEL_CHECK();
  float unbias = M > 1 ? (float)M / (float)(M - 1) : 1.f;
  bn_finalize_fwd_bands_k<<<C, 256, 0, s>>>(
      partial, grid, C, gamma, beta, 1.f / (float)M, eps, mean, invstd, scale,
      shift, running_mean, running_var, momentum, unbias);
  HIP_KERNEL_CHECK();
  bn_apply_k<<<bn_apply_grid(M, C8), 256, 0, s>>>(
      (const ushort8 *)x, (const ushort8 *)res, scale, shift, (ushort8 *)y,
      relu_mask, M, C8, relu);
  HIP_KERNEL_CHECK();
  return hipSuccess;
}

// Band-slab finalize: the conv-epilogue slab has ceil(M/64) entries (up to
// ~3.1k at ResNet stage-1 sizes) vs the partials path's ≤512 — the 32-lane
// lane8_sums loop went latency-bound there. 256 lanes per channel: 12
// iterations and a full-chip grid.
__global__ void bn_finalize_fwd_bands_k(
    const float *__restrict__ partial, int grid, int C,
    const float *__restrict__ gamma, const float *__restrict__ beta,
    float inv_m, float eps, float *__restrict__ mean,
    float *__restrict__ invstd, float *__restrict__ scale,
    float *__restrict__ shift, float *__restrict__ running_mean,
    float *__restrict__ running_var, float momentum, float unbias) {
  int c = blockIdx.x; // one block (256 threads) per channel
  if (c >= C) return;
  float s = 0.f, sq = 0.f;
  for (int g = threadIdx.x; g < grid; g += 256) {
    s += partial[(long)g * 2 * C + c];
    sq += partial[(long)g * 2 * C + C + c];
  }
  __shared__ float red[2][256 / WAVE];
  s = wave_sum(s);
  sq = wave_sum(sq);
  if ((threadIdx.x & (WAVE - 1)) == 0) {
    red[0][threadIdx.x / WAVE] = s;
    red[1][threadIdx.x / WAVE] = sq;
  }
  __syncthreads();
  if (threadIdx.x != 0) return;
  s = red[0][0] + red[0][1] + red[0][2] + red[0][3];
  sq = red[1][0] + red[1][1] + red[1][2] + red[1][3];
  float mu = s * inv_m;
  float var = fmaxf(sq * inv_m - mu * mu, 0.f);
  float is = rsqrtf(var + eps);
  mean[c] = mu;
  invstd[c] = is;
  float sc = gamma[c] * is;
  scale[c] = sc;
  shift[c] = beta[c] - mu * sc;
  if (running_mean) {
    running_mean[c] = running_mean[c] * (1.f - momentum) + mu * momentum;
    running_var[c] = running_var[c] * (1.f - momentum) + var * unbias * momentum;
  }
}

// Variant taking PRE-COMPUTED partials (the conv epilogue's BnStatsWriter
// slab, [pre_grid][2][C]) — skips bn_partials' full activation re-read.
extern "C" hipError_t bn_fwd_train_pre_launch(
    const float *pre_partial, int pre_grid, const void *x, const void *res,
    const float *gamma, const float *beta, float eps, int relu, void *y,
    uint8_t *relu_mask, float *mean, float *invstd, float *scale,
    float *shift, float *running_mean, float *running_var, float momentum,
    long M, int C, hipStream_t s) {
  int C8 = C / 8;
  if (C8 < 1 || C8 > 256) return hipErrorInvalidValue;
  float unbias = M > 1 ? (float)M / (float)(M - 1) : 1.f;
  bn_finalize_fwd_bands_k<<<C, 256, 0, s>>>(
      pre_partial, pre_grid, C, gamma, beta, 1.f / (float)M, eps, mean,
      invstd, scale, shift, running_mean, running_var, momentum, unbias);
  HIP_KERNEL_CHECK();
  bn_apply_k<<<bn_apply_grid(M, C8), 256, 0, s>>>(
      (const ushort8 *)x, (const ushort8 *)res, scale, shift, (ushort8 *)y,
      relu_mask, M, C8, relu);
  HIP_KERNEL_CHECK();
  return hipSuccess;
}

extern "C" hipError_t bn_fwd_eval_launch(const void *x, const float *scale,
                                         const float *shift, int relu, void *y,
                                         long M, int C, hipStream_t s) {
  int C8 = C / 8;
  if (C8 < 1 || C8 > 256) return hipErrorInvalidValue;
  int grid, rpb;
  bn_geom(M, C8, grid, rpb);
  bn_apply_k<<<bn_apply_grid(M, C8), 256, 0, s>>>(
      (const ushort8 *)x, nullptr, scale, shift, (ushort8 *)y, nullptr, M, C8,
      relu);
  HIP_KERNEL_CHECK();
  return hipSuccess;
}

extern "C" hipError_t bn_bwd_launch(const void *dy, const void *x,
                                    const void *y, const uint8_t *relu_mask,
                                    const float *gamma, const float *mean,
                                    const float *invstd, int relu, void *dx,
                                    float *dgamma, float *dbeta, float *k1,
                                    float *k2, float *k3, float *partial,
                                    long M, int C, hipStream_t s) {
  int C8 = C / 8;
  if (C8 < 1 || C8 > 256) return hipErrorInvalidValue;
  int grid, rpb;
  bn_geom(M, C8, grid, rpb);
  bn_partials_k<1><<<grid, 256, 0, s>>>((const ushort8 *)x, (const ushort8 *)dy,
                                        (const ushort8 *)y, relu_mask, mean,
                                        invstd, partial, M, C8, relu);
  HIP_KERNEL_CHECK();
  bn_finalize_bwd_bands_k<<<C, 256, 0, s>>>(
      partial, grid, C, gamma, invstd, 1.f / (float)M, dbeta, dgamma, k1, k2, k3);
  HIP_KERNEL_CHECK();
  bn_bwd_apply_k<<<bn_apply_grid(M, C8), 256, 0, s>>>(
      (const ushort8 *)dy, (const ushort8 *)x, (const ushort8 *)y, relu_mask,
      mean, invstd, k1, k2, k3, (ushort8 *)dx, M, C8, relu);
  HIP_KERNEL_CHECK();
  return hipSuccess;
}
