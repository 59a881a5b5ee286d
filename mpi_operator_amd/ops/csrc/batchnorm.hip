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
// The finalize runs inside the LAST-ARRIVING block of this kernel (plain
// slab stores -> agent release fence -> relaxed ticket; the last block takes
// an agent acquire and reduces the L2-hot slab) — one launch instead of
// two per BN direction. Counter is a persistent device int, reset by the
// last arriver so the next launch (or graph replay) starts from 0.
template <int WHAT> // 0: fwd stats (sum, sumsq); 1: bwd stats (dy*m, dy*m*xhat)
__global__ void bn_partials_k(const ushort8 *__restrict__ x,
                              const ushort8 *__restrict__ dy,
                              const ushort8 *__restrict__ y,
                              const float *__restrict__ mean_in,
                              const float *__restrict__ invstd_in,
                              float *__restrict__ partial, // [grid][2][C]
                              long M, int C8, int relu, int *__restrict__ cnt,
                              const float *__restrict__ gamma,
                              const float *__restrict__ beta, float eps,
                              float *__restrict__ mean_out,
                              float *__restrict__ invstd_out,
                              float *__restrict__ scale,
                              float *__restrict__ shift,
                              float *__restrict__ running_mean,
                              float *__restrict__ running_var, float momentum,
                              float *__restrict__ dbeta,
                              float *__restrict__ dgamma,
                              float *__restrict__ k1, float *__restrict__ k2,
                              float *__restrict__ k3) {
  const float *mean = mean_in;
  const float *invstd = invstd_in;
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
          ushort8 vy = y[off];
          ushort8 vy2 = has2 ? y[off2] : ushort8{0, 0, 0, 0, 0, 0, 0, 0};
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            if (!(bf2f(vy[j]) > 0.f)) fdy[j] = 0.f;
            if (!(bf2f(vy2[j]) > 0.f)) fdy2[j] = 0.f;
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
    // slab publish with WRITE-THROUGH (sc1) stores: an agent release fence
    // here (buffer_wbl2) flushed every XCD's whole dirty L2 per block and
    // made the WHOLE STEP 3.4x slower — sc1 stores skip L1/L2 dirty state,
    // and the sc1 reader needs no acquire (guide G16 publish forms)
    float *p0 = partial + (long)blockIdx.x * 2 * C + cb * 8;
    float *p1 = p0 + C;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      __hip_atomic_store(&p0[j], a0[j], __ATOMIC_RELAXED,
                         __HIP_MEMORY_SCOPE_AGENT);
      __hip_atomic_store(&p1[j], a1[j], __ATOMIC_RELAXED,
                         __HIP_MEMORY_SCOPE_AGENT);
    }
  }
  // ---- last-arriver finalize (in-launch combine) ----
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  int *flag = (int *)&lds[0][0]; // reuse the ONE shared object
  if (threadIdx.x == 0) {
    int old = __hip_atomic_fetch_add(cnt, 1, __ATOMIC_RELAXED,
                                     __HIP_MEMORY_SCOPE_AGENT);
    *flag = (old == (int)gridDim.x - 1) ? 1 : 0;
  }
  __syncthreads();
  if (*flag == 0) return;
  if (threadIdx.x == 0) {
    // ONE agent acquire for the whole read phase (drops this CU's L1; the
    // sc1-published slab is L2-resident) — per-element atomic loads cannot
    // be pipelined by the compiler and serialized ~1000 L2 round trips
    __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
    *cnt = 0; // next launch / graph replay starts clean
  }
  __syncthreads();
  int grid = gridDim.x;
  float inv_m = 1.f / (float)M;
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    float s0 = 0.f, s1 = 0.f;
    for (int g = 0; g < grid; ++g) {
      s0 += partial[(long)g * 2 * C + c];
      s1 += partial[(long)g * 2 * C + C + c];
    }
    if (WHAT == 0) {
      float mu = s0 * inv_m;
      float var = fmaxf(s1 * inv_m - mu * mu, 0.f);
      float is = rsqrtf(var + eps);
      mean_out[c] = mu;
      invstd_out[c] = is;
      float sc = gamma[c] * is;
      scale[c] = sc;
      shift[c] = beta[c] - mu * sc;
      if (running_mean) {
        float unbias = M > 1 ? (float)M / (float)(M - 1) : 1.f;
        running_mean[c] = running_mean[c] * (1.f - momentum) + mu * momentum;
        running_var[c] =
            running_var[c] * (1.f - momentum) + var * unbias * momentum;
      }
    } else {
      dbeta[c] = s0;
      dgamma[c] = s1;
      float g_is = gamma[c] * invstd[c];
      k1[c] = g_is;
      k2[c] = g_is * s0 * inv_m;
      k3[c] = g_is * s1 * inv_m;
    }
  }
}

// apply scale/shift (+optional residual add) (+ReLU): fwd-train, fwd-eval
// and the bottleneck-join fusion (res != nullptr folds the skip connection
// into this pass — one fewer full activation read+write per block).
__global__ void bn_apply_k(const ushort8 *__restrict__ x,
                           const ushort8 *__restrict__ res,
                           const float *__restrict__ scale,
                           const float *__restrict__ shift,
                           ushort8 *__restrict__ y, long M, int C8, int relu) {
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
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      f[j] = f[j] * sc[j] + sh[j] + g[j];
      f2[j] = f2[j] * sc[j] + sh[j] + g2[j];
      if (relu) {
        f[j] = fmaxf(f[j], 0.f);
        f2[j] = fmaxf(f2[j], 0.f);
      }
    }
    y[off] = f8_to_bf8(f);
    if (r2 < M) y[off2] = f8_to_bf8(f2);
  }
}

// dx = k1*dy·mask - k2 - k3*xhat, xhat = (x-mean)*invstd
__global__ void bn_bwd_apply_k(const ushort8 *__restrict__ dy,
                               const ushort8 *__restrict__ x,
                               const ushort8 *__restrict__ y,
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
      ushort8 vy = y[off];
      ushort8 vy2 = r2 < M ? y[off2] : ushort8{0, 0, 0, 0, 0, 0, 0, 0};
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        if (!(bf2f(vy[j]) > 0.f)) fdy[j] = 0.f;
        if (!(bf2f(vy2[j]) > 0.f)) fdy2[j] = 0.f;
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

// Partials grid is capped by the slab ([512][2][C]); the elementwise
// apply kernels have no slab and get a bandwidth-sized grid (256 CUs want
// >=2 blocks/CU in flight to cover HBM latency).
static void bn_geom(long M, int C8, int &grid, int &rows_per_block) {
  rows_per_block = 256 / C8;
  long g = (M + rows_per_block - 1) / rows_per_block;
  grid = (int)(g > 512 ? 512 : (g < 1 ? 1 : g));
}

static int bn_apply_grid(long M, int C8) {
  int rpb = 256 / C8;
  long g = (M + rpb - 1) / rpb;
  return (int)(g > 4096 ? 4096 : (g < 1 ? 1 : g));
}

extern "C" hipError_t bn_fwd_train_launch(
    const void *x, const void *res, const float *gamma, const float *beta,
    float eps, int relu, void *y, float *mean, float *invstd, float *scale,
    float *shift, float *partial, int *cnt, float *running_mean,
    float *running_var, float momentum, long M, int C, hipStream_t s) {
  int C8 = C / 8;
  if (C8 < 1 || C8 > 256) return hipErrorInvalidValue;
  int grid, rpb;
  bn_geom(M, C8, grid, rpb);
  bn_partials_k<0><<<grid, 256, 0, s>>>(
      (const ushort8 *)x, nullptr, nullptr, nullptr, nullptr, partial, M, C8,
      0, cnt, gamma, beta, eps, mean, invstd, scale, shift, running_mean,
      running_var, momentum, nullptr, nullptr, nullptr, nullptr, nullptr);
  HIP_KERNEL_CHECK();
  bn_apply_k<<<bn_apply_grid(M, C8), 256, 0, s>>>(
      (const ushort8 *)x, (const ushort8 *)res, scale, shift, (ushort8 *)y, M,
      C8, relu);
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
      (const ushort8 *)x, nullptr, scale, shift, (ushort8 *)y, M, C8, relu);
  HIP_KERNEL_CHECK();
  return hipSuccess;
}

extern "C" hipError_t bn_bwd_launch(const void *dy, const void *x,
                                    const void *y, const float *gamma,
                                    const float *mean, const float *invstd,
                                    int relu, void *dx, float *dgamma,
                                    float *dbeta, float *k1, float *k2,
                                    float *k3, float *partial, int *cnt,
                                    long M, int C, hipStream_t s) {
  int C8 = C / 8;
  if (C8 < 1 || C8 > 256) return hipErrorInvalidValue;
  int grid, rpb;
  bn_geom(M, C8, grid, rpb);
  bn_partials_k<1><<<grid, 256, 0, s>>>(
      (const ushort8 *)x, (const ushort8 *)dy, (const ushort8 *)y, mean,
      invstd, partial, M, C8, relu, cnt, gamma, nullptr, 0.f, nullptr, nullptr,
      nullptr, nullptr, nullptr, nullptr, 0.f, dbeta, dgamma, k1, k2, k3);
  HIP_KERNEL_CHECK();
  bn_bwd_apply_k<<<bn_apply_grid(M, C8), 256, 0, s>>>(
      (const ushort8 *)dy, (const ushort8 *)x, (const ushort8 *)y, mean,
      invstd, k1, k2, k3, (ushort8 *)dx, M, C8, relu);
  HIP_KERNEL_CHECK();
  return hipSuccess;
}
