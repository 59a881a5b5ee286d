// Fused LayerNorm for bf16 [M rows][N features], fp32 gamma/beta and
// statistics — the transformer-side normalization (BERT runs 50+ LN
// fwd+bwd pairs per step; the eager torch path was ~20% of a BERT step).
//
// One WAVE owns one row: per-lane ushort8 chunks (octets strided by 64
// lanes), row sums via 6-step shfl_xor reduction — no LDS on the hot path.
// Backward's dgamma/dbeta use per-block fp32 partial slabs reduced by the
// existing splitk_reduce kernel (atomics on N addresses would serialize).
#include "common.h"

// y = (x - mean) * rstd * gamma + beta;  saves mean/rstd per row
__global__ void ln_fwd_k(const ushort8 *__restrict__ x,
                         const float *__restrict__ gamma,
                         const float *__restrict__ beta,
                         ushort8 *__restrict__ y, float *__restrict__ mean,
                         float *__restrict__ rstd, long M, int C8, float eps) {
  int lane = threadIdx.x & 63;
  int wave = threadIdx.x >> 6;
  int waves = blockDim.x >> 6;
  int N = C8 * 8;
  float inv_n = 1.f / (float)N;
  for (long row = (long)blockIdx.x * waves + wave; row < M;
       row += (long)gridDim.x * waves) {
    const ushort8 *xr = x + row * C8;
    float s = 0.f, sq = 0.f;
    ushort8 cx[2]; // row cached in registers (2 octets/lane at N<=1024):
                   // the stats pass otherwise re-reads the whole row
    for (int c = lane; c < C8; c += 64) {
      int ci = (c - lane) >> 6;
      ushort8 v = xr[c];
      if (ci < 2) cx[ci] = v;
      float f[8];
      bf8_to_f8(v, f);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        s += f[j];
        sq += f[j] * f[j];
      }
    }
    s = wave_sum(s);
    sq = wave_sum(sq);
    float mu = s * inv_n;
    float var = fmaxf(sq * inv_n - mu * mu, 0.f);
    float rs = rsqrtf(var + eps);
    if (lane == 0) {
      mean[row] = mu;
      rstd[row] = rs;
    }
    ushort8 *yr = y + row * C8;
    for (int c = lane; c < C8; c += 64) {
      int ci = (c - lane) >> 6;
      float f[8];
      bf8_to_f8(ci < 2 ? cx[ci] : xr[c], f);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        f[j] = (f[j] - mu) * rs * gamma[c * 8 + j] + beta[c * 8 + j];
      yr[c] = f8_to_bf8(f);
    }
  }
}

// dx = rstd * (g*dy - mean(g*dy) - xhat * mean(g*dy*xhat));
// partial[block][0][N] += dy*xhat (dgamma), partial[block][1][N] += dy (dbeta)
__global__ void ln_bwd_k(const ushort8 *__restrict__ dy,
                         const ushort8 *__restrict__ x,
                         const float *__restrict__ gamma,
                         const float *__restrict__ mean,
                         const float *__restrict__ rstd,
                         ushort8 *__restrict__ dx,
                         float *__restrict__ partial, long M, int C8) {
  int lane = threadIdx.x & 63;
  int wave = threadIdx.x >> 6;
  int waves = blockDim.x >> 6;
  int N = C8 * 8;
  float inv_n = 1.f / (float)N;
  // per-thread fp32 col partials (up to 4 octets per lane at N<=2048)
  float pg[4][8] = {}, pb[4][8] = {};
  for (long row = (long)blockIdx.x * waves + wave; row < M;
       row += (long)gridDim.x * waves) {
    const ushort8 *xr = x + row * C8;
    const ushort8 *dr = dy + row * C8;
    float mu = mean[row], rs = rstd[row];
    float s1 = 0.f, s2 = 0.f;
    ushort8 cx[2], cd[2]; // row cached (2 octets/lane at N<=1024): the dx
                          // pass was re-reading x AND dy — the measured gap
                          // that kept fused LN behind torch native
    for (int c = lane; c < C8; c += 64) {
      int ci = (c - lane) >> 6;
      ushort8 vx = xr[c], vd = dr[c];
      if (ci < 2) {
        cx[ci] = vx;
        cd[ci] = vd;
      }
      float fx[8], fd[8];
      bf8_to_f8(vx, fx);
      bf8_to_f8(vd, fd);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float xh = (fx[j] - mu) * rs;
        float gd = gamma[c * 8 + j] * fd[j];
        s1 += gd;
        s2 += gd * xh;
        if (ci < 4) {
          pg[ci][j] += fd[j] * xh;
          pb[ci][j] += fd[j];
        }
      }
    }
    s1 = wave_sum(s1) * inv_n;
    s2 = wave_sum(s2) * inv_n;
    ushort8 *dxr = dx + row * C8;
    for (int c = lane; c < C8; c += 64) {
      int ci = (c - lane) >> 6;
      float fx[8], fd[8];
      bf8_to_f8(ci < 2 ? cx[ci] : xr[c], fx);
      bf8_to_f8(ci < 2 ? cd[ci] : dr[c], fd);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float xh = (fx[j] - mu) * rs;
        fd[j] = rs * (gamma[c * 8 + j] * fd[j] - s1 - xh * s2);
      }
      dxr[c] = f8_to_bf8(fd);
    }
  }
  // fold this block's per-thread partials into its slab rows via LDS
  __shared__ float lds[1024 * 8]; // sized for the 16-wave backward block
  float *slab_g = partial + (long)blockIdx.x * 2 * N;
  float *slab_b = slab_g + N;
#pragma unroll
  for (int ci = 0; ci < 4; ++ci) {
    int c = ci * 64 + lane;
    // reduce across the block's waves one octet-bank at a time
#pragma unroll
    for (int j = 0; j < 8; ++j) lds[threadIdx.x * 8 + j] = pg[ci][j];
    __syncthreads();
    if (wave == 0 && c < C8) {
      float acc[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) acc[j] = lds[lane * 8 + j];
      for (int w = 1; w < waves; ++w)
#pragma unroll
        for (int j = 0; j < 8; ++j) acc[j] += lds[(w * 64 + lane) * 8 + j];
#pragma unroll
      for (int j = 0; j < 8; ++j) slab_g[c * 8 + j] = acc[j];
    }
    __syncthreads();
#pragma unroll
    for (int j = 0; j < 8; ++j) lds[threadIdx.x * 8 + j] = pb[ci][j];
    __syncthreads();
    if (wave == 0 && c < C8) {
      float acc[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) acc[j] = lds[lane * 8 + j];
      for (int w = 1; w < waves; ++w)
#pragma unroll
        for (int j = 0; j < 8; ++j) acc[j] += lds[(w * 64 + lane) * 8 + j];
#pragma unroll
      for (int j = 0; j < 8; ++j) slab_b[c * 8 + j] = acc[j];
    }
    __syncthreads();
  }
}

// Residual-join + LayerNorm in one pass: s = a + b (written — backward
// reads it as the LN input), y = LN(s). Replaces the separate residual
// add's extra read+write of the [M][N] activation (BERT runs 2 joins per
// layer per direction).
__global__ void ln_fwd_add_k(const ushort8 *__restrict__ a,
                             const ushort8 *__restrict__ b,
                             const float *__restrict__ gamma,
                             const float *__restrict__ beta,
                             ushort8 *__restrict__ sum_out,
                             ushort8 *__restrict__ y, float *__restrict__ mean,
                             float *__restrict__ rstd, long M, int C8,
                             float eps) {
  int lane = threadIdx.x & 63;
  int wave = threadIdx.x >> 6;
  int waves = blockDim.x >> 6;
  int N = C8 * 8;
  float inv_n = 1.f / (float)N;
  for (long row = (long)blockIdx.x * waves + wave; row < M;
       row += (long)gridDim.x * waves) {
    const ushort8 *ar = a + row * C8;
    const ushort8 *br = b + row * C8;
    ushort8 *sr = sum_out + row * C8;
    float s = 0.f, sq = 0.f;
    ushort8 cs[2];
    for (int c = lane; c < C8; c += 64) {
      int ci = (c - lane) >> 6;
      float fa[8], fb[8];
      bf8_to_f8(ar[c], fa);
      bf8_to_f8(br[c], fb);
      ushort8 v;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        // the SAVED sum is bf16-rounded; stats must match what backward
        // will read, so accumulate on the rounded value
        v[j] = f2bf(fa[j] + fb[j]);
        float f = bf2f(v[j]);
        s += f;
        sq += f * f;
      }
      sr[c] = v;
      if (ci < 2) cs[ci] = v;
    }
    s = wave_sum(s);
    sq = wave_sum(sq);
    float mu = s * inv_n;
    float var = fmaxf(sq * inv_n - mu * mu, 0.f);
    float rs = rsqrtf(var + eps);
    if (lane == 0) {
      mean[row] = mu;
      rstd[row] = rs;
    }
    ushort8 *yr = y + row * C8;
    for (int c = lane; c < C8; c += 64) {
      int ci = (c - lane) >> 6;
      float f[8];
      bf8_to_f8(ci < 2 ? cs[ci] : sr[c], f);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        f[j] = (f[j] - mu) * rs * gamma[c * 8 + j] + beta[c * 8 + j];
      yr[c] = f8_to_bf8(f);
    }
  }
}

static int ln_grid(long M, int waves) {
  long g = (M + waves - 1) / waves;
  if (g > 1024) g = 1024;
  if (g < 1) g = 1;
  return (int)g;
}

extern "C" hipError_t ln_fwd(const void *x, const float *gamma,
                             const float *beta, void *y, float *mean,
                             float *rstd, long M, int N, float eps,
                             hipStream_t s) {
  if (N % 8) return hipErrorInvalidValue;
  ln_fwd_k<<<ln_grid(M, 4), 256, 0, s>>>((const ushort8 *)x, gamma, beta,
                                         (ushort8 *)y, mean, rstd, M, N / 8,
                                         eps);
  HIP_KERNEL_CHECK();
  return hipSuccess;
}

extern "C" hipError_t ln_fwd_add(const void *a, const void *b,
                                 const float *gamma, const float *beta,
                                 void *sum_out, void *y, float *mean,
                                 float *rstd, long M, int N, float eps,
                                 hipStream_t s) {
  if (N % 8) return hipErrorInvalidValue;
  ln_fwd_add_k<<<ln_grid(M, 4), 256, 0, s>>>(
      (const ushort8 *)a, (const ushort8 *)b, gamma, beta,
      (ushort8 *)sum_out, (ushort8 *)y, mean, rstd, M, N / 8, eps);
  HIP_KERNEL_CHECK();
  return hipSuccess;
}

extern "C" hipError_t splitk_reduce(const float *, int, long, void *, int,
                                    hipStream_t);

// partial must hold [grid][2][N] fp32; dgamma/dbeta are fp32 [N] each,
// reduced here via splitk_reduce over the slab
extern "C" hipError_t ln_bwd(const void *dy, const void *x, const float *gamma,
                             const float *mean, const float *rstd, void *dx,
                             float *partial, float *dgamma_dbeta, long M,
                             int N, int *grid_out, hipStream_t s) {
  if (N % 8 || N > 2048) return hipErrorInvalidValue;
  // ≤32 slab rows (16-wave blocks keep the dx pass at ~512 waves): the
  // dgamma/dbeta splitk_reduce over [grid][2N] runs 2 blocks at this len —
  // with the old 1024-slab grid it serialized 512-deep per element and was
  // 23% of a fused-LN BERT step (profiled; fused LN lost to torch because
  // of THIS kernel, not the LN passes themselves)
  int grid = ln_grid(M, 16);
  if (grid > 32) grid = 32;
  if (grid_out) *grid_out = grid;
  ln_bwd_k<<<grid, 1024, 0, s>>>((const ushort8 *)dy, (const ushort8 *)x,
                                 gamma, mean, rstd, (ushort8 *)dx, partial, M,
                                 N / 8);
  HIP_KERNEL_CHECK();
  // dgamma_dbeta[0..N) = dgamma, [N..2N) = dbeta
  return splitk_reduce(partial, grid, 2L * N, dgamma_dbeta, 0, s);
}
