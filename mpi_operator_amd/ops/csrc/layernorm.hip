// Fused LayerNorm for bf16 [M rows][N features], fp32 gamma/beta and
// statistics — the transformer-side normalization (BERT runs 50+ LN
// fwd+bwd pairs per step; the eager torch path was ~20% of a BERT step).
//
// One WAVE owns one row: per-lane ushort8 chunks (octets strided by 64
// lanes), row sums via 6-step shfl_xor reduction — no LDS on the hot path.
// Backward's dgamma/dbeta use per-block fp32 partial slabs reduced by the
// existing splitk_reduce kernel (atomics on N addresses would serialize).
#include "common.h"

// y = (x - mean) * rstd * gamma + beta;  saves mean/rstd per row.
// C8T: compile-time C8 specialization (0 = runtime). The runtime-bounded
// per-lane octet loops don't unroll (same class of loss as the splitk /
// attention / SGD kernels), leaving only 2 loads in flight per wave.
template <int C8T>
__global__ void ln_fwd_k(const ushort8 *__restrict__ x,
                         const float *__restrict__ gamma,
                         const float *__restrict__ beta,
                         ushort8 *__restrict__ y, float *__restrict__ mean,
                         float *__restrict__ rstd, long M, int C8rt,
                         float eps) {
  const int C8 = C8T ? C8T : C8rt;
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

// dx = rstd * (g*dy - mean(g*dy) - xhat * mean(g*dy*xhat)).
// SLAB-FREE: the round-1 kernel folded dgamma/dbeta slabs at block end,
// which capped the grid at 32 blocks (512 waves) and left the dx pass
// latency-bound — the measured reason fused LN lost to torch native.
// dgamma/dbeta moved to the column-parallel kernel below.
template <int C8T>
__global__ void ln_bwd_dx_k(const ushort8 *__restrict__ dy,
                            const ushort8 *__restrict__ x,
                            const float *__restrict__ gamma,
                            const float *__restrict__ mean,
                            const float *__restrict__ rstd,
                            ushort8 *__restrict__ dx, long M, int C8rt) {
  const int C8 = C8T ? C8T : C8rt;
  int lane = threadIdx.x & 63;
  int wave = threadIdx.x >> 6;
  int waves = blockDim.x >> 6;
  int N = C8 * 8;
  float inv_n = 1.f / (float)N;
  for (long row = (long)blockIdx.x * waves + wave; row < M;
       row += (long)gridDim.x * waves) {
    const ushort8 *xr = x + row * C8;
    const ushort8 *dr = dy + row * C8;
    float mu = mean[row], rs = rstd[row];
    float s1 = 0.f, s2 = 0.f;
    ushort8 cx[2], cd[2]; // row cached in registers at N<=1024
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
}

// 2-row-interleaved dx variant (N <= 1024 so both rows' octets fit the
// register cache): each wave owns rows (r, r + stride) with fully
// independent load/reduce chains interleaved in one body — the 1-row
// kernel ran one 2-octet load pair per wave between shfl-reduction
// chains and measured ~3.4 TB/s (latency-, not bandwidth-bound).
// MPIAMD_LN2 gates the route (A/B).
__global__ void ln_bwd_dx2_k(const ushort8 *__restrict__ dy,
                             const ushort8 *__restrict__ x,
                             const float *__restrict__ gamma,
                             const float *__restrict__ mean,
                             const float *__restrict__ rstd,
                             ushort8 *__restrict__ dx, long M, int C8) {
  int lane = threadIdx.x & 63;
  int wave = threadIdx.x >> 6;
  int waves = blockDim.x >> 6;
  int N = C8 * 8;
  float inv_n = 1.f / (float)N;
  long stride = (long)gridDim.x * waves; // row pair: (r, r + stride)
  for (long rA = (long)blockIdx.x * waves + wave; rA < M; rA += 2 * stride) {
    long rB = rA + stride;
    bool hasB = rB < M;
    const ushort8 *xA = x + rA * C8, *dA = dy + rA * C8;
    const ushort8 *xB = x + (hasB ? rB : rA) * C8;
    const ushort8 *dB = dy + (hasB ? rB : rA) * C8;
    float muA = mean[rA], rsA = rstd[rA];
    float muB = mean[hasB ? rB : rA], rsB = rstd[hasB ? rB : rA];
    float s1A = 0.f, s2A = 0.f, s1B = 0.f, s2B = 0.f;
    ushort8 cxA[2], cdA[2], cxB[2], cdB[2];
    for (int c = lane; c < C8; c += 64) {
      int ci = (c - lane) >> 6;
      ushort8 vxA = xA[c], vdA = dA[c], vxB = xB[c], vdB = dB[c];
      if (ci < 2) {
        cxA[ci] = vxA; cdA[ci] = vdA; cxB[ci] = vxB; cdB[ci] = vdB;
      }
      float fxA[8], fdA[8], fxB[8], fdB[8];
      bf8_to_f8(vxA, fxA); bf8_to_f8(vdA, fdA);
      bf8_to_f8(vxB, fxB); bf8_to_f8(vdB, fdB);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float g = gamma[c * 8 + j];
        float gdA = g * fdA[j], gdB = g * fdB[j];
        s1A += gdA; s2A += gdA * ((fxA[j] - muA) * rsA);
        s1B += gdB; s2B += gdB * ((fxB[j] - muB) * rsB);
      }
    }
    s1A = wave_sum(s1A) * inv_n; s2A = wave_sum(s2A) * inv_n;
    s1B = wave_sum(s1B) * inv_n; s2B = wave_sum(s2B) * inv_n;
    ushort8 *oA = dx + rA * C8;
    ushort8 *oB = dx + (hasB ? rB : rA) * C8;
    for (int c = lane; c < C8; c += 64) {
      int ci = (c - lane) >> 6;
      float fxA[8], fdA[8], fxB[8], fdB[8];
      bf8_to_f8(ci < 2 ? cxA[ci] : xA[c], fxA);
      bf8_to_f8(ci < 2 ? cdA[ci] : dA[c], fdA);
      bf8_to_f8(ci < 2 ? cxB[ci] : xB[c], fxB);
      bf8_to_f8(ci < 2 ? cdB[ci] : dB[c], fdB);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float g = gamma[c * 8 + j];
        fdA[j] = rsA * (g * fdA[j] - s1A - ((fxA[j] - muA) * rsA) * s2A);
        fdB[j] = rsB * (g * fdB[j] - s1B - ((fxB[j] - muB) * rsB) * s2B);
      }
      oA[c] = f8_to_bf8(fdA);
      if (hasB) oB[c] = f8_to_bf8(fdB);
    }
  }
}

// dgamma/dbeta partials, bn_partials-style: thread owns a channel octet,
// strides rows (coalesced 16 B/lane), per-row mean/rstd scalar loads;
// [grid][2][N] fp32 slabs reduced by splitk_reduce.
__global__ void ln_gb_partials_k(const ushort8 *__restrict__ dy,
                                 const ushort8 *__restrict__ x,
                                 const float *__restrict__ mean,
                                 const float *__restrict__ rstd,
                                 float *__restrict__ partial, long M, int C8) {
  int C = C8 * 8;
  int cb = threadIdx.x % C8;
  int row_lane = threadIdx.x / C8;
  int rows_per_block = blockDim.x / C8;
  float a0[8] = {0}, a1[8] = {0}; // dbeta, dgamma
  if (row_lane < rows_per_block) {
    long stride = (long)gridDim.x * rows_per_block;
    for (long row = (long)blockIdx.x * rows_per_block + row_lane; row < M;
         row += stride) {
      long off = row * C8 + cb;
      float mu = mean[row], rs = rstd[row];
      float fx[8], fd[8];
      bf8_to_f8(x[off], fx);
      bf8_to_f8(dy[off], fd);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        a0[j] += fd[j];
        a1[j] += fd[j] * ((fx[j] - mu) * rs);
      }
    }
  }
  __shared__ float lds[2][256 * 8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    lds[0][threadIdx.x * 8 + j] = a1[j]; // dgamma first (slab order)
    lds[1][threadIdx.x * 8 + j] = a0[j];
  }
  __syncthreads();
  if (row_lane == 0) {
    for (int rl = 1; rl < rows_per_block; ++rl)
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        a1[j] += lds[0][(rl * C8 + cb) * 8 + j];
        a0[j] += lds[1][(rl * C8 + cb) * 8 + j];
      }
    float *pg = partial + (long)blockIdx.x * 2 * C + cb * 8;
    float *pb = pg + C;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      pg[j] = a1[j];
      pb[j] = a0[j];
    }
  }
}

// Residual-join + LayerNorm in one pass: s = a + b (written — backward
// reads it as the LN input), y = LN(s). Replaces the separate residual
// add's extra read+write of the [M][N] activation (BERT runs 2 joins per
// layer per direction).
template <int C8T>
__global__ void ln_fwd_add_k(const ushort8 *__restrict__ a,
                             const ushort8 *__restrict__ b,
                             const float *__restrict__ gamma,
                             const float *__restrict__ beta,
                             ushort8 *__restrict__ sum_out,
                             ushort8 *__restrict__ y, float *__restrict__ mean,
                             float *__restrict__ rstd, long M, int C8rt,
                             float eps) {
  const int C8 = C8T ? C8T : C8rt;
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

// Per-channel slab reduce: one 256-thread block per channel strides the
// grid entries (splitk_reduce at len 2·N with 512 slabs ran 4 blocks ×
// 512-deep serial — 35 µs/call and 17% of a fused-LN BERT step).
__global__ void ln_gb_reduce_k(const float *__restrict__ partial,
                               float *__restrict__ out, int grid, int N2) {
  int c = blockIdx.x;
  if (c >= N2) return;
  float a = 0.f;
  for (int g = threadIdx.x; g < grid; g += 256)
    a += partial[(long)g * N2 + c];
  __shared__ float red[256 / WAVE];
  a = wave_sum(a);
  if ((threadIdx.x & (WAVE - 1)) == 0) red[threadIdx.x / WAVE] = a;
  __syncthreads();
  if (threadIdx.x == 0)
    out[c] = red[0] + red[1] + red[2] + red[3];
}

// LN C8 specialization: same-box A/B at BERT-Large bs32 showed the
// compile-time octet loops LOSE 2.4% (1384 vs 1415; ln_bwd_dx 15.5->24.1
// us) — the unrolled form costs more than the runtime loop at M=4096
// despite lower VGPRs. Default OFF; MPIAMD_LN_SPEC=1 enables for
// small-shape A/Bs.
static bool ln_spec() {
  static const bool on = [] {
    const char *e = getenv("MPIAMD_LN_SPEC");
    return e && e[0] == '1';
  }();
  return on;
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
  if (ln_spec() && N == 1024)
    ln_fwd_k<128><<<ln_grid(M, 4), 256, 0, s>>>(
        (const ushort8 *)x, gamma, beta, (ushort8 *)y, mean, rstd, M, N / 8,
        eps);
  else if (ln_spec() && N == 768)
    ln_fwd_k<96><<<ln_grid(M, 4), 256, 0, s>>>(
        (const ushort8 *)x, gamma, beta, (ushort8 *)y, mean, rstd, M, N / 8,
        eps);
  else
    ln_fwd_k<0><<<ln_grid(M, 4), 256, 0, s>>>(
        (const ushort8 *)x, gamma, beta, (ushort8 *)y, mean, rstd, M, N / 8,
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
  if (ln_spec() && N == 1024)
    ln_fwd_add_k<128><<<ln_grid(M, 4), 256, 0, s>>>(
        (const ushort8 *)a, (const ushort8 *)b, gamma, beta,
        (ushort8 *)sum_out, (ushort8 *)y, mean, rstd, M, N / 8, eps);
  else if (ln_spec() && N == 768)
    ln_fwd_add_k<96><<<ln_grid(M, 4), 256, 0, s>>>(
        (const ushort8 *)a, (const ushort8 *)b, gamma, beta,
        (ushort8 *)sum_out, (ushort8 *)y, mean, rstd, M, N / 8, eps);
  else
    ln_fwd_add_k<0><<<ln_grid(M, 4), 256, 0, s>>>(
        (const ushort8 *)a, (const ushort8 *)b, gamma, beta,
        (ushort8 *)sum_out, (ushort8 *)y, mean, rstd, M, N / 8, eps);
  HIP_KERNEL_CHECK();
  return hipSuccess;
}

extern "C" hipError_t splitk_reduce(const float *, int, long, void *, int,
                                    hipStream_t);

// partial must hold [grid][2][N] fp32 (grid from *grid_out contract:
// caller passes the allocation's row count); dgamma/dbeta are fp32 [N]
// each, reduced via splitk_reduce over the slab
extern "C" hipError_t ln_bwd(const void *dy, const void *x, const float *gamma,
                             const float *mean, const float *rstd, void *dx,
                             float *partial, float *dgamma_dbeta, long M,
                             int N, int *grid_out, hipStream_t s) {
  if (N % 8 || N > 2048) return hipErrorInvalidValue;
  int C8 = N / 8;
  static const bool ln2 = [] { // MPIAMD_LN2: 2-row-interleaved dx (A/B)
    const char *e = getenv("MPIAMD_LN2");
    return e && e[0] == '1';
  }();
  if (ln2 && N <= 1024) {
    long waves2 = (M + 1) / 2;
    int grid2 = (int)((waves2 + 3) / 4);
    if (grid2 > 1024) grid2 = 1024;
    if (grid2 < 1) grid2 = 1;
    ln_bwd_dx2_k<<<grid2, 256, 0, s>>>((const ushort8 *)dy,
                                       (const ushort8 *)x, gamma, mean, rstd,
                                       (ushort8 *)dx, M, C8);
  } else if (ln_spec() && N == 1024)
    ln_bwd_dx_k<128><<<ln_grid(M, 4), 256, 0, s>>>(
        (const ushort8 *)dy, (const ushort8 *)x, gamma, mean, rstd,
        (ushort8 *)dx, M, C8);
  else if (ln_spec() && N == 768)
    ln_bwd_dx_k<96><<<ln_grid(M, 4), 256, 0, s>>>(
        (const ushort8 *)dy, (const ushort8 *)x, gamma, mean, rstd,
        (ushort8 *)dx, M, C8);
  else
    ln_bwd_dx_k<0><<<ln_grid(M, 4), 256, 0, s>>>(
        (const ushort8 *)dy, (const ushort8 *)x, gamma, mean, rstd,
        (ushort8 *)dx, M, C8);
  HIP_KERNEL_CHECK();
  int rpb = 256 / C8;
  long g = (M + rpb - 1) / rpb;
  int grid = (int)(g > 512 ? 512 : (g < 1 ? 1 : g));
  if (grid_out) *grid_out = grid;
  ln_gb_partials_k<<<grid, 256, 0, s>>>((const ushort8 *)dy,
                                        (const ushort8 *)x, mean, rstd,
                                        partial, M, C8);
  HIP_KERNEL_CHECK();
  // dgamma_dbeta[0..N) = dgamma, [N..2N) = dbeta
  ln_gb_reduce_k<<<2 * N, 256, 0, s>>>(partial, dgamma_dbeta, grid, 2 * N);
  HIP_KERNEL_CHECK();
  return hipSuccess;
}
