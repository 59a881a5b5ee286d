// Max-pool fwd (+argmax u8) and gather-style bwd for NHWC bf16.
// Reference op: tf_cnn_benchmarks ResNet stem max-pool.
#include "common.h"

// fwd: one lane-task per (output pixel, 8-channel group); 16 B loads.
__global__ void maxpool_fwd_k(const ushort8 *__restrict__ x,
                              ushort8 *__restrict__ y, uint8_t *__restrict__ idx,
                              int N, int H, int W, int HO, int WO, int C8,
                              int K, int stride, int pad) {
  long M = (long)N * HO * WO;
  for (long t = blockIdx.x * blockDim.x + threadIdx.x; t < M * C8;
       t += (long)gridDim.x * blockDim.x) {
    int cb = t % C8;
    long m = t / C8;
    int wo = m % WO;
    int ho = (m / WO) % HO;
    int n = m / ((long)WO * HO);
    float best[8];
    uint8_t bidx[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) { best[j] = -3.4e38f; bidx[j] = 0; }
    for (int r = 0; r < K; ++r) {
      int h = ho * stride + r - pad;
      if (h < 0 || h >= H) continue;
      for (int s = 0; s < K; ++s) {
        int w = wo * stride + s - pad;
        if (w < 0 || w >= W) continue;
        ushort8 v = x[((long)(n * H + h) * W + w) * C8 + cb];
        float f[8];
        bf8_to_f8(v, f);
        uint8_t pos = (uint8_t)(r * K + s);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          if (f[j] > best[j]) { best[j] = f[j]; bidx[j] = pos; }
      }
    }
    y[t] = f8_to_bf8(best);
#pragma unroll
    for (int j = 0; j < 8; ++j) idx[t * 8 + j] = bidx[j];
  }
}

// bwd: gather over the <=ceil(K/stride)^2 windows covering each input pixel.
__global__ void maxpool_bwd_k(const ushort8 *__restrict__ dy,
                              const uint8_t *__restrict__ idx,
                              ushort8 *__restrict__ dx, int N, int H, int W,
                              int HO, int WO, int C8, int K, int stride,
                              int pad) {
  long M = (long)N * H * W;
  for (long t = blockIdx.x * blockDim.x + threadIdx.x; t < M * C8;
       t += (long)gridDim.x * blockDim.x) {
    int cb = t % C8;
    long m = t / C8;
    int w = m % W;
    int h = (m / W) % H;
    int n = m / ((long)W * H);
    float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    int ho_lo = (h + pad - K + stride) / stride; // ceil((h+pad-K+1)/stride)
    if (ho_lo < 0) ho_lo = 0;
    int ho_hi = (h + pad) / stride;
    if (ho_hi >= HO) ho_hi = HO - 1;
    int wo_lo = (w + pad - K + stride) / stride;
    if (wo_lo < 0) wo_lo = 0;
    int wo_hi = (w + pad) / stride;
    if (wo_hi >= WO) wo_hi = WO - 1;
    for (int ho = ho_lo; ho <= ho_hi; ++ho) {
      int r = h + pad - ho * stride;
      if (r < 0 || r >= K) continue;
      for (int wo = wo_lo; wo <= wo_hi; ++wo) {
        int s = w + pad - wo * stride;
        if (s < 0 || s >= K) continue;
        long o = ((long)(n * HO + ho) * WO + wo) * C8 + cb;
        uint8_t pos = (uint8_t)(r * K + s);
        ushort8 v = dy[o];
        const uint8_t *ip = idx + o * 8;
#pragma unroll
        for (int j = 0; j < 8; ++j)
          if (ip[j] == pos) acc[j] += bf2f(v[j]);
      }
    }
    dx[t] = f8_to_bf8(acc);
  }
}

static inline int pool_grid(long lane_tasks) {
  long b = (lane_tasks + 255) / 256;
  return (int)(b > 2048 ? 2048 : (b < 1 ? 1 : b));
}

extern "C" hipError_t maxpool_fwd_launch(const void *x, void *y, uint8_t *idx,
                                         int N, int H, int W, int HO, int WO,
                                         int C, int K, int stride, int pad,
                                         hipStream_t s) {
  int C8 = C / 8;
  long tasks = (long)N * HO * WO * C8;
  maxpool_fwd_k<<<pool_grid(tasks), 256, 0, s>>>(
      (const ushort8 *)x, (ushort8 *)y, idx, N, H, W, HO, WO, C8, K, stride, pad);
  HIP_KERNEL_CHECK();
  return hipSuccess;
}

extern "C" hipError_t maxpool_bwd_launch(const void *dy, const uint8_t *idx,
                                         void *dx, int N, int H, int W, int HO,
                                         int WO, int C, int K, int stride,
                                         int pad, hipStream_t s) {
  int C8 = C / 8;
  long tasks = (long)N * H * W * C8;
  maxpool_bwd_k<<<pool_grid(tasks), 256, 0, s>>>(
      (const ushort8 *)dy, idx, (ushort8 *)dx, N, H, W, HO, WO, C8, K, stride, pad);
  HIP_KERNEL_CHECK();
  return hipSuccess;
}
