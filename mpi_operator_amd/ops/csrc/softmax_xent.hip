// Fused softmax + cross-entropy (mean reduction) for the classifier head.
// logits bf16 [B,V] → loss fp32 scalar + probs fp32 [B,V] (saved for bwd).
#include "common.h"

__global__ void softmax_xent_fwd_k(const uint16_t *__restrict__ logits,
                                   const long *__restrict__ target,
                                   float *__restrict__ probs,
                                   float *__restrict__ loss, int B, int V,
                                   float inv_b) {
  int b = blockIdx.x;
  const uint16_t *row = logits + (long)b * V;
  float *prow = probs + (long)b * V;
  __shared__ float red[256 / WAVE];
  // max
  float mx = -3.4e38f;
  for (int v = threadIdx.x; v < V; v += blockDim.x) mx = fmaxf(mx, bf2f(row[v]));
  mx = wave_max(mx);
  if ((threadIdx.x & (WAVE - 1)) == 0) red[threadIdx.x / WAVE] = mx;
  __syncthreads();
  mx = fmaxf(fmaxf(red[0], red[1]), fmaxf(red[2], red[3]));
  // sum exp
  float sum = 0;
  for (int v = threadIdx.x; v < V; v += blockDim.x) {
    float e = __expf(bf2f(row[v]) - mx);
    prow[v] = e; // un-normalized for now
    sum += e;
  }
  sum = wave_sum(sum);
  __syncthreads();
  if ((threadIdx.x & (WAVE - 1)) == 0) red[threadIdx.x / WAVE] = sum;
  __syncthreads();
  sum = red[0] + red[1] + red[2] + red[3];
  float inv_sum = 1.f / sum;
  for (int v = threadIdx.x; v < V; v += blockDim.x) prow[v] *= inv_sum;
  if (threadIdx.x == 0) {
    float xt = bf2f(row[target[b]]);
    atomicAdd(loss, (logf(sum) + mx - xt) * inv_b);
  }
}

// dscale is read from DEVICE memory (the incoming dloss scalar): reading it
// on the host would force a per-step D2H sync and break hipGraph capture.
__global__ void softmax_xent_bwd_k(const float *__restrict__ probs,
                                   const long *__restrict__ target,
                                   uint16_t *__restrict__ dlogits, int B, int V,
                                   const float *__restrict__ dscale,
                                   float inv_b) {
  float scale = *dscale * inv_b;
  for (long t = blockIdx.x * blockDim.x + threadIdx.x; t < (long)B * V;
       t += (long)gridDim.x * blockDim.x) {
    int b = t / V;
    int v = t % V;
    float p = probs[t];
    if (v == (int)target[b]) p -= 1.f;
    dlogits[t] = f2bf(p * scale);
  }
}

extern "C" hipError_t softmax_xent_fwd_launch(const void *logits,
                                              const long *target, float *probs,
                                              float *loss, int B, int V,
                                              hipStream_t s) {
  hipMemsetAsync(loss, 0, 4, s);
  softmax_xent_fwd_k<<<B, 256, 0, s>>>((const uint16_t *)logits, target, probs,
                                       loss, B, V, 1.f / B);
  HIP_KERNEL_CHECK();
  return hipSuccess;
}

extern "C" hipError_t softmax_xent_bwd_launch(const float *probs,
                                              const long *target, void *dlogits,
                                              int B, int V, const float *dscale,
                                              hipStream_t s) {
  long tasks = (long)B * V;
  long blocks = (tasks + 255) / 256;
  if (blocks > 2048) blocks = 2048;
  // loss is a batch MEAN: dlogits = (p - onehot) * dloss / B
  softmax_xent_bwd_k<<<(int)blocks, 256, 0, s>>>(probs, target,
                                                 (uint16_t *)dlogits, B, V,
                                                 dscale, 1.f / B);
  HIP_KERNEL_CHECK();
  return hipSuccess;
}
