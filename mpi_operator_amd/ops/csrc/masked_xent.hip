// Masked softmax-cross-entropy for the MLM head (vocab-scale V, rows with
// target == ignore_index contribute nothing; mean over VALID rows — torch
// F.cross_entropy(ignore_index=...) semantics).
//
// MI355X-first memory plan: at BERT-Large scale the logits are
// [B·S=4096][30522] — the torch path materializes an fp32 cast (+500 MB
// write, +500 MB read), saves fp32 probs, and re-reads them in backward.
// Here forward reads the bf16 logits ONCE and saves only two fp32 stats
// per row (max, inv_sum); backward re-derives probabilities from the bf16
// logits — ~4× less HBM traffic on an 8 TB/s-bound op.
//
// fwd:  loss_sum += logsumexp(row) - x[target];  count += 1   (valid rows)
// bwd:  dlogits = (exp(x - mx)·inv_sum - onehot) · dloss / count
#include "common.h"

__global__ void masked_xent_fwd_k(const uint16_t *__restrict__ logits,
                                  const long *__restrict__ target,
                                  float *__restrict__ stats, // [B][2]
                                  float *__restrict__ out,   // [2]: sum, count
                                  int B, int V, long ld, long ignore_index) {
  int b = blockIdx.x;
  long t = target[b];
  const uint16_t *row = logits + (long)b * ld; // ld >= V (padded-vocab rows)
  // 16 B/lane vectorized row passes (scalar u16 strided loads measured
  // 254 µs/call at vocab scale — 8x off bandwidth). Ragged V means row
  // bases are only 4-B aligned: scalar head up to the first 16-B
  // boundary, vector main, scalar tail.
  int head = (int)((16 - ((unsigned long)row & 15)) & 15) / 2;
  if (head > V) head = V;
  const ushort8 *row8 = (const ushort8 *)(row + head);
  int V8 = (V - head) / 8;
  int tail0 = head + V8 * 8;
  // ONLINE max+sum in one pass over the row (the two-pass form read the
  // 250 MB vocab-scale logits twice — pure HBM waste): per-thread running
  // (mx, sum) with a rescale when the max moves, then a wave/block combine
  // that rescales each partial sum to the global max. Same logsumexp.
  __shared__ float redm[256 / WAVE], reds[256 / WAVE];
  float mx = -3.4e38f, sum = 0.f;
  for (int v = threadIdx.x; v < V8; v += blockDim.x) {
    float f[8];
    bf8_to_f8(row8[v], f);
    float m8 = f[0];
#pragma unroll
    for (int j = 1; j < 8; ++j) m8 = fmaxf(m8, f[j]);
    if (m8 > mx) {
      sum *= __expf(mx - m8);
      mx = m8;
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) sum += __expf(f[j] - mx);
  }
  if (threadIdx.x < head) {
    float f = bf2f(row[threadIdx.x]);
    if (f > mx) {
      sum *= __expf(mx - f);
      mx = f;
    }
    sum += __expf(f - mx);
  }
  for (int v = tail0 + threadIdx.x; v < V; v += blockDim.x) {
    float f = bf2f(row[v]);
    if (f > mx) {
      sum *= __expf(mx - f);
      mx = f;
    }
    sum += __expf(f - mx);
  }
  float wm = wave_max(mx);
  sum *= __expf(mx - wm);
  sum = wave_sum(sum);
  if ((threadIdx.x & (WAVE - 1)) == 0) {
    redm[threadIdx.x / WAVE] = wm;
    reds[threadIdx.x / WAVE] = sum;
  }
  __syncthreads();
  mx = fmaxf(fmaxf(redm[0], redm[1]), fmaxf(redm[2], redm[3]));
  sum = reds[0] * __expf(redm[0] - mx) + reds[1] * __expf(redm[1] - mx) +
        reds[2] * __expf(redm[2] - mx) + reds[3] * __expf(redm[3] - mx);
  if (threadIdx.x == 0) {
    stats[b * 2] = mx;
    stats[b * 2 + 1] = 1.f / sum;
    if (t != ignore_index) {
      atomicAdd(&out[0], logf(sum) + mx - bf2f(row[t]));
      atomicAdd(&out[1], 1.f);
    }
  }
}

__global__ void masked_xent_bwd_k(const uint16_t *__restrict__ logits,
                                  const long *__restrict__ target,
                                  const float *__restrict__ stats,
                                  const float *__restrict__ out, // [2]
                                  const float *__restrict__ dscale,
                                  uint16_t *__restrict__ dlogits, int B,
                                  int V, long ld, long ignore_index) {
  int b = blockIdx.x;
  long t = target[b];
  const uint16_t *row = logits + (long)b * ld;
  uint16_t *drow = dlogits + (long)b * ld;
  // pad columns [V, ld) are written ZERO: the downstream dx GEMM reduces
  // over the padded vocab dim and relies on zero pads to mask its
  // zeros-page-granule edge (gemm.hip tn_cols_ok contract)
  for (int v = V + (int)threadIdx.x; v < ld; v += blockDim.x) drow[v] = 0;
  if (t == ignore_index) {
    for (int v = threadIdx.x; v < V; v += blockDim.x) drow[v] = 0;
    return;
  }
  float cnt = out[1];
  float scale = *dscale / (cnt > 0.f ? cnt : 1.f);
  float mx = stats[b * 2], inv_sum = stats[b * 2 + 1];
  // same alignment discipline as forward (row bases are 4-B aligned at
  // ragged V; dlogits shares the layout so head/tail apply to both)
  int head = (int)((16 - ((unsigned long)row & 15)) & 15) / 2;
  if (head > V) head = V;
  const ushort8 *row8 = (const ushort8 *)(row + head);
  ushort8 *drow8 = (ushort8 *)(drow + head);
  int V8 = (V - head) / 8;
  int tail0 = head + V8 * 8;
  for (int v = threadIdx.x; v < V8; v += blockDim.x) {
    float f[8];
    bf8_to_f8(row8[v], f);
    ushort8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float p = __expf(f[j] - mx) * inv_sum;
      if (head + v * 8 + j == (int)t) p -= 1.f;
      o[j] = f2bf(p * scale);
    }
    drow8[v] = o;
  }
  if (threadIdx.x < head) {
    int v = threadIdx.x;
    float p = __expf(bf2f(row[v]) - mx) * inv_sum;
    if (v == (int)t) p -= 1.f;
    drow[v] = f2bf(p * scale);
  }
  for (int v = tail0 + threadIdx.x; v < V; v += blockDim.x) {
    float p = __expf(bf2f(row[v]) - mx) * inv_sum;
    if (v == (int)t) p -= 1.f;
    drow[v] = f2bf(p * scale);
  }
}

extern "C" hipError_t masked_xent_fwd_launch(const void *logits,
                                             const long *target, float *stats,
                                             float *out, int B, int V, long ld,
                                             long ignore_index,
                                             hipStream_t s) {
  hipMemsetAsync(out, 0, 8, s);
  masked_xent_fwd_k<<<B, 256, 0, s>>>((const uint16_t *)logits, target, stats,
                                      out, B, V, ld, ignore_index);
  HIP_KERNEL_CHECK();
  return hipSuccess;
}

extern "C" hipError_t masked_xent_bwd_launch(const void *logits,
                                             const long *target,
                                             const float *stats,
                                             const float *out,
                                             const float *dscale, void *dlogits,
                                             int B, int V, long ld,
                                             long ignore_index, hipStream_t s) {
  masked_xent_bwd_k<<<B, 256, 0, s>>>((const uint16_t *)logits, target, stats,
                                      out, dscale, (uint16_t *)dlogits, B, V,
                                      ld, ignore_index);
  HIP_KERNEL_CHECK();
  return hipSuccess;
}
