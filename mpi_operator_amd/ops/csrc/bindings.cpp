// PyTorch bindings for the gfx950 kernels. The only TU that includes torch
// headers — kernel TUs are pure HIP and are linked via the extern "C"
// launchers declared below.
#include <torch/extension.h>

#include <ATen/hip/impl/HIPGuardImplMasqueradingAsCUDA.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>
#include <hip/hip_runtime.h>

#include <vector>

namespace {

using at::Tensor;

// Late-stage convs produce few 128x128 output tiles (7x7 spatial ~100
// blocks on 256 CUs): split the reduce dim so fwd/dgrad fill the chip,
// paying one fp32 slab + reduce pass.
int conv_splits(long M, int Ncols, int K) {
  static const int force = [] {
    const char *e = getenv("MPIAMD_CONV_SPLITS");
    return e ? atoi(e) : -1;
  }();
  if (force >= 1) return force;
  long tiles = ((M + 127) / 128) * ((Ncols + 127) / 128);
  int nk = (K + 63) / 64;
  if (tiles >= 256 || nk < 16) return 1;
  long s = 512 / tiles;
  if (s > nk / 8) s = nk / 8;
  if (s > 16) s = 16;
  return s < 1 ? 1 : (int)s;
}

#define CHK(call)                                                              \
  do {                                                                         \
    hipError_t e_ = (call);                                                    \
    TORCH_CHECK(e_ == hipSuccess, "HIP kernel failure: ", hipGetErrorString(e_)); \
  } while (0)

hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
}
using HIPDeviceGuard = c10::hip::HIPGuardMasqueradingAsCUDA;

void check_cl_bf16(const Tensor &t, const char *name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == at::kBFloat16, name, " must be bf16");
  TORCH_CHECK(t.dim() == 4 && t.is_contiguous(at::MemoryFormat::ChannelsLast),
              name, " must be 4-D channels_last");
}

Tensor empty_cl_bf16(int64_t n, int64_t c, int64_t h, int64_t w, const Tensor &like) {
  // memory_format at allocation — .contiguous() on a fresh empty COPIES.
  return at::empty({n, c, h, w}, like.options().dtype(at::kBFloat16),
                   at::MemoryFormat::ChannelsLast);
}

} // namespace

// ---- extern "C" launchers from the .hip TUs ----
extern "C" {
hipError_t add_relu_fwd(const void *, const void *, void *, long, hipStream_t);
hipError_t add_bf16(const void *, const void *, void *, long, hipStream_t);
hipError_t add_relu_bwd(const void *, const void *, void *, long, hipStream_t);
hipError_t add_relu_bwd_add(const void *, const void *, const void *, void *,
                            long, hipStream_t);
hipError_t add_relu_bwd_add_mask(const void *, const void *, const void *,
                                 void *, long, hipStream_t);
hipError_t gap_fwd(const void *, void *, int, int, int, hipStream_t);
hipError_t gap_bwd(const void *, void *, int, int, int, hipStream_t);
hipError_t bias_add(void *, const float *, long, int, hipStream_t);
hipError_t colsum_bf16(const void *, float *, float *, long, int, long, hipStream_t);
int colsum_chunks(long, int);
struct SgdDesc {
  const void *grad;
  float *master;
  float *mom;
  uint16_t *out;
  long numel;
};
hipError_t sgd_step_launch(const SgdDesc *, int, int, float, float, float, int,
                           hipStream_t);
int bn_grid_cap(int);
hipError_t bn_fwd_train_launch(const void *, const void *, const float *,
                               const float *, float, int, void *, uint8_t *,
                               float *, float *, float *, float *, float *,
                               float *, float *, float, long, int,
                               hipStream_t);
hipError_t bn_fwd_eval_launch(const void *, const float *, const float *, int,
                              void *, long, int, hipStream_t);
hipError_t bn_bwd_launch(const void *, const void *, const void *,
                         const uint8_t *, const float *, const float *,
                         const float *, int, void *, float *, float *,
                         float *, float *, float *, float *, long, int,
                         hipStream_t);
hipError_t maxpool_fwd_launch(const void *, void *, uint8_t *, int, int, int,
                              int, int, int, int, int, int, hipStream_t);
hipError_t maxpool_bwd_launch(const void *, const uint8_t *, void *, int, int,
                              int, int, int, int, int, int, int, hipStream_t);
hipError_t softmax_xent_fwd_launch(const void *, const long *, float *, float *,
                                   int, int, hipStream_t);
hipError_t masked_xent_fwd_launch(const void *, const long *, float *, float *,
                                  int, int, long, long, hipStream_t);
hipError_t masked_xent_bwd_launch(const void *, const long *, const float *,
                                  const float *, const float *, void *, int,
                                  int, long, long, hipStream_t);
hipError_t attn_fwd_launch(const void *, void *, void *, const float *, int,
                           int, int, float, hipStream_t);
hipError_t attn_bwd_launch(const void *, const void *, const void *, void *,
                           int, int, int, float, hipStream_t);
hipError_t softmax_xent_bwd_launch(const float *, const long *, void *, int,
                                   int, const float *, hipStream_t);
hipError_t gemm_nt(const void *, const void *, void *, int, int, int, long,
                   long, long, int, hipStream_t);
hipError_t gemm_nt_bias(const void *, const void *, const float *, void *,
                        int, int, int, long, long, long, hipStream_t);
hipError_t gemm_nt_bias_sk(const void *, const void *, const float *, float *,
                           void *, int, int, int, long, long, long, int,
                           hipStream_t);
int gemm_fwd_splits(int, int, int);
hipError_t gemm_nt_tn(const void *, const void *, void *, int, int, int, long,
                      long, long, int, hipStream_t);
hipError_t gemm_nt_tn_acc(const void *, const void *, void *, int, int, int,
                          long, long, long, hipStream_t);
hipError_t gemm_nt_gelu_bias(const void *, const void *, const float *,
                             void *, void *, int, int, int, long, long, long,
                             hipStream_t);
hipError_t gemm_nt_tn_gelubwd(const void *, const void *, const void *,
                              void *, int, int, int, long, long, long,
                              void *, hipStream_t);
int gemm_gelubwd_wants_wt(int, int, int);
hipError_t gemm_tn_tn(const void *, const void *, void *, int, int, int, long,
                      long, long, int, hipStream_t);
int gemm_tn_tn_splits(int, int, int);
hipError_t gemm_tn_tn_sk(const void *, const void *, float *, void *, int,
                         int, int, long, long, long, int, int, hipStream_t);
hipError_t gemm_nt_tn_sk(const void *, const void *, float *, void *, int,
                         int, int, long, long, long, int, hipStream_t);
int gemm_nt_tn_splits(int, int, int);
hipError_t conv_fwd(const void *, const void *, void *, int, int, int, int,
                    int, int, int, int, int, int, int, int, float *,
                    hipStream_t);
hipError_t conv_dgrad(const void *, const void *, void *, int, int, int, int,
                      int, int, int, int, int, int, int, int, float *,
                      hipStream_t);
hipError_t conv_fwd_bn(const void *, const void *, void *, int, int, int,
                       int, int, int, int, int, int, int, int, float *,
                       hipStream_t);
hipError_t bn_fwd_train_pre_launch(const float *, int, const void *,
                                   const void *, const float *, const float *,
                                   float, int, void *, uint8_t *, float *,
                                   float *, float *, float *, float *,
                                   float *, float, long, int, hipStream_t);
hipError_t conv_dgrad_1x1_acc(const void *, const void *, void *, long, int,
                              int, hipStream_t);
hipError_t conv_wgrad_implicit(const void *, const void *, float *, void *,
                               int, int, int, int, int, int, int, int, int,
                               int, int, int, int, hipStream_t);
hipError_t mfma_probe(const void *, const void *, float *, hipStream_t);
hipError_t ln_fwd(const void *, const float *, const float *, void *, float *,
                  float *, long, int, float, hipStream_t);
hipError_t ln_fwd_add(const void *, const void *, const float *,
                      const float *, void *, void *, float *, float *, long,
                      int, float, hipStream_t);
hipError_t ln_bwd(const void *, const void *, const float *, const float *,
                  const float *, void *, float *, float *, long, int, int *,
                  hipStream_t);
}

// ------------------------- conv -------------------------
static Tensor conv2d_fwd(const Tensor &x, const Tensor &w, int64_t stride,
                         int64_t pad) {
  check_cl_bf16(x, "x");
  check_cl_bf16(w, "w");
  const HIPDeviceGuard guard(x.device());
  int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  int Kout = w.size(0), R = w.size(2), S = w.size(3);
  TORCH_CHECK(w.size(1) == C, "conv channel mismatch");
  TORCH_CHECK(C % 8 == 0, "conv requires C%8==0 (pad the stem input)");
  int HO = (H + 2 * (int)pad - R) / (int)stride + 1;
  int WO = (W + 2 * (int)pad - S) / (int)stride + 1;
  Tensor y = empty_cl_bf16(N, Kout, HO, WO, x);
  long M = (long)N * HO * WO;
  int splits = conv_splits(M, Kout, R * S * C);
  Tensor partial;
  float *pp = nullptr;
  if (splits > 1) {
    partial = at::empty({(long)splits, M, (long)Kout},
                        x.options().dtype(at::kFloat));
    pp = partial.data_ptr<float>();
  }
  CHK(conv_fwd(x.data_ptr(), w.data_ptr(), y.data_ptr(), N, H, W, C, Kout, R,
               S, (int)stride, (int)pad, HO, WO, splits, pp, cur_stream()));
  return y;
}

static Tensor conv2d_dgrad(const Tensor &dy, const Tensor &w, int64_t H,
                           int64_t W, int64_t stride, int64_t pad) {
  check_cl_bf16(dy, "dy");
  check_cl_bf16(w, "w");
  const HIPDeviceGuard guard(dy.device());
  int N = dy.size(0), Kout = dy.size(1), HO = dy.size(2), WO = dy.size(3);
  int C = w.size(1), R = w.size(2), S = w.size(3);
  TORCH_CHECK(Kout % 8 == 0, "dgrad requires out-channels %8==0");
  // conv_dgrad routes every stride != 1 to the parity-2 decomposition
  // (conv.hip conv_dgrad_s2), which is stride-2-only by construction.
  TORCH_CHECK(stride == 1 || stride == 2,
              "conv2d_dgrad supports stride 1 or 2 only, got ", stride);
  Tensor dx = empty_cl_bf16(N, C, H, W, dy);
  long M = (long)N * H * W;
  int splits = stride == 1 ? conv_splits(M, C, R * S * Kout) : 1;
  Tensor partial;
  float *pp = nullptr;
  if (splits > 1) {
    partial = at::empty({(long)splits, M, (long)C},
                        dy.options().dtype(at::kFloat));
    pp = partial.data_ptr<float>();
  }
  CHK(conv_dgrad(dy.data_ptr(), w.data_ptr(), dx.data_ptr(), N, (int)H,
                 (int)W, C, Kout, R, S, (int)stride, (int)pad, HO, WO, splits,
                 pp, cur_stream()));
  return dx;
}

static Tensor conv2d_wgrad(const Tensor &x, const Tensor &dy, int64_t R,
                           int64_t S, int64_t stride, int64_t pad) {
  check_cl_bf16(x, "x");
  check_cl_bf16(dy, "dy");
  const HIPDeviceGuard guard(x.device());
  int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  int Kout = dy.size(1), HO = dy.size(2), WO = dy.size(3);
  int RSC = (int)R * S * C;
  int tiles = ((Kout + 127) / 128) * ((RSC + 127) / 128);
  long M = (long)N * HO * WO;
  int nk = (int)((M + 63) / 64);
  // fill the chip (tiles×splits ≈ 512 blocks = 2 WG/CU) but keep ≥8 k-tiles
  // per split so each block amortizes its pipeline prologue; a flat 1024
  // cap overshot mid shapes (b2 1x1 dw: 256 splits of nk=3 ran at 94 TF
  // vs 167 at 64 splits)
  // target-block budget was tuned on the 4-wave pipeline (2 WG/CU); the
  // 8-wave kernel fills at half the workgroups — MPIAMD_WGRAD_BUDGET
  // overrides for A/B (default 512)
  static const int budget = [] {
    const char *e = getenv("MPIAMD_WGRAD_BUDGET");
    int v = e ? atoi(e) : 0;
    return (v >= 64 && v <= 2048) ? v : 512;
  }();
  int splits = std::max(budget / tiles, 1);
  splits = std::min(splits, std::max(nk / 8, 1));
  splits = std::min(splits, 256);
  // multiple of 8 so the split-major grid keeps a panel's N-tile sharers
  // on one XCD ((split + splits*tile) % 8 must not walk with tile)
  if (splits >= 8) splits &= ~7;
  Tensor partial = at::empty({(long)splits, (long)Kout, (long)RSC},
                             x.options().dtype(at::kFloat));
  // dw bf16 (fp32-accumulated in the split-K slabs, rounded once at the
  // reduce) with channels_last memory [Kout][R][S][C]
  Tensor dw = at::empty({(int64_t)Kout, (int64_t)C, R, S},
                        x.options().dtype(at::kBFloat16),
                        at::MemoryFormat::ChannelsLast);
  CHK(conv_wgrad_implicit(dy.data_ptr(), x.data_ptr(),
                          partial.data_ptr<float>(), dw.data_ptr(), N, H, W, C,
                          Kout, (int)R, (int)S, (int)stride, (int)pad, HO, WO,
                          splits, 1, cur_stream()));
  return dw;
}

// c = a + b, bf16 16B-vectorized (the residual-join gradient sum)
static Tensor add_bf16_b(const Tensor &a, const Tensor &b) {
  const HIPDeviceGuard guard(a.device());
  TORCH_CHECK(a.numel() == b.numel() && a.numel() % 8 == 0);
  Tensor c = at::empty_like(a);
  CHK(add_bf16(a.data_ptr(), b.data_ptr(), c.data_ptr(), a.numel(), cur_stream()));
  return c;
}

// dx_acc += dgrad(dy, w) for a 1x1 stride-1 conv (bottleneck conv1)
static void conv2d_dgrad_acc(const Tensor &dy, const Tensor &w, Tensor &dx_acc) {
  check_cl_bf16(dy, "dy");
  check_cl_bf16(w, "w");
  check_cl_bf16(dx_acc, "dx_acc");
  const HIPDeviceGuard guard(dy.device());
  TORCH_CHECK(w.size(2) == 1 && w.size(3) == 1, "acc dgrad is 1x1-only");
  long M = (long)dy.size(0) * dy.size(2) * dy.size(3);
  CHK(conv_dgrad_1x1_acc(dy.data_ptr(), w.data_ptr(), dx_acc.data_ptr(), M,
                         (int)w.size(1), (int)dy.size(1), cur_stream()));
}

// ------------------------- batchnorm -------------------------
// res (optional, may be undefined): residual tensor folded into the apply
// pass — y = [relu](bn(x) + res), the bottleneck-join fusion.
// conv forward + fused BN partial stats: returns (y, slab) — slab empty
// when the shape takes a split-K route (stats would see partial sums).
static std::vector<Tensor> conv2d_fwd_bn(const Tensor &x, const Tensor &w,
                                         int64_t stride, int64_t pad) {
  check_cl_bf16(x, "x");
  check_cl_bf16(w, "w");
  const HIPDeviceGuard guard(x.device());
  int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  int Kout = w.size(0), R = w.size(2), S = w.size(3);
  int HO = (H + 2 * (int)pad - R) / (int)stride + 1;
  int WO = (W + 2 * (int)pad - S) / (int)stride + 1;
  long M = (long)N * HO * WO;
  int splits = conv_splits(M, Kout, R * S * C);
  auto f32 = x.options().dtype(at::kFloat);
  if (splits > 1 || Kout % 8 != 0) {
    Tensor y = conv2d_fwd(x, w, stride, pad);
    return {y, at::empty({0}, f32)};
  }
  Tensor y = empty_cl_bf16(N, Kout, HO, WO, x);
  long bands = (M + 63) / 64;
  Tensor slab = at::empty({bands, 2, (long)Kout}, f32);
  CHK(conv_fwd_bn(x.data_ptr(), w.data_ptr(), y.data_ptr(), N, H, W, C, Kout,
                  R, S, (int)stride, (int)pad, HO, WO,
                  slab.data_ptr<float>(), cur_stream()));
  return {y, slab};
}

// bn_fwd_train consuming the conv epilogue's stats slab (skips partials)
static std::vector<Tensor> bn_fwd_train_pre(
    const Tensor &x, const Tensor &slab, const Tensor &gamma,
    const Tensor &beta, double eps, bool relu, const Tensor &running_mean,
    const Tensor &running_var, double momentum, const Tensor &res) {
  check_cl_bf16(x, "x");
  const HIPDeviceGuard guard(x.device());
  int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  long M = (long)N * H * W;
  TORCH_CHECK(slab.numel() == ((M + 63) / 64) * 2 * (long)C, "slab shape");
  auto f32 = x.options().dtype(at::kFloat);
  Tensor y = empty_cl_bf16(N, C, H, W, x);
  Tensor mean = at::empty({C}, f32), invstd = at::empty({C}, f32);
  Tensor scale = at::empty({C}, f32), shift = at::empty({C}, f32);
  float *rm = running_mean.defined() && running_mean.numel() == C
                  ? running_mean.data_ptr<float>() : nullptr;
  float *rv = rm ? running_var.data_ptr<float>() : nullptr;
  const void *resp = nullptr;
  if (res.defined() && res.numel() > 0) {
    check_cl_bf16(res, "res");
    resp = res.data_ptr();
  }
  Tensor mask;
  uint8_t *mp = nullptr;
  if (relu) { // 1 bit/element relu mask: backward skips the y re-read
    mask = at::empty({M, (long)C / 8}, x.options().dtype(at::kByte));
    mp = mask.data_ptr<uint8_t>();
  } else {
    mask = at::empty({0}, x.options().dtype(at::kByte));
  }
  CHK(bn_fwd_train_pre_launch(
      slab.data_ptr<float>(), (int)((M + 63) / 64), x.data_ptr(), resp,
      gamma.data_ptr<float>(), beta.data_ptr<float>(), (float)eps,
      relu ? 1 : 0, y.data_ptr(), mp, mean.data_ptr<float>(),
      invstd.data_ptr<float>(), scale.data_ptr<float>(),
      shift.data_ptr<float>(), rm, rv, (float)momentum, M, C, cur_stream()));
  return {y, mean, invstd, mask};
}

static std::vector<Tensor> bn_fwd_train(const Tensor &x, const Tensor &gamma,
                                        const Tensor &beta, double eps,
                                        bool relu, const Tensor &running_mean,
                                        const Tensor &running_var,
                                        double momentum, const Tensor &res) {
  check_cl_bf16(x, "x");
  const HIPDeviceGuard guard(x.device());
  int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  long M = (long)N * H * W;
  auto f32 = x.options().dtype(at::kFloat);
  Tensor y = empty_cl_bf16(N, C, H, W, x);
  Tensor mean = at::empty({C}, f32), invstd = at::empty({C}, f32);
  Tensor scale = at::empty({C}, f32), shift = at::empty({C}, f32);
  Tensor partial = at::empty({(long)bn_grid_cap(C / 8) * 2 * C}, f32);
  float *rm = running_mean.defined() && running_mean.numel() == C
                  ? running_mean.data_ptr<float>() : nullptr;
  float *rv = rm ? running_var.data_ptr<float>() : nullptr;
  const void *resp = nullptr;
  if (res.defined() && res.numel() > 0) {
    check_cl_bf16(res, "res");
    resp = res.data_ptr();
  }
  Tensor mask;
  uint8_t *mp = nullptr;
  if (relu) { // 1 bit/element relu mask: backward skips the y re-read
    mask = at::empty({M, (long)C / 8}, x.options().dtype(at::kByte));
    mp = mask.data_ptr<uint8_t>();
  } else {
    mask = at::empty({0}, x.options().dtype(at::kByte));
  }
  CHK(bn_fwd_train_launch(x.data_ptr(), resp, gamma.data_ptr<float>(),
                          beta.data_ptr<float>(), (float)eps, relu ? 1 : 0,
                          y.data_ptr(), mp, mean.data_ptr<float>(),
                          invstd.data_ptr<float>(), scale.data_ptr<float>(),
                          shift.data_ptr<float>(), partial.data_ptr<float>(),
                          rm, rv, (float)momentum, M, C, cur_stream()));
  return {y, mean, invstd, mask};
}

static Tensor bn_fwd_eval(const Tensor &x, const Tensor &scale,
                          const Tensor &shift, bool relu) {
  check_cl_bf16(x, "x");
  const HIPDeviceGuard guard(x.device());
  int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  Tensor y = empty_cl_bf16(N, C, H, W, x);
  CHK(bn_fwd_eval_launch(x.data_ptr(), scale.contiguous().data_ptr<float>(),
                         shift.contiguous().data_ptr<float>(), relu ? 1 : 0,
                         y.data_ptr(), (long)N * H * W, C, cur_stream()));
  return y;
}

static std::vector<Tensor> bn_bwd(const Tensor &dy, const Tensor &x,
                                  const Tensor &y, const Tensor &gamma,
                                  const Tensor &mean, const Tensor &invstd,
                                  bool relu, const Tensor &mask) {
  check_cl_bf16(dy, "dy");
  check_cl_bf16(x, "x");
  const HIPDeviceGuard guard(x.device());
  int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  long M = (long)N * H * W;
  auto f32 = x.options().dtype(at::kFloat);
  Tensor dx = empty_cl_bf16(N, C, H, W, x);
  Tensor dgamma = at::empty({C}, f32), dbeta = at::empty({C}, f32);
  Tensor k1 = at::empty({C}, f32), k2 = at::empty({C}, f32), k3 = at::empty({C}, f32);
  Tensor partial = at::empty({(long)bn_grid_cap(C / 8) * 2 * C}, f32);
  const uint8_t *mp =
      mask.defined() && mask.numel() > 0 ? mask.data_ptr<uint8_t>() : nullptr;
  CHK(bn_bwd_launch(dy.data_ptr(), x.data_ptr(), y.data_ptr(), mp,
                    gamma.data_ptr<float>(), mean.data_ptr<float>(),
                    invstd.data_ptr<float>(), relu ? 1 : 0, dx.data_ptr(),
                    dgamma.data_ptr<float>(), dbeta.data_ptr<float>(),
                    k1.data_ptr<float>(), k2.data_ptr<float>(),
                    k3.data_ptr<float>(), partial.data_ptr<float>(), M, C,
                    cur_stream()));
  return {dx, dgamma, dbeta};
}

// ------------------------- pooling -------------------------
static std::vector<Tensor> maxpool_fwd(const Tensor &x, int64_t K,
                                       int64_t stride, int64_t pad) {
  check_cl_bf16(x, "x");
  const HIPDeviceGuard guard(x.device());
  int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  int HO = (H + 2 * (int)pad - (int)K) / (int)stride + 1;
  int WO = (W + 2 * (int)pad - (int)K) / (int)stride + 1;
  Tensor y = empty_cl_bf16(N, C, HO, WO, x);
  Tensor idx = at::empty({N, HO, WO, C}, x.options().dtype(at::kByte));
  CHK(maxpool_fwd_launch(x.data_ptr(), y.data_ptr(), idx.data_ptr<uint8_t>(),
                         N, H, W, HO, WO, C, (int)K, (int)stride, (int)pad,
                         cur_stream()));
  return {y, idx};
}

static Tensor maxpool_bwd(const Tensor &dy, const Tensor &idx, int64_t H,
                          int64_t W, int64_t K, int64_t stride, int64_t pad) {
  check_cl_bf16(dy, "dy");
  const HIPDeviceGuard guard(dy.device());
  int N = dy.size(0), C = dy.size(1), HO = dy.size(2), WO = dy.size(3);
  Tensor dx = empty_cl_bf16(N, C, H, W, dy);
  CHK(maxpool_bwd_launch(dy.data_ptr(), idx.data_ptr<uint8_t>(), dx.data_ptr(),
                         N, (int)H, (int)W, HO, WO, C, (int)K, (int)stride,
                         (int)pad, cur_stream()));
  return dx;
}

// ------------------------- gap -------------------------
static Tensor gap_fwd_b(const Tensor &x) {
  check_cl_bf16(x, "x");
  const HIPDeviceGuard guard(x.device());
  int N = x.size(0), C = x.size(1), HW = x.size(2) * x.size(3);
  Tensor y = at::empty({N, C}, x.options());
  CHK(gap_fwd(x.data_ptr(), y.data_ptr(), N, HW, C, cur_stream()));
  return y;
}

static Tensor gap_bwd_b(const Tensor &dy, int64_t H, int64_t W) {
  TORCH_CHECK(dy.is_cuda() && dy.scalar_type() == at::kBFloat16);
  const HIPDeviceGuard guard(dy.device());
  Tensor dyc = dy.contiguous();
  int N = dy.size(0), C = dy.size(1);
  Tensor dx = empty_cl_bf16(N, C, H, W, dy);
  CHK(gap_bwd(dyc.data_ptr(), dx.data_ptr(), N, (int)(H * W), C, cur_stream()));
  return dx;
}

// ------------------------- layernorm -------------------------
static std::vector<Tensor> layernorm_fwd(const Tensor &x, const Tensor &gamma,
                                         const Tensor &beta, double eps) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16);
  const HIPDeviceGuard guard(x.device());
  Tensor xc = x.contiguous();
  long M = xc.numel() / xc.size(-1);
  int N = xc.size(-1);
  auto f32 = x.options().dtype(at::kFloat);
  Tensor y = at::empty_like(xc);
  Tensor mean = at::empty({M}, f32), rstd = at::empty({M}, f32);
  CHK(ln_fwd(xc.data_ptr(), gamma.data_ptr<float>(), beta.data_ptr<float>(),
             y.data_ptr(), mean.data_ptr<float>(), rstd.data_ptr<float>(), M,
             N, (float)eps, cur_stream()));
  return {y, mean, rstd};
}

// y = LN(a + b) with the sum materialized for backward (one pass instead
// of a separate residual add)
static std::vector<Tensor> layernorm_add_fwd(const Tensor &a, const Tensor &b,
                                             const Tensor &gamma,
                                             const Tensor &beta, double eps) {
  const HIPDeviceGuard guard(a.device());
  Tensor ac = a.contiguous(), bc = b.contiguous();
  TORCH_CHECK(ac.sizes() == bc.sizes());
  long M = ac.numel() / ac.size(-1);
  int N = ac.size(-1);
  auto f32 = a.options().dtype(at::kFloat);
  Tensor sum = at::empty_like(ac);
  Tensor y = at::empty_like(ac);
  Tensor mean = at::empty({M}, f32), rstd = at::empty({M}, f32);
  CHK(ln_fwd_add(ac.data_ptr(), bc.data_ptr(), gamma.data_ptr<float>(),
                 beta.data_ptr<float>(), sum.data_ptr(), y.data_ptr(),
                 mean.data_ptr<float>(), rstd.data_ptr<float>(), M, N,
                 (float)eps, cur_stream()));
  return {y, sum, mean, rstd};
}

static std::vector<Tensor> layernorm_bwd(const Tensor &dy, const Tensor &x,
                                         const Tensor &gamma,
                                         const Tensor &mean,
                                         const Tensor &rstd) {
  const HIPDeviceGuard guard(x.device());
  Tensor dyc = dy.contiguous(), xc = x.contiguous();
  long M = xc.numel() / xc.size(-1);
  int N = xc.size(-1);
  auto f32 = x.options().dtype(at::kFloat);
  // slab rows must cover ln_bwd's stats grid (≤512 by construction)
  Tensor partial = at::empty({512, 2L * N}, f32);
  Tensor dgb = at::empty({2L * N}, f32);
  Tensor dx = at::empty_like(xc);
  CHK(ln_bwd(dyc.data_ptr(), xc.data_ptr(), gamma.data_ptr<float>(),
             mean.data_ptr<float>(), rstd.data_ptr<float>(), dx.data_ptr(),
             partial.data_ptr<float>(), dgb.data_ptr<float>(), M, N, nullptr,
             cur_stream()));
  return {dx, dgb.narrow(0, 0, N), dgb.narrow(0, N, N)};
}

// ------------------------- linear -------------------------
static Tensor linear_fwd(const Tensor &x, const Tensor &w, const Tensor &b) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16);
  const HIPDeviceGuard guard(x.device());
  Tensor xc = x.contiguous(), wc = w.contiguous();
  int M = xc.size(0), K = xc.size(1), N = wc.size(0);
  TORCH_CHECK(K % 8 == 0, "linear requires in_features %8==0");
  Tensor y = at::empty({M, N}, xc.options());
  // bias folded into the GEMM epilogue (fp32; accept bf16 bias (BERT) or
  // fp32 — the classifier keeps an fp32 bias for exactness)
  Tensor bf = b.scalar_type() == at::kFloat ? b.contiguous()
                                            : b.to(at::kFloat).contiguous();
  int fsp = gemm_fwd_splits(M, N, K);
  Tensor part;
  float *pp = nullptr;
  if (fsp > 1) {
    part = at::empty({fsp, (long)M * N}, x.options().dtype(at::kFloat));
    pp = part.data_ptr<float>();
  }
  CHK(gemm_nt_bias_sk(xc.data_ptr(), wc.data_ptr(), bf.data_ptr<float>(), pp,
                      y.data_ptr(), M, N, K, K, K, N, fsp, cur_stream()));
  return y;
}

static std::vector<Tensor> linear_bwd(const Tensor &dy, const Tensor &x,
                                      const Tensor &w) {
  const HIPDeviceGuard guard(x.device());
  Tensor dyc = dy.contiguous(), xc = x.contiguous(), wc = w.contiguous();
  int M = xc.size(0), K = xc.size(1), N = wc.size(0);
  auto f32 = x.options().dtype(at::kFloat);
  // dx = dy @ w: A = dy [M][N] k-contiguous; B = w [N rows][K cols]
  // k-strided (TN-staged in LDS — no transpose buffer). The NT operand's
  // reduce dim must be %8 (16 B load granule): zero-pad dy's N if ragged.
  long Np = (N + 7) / 8 * 8;
  Tensor dyp = dyc;
  if (Np != N) {
    dyp = at::zeros({M, Np}, dyc.options());
    dyp.narrow(1, 0, N).copy_(dyc);
  }
  Tensor dx = at::empty({M, K}, xc.options());
  int dx_splits = gemm_nt_tn_splits(M, K, N);
  Tensor dxp = dx_splits > 1 ? at::empty({dx_splits, (long)M * K}, f32) : dx;
  CHK(gemm_nt_tn_sk(dyp.data_ptr(), wc.data_ptr(),
                    dx_splits > 1 ? dxp.data_ptr<float>() : nullptr,
                    dx.data_ptr(), M, K, N, Np, K, K, dx_splits,
                    cur_stream()));
  // dw = dy^T @ x: both operands k-strided (k = batch row m) → fp32.
  // split-K when the [N][K] tile grid underfills the chip (K_reduce = M).
  // The PADDED dyp (lda = Np) keeps the vocab-ragged head on the pipe
  // route (tn_cols_ok's padded-stride contract).
  // dw in the weight's own dtype: bf16 params take the rounding inside the
  // GEMM epilogue / split-K reduce instead of an extra fp32 tensor + torch
  // cast kernel per weight per step
  bool wbf = wc.scalar_type() == at::kBFloat16;
  Tensor dw = at::empty({N, K}, wbf ? wc.options() : f32);
  int dw_splits = gemm_tn_tn_splits(N, K, M);
  Tensor dwp = dw_splits > 1
                   ? at::empty({dw_splits, (long)N * K}, f32)
                   : dw;
  CHK(gemm_tn_tn_sk(dyp.data_ptr(),
                    xc.data_ptr(),
                    dw_splits > 1 ? dwp.data_ptr<float>() : nullptr,
                    dw.data_ptr(), N, K, M, Np, K, K, dw_splits, wbf ? 1 : 0,
                    cur_stream()));
  // every colsum path writes all N entries — no zero-init needed
  Tensor db = at::empty({N}, f32);
  int chunks = colsum_chunks(M, N);
  Tensor dbp = chunks > 0 ? at::empty({chunks, (long)N}, f32) : db;
  CHK(colsum_bf16(dyc.data_ptr(), dbp.data_ptr<float>(),
                  db.data_ptr<float>(), M, N, N, cur_stream()));
  return {dx, dw, db};
}

// dx only (side-stream wgrad mode: the main stream runs the dgrad chain
// while linear_wgrad_only runs concurrently on a second stream)
static Tensor linear_dgrad(const Tensor &dy, const Tensor &w) {
  const HIPDeviceGuard guard(dy.device());
  Tensor dyc = dy.contiguous(), wc = w.contiguous();
  int M = dyc.size(0), N = wc.size(0), K = wc.size(1);
  auto f32 = dy.options().dtype(at::kFloat);
  long Np = (N + 7) / 8 * 8;
  Tensor dyp = dyc;
  if (Np != N) {
    dyp = at::zeros({M, Np}, dyc.options());
    dyp.narrow(1, 0, N).copy_(dyc);
  }
  Tensor dx = at::empty({M, K}, dyc.options());
  int dx_splits = gemm_nt_tn_splits(M, K, N);
  Tensor dxp = dx_splits > 1 ? at::empty({dx_splits, (long)M * K}, f32) : dx;
  CHK(gemm_nt_tn_sk(dyp.data_ptr(), wc.data_ptr(),
                    dx_splits > 1 ? dxp.data_ptr<float>() : nullptr,
                    dx.data_ptr(), M, K, N, Np, K, K, dx_splits,
                    cur_stream()));
  return dx;
}

// dx accumulated in place into `acc` (acc += dy @ w) — residual-join
// backward without the separate add pass
static Tensor linear_dgrad_acc(const Tensor &dy, const Tensor &w,
                               Tensor acc) {
  const HIPDeviceGuard guard(dy.device());
  Tensor dyc = dy.contiguous(), wc = w.contiguous();
  int M = dyc.size(0), N = wc.size(0), K = wc.size(1);
  TORCH_CHECK(N % 8 == 0, "linear_dgrad_acc requires out_features %8");
  TORCH_CHECK(acc.is_contiguous() && acc.size(0) == M && acc.size(1) == K);
  CHK(gemm_nt_tn_acc(dyc.data_ptr(), wc.data_ptr(), acc.data_ptr(), M, K, N,
                     N, K, K, cur_stream()));
  return acc;
}

// dw/db only (FFN backward computes fc2's dx separately with the fused
// dgelu epilogue — re-running the full linear_bwd would pay that GEMM twice)
static std::vector<Tensor> linear_wgrad_only(const Tensor &dy,
                                             const Tensor &x) {
  const HIPDeviceGuard guard(x.device());
  Tensor dyc = dy.contiguous(), xc = x.contiguous();
  int M = xc.size(0), K = xc.size(1), N = dyc.size(1);
  auto f32 = x.options().dtype(at::kFloat);
  bool wbf = xc.scalar_type() == at::kBFloat16; // dw matches the weight dtype
  Tensor dw = at::empty({N, K}, wbf ? xc.options() : f32);
  int dw_splits = gemm_tn_tn_splits(N, K, M);
  Tensor dwp =
      dw_splits > 1 ? at::empty({dw_splits, (long)N * K}, f32) : dw;
  CHK(gemm_tn_tn_sk(dyc.data_ptr(), xc.data_ptr(),
                    dw_splits > 1 ? dwp.data_ptr<float>() : nullptr,
                    dw.data_ptr(), N, K, M, N, K, K, dw_splits, wbf ? 1 : 0,
                    cur_stream()));
  Tensor db = at::empty({N}, f32);
  int chunks = colsum_chunks(M, N);
  Tensor dbp = chunks > 0 ? at::empty({chunks, (long)N}, f32) : db;
  CHK(colsum_bf16(dyc.data_ptr(), dbp.data_ptr<float>(),
                  db.data_ptr<float>(), M, N, N, cur_stream()));
  return {dw, db};
}

// ------------------------- fused attention -------------------------
// qkv: [B, S, 3, H, 64] (the fused projection output, untransposed);
// returns (ctx [B, S, H*64], probs [B, H, S, S] for the torch backward)
static std::vector<Tensor> attn_fwd(const Tensor &qkv, int64_t heads,
                                    double scale,
                                    const c10::optional<Tensor> &mask,
                                    bool want_probs) {
  TORCH_CHECK(qkv.is_cuda() && qkv.scalar_type() == at::kBFloat16);
  TORCH_CHECK(qkv.dim() == 5 && qkv.size(2) == 3 && qkv.size(4) == 64,
              "qkv must be [B,S,3,H,64]");
  TORCH_CHECK(qkv.is_contiguous());
  const HIPDeviceGuard guard(qkv.device());
  int B = qkv.size(0), S = qkv.size(1), H = qkv.size(3);
  TORCH_CHECK(H == heads);
  TORCH_CHECK(S <= 256, "fused attention supports seq <= 256");
  Tensor ctx = at::empty({B, S, (long)H * 64}, qkv.options());
  Tensor probs;
  const float *mp = nullptr;
  Tensor maskc;
  if (mask.has_value()) {
    maskc = mask->to(at::kFloat).contiguous();
    TORCH_CHECK(maskc.numel() == (long)B * S, "mask must be [B,S] additive");
    mp = maskc.data_ptr<float>();
  }
  void *pp = nullptr;
  if (want_probs) {
    probs = at::empty({B, (long)H, S, S}, qkv.options());
    pp = probs.data_ptr();
  }
  CHK(attn_fwd_launch(qkv.data_ptr(), ctx.data_ptr(), pp, mp, B, S, H,
                      (float)scale, cur_stream()));
  if (!want_probs) probs = at::empty({0}, qkv.options());
  return {ctx, probs};
}

static Tensor attn_bwd(const Tensor &qkv, const Tensor &dctx,
                       const Tensor &probs, double scale) {
  TORCH_CHECK(qkv.is_cuda() && qkv.scalar_type() == at::kBFloat16);
  const HIPDeviceGuard guard(qkv.device());
  int B = qkv.size(0), S = qkv.size(1), H = qkv.size(3);
  TORCH_CHECK(S == 128 && qkv.size(4) == 64, "fused attn bwd: S=128, D=64");
  TORCH_CHECK(qkv.is_contiguous() && probs.is_contiguous());
  Tensor dc = dctx.contiguous();
  Tensor dqkv = at::empty_like(qkv);
  CHK(attn_bwd_launch(qkv.data_ptr(), dc.data_ptr(), probs.data_ptr(),
                      dqkv.data_ptr(), B, S, H, (float)scale, cur_stream()));
  return dqkv;
}

// ------------------------- fused FFN (GELU) -------------------------
// fc1 forward with the bias+GELU in the epilogue: returns (g, h_pre)
static std::vector<Tensor> linear_gelu_fwd(const Tensor &x, const Tensor &w,
                                           const Tensor &b) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16 && x.dim() == 2);
  TORCH_CHECK(x.is_contiguous() && w.is_contiguous());
  const HIPDeviceGuard guard(x.device());
  int M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(w.size(1) == K);
  Tensor bias = b.to(at::kFloat).contiguous();
  Tensor g = at::empty({M, N}, x.options());
  Tensor deriv = at::empty({M, N}, x.options()); // gelu'(h), for backward
  CHK(gemm_nt_gelu_bias(x.data_ptr(), w.data_ptr(), bias.data_ptr<float>(),
                        deriv.data_ptr(), g.data_ptr(), M, N, K, K, K, N,
                        cur_stream()));
  return {g, deriv};
}

// fc2-dx with dgelu fused: dh = (dy·w2) ⊙ gelu'(h_pre)
static Tensor linear_gelu_dgrad(const Tensor &dy, const Tensor &w,
                                const Tensor &pre) {
  TORCH_CHECK(dy.is_cuda() && dy.scalar_type() == at::kBFloat16);
  TORCH_CHECK(dy.is_contiguous() && w.is_contiguous() && pre.is_contiguous());
  const HIPDeviceGuard guard(dy.device());
  int M = dy.size(0), K = dy.size(1), N = w.size(1);
  TORCH_CHECK(pre.size(0) == M && pre.size(1) == N);
  Tensor dh = at::empty({M, N}, dy.options());
  Tensor wt; // w^T scratch: unlocks the NT pipe256 route on full-tile shapes
  void *wtp = nullptr;
  if (gemm_gelubwd_wants_wt(M, N, K)) {
    wt = at::empty({N, K}, dy.options());
    wtp = wt.data_ptr();
  }
  CHK(gemm_nt_tn_gelubwd(dy.data_ptr(), w.data_ptr(), pre.data_ptr(),
                         dh.data_ptr(), M, N, K, K, N, N, wtp, cur_stream()));
  return dh;
}

// ------------------------- softmax xent -------------------------
static std::vector<Tensor> softmax_xent_fwd(const Tensor &logits,
                                            const Tensor &target) {
  TORCH_CHECK(logits.is_cuda() && logits.scalar_type() == at::kBFloat16);
  const HIPDeviceGuard guard(logits.device());
  int B = logits.size(0), V = logits.size(1);
  auto f32 = logits.options().dtype(at::kFloat);
  Tensor probs = at::empty({B, V}, f32);
  Tensor loss = at::empty({}, f32);
  CHK(softmax_xent_fwd_launch(logits.data_ptr(), target.data_ptr<long>(),
                              probs.data_ptr<float>(), loss.data_ptr<float>(),
                              B, V, cur_stream()));
  return {loss, probs};
}

static Tensor softmax_xent_bwd(const Tensor &probs, const Tensor &target,
                               const Tensor &dloss) {
  const HIPDeviceGuard guard(probs.device());
  int B = probs.size(0), V = probs.size(1);
  TORCH_CHECK(dloss.is_cuda() && dloss.scalar_type() == at::kFloat &&
              dloss.numel() == 1, "dloss must be a device fp32 scalar");
  Tensor d = at::empty({B, V}, probs.options().dtype(at::kBFloat16));
  CHK(softmax_xent_bwd_launch(probs.data_ptr<float>(), target.data_ptr<long>(),
                              d.data_ptr(), B, V, dloss.data_ptr<float>(),
                              cur_stream()));
  return d;
}

// MLM masked CE: fwd reads bf16 logits once, saves [B][2] stats + the
// (loss_sum, valid_count) pair; bwd recomputes probs from the bf16 logits
// (no fp32 probs materialization — 4× less HBM at vocab scale).
static std::vector<Tensor> masked_xent_fwd(const Tensor &logits,
                                           const Tensor &target,
                                           int64_t ignore_index) {
  TORCH_CHECK(logits.is_cuda() && logits.scalar_type() == at::kBFloat16);
  TORCH_CHECK(logits.is_contiguous());
  const HIPDeviceGuard guard(logits.device());
  int B = logits.size(0), V = logits.size(1);
  auto f32 = logits.options().dtype(at::kFloat);
  Tensor stats = at::empty({B, 2}, f32);
  Tensor out = at::empty({2}, f32); // loss_sum, valid_count
  CHK(masked_xent_fwd_launch(logits.data_ptr(), target.data_ptr<long>(),
                             stats.data_ptr<float>(), out.data_ptr<float>(),
                             B, V, V, ignore_index, cur_stream()));
  return {out, stats};
}

static Tensor masked_xent_bwd(const Tensor &logits, const Tensor &target,
                              const Tensor &stats, const Tensor &out,
                              const Tensor &dloss, int64_t ignore_index) {
  const HIPDeviceGuard guard(logits.device());
  int B = logits.size(0), V = logits.size(1);
  TORCH_CHECK(dloss.is_cuda() && dloss.scalar_type() == at::kFloat &&
              dloss.numel() == 1, "dloss must be a device fp32 scalar");
  Tensor d = at::empty({B, V}, logits.options());
  CHK(masked_xent_bwd_launch(logits.data_ptr(), target.data_ptr<long>(),
                             stats.data_ptr<float>(), out.data_ptr<float>(),
                             dloss.data_ptr<float>(), d.data_ptr(), B, V, V,
                             ignore_index, cur_stream()));
  return d;
}

// --------------- fused MLM head: decoder GEMM + masked CE ---------------
// The decoder writes logits into a [M][Vp] padded-vocab buffer (Vp =
// roundup(V, 8)); CE reads rows with stride Vp; backward re-derives
// probabilities into a padded dlogits whose pad columns the kernel zeroes,
// so the dx/dw GEMMs and db colsum consume it DIRECTLY (lda = Vp) — no
// per-step [M][Vp] zero-fill + strided pad copy (prof8: ~180 us/step), and
// no unpadded fp32/bf16 logits round-trips. Reference role: the MLM
// pretraining head of the out-of-tree BERT images.
static std::vector<Tensor> mlm_head_fwd(const Tensor &h, const Tensor &w,
                                        const Tensor &b, const Tensor &target,
                                        int64_t ignore_index) {
  TORCH_CHECK(h.is_cuda() && h.scalar_type() == at::kBFloat16 && h.dim() == 2);
  TORCH_CHECK(h.is_contiguous() && w.is_contiguous());
  TORCH_CHECK(b.scalar_type() == at::kFloat, "mlm bias must be fp32");
  const HIPDeviceGuard guard(h.device());
  int M = h.size(0), K = h.size(1), V = w.size(0);
  TORCH_CHECK(w.size(1) == K && target.numel() == M);
  long Vp = ((long)V + 7) / 8 * 8;
  auto f32 = h.options().dtype(at::kFloat);
  Tensor logits = at::empty({M, Vp}, h.options());
  Tensor bc = b.contiguous();
  CHK(gemm_nt_bias(h.data_ptr(), w.data_ptr(), bc.data_ptr<float>(),
                   logits.data_ptr(), M, V, K, K, K, Vp, cur_stream()));
  Tensor stats = at::empty({M, 2}, f32);
  Tensor out = at::empty({2}, f32); // loss_sum, valid_count
  CHK(masked_xent_fwd_launch(logits.data_ptr(), target.data_ptr<long>(),
                             stats.data_ptr<float>(), out.data_ptr<float>(),
                             M, V, Vp, ignore_index, cur_stream()));
  return {out, logits, stats};
}

static std::vector<Tensor> mlm_head_bwd(const Tensor &logits, int64_t V64,
                                        const Tensor &target,
                                        const Tensor &stats, const Tensor &out,
                                        const Tensor &dloss, const Tensor &h,
                                        const Tensor &w, int64_t ignore_index) {
  const HIPDeviceGuard guard(h.device());
  int M = logits.size(0), V = (int)V64, K = h.size(1);
  long Vp = logits.size(1);
  TORCH_CHECK(dloss.is_cuda() && dloss.scalar_type() == at::kFloat &&
              dloss.numel() == 1, "dloss must be a device fp32 scalar");
  auto f32 = h.options().dtype(at::kFloat);
  Tensor dlog = at::empty_like(logits);
  CHK(masked_xent_bwd_launch(logits.data_ptr(), target.data_ptr<long>(),
                             stats.data_ptr<float>(), out.data_ptr<float>(),
                             dloss.data_ptr<float>(), dlog.data_ptr(), M, V,
                             Vp, ignore_index, cur_stream()));
  // dh = dlog . w (reduce over padded vocab; pads are zero)
  Tensor dh = at::empty({M, K}, h.options());
  int dx_splits = gemm_nt_tn_splits(M, K, V);
  Tensor dhp = dx_splits > 1 ? at::empty({dx_splits, (long)M * K}, f32) : dh;
  CHK(gemm_nt_tn_sk(dlog.data_ptr(), w.data_ptr(),
                    dx_splits > 1 ? dhp.data_ptr<float>() : nullptr,
                    dh.data_ptr(), M, K, V, Vp, K, K, dx_splits,
                    cur_stream()));
  // dw (tied embedding grad) in the weight dtype straight from the GEMM
  bool wbf = w.scalar_type() == at::kBFloat16;
  Tensor dw = at::empty({(long)V, K}, wbf ? w.options() : f32);
  int dw_splits = gemm_tn_tn_splits(V, K, M);
  Tensor dwp = dw_splits > 1 ? at::empty({dw_splits, (long)V * K}, f32) : dw;
  CHK(gemm_tn_tn_sk(dlog.data_ptr(), h.data_ptr(),
                    dw_splits > 1 ? dwp.data_ptr<float>() : nullptr,
                    dw.data_ptr(), V, K, M, Vp, K, K, dw_splits, wbf ? 1 : 0,
                    cur_stream()));
  // db over the PADDED width: pads are zero, so summing Vp columns keeps
  // the %8-vectorized colsum8 slab path (the ragged V route ran 2-B scalar
  // loads at ~4x off bandwidth); the bias grad is the leading V entries
  Tensor db_pad = at::empty({Vp}, f32);
  int chunks = colsum_chunks(M, (int)Vp);
  Tensor dbp = chunks > 0 ? at::empty({chunks, Vp}, f32) : db_pad;
  CHK(colsum_bf16(dlog.data_ptr(), dbp.data_ptr<float>(),
                  db_pad.data_ptr<float>(), M, (int)Vp, Vp, cur_stream()));
  return {dh, dw, db_pad.narrow(0, 0, V)};
}

// ------------------------- add-relu -------------------------
static Tensor add_relu_fwd_b(const Tensor &a, const Tensor &b) {
  check_cl_bf16(a, "a");
  const HIPDeviceGuard guard(a.device());
  Tensor y = at::empty_like(a);
  CHK(add_relu_fwd(a.data_ptr(), b.data_ptr(), y.data_ptr(), a.numel(), cur_stream()));
  return y;
}

static Tensor add_relu_bwd_b(const Tensor &dy, const Tensor &y) {
  const HIPDeviceGuard guard(dy.device());
  Tensor dx = at::empty_like(dy);
  CHK(add_relu_bwd(dy.data_ptr(), y.data_ptr(), dx.data_ptr(), dy.numel(), cur_stream()));
  return dx;
}

static Tensor add_relu_bwd_add_b(const Tensor &dy, const Tensor &y,
                                 const Tensor &dx0) {
  const HIPDeviceGuard guard(dy.device());
  TORCH_CHECK(dy.numel() == y.numel() && dy.numel() == dx0.numel() &&
              dy.numel() % 8 == 0);
  Tensor out = at::empty_like(dx0);
  CHK(add_relu_bwd_add(dy.data_ptr(), y.data_ptr(), dx0.data_ptr(),
                       out.data_ptr(), dy.numel(), cur_stream()));
  return out;
}

// join backward off the bn relu-mask (1 byte per 8 channels): dxt =
// dx0 + dy·mask — drops the full y re-read
static Tensor add_relu_bwd_add_mask_b(const Tensor &dy, const Tensor &mask,
                                      const Tensor &dx0) {
  const HIPDeviceGuard guard(dy.device());
  TORCH_CHECK(dy.numel() == dx0.numel() && dy.numel() % 8 == 0 &&
              mask.numel() == dy.numel() / 8 &&
              mask.scalar_type() == at::kByte);
  Tensor out = at::empty_like(dx0);
  CHK(add_relu_bwd_add_mask(dy.data_ptr(), mask.data_ptr(), dx0.data_ptr(),
                            out.data_ptr(), dy.numel(), cur_stream()));
  return out;
}

// ------------------------- sgd -------------------------
static void sgd_step(std::vector<Tensor> masters, std::vector<Tensor> grads,
                     std::vector<Tensor> momenta, std::vector<Tensor> outs,
                     double lr, double mu, double wd, bool nesterov) {
  TORCH_CHECK(!masters.empty());
  const HIPDeviceGuard guard(masters[0].device());
  std::vector<SgdDesc> by_dtype[2]; // 0: f32 grads, 1: bf16 grads
  for (size_t i = 0; i < masters.size(); ++i) {
    SgdDesc d;
    d.grad = grads[i].data_ptr();
    d.master = masters[i].data_ptr<float>();
    d.mom = momenta[i].data_ptr<float>();
    d.out = outs[i].numel() > 0 ? (uint16_t *)outs[i].data_ptr() : nullptr;
    d.numel = masters[i].numel();
    by_dtype[grads[i].scalar_type() == at::kBFloat16 ? 1 : 0].push_back(d);
  }
  for (int t = 0; t < 2; ++t)
    if (!by_dtype[t].empty())
      CHK(sgd_step_launch(by_dtype[t].data(), (int)by_dtype[t].size(), t,
                          (float)lr, (float)mu, (float)wd, nesterov ? 1 : 0,
                          cur_stream()));
}

// ------------------------- raw gemm (tests) -------------------------
static Tensor gemm_nt_b(const Tensor &a, const Tensor &b, bool c_f32) {
  const HIPDeviceGuard guard(a.device());
  Tensor ac = a.contiguous(), bc = b.contiguous();
  int M = ac.size(0), K = ac.size(1), N = bc.size(0);
  TORCH_CHECK(bc.size(1) == K);
  TORCH_CHECK(K % 8 == 0, "gemm_nt requires K%8==0");
  Tensor c = at::empty({M, N}, ac.options().dtype(c_f32 ? at::kFloat : at::kBFloat16));
  CHK(gemm_nt(ac.data_ptr(), bc.data_ptr(), c.data_ptr(), M, N, K, K, K, N,
              c_f32 ? 1 : 0, cur_stream()));
  return c;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("conv2d_fwd", &conv2d_fwd);
  m.def("conv2d_dgrad", &conv2d_dgrad);
  m.def("conv2d_fwd_bn", &conv2d_fwd_bn);
  m.def("bn_fwd_train_pre", &bn_fwd_train_pre);
  m.def("conv2d_wgrad", &conv2d_wgrad);
  m.def("conv2d_dgrad_acc", &conv2d_dgrad_acc);
  m.def("add_bf16", &add_bf16_b);
  m.def("bn_fwd_train", &bn_fwd_train);
  m.def("bn_fwd_eval", &bn_fwd_eval);
  m.def("bn_bwd", &bn_bwd);
  m.def("maxpool_fwd", &maxpool_fwd);
  m.def("maxpool_bwd", &maxpool_bwd);
  m.def("gap_fwd", &gap_fwd_b);
  m.def("gap_bwd", &gap_bwd_b);
  m.def("layernorm_fwd", &layernorm_fwd);
  m.def("layernorm_bwd", &layernorm_bwd);
  m.def("layernorm_add_fwd", &layernorm_add_fwd);
  m.def("linear_fwd", &linear_fwd);
  m.def("linear_bwd", &linear_bwd);
  m.def("linear_dgrad", &linear_dgrad);
  m.def("linear_dgrad_acc", &linear_dgrad_acc);
  m.def("softmax_xent_fwd", &softmax_xent_fwd);
  m.def("masked_xent_fwd", &masked_xent_fwd);
  m.def("mlm_head_fwd", &mlm_head_fwd);
  m.def("mlm_head_bwd", &mlm_head_bwd);
  m.def("linear_gelu_fwd", &linear_gelu_fwd);
  m.def("attn_fwd", &attn_fwd);
  m.def("attn_bwd", &attn_bwd);
  m.def("linear_wgrad_only", &linear_wgrad_only);
  m.def("linear_gelu_dgrad", &linear_gelu_dgrad);
  m.def("masked_xent_bwd", &masked_xent_bwd);
  m.def("softmax_xent_bwd", &softmax_xent_bwd);
  m.def("add_relu_fwd", &add_relu_fwd_b);
  m.def("add_relu_bwd", &add_relu_bwd_b);
  m.def("add_relu_bwd_add", &add_relu_bwd_add_b);
  m.def("add_relu_bwd_add_mask", &add_relu_bwd_add_mask_b);
  m.def("sgd_step", &sgd_step);
  m.def("gemm_nt", &gemm_nt_b);
  m.def("mfma_probe", [](const Tensor &a, const Tensor &b) {
    const HIPDeviceGuard guard(a.device());
    Tensor d = at::empty({16, 16}, a.options().dtype(at::kFloat));
    CHK(mfma_probe(a.contiguous().data_ptr(), b.contiguous().data_ptr(),
                   d.data_ptr<float>(), cur_stream()));
    return d;
  });
}
