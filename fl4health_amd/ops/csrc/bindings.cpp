// Python bindings for the CDNA4 flat-buffer FL kernels (see flat_ops.hip).
#include <torch/extension.h>
#include <cstdlib>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

extern "C" {
void launch_axpby(float*, const float*, float, float, int64_t, hipStream_t);
void launch_prox_sgd(float*, const void*, int, const float*, float*, unsigned short*, float,
                     float, const float*, float, float, int, int64_t, hipStream_t);
void launch_scaffold_sgd(float*, const float*, const float*, const float*, float, float, int64_t,
                         hipStream_t);
void launch_scaffold_variate(float*, float*, const float*, const float*, const float*, float,
                             int64_t, hipStream_t);
void launch_server_opt(float*, const float*, float*, float*, float*, int, float, float, float,
                       float, float, int64_t, hipStream_t);
void launch_reduce(const float*, const float*, double*, double*, int, int64_t, hipStream_t);
void launch_clip_delta(float*, const float*, const float*, const double*, float, float*, int64_t,
                       hipStream_t);
void launch_gaussian_noise(float*, float, float, uint64_t, uint64_t, int64_t, hipStream_t);
void launch_bernoulli_mask(const float*, const float*, float*, float*, uint64_t, uint64_t, int,
                           int64_t, hipStream_t);
void launch_per_sample_sqnorm(const float*, float*, int64_t, int64_t, hipStream_t);
void launch_clip_rowsum(const float*, const float*, float*, float, int64_t, int64_t, hipStream_t);
void launch_clip_rowsum_noise(const float*, const float*, float*, float, float, uint64_t,
                              uint64_t, int64_t, int64_t, hipStream_t);
void launch_confusion(const int64_t*, const int64_t*, unsigned long long*, int, int64_t,
                      hipStream_t);
void launch_weighted_sum_rows(const float*, const float*, float*, int, int64_t, hipStream_t);
void launch_bn_fwd(const void*, void*, float*, float*, float*, const float*, const float*, float*,
                   float*, int64_t*, float, float, int64_t, int, int, int, int, const void*,
                   hipStream_t);
void launch_conv3x3_fwd_kb32(const void*, const void*, const float*, void*, int, int, int, int,
                             int, hipStream_t);
void launch_pack_kb32(const void*, void*, int, int, int, hipStream_t);
void launch_conv3x3_fwd_kzloop(const void*, const void*, const float*, void*, int, int, int, int,
                               int, hipStream_t);
void launch_conv3x3_wrw(const void*, const void*, void*, float*, void*, int, int, int, int, int,
                        hipStream_t);
void launch_coo_count(const float*, float, int64_t, int64_t, int32_t*, hipStream_t);
void launch_coo_write(const float*, const float*, float, const int32_t*, float*, int64_t*,
                      const int64_t*, int, int64_t, int64_t, int64_t, hipStream_t);
void launch_moon_contrastive(const float*, const float*, const float*, int, int, int64_t, float,
                             float*, float*, hipStream_t);
void launch_in3d_fwd(const void*, void*, float*, float*, float*, const float*, const float*, int,
                     int, int64_t, int, float, float, hipStream_t);
void launch_in3d_bwd(const void*, const void*, void*, float*, float*, float*, float*, float*,
                     const float*, const float*, const float*, const float*, int, int, int64_t,
                     int, float, hipStream_t);
void launch_conv3x3_fwd(const void*, const void*, const float*, void*, int, int, int, int, int,
                        hipStream_t);
void launch_mkmmd_sums(const float*, const float*, double*, float*, int, int64_t, int64_t, int,
                       hipStream_t);
void launch_mkmmd_backward(const float*, const float*, const float*, float*, int, int64_t,
                           int64_t, int, hipStream_t);
void launch_bn_bwd(const void*, const void*, void*, float*, const float*, const float*,
                   const float*, const float*, float*, float*, float*, float*, int64_t, int, int,
                   int, int, const void*, void*, hipStream_t);
}

namespace {

void check_f32(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
  TORCH_CHECK(t.scalar_type() == torch::kFloat32, name, " must be fp32");
}

hipStream_t stream() { return c10::hip::getCurrentHIPStream().stream(); }

void axpby_(torch::Tensor y, torch::Tensor x, double a, double b) {
  check_f32(y, "y");
  check_f32(x, "x");
  TORCH_CHECK(x.numel() == y.numel(), "size mismatch");
  launch_axpby(y.data_ptr<float>(), x.data_ptr<float>(), (float)a, (float)b, y.numel(), stream());
}

void prox_sgd_step_(torch::Tensor p, torch::Tensor g, c10::optional<torch::Tensor> w0,
                    c10::optional<torch::Tensor> mbuf, c10::optional<torch::Tensor> mirror,
                    double lr, double mu, c10::optional<torch::Tensor> mu_dev, double momentum,
                    double weight_decay, bool nesterov) {
  check_f32(p, "p");
  TORCH_CHECK(g.is_cuda() && g.is_contiguous(), "g must be contiguous GPU");
  int grad_bf16 = g.scalar_type() == torch::kBFloat16 ? 1 : 0;
  if (!grad_bf16) TORCH_CHECK(g.scalar_type() == torch::kFloat32, "g must be fp32 or bf16");
  TORCH_CHECK(g.numel() == p.numel(), "grad size mismatch");
  const float* w0p = nullptr;
  float* mp = nullptr;
  const float* mudp = nullptr;
  unsigned short* mirp = nullptr;
  if (w0.has_value()) { check_f32(*w0, "w0"); w0p = w0->data_ptr<float>(); }
  if (mbuf.has_value()) { check_f32(*mbuf, "mbuf"); mp = mbuf->data_ptr<float>(); }
  if (mu_dev.has_value()) { check_f32(*mu_dev, "mu_dev"); mudp = mu_dev->data_ptr<float>(); }
  if (mirror.has_value()) {
    TORCH_CHECK(mirror->scalar_type() == torch::kBFloat16 && mirror->numel() == p.numel());
    mirp = reinterpret_cast<unsigned short*>(mirror->data_ptr());
  }
  launch_prox_sgd(p.data_ptr<float>(), g.data_ptr(), grad_bf16, w0p, mp, mirp, (float)lr,
                  (float)mu, mudp, (float)momentum, (float)weight_decay, nesterov ? 1 : 0,
                  p.numel(), stream());
}

void scaffold_sgd_step_(torch::Tensor p, torch::Tensor g, torch::Tensor c, torch::Tensor ci,
                        double lr, double weight_decay) {
  check_f32(p, "p"); check_f32(g, "g"); check_f32(c, "c"); check_f32(ci, "ci");
  launch_scaffold_sgd(p.data_ptr<float>(), g.data_ptr<float>(), c.data_ptr<float>(),
                      ci.data_ptr<float>(), (float)lr, (float)weight_decay, p.numel(), stream());
}

void scaffold_variate_update_(torch::Tensor ci, torch::Tensor dci, torch::Tensor c,
                              torch::Tensor x_start, torch::Tensor y_end, double inv_klr) {
  check_f32(ci, "ci"); check_f32(dci, "dci"); check_f32(c, "c");
  check_f32(x_start, "x_start"); check_f32(y_end, "y_end");
  launch_scaffold_variate(ci.data_ptr<float>(), dci.data_ptr<float>(), c.data_ptr<float>(),
                          x_start.data_ptr<float>(), y_end.data_ptr<float>(), (float)inv_klr,
                          ci.numel(), stream());
}

void server_opt_step_(torch::Tensor x, torch::Tensor delta, torch::Tensor m, torch::Tensor v,
                      torch::Tensor dt, int64_t kind, double b1, double b2, double b3, double lr,
                      double tau) {
  check_f32(x, "x"); check_f32(delta, "delta"); check_f32(m, "m"); check_f32(v, "v");
  check_f32(dt, "dt");
  launch_server_opt(x.data_ptr<float>(), delta.data_ptr<float>(), m.data_ptr<float>(),
                    v.data_ptr<float>(), dt.data_ptr<float>(), (int)kind, (float)b1, (float)b2,
                    (float)b3, (float)lr, (float)tau, x.numel(), stream());
}

torch::Tensor reduce_op(torch::Tensor x, c10::optional<torch::Tensor> y, int64_t mode) {
  check_f32(x, "x");
  const float* yp = nullptr;
  if (y.has_value()) {
    check_f32(*y, "y");
    TORCH_CHECK(y->numel() == x.numel(), "size mismatch");
    yp = y->data_ptr<float>();
  }
  auto opts = torch::TensorOptions().dtype(torch::kFloat64).device(x.device());
  auto partial = torch::empty({1024}, opts);
  auto out = torch::empty({1}, opts);
  launch_reduce(x.data_ptr<float>(), yp, partial.data_ptr<double>(), out.data_ptr<double>(),
                (int)mode, x.numel(), stream());
  return out;
}

torch::Tensor clip_delta(torch::Tensor w, torch::Tensor w0, torch::Tensor sqnorm,
                         double clip_bound, c10::optional<torch::Tensor> clip_bit) {
  check_f32(w, "w"); check_f32(w0, "w0");
  TORCH_CHECK(sqnorm.scalar_type() == torch::kFloat64 && sqnorm.is_cuda(), "sqnorm f64 gpu");
  auto out = torch::empty_like(w);
  float* bitp = nullptr;
  if (clip_bit.has_value()) { check_f32(*clip_bit, "clip_bit"); bitp = clip_bit->data_ptr<float>(); }
  launch_clip_delta(out.data_ptr<float>(), w.data_ptr<float>(), w0.data_ptr<float>(),
                    sqnorm.data_ptr<double>(), (float)clip_bound, bitp, w.numel(), stream());
  return out;
}

void gaussian_noise_(torch::Tensor x, double a, double sigma, int64_t seed, int64_t offset) {
  check_f32(x, "x");
  launch_gaussian_noise(x.data_ptr<float>(), (float)a, (float)sigma, (uint64_t)seed,
                        (uint64_t)offset, x.numel(), stream());
}

std::vector<torch::Tensor> bernoulli_mask(torch::Tensor scores, c10::optional<torch::Tensor> w,
                                          int64_t seed, int64_t offset, bool apply_sigmoid) {
  check_f32(scores, "scores");
  auto mask = torch::empty_like(scores);
  torch::Tensor weff;
  const float* wp = nullptr;
  float* weffp = nullptr;
  if (w.has_value()) {
    check_f32(*w, "w");
    weff = torch::empty_like(*w);
    wp = w->data_ptr<float>();
    weffp = weff.data_ptr<float>();
  }
  launch_bernoulli_mask(scores.data_ptr<float>(), wp, mask.data_ptr<float>(), weffp,
                        (uint64_t)seed, (uint64_t)offset, apply_sigmoid ? 1 : 0, scores.numel(),
                        stream());
  if (w.has_value()) return {mask, weff};
  return {mask};
}

void per_sample_sqnorm_(torch::Tensor g, torch::Tensor out) {
  check_f32(g, "g");
  check_f32(out, "out");
  int64_t B = g.size(0);
  int64_t D = g.numel() / B;
  TORCH_CHECK(out.numel() == B, "out must be [B]");
  launch_per_sample_sqnorm(g.data_ptr<float>(), out.data_ptr<float>(), B, D, stream());
}

void clip_rowsum_(torch::Tensor g, torch::Tensor sqnorms, torch::Tensor out, double clip_bound) {
  check_f32(g, "g"); check_f32(sqnorms, "sqnorms"); check_f32(out, "out");
  int64_t B = g.size(0);
  int64_t D = g.numel() / B;
  TORCH_CHECK(out.numel() == D, "out must be [D]");
  launch_clip_rowsum(g.data_ptr<float>(), sqnorms.data_ptr<float>(), out.data_ptr<float>(),
                     (float)clip_bound, B, D, stream());
}

// fused clipped-rowsum + DP Gaussian noise (one pass; Philox stream matches
// gaussian_noise_ for the same seed/offset)
void clip_rowsum_noise_(torch::Tensor g, torch::Tensor sqnorms, torch::Tensor out,
                        double clip_bound, double sigma, int64_t seed, int64_t offset) {
  check_f32(g, "g"); check_f32(sqnorms, "sqnorms"); check_f32(out, "out");
  int64_t B = g.size(0);
  int64_t D = g.numel() / B;
  TORCH_CHECK(out.numel() == D, "out must be [D]");
  launch_clip_rowsum_noise(g.data_ptr<float>(), sqnorms.data_ptr<float>(), out.data_ptr<float>(),
                           (float)clip_bound, (float)sigma, (uint64_t)seed, (uint64_t)offset, B, D,
                           stream());
}

void confusion_counts_(torch::Tensor preds, torch::Tensor targets, torch::Tensor out) {
  TORCH_CHECK(preds.is_cuda() && preds.scalar_type() == torch::kInt64 && preds.is_contiguous());
  TORCH_CHECK(targets.is_cuda() && targets.scalar_type() == torch::kInt64 && targets.is_contiguous());
  TORCH_CHECK(out.is_cuda() && out.scalar_type() == torch::kInt64 && out.is_contiguous());
  int C = (int)out.size(0);
  TORCH_CHECK(out.size(1) == 4, "out must be [C,4]");
  launch_confusion(preds.data_ptr<int64_t>(), targets.data_ptr<int64_t>(),
                   reinterpret_cast<unsigned long long*>(out.data_ptr<int64_t>()), C,
                   preds.numel(), stream());
}

torch::Tensor weighted_sum_rows(torch::Tensor stack, torch::Tensor w) {
  check_f32(stack, "stack");
  check_f32(w, "w");
  int K = (int)stack.size(0);
  int64_t n = stack.numel() / K;
  TORCH_CHECK(w.numel() == K, "w must be [K]");
  auto out = torch::empty({n}, stack.options());
  launch_weighted_sum_rows(stack.data_ptr<float>(), w.data_ptr<float>(), out.data_ptr<float>(), K,
                           n, stream());
  return out;
}

// Partial-group (reduce BLOCK) count Gb: a pure function of the shape
// (deterministic). Each reduce block is 128 channels x 8 row-stripes
// (BNRW in bn_ops.hip), so thread occupancy matches the old 1-stripe
// 4096-block grid while the partial array (and the finalize kernel's read
// volume) shrinks 8x.
int bn_groups(int64_t R, int C) {
  static int target = [] {
    const char* e = getenv("FL4_BN_GB");
    return e ? atoi(e) : 512;
  }();
  int cblocks = (C + 127) / 128;
  int64_t g = target / cblocks;
  if (g > (R + 31) / 32) g = (R + 31) / 32;  // >= ~4 rows per stripe
  if (g < 1) g = 1;
  if (g > 1024) g = 1024;
  return (int)g;
}

int bn_dtype_of(const torch::Tensor& t) {
  if (t.scalar_type() == torch::kBFloat16) return 1;
  TORCH_CHECK(t.scalar_type() == torch::kFloat32, "bn ops support fp32/bf16");
  return 0;
}

// x: NHWC-flattened [R, C] contiguous. Returns (y, save_mean, save_invstd).
// wrw: dW for 3x3/s1/p1 NHWC bf16 (C,K in {64,128}; W=32 or 16 - the ResNet
// layer-1/layer-2 families; >64-channel cases run as 64x64 sub-slices).
// Returns bf16 [K, C, 3, 3] in channels_last memory (matches what
// aten::convolution_backward hands back for channels_last convs).
torch::Tensor conv3x3_wrw(torch::Tensor x, torch::Tensor dy) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 4, "x must be NHWC contiguous");
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous() && dy.dim() == 4, "dy must be NHWC contiguous");
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16 && dy.scalar_type() == torch::kBFloat16);
  int Nn = (int)x.size(0), H = (int)x.size(1), W = (int)x.size(2), C = (int)x.size(3);
  int K = (int)dy.size(3);
  TORCH_CHECK((C == 64 || C == 128) && (K == 64 || K == 128), "wrw supports C,K in {64,128}");
  TORCH_CHECK((W == 32 && H % 4 == 0) || (W == 16 && H % 8 == 0), "wrw supports W=32 (BH 4) or W=16 (BH 8)");
  auto bopts = torch::TensorOptions().dtype(torch::kBFloat16).device(x.device());
  // one 64x64 sub-slice at a time: partial/mid are reused across sub-launches
  auto partial = torch::empty({256LL * 9 * 64 * 64}, bopts);
  auto mid = torch::empty({16LL * 9 * 64 * 64},
                          torch::TensorOptions().dtype(torch::kFloat32).device(x.device()));
  auto dw = torch::empty_strided({K, C, 3, 3}, {(int64_t)9 * C, 1, (int64_t)3 * C, C}, bopts);
  launch_conv3x3_wrw(x.data_ptr(), dy.data_ptr(), partial.data_ptr(), mid.data_ptr<float>(),
                     dw.data_ptr(), Nn, H, W, C, K, stream());
  return dw;
}

std::vector<torch::Tensor> bn_fwd_train(torch::Tensor x, c10::optional<torch::Tensor> gamma,
                                        c10::optional<torch::Tensor> beta,
                                        c10::optional<torch::Tensor> running_mean,
                                        c10::optional<torch::Tensor> running_var, double momentum,
                                        double eps, bool fuse_relu,
                                        c10::optional<torch::Tensor> res,
                                        c10::optional<torch::Tensor> num_batches_tracked) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 2, "x must be [R, C] contiguous");
  int64_t R = x.size(0);
  int C = (int)x.size(1);
  int dtype = bn_dtype_of(x);
  int G = bn_groups(R, C);
  auto fopts = torch::TensorOptions().dtype(torch::kFloat32).device(x.device());
  auto y = torch::empty_like(x);
  auto partial = torch::empty({2LL * G * C}, fopts);
  auto mean = torch::empty({C}, fopts);
  auto invstd = torch::empty({C}, fopts);
  launch_bn_fwd(
      x.data_ptr(), y.data_ptr(), partial.data_ptr<float>(), mean.data_ptr<float>(),
      invstd.data_ptr<float>(),
      gamma.has_value() ? gamma->data_ptr<float>() : nullptr,
      beta.has_value() ? beta->data_ptr<float>() : nullptr,
      running_mean.has_value() ? running_mean->data_ptr<float>() : nullptr,
      running_var.has_value() ? running_var->data_ptr<float>() : nullptr,
      num_batches_tracked.has_value() ? num_batches_tracked->data_ptr<int64_t>() : nullptr,
      (float)momentum, (float)eps, R, C, G, dtype, fuse_relu ? 1 : 0,
      res.has_value() ? res->data_ptr() : nullptr, stream());
  return {y, mean, invstd};
}

std::vector<torch::Tensor> bn_bwd(torch::Tensor x, torch::Tensor dy, torch::Tensor mean,
                                  torch::Tensor invstd, c10::optional<torch::Tensor> gamma,
                                  c10::optional<torch::Tensor> beta, bool fuse_relu,
                                  c10::optional<torch::Tensor> res) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && dy.is_contiguous() && x.sizes() == dy.sizes());
  int64_t R = x.size(0);
  int C = (int)x.size(1);
  int dtype = bn_dtype_of(x);
  int G = bn_groups(R, C);
  auto fopts = torch::TensorOptions().dtype(torch::kFloat32).device(x.device());
  auto dx = torch::empty_like(x);
  auto partial = torch::empty({2LL * G * C}, fopts);
  auto sum_dy = torch::empty({C}, fopts);
  auto sum_dy_xhat = torch::empty({C}, fopts);
  auto dgamma = torch::empty({C}, fopts);
  auto dbeta = torch::empty({C}, fopts);
  torch::Tensor dres;
  void* dres_ptr = nullptr;
  if (res.has_value()) {
    dres = torch::empty_like(x);
    dres_ptr = dres.data_ptr();
  }
  launch_bn_bwd(x.data_ptr(), dy.data_ptr(), dx.data_ptr(), partial.data_ptr<float>(),
                mean.data_ptr<float>(), invstd.data_ptr<float>(),
                gamma.has_value() ? gamma->data_ptr<float>() : nullptr,
                beta.has_value() ? beta->data_ptr<float>() : nullptr, sum_dy.data_ptr<float>(),
                sum_dy_xhat.data_ptr<float>(), dgamma.data_ptr<float>(), dbeta.data_ptr<float>(),
                R, C, G, dtype, fuse_relu ? 1 : 0,
                res.has_value() ? res->data_ptr() : nullptr, dres_ptr, stream());
  if (res.has_value()) return {dx, dgamma, dbeta, dres};
  return {dx, dgamma, dbeta};
}

// Direct 3x3 s1 p1 NHWC bf16 conv forward on MFMA (round-2 conv prototype).
// x: [N, H, W, C] bf16 (channels-last); w: [9, C, K] bf16 prepacked
// (torch weight [K, C, 3, 3] -> permute(2, 3, 1, 0).reshape(9, C, K));
// returns [N, H, W, K] bf16.
torch::Tensor conv3x3_fwd(torch::Tensor x, torch::Tensor w,
                          c10::optional<torch::Tensor> bias) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16 && x.dim() == 4 && x.is_contiguous());
  TORCH_CHECK(w.is_cuda() && w.scalar_type() == torch::kBFloat16 && w.dim() == 3 && w.is_contiguous());
  TORCH_CHECK(w.size(0) == 9 && w.size(1) == x.size(3), "w must be [9, C, K]");
  int64_t N = x.size(0), H = x.size(1), W = x.size(2), C = x.size(3), K = w.size(2);
  TORCH_CHECK(W <= 32, "conv3x3_fwd prototype supports W <= 32");
  if (bias.has_value()) check_f32(*bias, "bias");
  auto y = torch::empty({N, H, W, K}, x.options());
  launch_conv3x3_fwd(x.data_ptr(), w.data_ptr(),
                     bias.has_value() ? bias->data_ptr<float>() : nullptr, y.data_ptr(),
                     (int)N, (int)H, (int)W, (int)C, (int)K, stream());
  return y;
}

// Variant C of the direct conv: KB = 32, both operands glds-pipelined.
// wimg: [K/32, C/64, 9, 32, 64] bf16 — the exact LDS image, packed host-side
// (ops/conv.py pack_weight_kb32). Requires C % 64 == 0 and K % 32 == 0.
torch::Tensor conv3x3_fwd_kb32(torch::Tensor x, torch::Tensor wimg,
                               c10::optional<torch::Tensor> bias) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16 && x.dim() == 4 && x.is_contiguous());
  TORCH_CHECK(wimg.is_cuda() && wimg.scalar_type() == torch::kBFloat16 && wimg.dim() == 5 &&
              wimg.is_contiguous());
  int64_t N = x.size(0), H = x.size(1), W = x.size(2), C = x.size(3);
  int64_t K = wimg.size(0) * 32;
  TORCH_CHECK(wimg.size(1) * 64 == C && wimg.size(2) == 9 && wimg.size(3) == 32 &&
              wimg.size(4) == 64, "wimg must be [K/32, C/64, 9, 32, 64]");
  TORCH_CHECK(W <= 32, "conv3x3 prototype supports W <= 32");
  if (bias.has_value()) check_f32(*bias, "bias");
  auto y = torch::empty({N, H, W, K}, x.options());
  launch_conv3x3_fwd_kb32(x.data_ptr(), wimg.data_ptr(),
                          bias.has_value() ? bias->data_ptr<float>() : nullptr, y.data_ptr(),
                          (int)N, (int)H, (int)W, (int)C, (int)K, stream());
  return y;
}

// Variant D: input-resident multi-kz (one block per tile loops every
// K-block; input staged once, weights pipelined). Same wimg pack as kb32.
torch::Tensor conv3x3_fwd_kzloop(torch::Tensor x, torch::Tensor wimg,
                                 c10::optional<torch::Tensor> bias) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16 && x.dim() == 4 && x.is_contiguous());
  TORCH_CHECK(wimg.is_cuda() && wimg.scalar_type() == torch::kBFloat16 && wimg.dim() == 5 &&
              wimg.is_contiguous());
  int64_t N = x.size(0), H = x.size(1), W = x.size(2), C = x.size(3);
  int64_t K = wimg.size(0) * 32;
  TORCH_CHECK(wimg.size(1) * 64 == C && wimg.size(2) == 9, "wimg must be [K/32, C/64, 9, 32, 64]");
  if (bias.has_value()) check_f32(*bias, "bias");
  auto y = torch::empty({N, H, W, K}, x.options());
  launch_conv3x3_fwd_kzloop(x.data_ptr(), wimg.data_ptr(),
                            bias.has_value() ? bias->data_ptr<float>() : nullptr, y.data_ptr(),
                            (int)N, (int)H, (int)W, (int)C, (int)K, stream());
  return y;
}

// Fused weight pack for conv3x3_fwd_kb32: [K, C, 3, 3] -> swizzled LDS-image
// slabs in ONE kernel (replaces a ~6-op torch chain per conv per step).
// bwd=true packs the bwd-data weights (roles swapped, taps rotated).
torch::Tensor pack_kb32(torch::Tensor w, bool bwd) {
  TORCH_CHECK(w.is_cuda() && w.scalar_type() == torch::kBFloat16 && w.dim() == 4 &&
              w.is_contiguous() && w.size(2) == 3 && w.size(3) == 3, "w must be [K, C, 3, 3] bf16");
  int K = (int)w.size(0), C = (int)w.size(1);
  int Kc = bwd ? C : K, Cc = bwd ? K : C;
  TORCH_CHECK(Kc % 32 == 0 && Cc % 64 == 0, "conv-role dims must be K%32==0, C%64==0");
  auto out = torch::empty({Kc / 32, Cc / 64, 9, 32, 64}, w.options());
  launch_pack_kb32(w.data_ptr(), out.data_ptr(), K, C, bwd ? 1 : 0, stream());
  return out;
}

// K11: deterministic GPU stream compaction — (values[score >= t],
// per-dim COO indices) in row-major order, matching nonzero()'s ordering.
std::vector<torch::Tensor> coo_compact(torch::Tensor values, torch::Tensor score,
                                       double threshold) {
  check_f32(values, "values");
  check_f32(score, "score");
  TORCH_CHECK(values.sizes() == score.sizes(), "values/score shape mismatch");
  int ndim = (int)values.dim();
  TORCH_CHECK(ndim >= 1 && ndim <= 8, "1..8 dims supported");
  int64_t n = values.numel();
  int64_t n_chunks = (n + 63) / 64;
  auto iopts = torch::TensorOptions().dtype(torch::kInt32).device(values.device());
  auto counts = torch::empty({n_chunks}, iopts);
  launch_coo_count(score.data_ptr<float>(), (float)threshold, n, n_chunks,
                   counts.data_ptr<int>(), stream());
  auto csum = counts.cumsum(0, torch::kInt32);
  auto offsets = csum - counts;  // exclusive
  int64_t nnz = csum.numel() ? csum[-1].item<int64_t>() : 0;  // ONE small sync
  auto values_out = torch::empty({nnz}, values.options());
  auto lopts = torch::TensorOptions().dtype(torch::kInt64).device(values.device());
  auto indices_out = torch::empty({ndim, nnz}, lopts);
  auto dims = torch::tensor(values.sizes().vec(), torch::TensorOptions().dtype(torch::kInt64))
                  .to(values.device());
  if (nnz > 0) {
    launch_coo_write(values.data_ptr<float>(), score.data_ptr<float>(), (float)threshold,
                     offsets.to(torch::kInt32).data_ptr<int>(), values_out.data_ptr<float>(),
                     indices_out.data_ptr<int64_t>(), dims.data_ptr<int64_t>(), ndim, n,
                     n_chunks, nnz, stream());
  }
  return {values_out, indices_out};
}

// Fused MOON contrastive loss + input gradient (K8): one workgroup per
// sample computes the cosine-similarity logits against [pos | negs], the
// softmax-CE loss, and dz in a single launch (partners are frozen snapshots).
std::vector<torch::Tensor> moon_contrastive(torch::Tensor z, torch::Tensor pos,
                                            torch::Tensor neg, double tau) {
  check_f32(z, "z");
  check_f32(pos, "pos");
  check_f32(neg, "neg");
  TORCH_CHECK(z.dim() == 2 && pos.sizes() == z.sizes(), "z/pos must be [B, D]");
  TORCH_CHECK(neg.dim() == 3 && neg.size(1) == z.size(0) && neg.size(2) == z.size(1),
              "neg must be [K, B, D]");
  int K = (int)neg.size(0);
  TORCH_CHECK(K + 1 <= 17, "at most 16 negative pairs");
  int B = (int)z.size(0);
  auto loss = torch::empty({B}, z.options());
  auto dz = torch::empty_like(z);
  launch_moon_contrastive(z.data_ptr<float>(), pos.data_ptr<float>(), neg.data_ptr<float>(), K, B,
                          z.size(1), (float)(1.0 / tau), loss.data_ptr<float>(),
                          dz.data_ptr<float>(), stream());
  return {loss, dz};
}

static int in3d_slices(int64_t planes, int64_t L) {
  // enough (plane, slice) blocks to fill 256 CUs; cap the partial buffer
  int S = (int)std::min<int64_t>(64, std::max<int64_t>(1, (1024 + planes - 1) / planes));
  while ((int64_t)S > 1 && (L + S - 1) / S < 4096) --S;  // keep chunks >= 4096 elems
  return std::max(S, 1);
}

// Fused InstanceNorm3d + LeakyReLU forward over NCDHW bf16 planes.
// Returns (y, mean[P], invstd[P]) — stats saved for backward.
std::vector<torch::Tensor> in3d_fwd(torch::Tensor x, torch::Tensor gamma, torch::Tensor beta,
                                    double eps, double slope) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 5 &&
              x.scalar_type() == torch::kBFloat16, "x must be [N,C,D,H,W] bf16 contiguous");
  check_f32(gamma, "gamma");
  check_f32(beta, "beta");
  int64_t N = x.size(0), C = x.size(1);
  int64_t L = x.size(2) * x.size(3) * x.size(4);
  int64_t P = N * C;
  int S = in3d_slices(P, L);
  auto opts = torch::TensorOptions().dtype(torch::kFloat32).device(x.device());
  auto mean = torch::empty({P}, opts);
  auto invstd = torch::empty({P}, opts);
  auto partial = torch::empty({P * S * 2}, opts);
  auto y = torch::empty_like(x);
  launch_in3d_fwd(x.data_ptr(), y.data_ptr(), mean.data_ptr<float>(), invstd.data_ptr<float>(),
                  partial.data_ptr<float>(), gamma.data_ptr<float>(), beta.data_ptr<float>(),
                  (int)P, (int)C, L, S, (float)eps, (float)slope, stream());
  return {y, mean, invstd};
}

// Backward: returns (dx, dgamma, dbeta).
std::vector<torch::Tensor> in3d_bwd(torch::Tensor x, torch::Tensor dy, torch::Tensor mean,
                                    torch::Tensor invstd, torch::Tensor gamma,
                                    torch::Tensor beta, double slope) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && dy.is_contiguous() &&
              x.scalar_type() == torch::kBFloat16 && dy.scalar_type() == torch::kBFloat16);
  int64_t N = x.size(0), C = x.size(1);
  int64_t L = x.size(2) * x.size(3) * x.size(4);
  int64_t P = N * C;
  int S = in3d_slices(P, L);
  auto opts = torch::TensorOptions().dtype(torch::kFloat32).device(x.device());
  auto partial = torch::empty({P * S * 2}, opts);
  auto s1 = torch::empty({P}, opts);
  auto s2 = torch::empty({P}, opts);
  auto dgamma = torch::empty({C}, opts);
  auto dbeta = torch::empty({C}, opts);
  auto dx = torch::empty_like(x);
  launch_in3d_bwd(x.data_ptr(), dy.data_ptr(), dx.data_ptr(), partial.data_ptr<float>(),
                  s1.data_ptr<float>(), s2.data_ptr<float>(), dgamma.data_ptr<float>(),
                  dbeta.data_ptr<float>(), mean.data_ptr<float>(), invstd.data_ptr<float>(),
                  gamma.data_ptr<float>(), beta.data_ptr<float>(), (int)P, (int)C, L, S,
                  (float)slope, stream());
  return {dx, dgamma, dbeta};
}

// Fused multi-bandwidth Gaussian-kernel sums over a pairwise-distance Gram
// (reference losses/mkmmd_loss.py:96-135).
torch::Tensor mkmmd_sums(torch::Tensor d, torch::Tensor gammas, bool skip_diag) {
  check_f32(d, "d");
  check_f32(gammas, "gammas");
  TORCH_CHECK(d.dim() == 2, "d must be [rows, cols]");
  int k = (int)gammas.numel();
  TORCH_CHECK(k >= 1 && k <= 32, "gammas must have 1..32 entries");
  int64_t rows = d.size(0), cols = d.size(1);
  if (skip_diag) TORCH_CHECK(rows == cols, "skip_diag needs a square Gram");
  auto partial = torch::zeros({(int64_t)k * 512}, d.options().dtype(torch::kFloat64));
  auto out = torch::empty({k}, d.options());
  launch_mkmmd_sums(d.data_ptr<float>(), gammas.data_ptr<float>(), partial.data_ptr<double>(),
                    out.data_ptr<float>(), k, rows, cols, skip_diag ? 1 : 0, stream());
  return out;
}

torch::Tensor mkmmd_backward(torch::Tensor d, torch::Tensor gammas, torch::Tensor coef,
                             bool skip_diag) {
  check_f32(d, "d");
  check_f32(gammas, "gammas");
  check_f32(coef, "coef");
  TORCH_CHECK(d.dim() == 2 && gammas.numel() == coef.numel(), "shape mismatch");
  auto dd = torch::empty_like(d);
  launch_mkmmd_backward(d.data_ptr<float>(), gammas.data_ptr<float>(), coef.data_ptr<float>(),
                        dd.data_ptr<float>(), (int)gammas.numel(), d.size(0), d.size(1),
                        skip_diag ? 1 : 0, stream());
  return dd;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("conv3x3_wrw", &conv3x3_wrw, "3x3 wrw (dW) for NHWC bf16");
  m.def("bn_fwd_train", &bn_fwd_train, "NHWC batchnorm training forward", py::arg("x"),
        py::arg("gamma"), py::arg("beta"), py::arg("running_mean"), py::arg("running_var"),
        py::arg("momentum"), py::arg("eps"), py::arg("fuse_relu"), py::arg("res"),
        py::arg("num_batches_tracked") = py::none());
  m.def("bn_bwd", &bn_bwd, "NHWC batchnorm backward");
  m.def("axpby_", &axpby_, "y = a*x + b*y (in-place)");
  m.def("prox_sgd_step_", &prox_sgd_step_, "fused proximal SGD step");
  m.def("scaffold_sgd_step_", &scaffold_sgd_step_, "fused SCAFFOLD-corrected SGD step");
  m.def("scaffold_variate_update_", &scaffold_variate_update_, "SCAFFOLD control variate update");
  m.def("server_opt_step_", &server_opt_step_, "fused FedOpt/Flash server step");
  m.def("reduce_op", &reduce_op, "deterministic reduction (0 sqnorm,1 sqdiff,2 dot,3 sum)");
  m.def("clip_delta", &clip_delta, "flat-clip weight delta");
  m.def("gaussian_noise_", &gaussian_noise_, "x = a*x + sigma*N(0,1), philox");
  m.def("bernoulli_mask", &bernoulli_mask, "bernoulli(sigmoid(scores)) mask (+masked weight)");
  m.def("per_sample_sqnorm_", &per_sample_sqnorm_, "accumulate per-sample grad sq norms");
  m.def("clip_rowsum_noise_", &clip_rowsum_noise_,
        "fused clipped rowsum + DP Gaussian noise (single pass over the grads)");
  m.def("clip_rowsum_", &clip_rowsum_, "clipped per-sample grad sum");
  m.def("confusion_counts_", &confusion_counts_, "streaming TP/FP/FN/TN counts");
  m.def("weighted_sum_rows", &weighted_sum_rows, "out = sum_k w[k]*stack[k]");
  m.def("conv3x3_fwd", &conv3x3_fwd, "direct 3x3 NHWC bf16 conv forward (MFMA)");
  m.def("conv3x3_fwd_kb32", &conv3x3_fwd_kb32,
        "direct 3x3 NHWC bf16 conv forward, KB=32 glds-pipelined variant");
  m.def("pack_kb32", &pack_kb32, "fused weight pack for conv3x3_fwd_kb32");
  m.def("conv3x3_fwd_kzloop", &conv3x3_fwd_kzloop,
        "direct 3x3 conv, input-resident multi-kz variant (KB=32)");
  m.def("coo_compact", &coo_compact,
        "deterministic stream compaction: values + per-dim COO indices above a threshold");
  m.def("moon_contrastive", &moon_contrastive,
        "fused MOON contrastive loss + dz (cosine logits, softmax-CE, label 0)");
  m.def("in3d_fwd", &in3d_fwd, "fused InstanceNorm3d + LeakyReLU forward (NCDHW bf16)");
  m.def("in3d_bwd", &in3d_bwd, "fused InstanceNorm3d + LeakyReLU backward");
  m.def("mkmmd_sums", &mkmmd_sums, "per-bandwidth Gaussian kernel sums over a Gram");
  m.def("mkmmd_backward", &mkmmd_backward, "dL/dGram for mkmmd_sums");
}
