// rocps._hipops — torch bindings for the hand-written CDNA4 kernels.
//
// Every function validates device/dtype/layout and launches on the
// current HIP stream. All per-channel/per-row small math runs in
// dedicated finalize kernels — host-side C-sized torch ops made the
// ResNet step CPU-bound (profiles/r50_b256_torch_profile_hostbound_before.txt).

#include <torch/extension.h>

#include <ATen/hip/impl/HIPGuardImplMasqueradingAsCUDA.h>
#include <ATen/hip/impl/HIPStreamMasqueradingAsCUDA.h>

#include "launchers.h"

namespace {

hipStream_t cur_stream(const at::Tensor& t) {
  return c10::hip::getCurrentHIPStreamMasqueradingAsCUDA(
      t.device().index()).stream();
}

void check_flat(const at::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

rbf16* bf16_ptr(const at::Tensor& t) {
  return reinterpret_cast<rbf16*>(t.data_ptr<at::BFloat16>());
}

}  // namespace

// ---------------------------------------------------------------------------
// fused optimizers (C8)
// ---------------------------------------------------------------------------

void fused_sgd(at::Tensor p, at::Tensor g, at::Tensor master, double lr,
               bool has_master) {
  check_flat(p, "params");
  check_flat(g, "grads");
  TORCH_CHECK(p.numel() == g.numel());
  c10::hip::OptionalHIPGuardMasqueradingAsCUDA guard(p.device());
  long n = p.numel();
  auto st = cur_stream(p);
  if (p.scalar_type() == at::kBFloat16) {
    TORCH_CHECK(g.scalar_type() == at::kBFloat16,
                "sgd: bf16 params need bf16 grads");
    TORCH_CHECK(!has_master || master.scalar_type() == at::kFloat);
    launch_sgd_bf16(bf16_ptr(p), bf16_ptr(g),
                    has_master ? master.data_ptr<float>() : nullptr,
                    (float)lr, n, has_master, st);
  } else {
    TORCH_CHECK(p.scalar_type() == at::kFloat, "sgd: fp32/bf16 only");
    TORCH_CHECK(g.scalar_type() == p.scalar_type(),
                "sgd: grad dtype must match param dtype in fp32 mode");
    TORCH_CHECK(!has_master, "sgd: master weights only with bf16 params");
    launch_sgd_f32(p.data_ptr<float>(), g.data_ptr<float>(),
                   p.data_ptr<float>(), (float)lr, n, false, st);
  }
}

void fused_momentum(at::Tensor p, at::Tensor g, at::Tensor master,
                    at::Tensor mom, double lr, double mu, bool nesterov,
                    bool has_master) {
  check_flat(p, "params");
  check_flat(g, "grads");
  check_flat(mom, "momentum");
  TORCH_CHECK(p.numel() == g.numel() && p.numel() == mom.numel());
  TORCH_CHECK(mom.scalar_type() == at::kFloat);
  c10::hip::OptionalHIPGuardMasqueradingAsCUDA guard(p.device());
  long n = p.numel();
  auto st = cur_stream(p);
  if (p.scalar_type() == at::kBFloat16) {
    TORCH_CHECK(g.scalar_type() == at::kBFloat16,
                "momentum: bf16 params need bf16 grads");
    launch_momentum_bf16(bf16_ptr(p), bf16_ptr(g),
                         has_master ? master.data_ptr<float>() : nullptr,
                         mom.data_ptr<float>(), (float)lr, (float)mu,
                         nesterov, n, has_master, st);
  } else {
    TORCH_CHECK(p.scalar_type() == at::kFloat, "momentum: fp32/bf16 only");
    TORCH_CHECK(g.scalar_type() == p.scalar_type(),
                "momentum: grad dtype must match param dtype in fp32 mode");
    TORCH_CHECK(!has_master,
                "momentum: master weights only with bf16 params");
    launch_momentum_f32(p.data_ptr<float>(), g.data_ptr<float>(),
                        p.data_ptr<float>(), mom.data_ptr<float>(), (float)lr,
                        (float)mu, nesterov, n, false, st);
  }
}

void fused_adam(at::Tensor p, at::Tensor g, at::Tensor master, at::Tensor m,
                at::Tensor v, double alpha, double b1, double b2, double eps,
                bool has_master) {
  check_flat(p, "params");
  check_flat(g, "grads");
  TORCH_CHECK(m.scalar_type() == at::kFloat && v.scalar_type() == at::kFloat);
  TORCH_CHECK(p.numel() == g.numel() && p.numel() == m.numel()
              && p.numel() == v.numel());
  c10::hip::OptionalHIPGuardMasqueradingAsCUDA guard(p.device());
  long n = p.numel();
  auto st = cur_stream(p);
  if (p.scalar_type() == at::kBFloat16) {
    TORCH_CHECK(g.scalar_type() == at::kBFloat16,
                "adam: bf16 params need bf16 grads");
    launch_adam_bf16(bf16_ptr(p), bf16_ptr(g),
                     has_master ? master.data_ptr<float>() : nullptr,
                     m.data_ptr<float>(), v.data_ptr<float>(), (float)alpha,
                     (float)b1, (float)b2, (float)eps, n, has_master, st);
  } else {
    TORCH_CHECK(p.scalar_type() == at::kFloat, "adam: fp32/bf16 only");
    TORCH_CHECK(g.scalar_type() == p.scalar_type(),
                "adam: grad dtype must match param dtype in fp32 mode");
    TORCH_CHECK(!has_master, "adam: master weights only with bf16 params");
    launch_adam_f32(p.data_ptr<float>(), g.data_ptr<float>(),
                    p.data_ptr<float>(), m.data_ptr<float>(),
                    v.data_ptr<float>(), (float)alpha, (float)b1, (float)b2,
                    (float)eps, n, false, st);
  }
}

// ---------------------------------------------------------------------------
// softmax-xent (C11)
// ---------------------------------------------------------------------------

std::vector<at::Tensor> softmax_xent_fwd(at::Tensor logits,
                                         at::Tensor labels) {
  TORCH_CHECK(logits.is_cuda() && logits.dim() == 2);
  TORCH_CHECK(labels.scalar_type() == at::kLong);
  auto lc = logits.contiguous();
  auto lab = labels.contiguous();
  long N = lc.size(0);
  int C = (int)lc.size(1);
  c10::hip::OptionalHIPGuardMasqueradingAsCUDA guard(lc.device());
  auto loss = at::empty({N}, lc.options().dtype(at::kFloat));
  auto dlogits = at::empty_like(lc);
  auto st = cur_stream(lc);
  if (lc.scalar_type() == at::kBFloat16) {
    launch_softmax_xent_bf16(bf16_ptr(lc), lab.data_ptr<long>(),
                             loss.data_ptr<float>(), bf16_ptr(dlogits), N, C,
                             st);
  } else {
    TORCH_CHECK(lc.scalar_type() == at::kFloat);
    launch_softmax_xent_f32(lc.data_ptr<float>(), lab.data_ptr<long>(),
                            loss.data_ptr<float>(), dlogits.data_ptr<float>(),
                            N, C, st);
  }
  return {loss, dlogits};
}

// ---------------------------------------------------------------------------
// fused BN(+ReLU) (C9)
// ---------------------------------------------------------------------------

namespace {

void bn_shapes(const at::Tensor& x, long* M, int* C) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4, "bn: NCHW 4-D GPU tensor");
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "bn: channels_last layout required");
  *C = (int)x.size(1);
  *M = x.numel() / *C;
}

void bn_stats(const at::Tensor& x, long M, int C, float* ws) {
  auto st = cur_stream(x);
  if (x.scalar_type() == at::kBFloat16)
    launch_bn_stats_bf16(bf16_ptr(x), ws, M, C, st);
  else
    launch_bn_stats_f32(x.data_ptr<float>(), ws, M, C, st);
}

at::Tensor bn_apply(const at::Tensor& x, const at::Tensor* res,
                    const float* scale, const float* shift, long M, int C,
                    bool relu) {
  auto y = at::empty_like(x);
  auto st = cur_stream(x);
  if (x.scalar_type() == at::kBFloat16)
    launch_bn_apply_bf16(bf16_ptr(x), res ? bf16_ptr(*res) : nullptr,
                         bf16_ptr(y), scale, shift, M, C, relu, st);
  else
    launch_bn_apply_f32(x.data_ptr<float>(),
                        res ? res->data_ptr<float>() : nullptr,
                        y.data_ptr<float>(), scale, shift, M, C, relu, st);
  return y;
}

void check_bn_params(const at::Tensor& w, const at::Tensor& b,
                     const at::Tensor& rm, const at::Tensor& rv) {
  TORCH_CHECK(w.scalar_type() == b.scalar_type()
                  && w.scalar_type() == rm.scalar_type()
                  && w.scalar_type() == rv.scalar_type(),
              "bn: weight/bias/running stats must share a dtype");
  TORCH_CHECK(w.is_contiguous() && b.is_contiguous() && rm.is_contiguous()
              && rv.is_contiguous());
}

}  // namespace

std::vector<at::Tensor> bn_relu_fwd_train(at::Tensor x, at::Tensor w,
                                          at::Tensor b, at::Tensor rm,
                                          at::Tensor rv, double momentum,
                                          double eps, bool relu,
                                          std::optional<at::Tensor> res) {
  long M;
  int C;
  bn_shapes(x, &M, &C);
  check_bn_params(w, b, rm, rv);
  if (res) {
    TORCH_CHECK(res->is_contiguous(at::MemoryFormat::ChannelsLast)
                && res->sizes() == x.sizes()
                && res->scalar_type() == x.scalar_type());
  }
  c10::hip::OptionalHIPGuardMasqueradingAsCUDA guard(x.device());
  auto opts = x.options().dtype(at::kFloat);
  int rb = bn_reduce_rb(M, C);
  auto ws = (C % 8 == 0) ? at::empty({(long)rb * 2 * C}, opts)
                         : at::zeros({(long)rb * 2 * C}, opts);
  auto out = at::empty({4 * C}, opts);      // mean | invstd | scale | shift
  bn_stats(x, M, C, ws.data_ptr<float>());
  auto st = cur_stream(x);
  if (w.scalar_type() == at::kBFloat16)
    launch_bn_finalize_train_bf16(ws.data_ptr<float>(), rb, bf16_ptr(w),
                                  bf16_ptr(b), bf16_ptr(rm), bf16_ptr(rv),
                                  (float)momentum, (float)eps, M,
                                  out.data_ptr<float>(), C, st);
  else
    launch_bn_finalize_train_f32(ws.data_ptr<float>(), rb,
                                 w.data_ptr<float>(),
                                 b.data_ptr<float>(), rm.data_ptr<float>(),
                                 rv.data_ptr<float>(), (float)momentum,
                                 (float)eps, M, out.data_ptr<float>(), C, st);
  auto y = bn_apply(x, res ? &*res : nullptr, out.data_ptr<float>() + 2 * C,
                    out.data_ptr<float>() + 3 * C, M, C, relu);
  // 4th return: the whole [4C] fold tensor — backward can recompute the
  // ReLU mask from x*scale+shift and skip reading y entirely.
  return {y, out.narrow(0, 0, C), out.narrow(0, C, C), out};
}

at::Tensor bn_relu_fwd_eval(at::Tensor x, at::Tensor w, at::Tensor b,
                            at::Tensor rm, at::Tensor rv, double eps,
                            bool relu, std::optional<at::Tensor> res) {
  long M;
  int C;
  bn_shapes(x, &M, &C);
  check_bn_params(w, b, rm, rv);
  c10::hip::OptionalHIPGuardMasqueradingAsCUDA guard(x.device());
  auto out = at::empty({2 * C}, x.options().dtype(at::kFloat));
  auto st = cur_stream(x);
  if (w.scalar_type() == at::kBFloat16)
    launch_bn_finalize_eval_bf16(bf16_ptr(w), bf16_ptr(b), bf16_ptr(rm),
                                 bf16_ptr(rv), (float)eps,
                                 out.data_ptr<float>(), C, st);
  else
    launch_bn_finalize_eval_f32(w.data_ptr<float>(), b.data_ptr<float>(),
                                rm.data_ptr<float>(), rv.data_ptr<float>(),
                                (float)eps, out.data_ptr<float>(), C, st);
  return bn_apply(x, res ? &*res : nullptr, out.data_ptr<float>(),
                  out.data_ptr<float>() + C, M, C, relu);
}

std::vector<at::Tensor> bn_relu_bwd(at::Tensor x, at::Tensor dy, at::Tensor y,
                                    at::Tensor w, at::Tensor mean,
                                    at::Tensor invstd, bool relu,
                                    bool has_res,
                                    std::optional<at::Tensor> fold4c) {
  // fold4c = the forward's [4C] {mean|invstd|scale|shift} tensor; with it
  // (and no fused residual) the ReLU mask recomputes from x — y unread.
  bool use_y = !(relu && !has_res && fold4c.has_value());
  const float* sc = nullptr;
  const float* sh = nullptr;
  if (!use_y) {
    int Cf = (int)x.size(1);
    sc = fold4c->data_ptr<float>() + 2 * Cf;
    sh = fold4c->data_ptr<float>() + 3 * Cf;
  }
  long M;
  int C;
  bn_shapes(x, &M, &C);
  TORCH_CHECK(dy.is_contiguous(at::MemoryFormat::ChannelsLast));
  TORCH_CHECK(mean.is_contiguous() && invstd.is_contiguous());
  c10::hip::OptionalHIPGuardMasqueradingAsCUDA guard(x.device());
  auto opts = x.options().dtype(at::kFloat);
  int rb = bn_reduce_rb(M, C);
  auto s12 = (C % 8 == 0) ? at::empty({(long)rb * 2 * C}, opts)
                          : at::zeros({(long)rb * 2 * C}, opts);
  auto abd = at::empty({3 * C}, opts);
  auto dgb = at::empty({2 * C}, x.options().dtype(w.scalar_type()));
  auto st = cur_stream(x);
  if (x.scalar_type() == at::kBFloat16)
    launch_bn_bwd_reduce_bf16(bf16_ptr(x), bf16_ptr(dy), bf16_ptr(y),
                              mean.data_ptr<float>(),
                              invstd.data_ptr<float>(), sc, sh,
                              s12.data_ptr<float>(), M, C, relu, use_y, st);
  else
    launch_bn_bwd_reduce_f32(x.data_ptr<float>(), dy.data_ptr<float>(),
                             y.data_ptr<float>(), mean.data_ptr<float>(),
                             invstd.data_ptr<float>(), sc, sh,
                             s12.data_ptr<float>(), M, C, relu, use_y, st);
  if (w.scalar_type() == at::kBFloat16)
    launch_bn_bwd_finalize_bf16(s12.data_ptr<float>(), rb, bf16_ptr(w),
                                mean.data_ptr<float>(),
                                invstd.data_ptr<float>(), M,
                                abd.data_ptr<float>(), bf16_ptr(dgb),
                                bf16_ptr(dgb) + C, C, st);
  else
    launch_bn_bwd_finalize_f32(s12.data_ptr<float>(), rb,
                               w.data_ptr<float>(), mean.data_ptr<float>(),
                               invstd.data_ptr<float>(), M,
                               abd.data_ptr<float>(), dgb.data_ptr<float>(),
                               dgb.data_ptr<float>() + C, C, st);
  auto dx = at::empty_like(x);
  at::Tensor dres;
  if (has_res) dres = at::empty_like(x);
  if (x.scalar_type() == at::kBFloat16)
    launch_bn_bwd_apply_bf16(bf16_ptr(x), bf16_ptr(dy), bf16_ptr(y),
                             bf16_ptr(dx),
                             has_res ? bf16_ptr(dres) : nullptr,
                             abd.data_ptr<float>(), sc, sh, M, C, relu,
                             use_y, st);
  else
    launch_bn_bwd_apply_f32(x.data_ptr<float>(), dy.data_ptr<float>(),
                            y.data_ptr<float>(), dx.data_ptr<float>(),
                            has_res ? dres.data_ptr<float>() : nullptr,
                            abd.data_ptr<float>(), sc, sh, M, C, relu,
                            use_y, st);
  if (has_res)
    return {dx, dgb.narrow(0, 0, C), dgb.narrow(0, C, C), dres};
  return {dx, dgb.narrow(0, 0, C), dgb.narrow(0, C, C)};
}

// ---------------------------------------------------------------------------
// fused LayerNorm (bert_ops LN slice)
// ---------------------------------------------------------------------------

std::vector<at::Tensor> ln_fwd(at::Tensor x, at::Tensor gamma,
                               at::Tensor beta, double eps) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.is_contiguous());
  TORCH_CHECK(gamma.scalar_type() == at::kFloat
              && beta.scalar_type() == at::kFloat);
  long N = x.size(0);
  int D = (int)x.size(1);
  c10::hip::OptionalHIPGuardMasqueradingAsCUDA guard(x.device());
  auto y = at::empty_like(x);
  auto mean = at::empty({N}, x.options().dtype(at::kFloat));
  auto invstd = at::empty_like(mean);
  auto st = cur_stream(x);
  if (x.scalar_type() == at::kBFloat16)
    launch_ln_fwd_bf16(bf16_ptr(x), gamma.data_ptr<float>(),
                       beta.data_ptr<float>(), bf16_ptr(y),
                       mean.data_ptr<float>(), invstd.data_ptr<float>(), N, D,
                       (float)eps, st);
  else
    launch_ln_fwd_f32(x.data_ptr<float>(), gamma.data_ptr<float>(),
                      beta.data_ptr<float>(), y.data_ptr<float>(),
                      mean.data_ptr<float>(), invstd.data_ptr<float>(), N, D,
                      (float)eps, st);
  return {y, mean, invstd};
}

std::vector<at::Tensor> ln_bwd(at::Tensor x, at::Tensor dy, at::Tensor gamma,
                               at::Tensor mean, at::Tensor invstd) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && x.is_contiguous());
  auto dyc = dy.contiguous();
  long N = x.size(0);
  int D = (int)x.size(1);
  c10::hip::OptionalHIPGuardMasqueradingAsCUDA guard(x.device());
  auto dx = at::empty_like(x);
  auto dgamma = at::zeros({D}, x.options().dtype(at::kFloat));
  auto dbeta = at::zeros({D}, x.options().dtype(at::kFloat));
  auto st = cur_stream(x);
  if (x.scalar_type() == at::kBFloat16) {
    launch_ln_bwd_dx_bf16(bf16_ptr(x), bf16_ptr(dyc),
                          gamma.data_ptr<float>(), mean.data_ptr<float>(),
                          invstd.data_ptr<float>(), bf16_ptr(dx), N, D, st);
    launch_ln_bwd_dgb_bf16(bf16_ptr(x), bf16_ptr(dyc),
                           mean.data_ptr<float>(), invstd.data_ptr<float>(),
                           dgamma.data_ptr<float>(), dbeta.data_ptr<float>(),
                           N, D, st);
  } else {
    launch_ln_bwd_dx_f32(x.data_ptr<float>(), dyc.data_ptr<float>(),
                         gamma.data_ptr<float>(), mean.data_ptr<float>(),
                         invstd.data_ptr<float>(), dx.data_ptr<float>(), N, D,
                         st);
    launch_ln_bwd_dgb_f32(x.data_ptr<float>(), dyc.data_ptr<float>(),
                          mean.data_ptr<float>(), invstd.data_ptr<float>(),
                          dgamma.data_ptr<float>(), dbeta.data_ptr<float>(),
                          N, D, st);
  }
  return {dx, dgamma, dbeta};
}

// ---------------------------------------------------------------------------
// MFMA bf16 GEMM (C10): D = A @ W^T (+bias) (+gelu)
// ---------------------------------------------------------------------------

// variant: -1 auto (256-tile 8-phase pipeline where the shape allows,
// else the 128-tile kernel), 1 force 128, 2 force 256 (microbench A/B)
static void dispatch_gemm(const at::Tensor& A, const at::Tensor& W,
                          const float* bptr, at::Tensor& D, rbf16* pre,
                          long M, long N, long K, long epilogue,
                          long variant) {
  bool use256 = gemm256_supported((int)M, (int)N, (int)K);
  if (variant == 3 || variant == 4) {  // A/B: 3 = look-ahead schedule,
    // 4 = in-phase with the st_16x32 swizzle
    TORCH_CHECK(use256 && epilogue == 0,
                "variant 3/4: 256-tile shape, epilogue 0 only");
    launch_gemm_bf16_256(bf16_ptr(A), bf16_ptr(W), bptr, bf16_ptr(D), pre,
                         (int)M, (int)N, (int)K, 0, cur_stream(A),
                         variant == 3 ? 1 : 2);
    return;
  }
  // Measured heuristic (profiles/gemm_micro_r2c.txt): the 256-tile
  // pipeline runs at 1 workgroup/CU, so it needs a grid that fills the
  // 256 CUs and K-depth to amortize its prologue; on small grids or
  // shallow K — and on huge grids, where 2-3 blocks/CU of the smaller
  // tile win back the latency hiding — the 128-tile kernel is faster.
  if (use256 && variant == -1) {
    long blocks = (M / 256) * (N / 256);
    use256 = blocks >= 192 && blocks < 1024 && K >= 1024;
  }
  if (variant == 1) use256 = false;
  if (variant == 2)
    TORCH_CHECK(use256, "shape not supported by the 256-tile kernel");
  if (use256)
    launch_gemm_bf16_256(bf16_ptr(A), bf16_ptr(W), bptr, bf16_ptr(D), pre,
                         (int)M, (int)N, (int)K, (int)epilogue,
                         cur_stream(A));
  else
    launch_gemm_bf16(bf16_ptr(A), bf16_ptr(W), bptr, bf16_ptr(D), pre,
                     (int)M, (int)N, (int)K, (int)epilogue, cur_stream(A));
}

at::Tensor gemm_bf16(at::Tensor A, at::Tensor W,
                     std::optional<at::Tensor> bias, long epilogue,
                     long variant) {
  TORCH_CHECK(A.is_cuda() && A.dim() == 2 && A.is_contiguous());
  TORCH_CHECK(W.is_cuda() && W.dim() == 2 && W.is_contiguous());
  TORCH_CHECK(A.scalar_type() == at::kBFloat16
              && W.scalar_type() == at::kBFloat16);
  long M = A.size(0), K = A.size(1), N = W.size(0);
  TORCH_CHECK(W.size(1) == K, "K mismatch");
  TORCH_CHECK(M % 128 == 0 && N % 128 == 0 && K % 64 == 0,
              "gemm_bf16 needs M%128==0, N%128==0, K%64==0 (got ", M, "x",
              N, "x", K, ")");
  const float* bptr = nullptr;
  at::Tensor b32;
  if (epilogue >= 1) {
    TORCH_CHECK(bias.has_value(), "bias epilogue needs a bias tensor");
    b32 = bias->to(at::kFloat).contiguous();
    TORCH_CHECK(b32.numel() == N);
    bptr = b32.data_ptr<float>();
  }
  c10::hip::OptionalHIPGuardMasqueradingAsCUDA guard(A.device());
  auto D = at::empty({M, N}, A.options());
  dispatch_gemm(A, W, bptr, D, nullptr, M, N, K, epilogue, variant);
  return D;
}

at::Tensor conv1x1_wgrad(at::Tensor dY, at::Tensor X) {
  TORCH_CHECK(dY.is_cuda() && dY.dim() == 2 && dY.is_contiguous());
  TORCH_CHECK(X.is_cuda() && X.dim() == 2 && X.is_contiguous());
  TORCH_CHECK(dY.scalar_type() == at::kBFloat16
              && X.scalar_type() == at::kBFloat16);
  long M = dY.size(0), N = dY.size(1), K = X.size(1);
  TORCH_CHECK(X.size(0) == M, "M mismatch");
  TORCH_CHECK(wgrad_supported(M, (int)N, (int)K),
              "conv1x1_wgrad needs N%64==0 and K%64==0");
  c10::hip::OptionalHIPGuardMasqueradingAsCUDA guard(dY.device());
  auto dW = at::zeros({N, K}, dY.options().dtype(at::kFloat));
  launch_conv1x1_wgrad(bf16_ptr(dY), bf16_ptr(X), dW.data_ptr<float>(), M,
                       (int)N, (int)K, cur_stream(dY));
  return dW;
}

at::Tensor gelu_bwd(at::Tensor pre, at::Tensor dy) {
  TORCH_CHECK(pre.is_cuda() && pre.scalar_type() == at::kBFloat16);
  TORCH_CHECK(dy.sizes() == pre.sizes()
              && dy.scalar_type() == at::kBFloat16);
  auto p = pre.contiguous(), d = dy.contiguous();
  c10::hip::OptionalHIPGuardMasqueradingAsCUDA guard(p.device());
  auto g = at::empty_like(p);
  launch_gelu_bwd(bf16_ptr(p), bf16_ptr(d), bf16_ptr(g), p.numel(),
                  cur_stream(p));
  return g;
}

// epilogue-3 variant: returns (gelu(A@W^T + b), pre-activation) so the
// training backward reuses pre instead of recomputing it (ROADMAP §Perf 5)
std::tuple<at::Tensor, at::Tensor> gemm_bf16_gelu_pre(at::Tensor A,
                                                      at::Tensor W,
                                                      at::Tensor bias) {
  TORCH_CHECK(A.is_cuda() && A.dim() == 2 && A.is_contiguous());
  TORCH_CHECK(W.is_cuda() && W.dim() == 2 && W.is_contiguous());
  TORCH_CHECK(A.scalar_type() == at::kBFloat16
              && W.scalar_type() == at::kBFloat16);
  long M = A.size(0), K = A.size(1), N = W.size(0);
  TORCH_CHECK(W.size(1) == K, "K mismatch");
  TORCH_CHECK(M % 128 == 0 && N % 128 == 0 && K % 64 == 0,
              "gemm_bf16 needs M%128==0, N%128==0, K%64==0");
  auto b32 = bias.to(at::kFloat).contiguous();
  TORCH_CHECK(b32.numel() == N);
  c10::hip::OptionalHIPGuardMasqueradingAsCUDA guard(A.device());
  auto D = at::empty({M, N}, A.options());
  auto P = at::empty({M, N}, A.options());
  dispatch_gemm(A, W, b32.data_ptr<float>(), D, bf16_ptr(P), M, N, K, 3, -1);
  return {D, P};
}

// ---------------------------------------------------------------------------
// maxpool 3x3/s2/p1 (C9 pooling slice)
// ---------------------------------------------------------------------------

std::vector<at::Tensor> maxpool332_fwd(at::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4
              && x.is_contiguous(at::MemoryFormat::ChannelsLast));
  int N = (int)x.size(0), C = (int)x.size(1), H = (int)x.size(2),
      W = (int)x.size(3);
  TORCH_CHECK(C % 8 == 0, "maxpool332: C % 8 == 0");
  int OH = (H + 1) / 2, OW = (W + 1) / 2;
  c10::hip::OptionalHIPGuardMasqueradingAsCUDA guard(x.device());
  auto y = at::empty({N, C, OH, OW},
                     x.options().memory_format(at::MemoryFormat::ChannelsLast));
  auto idx = at::empty({N, OH, OW, C}, x.options().dtype(at::kByte));
  auto st = cur_stream(x);
  if (x.scalar_type() == at::kBFloat16)
    launch_maxpool332_fwd_bf16(bf16_ptr(x), bf16_ptr(y),
                               idx.data_ptr<unsigned char>(), N, C, H, W, OH,
                               OW, st);
  else
    launch_maxpool332_fwd_f32(x.data_ptr<float>(), y.data_ptr<float>(),
                              idx.data_ptr<unsigned char>(), N, C, H, W, OH,
                              OW, st);
  return {y, idx};
}

at::Tensor maxpool332_bwd(at::Tensor dy, at::Tensor idx, long H, long W) {
  TORCH_CHECK(dy.is_cuda() && dy.dim() == 4);
  auto dyc = dy.contiguous(at::MemoryFormat::ChannelsLast);
  int N = (int)dyc.size(0), C = (int)dyc.size(1), OH = (int)dyc.size(2),
      OW = (int)dyc.size(3);
  c10::hip::OptionalHIPGuardMasqueradingAsCUDA guard(dy.device());
  auto dx = at::empty({N, C, H, W},
                      dy.options().memory_format(at::MemoryFormat::ChannelsLast));
  auto st = cur_stream(dyc);
  if (dyc.scalar_type() == at::kBFloat16)
    launch_maxpool332_bwd_bf16(bf16_ptr(dyc), idx.data_ptr<unsigned char>(),
                               bf16_ptr(dx), N, C, (int)H, (int)W, OH, OW,
                               st);
  else
    launch_maxpool332_bwd_f32(dyc.data_ptr<float>(),
                              idx.data_ptr<unsigned char>(),
                              dx.data_ptr<float>(), N, C, (int)H, (int)W, OH,
                              OW, st);
  return dx;
}

// ---------------------------------------------------------------------------

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "rocps hand-written CDNA4 (gfx950) kernels";
  m.def("fused_sgd", &fused_sgd);
  m.def("fused_momentum", &fused_momentum);
  m.def("fused_adam", &fused_adam);
  m.def("softmax_xent_fwd", &softmax_xent_fwd);
  m.def("bn_relu_fwd_train", &bn_relu_fwd_train);
  m.def("bn_relu_fwd_eval", &bn_relu_fwd_eval);
  m.def("bn_relu_bwd", &bn_relu_bwd);
  m.def("ln_fwd", &ln_fwd);
  m.def("ln_bwd", &ln_bwd);
  m.def("gemm_bf16", &gemm_bf16, py::arg("A"), py::arg("W"),
        py::arg("bias"), py::arg("epilogue"),
        py::arg("variant") = -1);
  m.def("gemm_bf16_gelu_pre", &gemm_bf16_gelu_pre);
  m.def("gelu_bwd", &gelu_bwd);
  m.def("conv1x1_wgrad", &conv1x1_wgrad);
  m.def("maxpool332_fwd", &maxpool332_fwd);
  m.def("maxpool332_bwd", &maxpool332_bwd);
}
