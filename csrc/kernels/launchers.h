// Declarations of the HIP kernel launchers (defined in *.hip TUs).
#pragma once

// Host-compilable: only the HIP C API (g++ cannot parse the device
// headers), bf16 pointers cross the ABI as raw unsigned short*.
#include <hip/hip_runtime_api.h>

using rbf16 = unsigned short;

// fused_optimizers.hip
void launch_sgd_f32(float* p, const float* g, float* m, float lr, long n,
                    bool has_master, hipStream_t stream);
void launch_sgd_bf16(rbf16* p, const rbf16* g, float* m, float lr, long n,
                     bool has_master, hipStream_t stream);
void launch_momentum_f32(float* p, const float* g, float* m, float* v,
                         float lr, float mu, bool nesterov, long n,
                         bool has_master, hipStream_t stream);
void launch_momentum_bf16(rbf16* p, const rbf16* g, float* m, float* v,
                          float lr, float mu, bool nesterov, long n,
                          bool has_master, hipStream_t stream);
void launch_adam_f32(float* p, const float* g, float* mst, float* m, float* v,
                     float alpha, float b1, float b2, float eps, long n,
                     bool has_master, hipStream_t stream);
void launch_adam_bf16(rbf16* p, const rbf16* g, float* mst, float* m,
                      float* v, float alpha, float b1, float b2, float eps,
                      long n, bool has_master, hipStream_t stream);

// softmax_xent.hip
void launch_softmax_xent_f32(const float* logits, const long* labels,
                             float* loss, float* dlogits, long N, int C,
                             hipStream_t stream);
void launch_softmax_xent_bf16(const rbf16* logits, const long* labels,
                              float* loss, rbf16* dlogits, long N, int C,
                              hipStream_t stream);

// bn_relu.hip
void launch_bn_stats_f32(const float* x, float* ws, long M, int C,
                         hipStream_t st);
void launch_bn_stats_bf16(const rbf16* x, float* ws, long M, int C,
                          hipStream_t st);
void launch_bn_apply_f32(const float* x, const float* res, float* y,
                         const float* sc, const float* sh, long M, int C,
                         bool relu, hipStream_t st);
void launch_bn_apply_bf16(const rbf16* x, const rbf16* res, rbf16* y,
                          const float* sc, const float* sh, long M, int C,
                          bool relu, hipStream_t st);
void launch_bn_bwd_reduce_f32(const float* x, const float* dy, const float* y,
                              const float* mean, const float* invstd,
                              const float* scale, const float* shift,
                              float* s12, long M, int C, bool relu,
                              bool use_y, hipStream_t st);
void launch_bn_bwd_reduce_bf16(const rbf16* x, const rbf16* dy,
                               const rbf16* y, const float* mean,
                               const float* invstd, const float* scale,
                               const float* shift, float* s12, long M, int C,
                               bool relu, bool use_y, hipStream_t st);
void launch_bn_bwd_apply_f32(const float* x, const float* dy, const float* y,
                             float* dx, float* dres, const float* abd,
                             const float* scale, const float* shift, long M,
                             int C, bool relu, bool use_y, hipStream_t st);
void launch_bn_bwd_apply_bf16(const rbf16* x, const rbf16* dy, const rbf16* y,
                              rbf16* dx, rbf16* dres, const float* abd,
                              const float* scale, const float* shift, long M,
                              int C, bool relu, bool use_y, hipStream_t st);

// layernorm.hip
void launch_ln_fwd_f32(const float* x, const float* gamma, const float* beta,
                       float* y, float* mean, float* invstd, long N, int D,
                       float eps, hipStream_t st);
void launch_ln_fwd_bf16(const rbf16* x, const float* gamma, const float* beta,
                        rbf16* y, float* mean, float* invstd, long N, int D,
                        float eps, hipStream_t st);
void launch_ln_bwd_dx_f32(const float* x, const float* dy, const float* gamma,
                          const float* mean, const float* invstd, float* dx,
                          long N, int D, hipStream_t st);
void launch_ln_bwd_dx_bf16(const rbf16* x, const rbf16* dy,
                           const float* gamma, const float* mean,
                           const float* invstd, rbf16* dx, long N, int D,
                           hipStream_t st);
void launch_ln_bwd_dgb_f32(const float* x, const float* dy, const float* mean,
                           const float* invstd, float* dgamma, float* dbeta,
                           long N, int D, hipStream_t st);
void launch_ln_bwd_dgb_bf16(const rbf16* x, const rbf16* dy,
                            const float* mean, const float* invstd,
                            float* dgamma, float* dbeta, long N, int D,
                            hipStream_t st);

// bn_relu.hip finalize kernels
int bn_reduce_rb(long M, int C);  // partial rows the reduce kernels emit
void launch_bn_finalize_train_f32(const float* part, int rb, const float* w,
                                  const float* b, float* rm, float* rv,
                                  float momentum, float eps, long M,
                                  float* out, int C, hipStream_t st);
void launch_bn_finalize_train_bf16(const float* part, int rb, const rbf16* w,
                                   const rbf16* b, rbf16* rm, rbf16* rv,
                                   float momentum, float eps, long M,
                                   float* out, int C, hipStream_t st);
void launch_bn_finalize_eval_f32(const float* w, const float* b,
                                 const float* rm, const float* rv, float eps,
                                 float* out, int C, hipStream_t st);
void launch_bn_finalize_eval_bf16(const rbf16* w, const rbf16* b,
                                  const rbf16* rm, const rbf16* rv, float eps,
                                  float* out, int C, hipStream_t st);
void launch_bn_bwd_finalize_f32(const float* part, int rb, const float* w,
                                const float* mean, const float* invstd,
                                long M, float* abd, float* dgamma,
                                float* dbeta, int C, hipStream_t st);
void launch_bn_bwd_finalize_bf16(const float* part, int rb, const rbf16* w,
                                 const float* mean, const float* invstd,
                                 long M, float* abd, rbf16* dgamma,
                                 rbf16* dbeta, int C, hipStream_t st);

// wgrad.hip (conv1x1 weight gradient, tall-skinny reduction GEMM)
bool wgrad_supported(long M, int N, int K);
void launch_conv1x1_wgrad(const rbf16* dY, const rbf16* X, float* dW,
                          long M, int N, int K, hipStream_t st);

// gelu.hip
void launch_gelu_bwd(const rbf16* pre, const rbf16* dy, rbf16* g, long n,
                     hipStream_t st);

// gemm_bf16_256.hip (the 8-phase 256x256 pipeline)
bool gemm256_supported(int M, int N, int K);
void launch_gemm_bf16_256(const rbf16* A, const rbf16* W, const float* bias,
                          rbf16* D, rbf16* PRE, int M, int N, int K,
                          int epilogue, hipStream_t st, int sched = 0);

// gemm_bf16.hip
void launch_gemm_bf16(const rbf16* A, const rbf16* W, const float* bias,
                      rbf16* D, rbf16* PRE, int M, int N, int K,
                      int epilogue, hipStream_t st);

// maxpool.hip (3x3 stride-2 pad-1, NHWC)
void launch_maxpool332_fwd_f32(const float* x, float* y, unsigned char* idx,
                               int N, int C, int H, int W, int OH, int OW,
                               hipStream_t st);
void launch_maxpool332_fwd_bf16(const rbf16* x, rbf16* y, unsigned char* idx,
                                int N, int C, int H, int W, int OH, int OW,
                                hipStream_t st);
void launch_maxpool332_bwd_f32(const float* dy, const unsigned char* idx,
                               float* dx, int N, int C, int H, int W, int OH,
                               int OW, hipStream_t st);
void launch_maxpool332_bwd_bf16(const rbf16* dy, const unsigned char* idx,
                                rbf16* dx, int N, int C, int H, int W,
                                int OH, int OW, hipStream_t st);
