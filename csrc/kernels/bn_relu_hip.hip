#include "hip/hip_runtime.h"
// Fused BatchNorm(+ReLU) forward/backward on NHWC — C9's BN slice.
//
// Reference behavior: TF FusedBatchNormV3[Grad] + Relu[Grad]
// (tensorflow/core/kernels/fused_batch_norm_op.cc, relu_op.cc) on every
// conv output of ResNet/W-ResNet. Recast for CDNA4 NHWC:
//
//   fwd train: stats kernel (per-channel sum/sumsq, fp32, per-thread
//     register accumulation over a row range + 2 device atomicAdds per
//     thread) → per-channel scale/shift folded on the small C-sized
//     tensors → apply kernel y = relu(x*scale[c]+shift[c]) vectorized
//     8 channels/lane (bf16 ushort8 = 16 B, Guideline 13).
//   bwd: reduce kernel (sum_dy, sum_dy*xhat with the ReLU mask folded)
//     → per-channel constants → apply kernel dx = A[c]*dy_m + B[c]*x +
//     D[c]. dgamma/dbeta fall out of the reduction.
//
// Layout contract: x is channels_last → memory is (M=N*H*W) rows × C
// contiguous channels. Lanes map to adjacent channels (coalesced); the
// per-channel constant arrays are C-sized and L2-resident.

#include "common.h"
#include "launchers.h"

namespace {

// ---- per-channel reduction over rows (stats + bwd reduce) ----------------
//
// A 256-thread block is a (CT channels × RT=256/CT rows) tile so small-C
// layers (the C=64 stem has M=1.6M rows) still put ~256k threads on the
// chip. Per-thread register accumulation over a grid-strided row range,
// LDS tree-reduce across the RT row-threads of each channel, then one
// atomicAdd per channel per block.

template <typename T>
__global__ void bn_stats_kernel(const T* __restrict__ x,
                                float* __restrict__ gsum,
                                float* __restrict__ gsumsq, long M, int C,
                                int CT) {
  __shared__ float lsum[256];
  __shared__ float lss[256];
  int RT = blockDim.x / CT;
  int cl = threadIdx.x % CT;
  int rl = threadIdx.x / CT;
  int c = blockIdx.y * CT + cl;
  float s = 0.f, ss = 0.f;
  if (c < C) {
    long rstride = (long)gridDim.x * RT;
    for (long r = (long)blockIdx.x * RT + rl; r < M; r += rstride) {
      float v = to_f32<T>(x[r * C + c]);
      s += v;
      ss = fmaf(v, v, ss);
    }
  }
  lsum[threadIdx.x] = s;
  lss[threadIdx.x] = ss;
  __syncthreads();
  for (int step = (RT >> 1) * CT; step >= CT; step >>= 1) {
    if (threadIdx.x < step) {
      lsum[threadIdx.x] += lsum[threadIdx.x + step];
      lss[threadIdx.x] += lss[threadIdx.x + step];
    }
    __syncthreads();
  }
  if (rl == 0 && c < C) {
    atomicAdd(&gsum[c], lsum[cl]);
    atomicAdd(&gsumsq[c], lss[cl]);
  }
}

template <typename T, bool RELU>
__global__ void bn_bwd_reduce_kernel(const T* __restrict__ x,
                                     const T* __restrict__ dy,
                                     const T* __restrict__ y,
                                     const float* __restrict__ mean,
                                     const float* __restrict__ invstd,
                                     float* __restrict__ gsum_dy,
                                     float* __restrict__ gsum_dyxh,
                                     long M, int C, int CT) {
  __shared__ float l1[256];
  __shared__ float l2[256];
  int RT = blockDim.x / CT;
  int cl = threadIdx.x % CT;
  int rl = threadIdx.x / CT;
  int c = blockIdx.y * CT + cl;
  float s1 = 0.f, s2 = 0.f;
  if (c < C) {
    float mu = mean[c], is = invstd[c];
    long rstride = (long)gridDim.x * RT;
    for (long r = (long)blockIdx.x * RT + rl; r < M; r += rstride) {
      long i = r * C + c;
      float d = to_f32<T>(dy[i]);
      if (RELU) d = to_f32<T>(y[i]) > 0.f ? d : 0.f;
      float xh = (to_f32<T>(x[i]) - mu) * is;
      s1 += d;
      s2 = fmaf(d, xh, s2);
    }
  }
  l1[threadIdx.x] = s1;
  l2[threadIdx.x] = s2;
  __syncthreads();
  for (int step = (RT >> 1) * CT; step >= CT; step >>= 1) {
    if (threadIdx.x < step) {
      l1[threadIdx.x] += l1[threadIdx.x + step];
      l2[threadIdx.x] += l2[threadIdx.x + step];
    }
    __syncthreads();
  }
  if (rl == 0 && c < C) {
    atomicAdd(&gsum_dy[c], l1[cl]);
    atomicAdd(&gsum_dyxh[c], l2[cl]);
  }
}

// ---- elementwise applies --------------------------------------------------
struct alignas(16) u16x8 { unsigned short v[8]; };

template <typename T, bool RELU, bool VEC8>
__global__ void bn_apply_kernel(const T* __restrict__ x, T* __restrict__ y,
                                const float* __restrict__ scale,
                                const float* __restrict__ shift,
                                long total, int C) {
  long tid = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  if (VEC8) {
    long nvec = total / 8;
    for (long i = tid; i < nvec; i += stride) {
      long base = i * 8;
      int c0 = (int)(base % C);
      float out[8];
#pragma unroll
      for (int k = 0; k < 8; k++) {
        float v = to_f32<T>(x[base + k]);
        int c = c0 + k;  // C % 8 == 0 → the 8 lanes stay inside one row
        v = fmaf(v, scale[c], shift[c]);
        out[k] = RELU ? fmaxf(v, 0.f) : v;
      }
#pragma unroll
      for (int k = 0; k < 8; k++) y[base + k] = from_f32<T>(out[k]);
    }
  } else {
    for (long i = tid; i < total; i += stride) {
      int c = (int)(i % C);
      float v = fmaf(to_f32<T>(x[i]), scale[c], shift[c]);
      y[i] = from_f32<T>(RELU ? fmaxf(v, 0.f) : v);
    }
  }
}

template <typename T, bool RELU>
__global__ void bn_bwd_apply_kernel(const T* __restrict__ x,
                                    const T* __restrict__ dy,
                                    const T* __restrict__ y,
                                    T* __restrict__ dx,
                                    const float* __restrict__ A,
                                    const float* __restrict__ B,
                                    const float* __restrict__ D,
                                    long total, int C) {
  long tid = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = tid; i < total; i += stride) {
    int c = (int)(i % C);
    float d = to_f32<T>(dy[i]);
    if (RELU) d = to_f32<T>(y[i]) > 0.f ? d : 0.f;
    float r = fmaf(A[c], d, fmaf(B[c], to_f32<T>(x[i]), D[c]));
    dx[i] = from_f32<T>(r);
  }
}

}  // namespace

// ---- launchers ------------------------------------------------------------
// ---- per-channel finalize kernels (C threads) -----------------------------
// Fold the tiny per-channel math into ONE launch instead of ~25 C-sized
// torch ops per BN call — that host overhead made ResNet-50 CPU-bound
// (profiles/r50_b256_torch_profile_hostbound_before.txt).

template <typename P>
__global__ void bn_finalize_train_kernel(
    const float* __restrict__ ws,  // [2C]: sum, sumsq
    const P* __restrict__ w, const P* __restrict__ b, P* __restrict__ rm,
    P* __restrict__ rv, float momentum, float eps, long M,
    float* __restrict__ out,  // [4C]: mean, invstd, scale, shift
    int C) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float mean = ws[c] / M;
  float var = fmaxf(ws[C + c] / M - mean * mean, 0.f);
  float invstd = rsqrtf(var + eps);
  float scale = to_f32<P>(w[c]) * invstd;
  float shift = to_f32<P>(b[c]) - mean * scale;
  out[c] = mean;
  out[C + c] = invstd;
  out[2 * C + c] = scale;
  out[3 * C + c] = shift;
  float unbias = M > 1 ? (float)M / (M - 1) : 1.f;
  rm[c] = from_f32<P>(fmaf(1.f - momentum, to_f32<P>(rm[c]),
                           momentum * mean));
  rv[c] = from_f32<P>(fmaf(1.f - momentum, to_f32<P>(rv[c]),
                           momentum * var * unbias));
}

template <typename P>
__global__ void bn_finalize_eval_kernel(const P* __restrict__ w,
                                        const P* __restrict__ b,
                                        const P* __restrict__ rm,
                                        const P* __restrict__ rv, float eps,
                                        float* __restrict__ out,  // [2C]
                                        int C) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float invstd = rsqrtf(to_f32<P>(rv[c]) + eps);
  float scale = to_f32<P>(w[c]) * invstd;
  out[c] = scale;
  out[C + c] = to_f32<P>(b[c]) - to_f32<P>(rm[c]) * scale;
}

template <typename P>
__global__ void bn_bwd_finalize_kernel(
    const float* __restrict__ s12,  // [2C]: sum_dy, sum_dy_xhat
    const P* __restrict__ w, const float* __restrict__ mean,
    const float* __restrict__ invstd, long M,
    float* __restrict__ abd,  // [3C]: A, B, D
    P* __restrict__ dgamma, P* __restrict__ dbeta, int C) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float s1 = s12[c], s2 = s12[C + c];
  float a = to_f32<P>(w[c]) * invstd[c];
  float B = -a * (s2 / M) * invstd[c];
  float D = -a * (s1 / M) + a * (s2 / M) * mean[c] * invstd[c];
  abd[c] = a;
  abd[C + c] = B;
  abd[2 * C + c] = D;
  dgamma[c] = from_f32<P>(s2);
  dbeta[c] = from_f32<P>(s1);
}

static int _ct_for(int C) {
  int ct = 8;
  while (ct < C && ct < 256) ct <<= 1;
  return ct;
}

static dim3 _reduce_grid(long M, int C, int CT) {
  int RT = 256 / CT;
  long rb = (M + RT - 1) / RT;
  if (rb > 1024) rb = 1024;
  if (rb < 1) rb = 1;
  return dim3((unsigned)rb, (unsigned)((C + CT - 1) / CT));
}

template <typename T>
static void _stats(const T* x, float* gsum, float* gsumsq, long M, int C,
                   hipStream_t stream) {
  int CT = _ct_for(C);
  hipLaunchKernelGGL((bn_stats_kernel<T>), _reduce_grid(M, C, CT), dim3(256),
                     0, stream, x, gsum, gsumsq, M, C, CT);
}
void launch_bn_stats_f32(const float* x, float* s, float* ss, long M, int C,
                         hipStream_t st) { _stats(x, s, ss, M, C, st); }
void launch_bn_stats_bf16(const rbf16* x, float* s, float* ss, long M, int C,
                          hipStream_t st) {
  _stats(reinterpret_cast<const bf16*>(x), s, ss, M, C, st);
}

template <typename T>
static void _apply(const T* x, T* y, const float* scale, const float* shift,
                   long total, int C, bool relu, hipStream_t stream) {
  int block = 256;
  int grid = rocps_grid_for((total + 7) / 8, block);
  bool vec8 = (C % 8 == 0);
  if (relu && vec8)
    hipLaunchKernelGGL((bn_apply_kernel<T, true, true>), dim3(grid),
                       dim3(block), 0, stream, x, y, scale, shift, total, C);
  else if (relu)
    hipLaunchKernelGGL((bn_apply_kernel<T, true, false>), dim3(grid),
                       dim3(block), 0, stream, x, y, scale, shift, total, C);
  else if (vec8)
    hipLaunchKernelGGL((bn_apply_kernel<T, false, true>), dim3(grid),
                       dim3(block), 0, stream, x, y, scale, shift, total, C);
  else
    hipLaunchKernelGGL((bn_apply_kernel<T, false, false>), dim3(grid),
                       dim3(block), 0, stream, x, y, scale, shift, total, C);
}
void launch_bn_apply_f32(const float* x, float* y, const float* sc,
                         const float* sh, long total, int C, bool relu,
                         hipStream_t st) { _apply(x, y, sc, sh, total, C, relu, st); }
void launch_bn_apply_bf16(const rbf16* x, rbf16* y, const float* sc,
                          const float* sh, long total, int C, bool relu,
                          hipStream_t st) {
  _apply(reinterpret_cast<const bf16*>(x), reinterpret_cast<bf16*>(y), sc, sh,
         total, C, relu, st);
}

template <typename T>
static void _bwd_reduce(const T* x, const T* dy, const T* y,
                        const float* mean, const float* invstd, float* s1,
                        float* s2, long M, int C, bool relu,
                        hipStream_t stream) {
  int CT = _ct_for(C);
  dim3 grid = _reduce_grid(M, C, CT);
  if (relu)
    hipLaunchKernelGGL((bn_bwd_reduce_kernel<T, true>), grid, dim3(256), 0,
                       stream, x, dy, y, mean, invstd, s1, s2, M, C, CT);
  else
    hipLaunchKernelGGL((bn_bwd_reduce_kernel<T, false>), grid, dim3(256), 0,
                       stream, x, dy, y, mean, invstd, s1, s2, M, C, CT);
}
void launch_bn_bwd_reduce_f32(const float* x, const float* dy, const float* y,
                              const float* mean, const float* invstd,
                              float* s1, float* s2, long M, int C, bool relu,
                              hipStream_t st) {
  _bwd_reduce(x, dy, y, mean, invstd, s1, s2, M, C, relu, st);
}
void launch_bn_bwd_reduce_bf16(const rbf16* x, const rbf16* dy,
                               const rbf16* y, const float* mean,
                               const float* invstd, float* s1, float* s2,
                               long M, int C, bool relu, hipStream_t st) {
  _bwd_reduce(reinterpret_cast<const bf16*>(x),
              reinterpret_cast<const bf16*>(dy),
              reinterpret_cast<const bf16*>(y), mean, invstd, s1, s2, M, C,
              relu, st);
}

template <typename T>
static void _bwd_apply(const T* x, const T* dy, const T* y, T* dx,
                       const float* A, const float* B, const float* D,
                       long total, int C, bool relu, hipStream_t stream) {
  int block = 256;
  int grid = rocps_grid_for(total, block);
  if (relu)
    hipLaunchKernelGGL((bn_bwd_apply_kernel<T, true>), dim3(grid),
                       dim3(block), 0, stream, x, dy, y, dx, A, B, D, total, C);
  else
    hipLaunchKernelGGL((bn_bwd_apply_kernel<T, false>), dim3(grid),
                       dim3(block), 0, stream, x, dy, y, dx, A, B, D, total, C);
}
void launch_bn_bwd_apply_f32(const float* x, const float* dy, const float* y,
                             float* dx, const float* A, const float* B,
                             const float* D, long total, int C, bool relu,
                             hipStream_t st) {
  _bwd_apply(x, dy, y, dx, A, B, D, total, C, relu, st);
}
void launch_bn_bwd_apply_bf16(const rbf16* x, const rbf16* dy,
                              const rbf16* y, rbf16* dx, const float* A,
                              const float* B, const float* D, long total,
                              int C, bool relu, hipStream_t st) {
  _bwd_apply(reinterpret_cast<const bf16*>(x),
             reinterpret_cast<const bf16*>(dy),
             reinterpret_cast<const bf16*>(y), reinterpret_cast<bf16*>(dx), A,
             B, D, total, C, relu, st);
}

// ---- finalize launchers ---------------------------------------------------
#define FINALIZE_GRID(C) dim3((unsigned)(((C) + 255) / 256)), dim3(256)

void launch_bn_finalize_train_f32(const float* ws, const float* w,
                                  const float* b, float* rm, float* rv,
                                  float momentum, float eps, long M,
                                  float* out, int C, hipStream_t st) {
  hipLaunchKernelGGL((bn_finalize_train_kernel<float>), FINALIZE_GRID(C), 0,
                     st, ws, w, b, rm, rv, momentum, eps, M, out, C);
}
void launch_bn_finalize_train_bf16(const float* ws, const rbf16* w,
                                   const rbf16* b, rbf16* rm, rbf16* rv,
                                   float momentum, float eps, long M,
                                   float* out, int C, hipStream_t st) {
  hipLaunchKernelGGL((bn_finalize_train_kernel<bf16>), FINALIZE_GRID(C), 0,
                     st, ws, reinterpret_cast<const bf16*>(w),
                     reinterpret_cast<const bf16*>(b),
                     reinterpret_cast<bf16*>(rm), reinterpret_cast<bf16*>(rv),
                     momentum, eps, M, out, C);
}
void launch_bn_finalize_eval_f32(const float* w, const float* b,
                                 const float* rm, const float* rv, float eps,
                                 float* out, int C, hipStream_t st) {
  hipLaunchKernelGGL((bn_finalize_eval_kernel<float>), FINALIZE_GRID(C), 0,
                     st, w, b, rm, rv, eps, out, C);
}
void launch_bn_finalize_eval_bf16(const rbf16* w, const rbf16* b,
                                  const rbf16* rm, const rbf16* rv, float eps,
                                  float* out, int C, hipStream_t st) {
  hipLaunchKernelGGL((bn_finalize_eval_kernel<bf16>), FINALIZE_GRID(C), 0,
                     st, reinterpret_cast<const bf16*>(w),
                     reinterpret_cast<const bf16*>(b),
                     reinterpret_cast<const bf16*>(rm),
                     reinterpret_cast<const bf16*>(rv), eps, out, C);
}
void launch_bn_bwd_finalize_f32(const float* s12, const float* w,
                                const float* mean, const float* invstd,
                                long M, float* abd, float* dgamma,
                                float* dbeta, int C, hipStream_t st) {
  hipLaunchKernelGGL((bn_bwd_finalize_kernel<float>), FINALIZE_GRID(C), 0, st,
                     s12, w, mean, invstd, M, abd, dgamma, dbeta, C);
}
void launch_bn_bwd_finalize_bf16(const float* s12, const rbf16* w,
                                 const float* mean, const float* invstd,
                                 long M, float* abd, rbf16* dgamma,
                                 rbf16* dbeta, int C, hipStream_t st) {
  hipLaunchKernelGGL((bn_bwd_finalize_kernel<bf16>), FINALIZE_GRID(C), 0, st,
                     s12, reinterpret_cast<const bf16*>(w), mean, invstd, M,
                     abd, reinterpret_cast<bf16*>(dgamma),
                     reinterpret_cast<bf16*>(dbeta), C);
}
