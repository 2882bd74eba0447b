#include "hip/hip_runtime.h"
// Fused multi-element optimizer updates — capability C8.
//
// Reference behavior: TF ApplyGradientDescent / ApplyMomentum / ApplyAdam
// (tensorflow/core/kernels/training_ops*.cc), one kernel per variable on
// the PS device. Here: ONE grid-stride launch per flat bucket/shard
// (the reduce-scattered segment that *is* the co-located PS shard),
// updating fp32 master weights and the bf16 replica view in the same
// pass — a single read of grad + single write of both.
//
// Memory-bound: bf16 grads load as ushort8 (16 B/lane), fp32 state as
// float4 pairs (Guideline 13). Engine-side buffers are padded to 64
// elements and 256-B aligned, so the vector path covers everything; a
// scalar tail handles unpadded (async-shard) calls.

#include "common.h"
#include "launchers.h"

namespace {

struct alignas(16) u16x8 { unsigned short v[8]; };
struct alignas(16) f32x4 { float v[4]; };

template <typename T>
__device__ __forceinline__ void load8(const T* p, float* out);

template <>
__device__ __forceinline__ void load8<bf16>(const bf16* p, float* out) {
  u16x8 raw = *reinterpret_cast<const u16x8*>(p);
#pragma unroll
  for (int k = 0; k < 8; k++) {
    unsigned int bits = (unsigned int)raw.v[k] << 16;
    out[k] = __uint_as_float(bits);
  }
}

template <>
__device__ __forceinline__ void load8<float>(const float* p, float* out) {
  f32x4 a = *reinterpret_cast<const f32x4*>(p);
  f32x4 b = *reinterpret_cast<const f32x4*>(p + 4);
#pragma unroll
  for (int k = 0; k < 4; k++) { out[k] = a.v[k]; out[4 + k] = b.v[k]; }
}

template <typename T>
__device__ __forceinline__ void store8(T* p, const float* in);

template <>
__device__ __forceinline__ void store8<bf16>(bf16* p, const float* in) {
  u16x8 raw;
#pragma unroll
  for (int k = 0; k < 8; k++)
    raw.v[k] = __hip_bfloat16_raw(__float2bfloat16(in[k])).x;
  *reinterpret_cast<u16x8*>(p) = raw;
}

template <>
__device__ __forceinline__ void store8<float>(float* p, const float* in) {
  f32x4 a, b;
#pragma unroll
  for (int k = 0; k < 4; k++) { a.v[k] = in[k]; b.v[k] = in[4 + k]; }
  *reinterpret_cast<f32x4*>(p) = a;
  *reinterpret_cast<f32x4*>(p + 4) = b;
}

__device__ __forceinline__ void loadf8(const float* p, float* out) {
  load8<float>(p, out);
}
__device__ __forceinline__ void storef8(float* p, const float* in) {
  store8<float>(p, in);
}

// --------------------------------------------------------------------------
template <typename T, bool HAS_MASTER>
__global__ void sgd_kernel(T* __restrict__ p, const T* __restrict__ g,
                           float* __restrict__ m, float lr, long n) {
  long tid = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  long nvec = n / 8;
  for (long i = tid; i < nvec; i += stride) {
    float gv[8], mv[8];
    load8<T>(g + i * 8, gv);
    if (HAS_MASTER) loadf8(m + i * 8, mv);
    else load8<T>(p + i * 8, mv);
#pragma unroll
    for (int k = 0; k < 8; k++) mv[k] = fmaf(-lr, gv[k], mv[k]);
    if (HAS_MASTER) storef8(m + i * 8, mv);
    store8<T>(p + i * 8, mv);
  }
  for (long j = nvec * 8 + tid; j < n; j += stride) {
    float mv = HAS_MASTER ? m[j] : to_f32<T>(p[j]);
    mv = fmaf(-lr, to_f32<T>(g[j]), mv);
    if (HAS_MASTER) m[j] = mv;
    p[j] = from_f32<T>(mv);
  }
}

template <typename T, bool HAS_MASTER, bool NESTEROV>
__global__ void momentum_kernel(T* __restrict__ p, const T* __restrict__ g,
                                float* __restrict__ m, float* __restrict__ v,
                                float lr, float mu, long n) {
  long tid = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  long nvec = n / 8;
  for (long i = tid; i < nvec; i += stride) {
    float gv[8], mv[8], vv[8];
    load8<T>(g + i * 8, gv);
    loadf8(v + i * 8, vv);
    if (HAS_MASTER) loadf8(m + i * 8, mv);
    else load8<T>(p + i * 8, mv);
#pragma unroll
    for (int k = 0; k < 8; k++) {
      vv[k] = fmaf(mu, vv[k], gv[k]);
      if (NESTEROV)
        mv[k] = mv[k] - lr * gv[k] - lr * mu * vv[k];
      else
        mv[k] = fmaf(-lr, vv[k], mv[k]);
    }
    storef8(v + i * 8, vv);
    if (HAS_MASTER) storef8(m + i * 8, mv);
    store8<T>(p + i * 8, mv);
  }
  for (long j = nvec * 8 + tid; j < n; j += stride) {
    float gv = to_f32<T>(g[j]);
    float vv = fmaf(mu, v[j], gv);
    float mv = HAS_MASTER ? m[j] : to_f32<T>(p[j]);
    mv = NESTEROV ? mv - lr * gv - lr * mu * vv : fmaf(-lr, vv, mv);
    v[j] = vv;
    if (HAS_MASTER) m[j] = mv;
    p[j] = from_f32<T>(mv);
  }
}

template <typename T, bool HAS_MASTER>
__global__ void adam_kernel(T* __restrict__ p, const T* __restrict__ g,
                            float* __restrict__ mst, float* __restrict__ m,
                            float* __restrict__ v, float alpha, float b1,
                            float b2, float eps, long n) {
  long tid = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  long nvec = n / 8;
  for (long i = tid; i < nvec; i += stride) {
    float gv[8], pv[8], mv[8], vv[8];
    load8<T>(g + i * 8, gv);
    loadf8(m + i * 8, mv);
    loadf8(v + i * 8, vv);
    if (HAS_MASTER) loadf8(mst + i * 8, pv);
    else load8<T>(p + i * 8, pv);
#pragma unroll
    for (int k = 0; k < 8; k++) {
      mv[k] = fmaf(b1, mv[k], (1.f - b1) * gv[k]);
      vv[k] = fmaf(b2, vv[k], (1.f - b2) * gv[k] * gv[k]);
      pv[k] -= alpha * mv[k] / (sqrtf(vv[k]) + eps);
    }
    storef8(m + i * 8, mv);
    storef8(v + i * 8, vv);
    if (HAS_MASTER) storef8(mst + i * 8, pv);
    store8<T>(p + i * 8, pv);
  }
  for (long j = nvec * 8 + tid; j < n; j += stride) {
    float gv = to_f32<T>(g[j]);
    float mv = fmaf(b1, m[j], (1.f - b1) * gv);
    float vv = fmaf(b2, v[j], (1.f - b2) * gv * gv);
    float pv = HAS_MASTER ? mst[j] : to_f32<T>(p[j]);
    pv -= alpha * mv / (sqrtf(vv) + eps);
    m[j] = mv;
    v[j] = vv;
    if (HAS_MASTER) mst[j] = pv;
    p[j] = from_f32<T>(pv);
  }
}

}  // namespace

// ---- C++ launchers (called from ext.cc) -----------------------------------
#define LAUNCH_BOTH(T, KERN, ...)                                           \
  do {                                                                      \
    int block = 256;                                                        \
    int grid = rocps_grid_for((n + 7) / 8, block);                          \
    if (has_master)                                                         \
      hipLaunchKernelGGL((KERN<T, true>), dim3(grid), dim3(block), 0,       \
                         stream, __VA_ARGS__);                              \
    else                                                                    \
      hipLaunchKernelGGL((KERN<T, false>), dim3(grid), dim3(block), 0,      \
                         stream, __VA_ARGS__);                              \
  } while (0)

void launch_sgd_f32(float* p, const float* g, float* m, float lr, long n,
                    bool has_master, hipStream_t stream) {
  LAUNCH_BOTH(float, sgd_kernel, p, g, m, lr, n);
}
void launch_sgd_bf16(rbf16* p_, const rbf16* g_, float* m, float lr, long n,
                     bool has_master, hipStream_t stream) {
  bf16* p = reinterpret_cast<bf16*>(p_);
  const bf16* g = reinterpret_cast<const bf16*>(g_);
  LAUNCH_BOTH(bf16, sgd_kernel, p, g, m, lr, n);
}

#define LAUNCH_MOM(T)                                                       \
  do {                                                                      \
    int block = 256;                                                        \
    int grid = rocps_grid_for((n + 7) / 8, block);                          \
    if (has_master && nesterov)                                             \
      hipLaunchKernelGGL((momentum_kernel<T, true, true>), dim3(grid),      \
                         dim3(block), 0, stream, p, g, m, v, lr, mu, n);    \
    else if (has_master)                                                    \
      hipLaunchKernelGGL((momentum_kernel<T, true, false>), dim3(grid),     \
                         dim3(block), 0, stream, p, g, m, v, lr, mu, n);    \
    else if (nesterov)                                                      \
      hipLaunchKernelGGL((momentum_kernel<T, false, true>), dim3(grid),     \
                         dim3(block), 0, stream, p, g, m, v, lr, mu, n);    \
    else                                                                    \
      hipLaunchKernelGGL((momentum_kernel<T, false, false>), dim3(grid),    \
                         dim3(block), 0, stream, p, g, m, v, lr, mu, n);    \
  } while (0)

void launch_momentum_f32(float* p, const float* g, float* m, float* v,
                         float lr, float mu, bool nesterov, long n,
                         bool has_master, hipStream_t stream) {
  LAUNCH_MOM(float);
}
void launch_momentum_bf16(rbf16* p_, const rbf16* g_, float* m, float* v,
                          float lr, float mu, bool nesterov, long n,
                          bool has_master, hipStream_t stream) {
  bf16* p = reinterpret_cast<bf16*>(p_);
  const bf16* g = reinterpret_cast<const bf16*>(g_);
  LAUNCH_MOM(bf16);
}

void launch_adam_f32(float* p, const float* g, float* mst, float* m, float* v,
                     float alpha, float b1, float b2, float eps, long n,
                     bool has_master, hipStream_t stream) {
  int block = 256;
  int grid = rocps_grid_for((n + 7) / 8, block);
  if (has_master)
    hipLaunchKernelGGL((adam_kernel<float, true>), dim3(grid), dim3(block), 0,
                       stream, p, g, mst, m, v, alpha, b1, b2, eps, n);
  else
    hipLaunchKernelGGL((adam_kernel<float, false>), dim3(grid), dim3(block),
                       0, stream, p, g, mst, m, v, alpha, b1, b2, eps, n);
}
void launch_adam_bf16(rbf16* p_, const rbf16* g_, float* mst, float* m,
                      float* v, float alpha, float b1, float b2, float eps,
                      long n, bool has_master, hipStream_t stream) {
  bf16* p = reinterpret_cast<bf16*>(p_);
  const bf16* g = reinterpret_cast<const bf16*>(g_);
  int block = 256;
  int grid = rocps_grid_for((n + 7) / 8, block);
  if (has_master)
    hipLaunchKernelGGL((adam_kernel<bf16, true>), dim3(grid), dim3(block), 0,
                       stream, p, g, mst, m, v, alpha, b1, b2, eps, n);
  else
    hipLaunchKernelGGL((adam_kernel<bf16, false>), dim3(grid), dim3(block), 0,
                       stream, p, g, mst, m, v, alpha, b1, b2, eps, n);
}
