// Shared helpers for rocps CDNA4 (gfx950) kernels.
//
// Wave width is 64 on CDNA4 (not 32) — every cross-lane idiom below is
// 64-wide. bf16 memory traffic is vectorized as ushort4/ushort8
// reinterprets (hipcc does not auto-vectorize scalar bf16 loads;
// cdna_hip_programming.md Guideline 13).
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64

using bf16 = __hip_bfloat16;

template <typename T>
__device__ __forceinline__ float to_f32(T v);
template <>
__device__ __forceinline__ float to_f32<float>(float v) { return v; }
template <>
__device__ __forceinline__ float to_f32<bf16>(bf16 v) {
  return __bfloat162float(v);
}

template <typename T>
__device__ __forceinline__ T from_f32(float v);
template <>
__device__ __forceinline__ float from_f32<float>(float v) { return v; }
template <>
__device__ __forceinline__ bf16 from_f32<bf16>(float v) {
  return __float2bfloat16(v);
}

// ---- block reductions (256 threads = 4 waves) -----------------------------
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1)
    v += __shfl_down(v, off, WAVE);
  return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1)
    v = fmaxf(v, __shfl_down(v, off, WAVE));
  return v;
}

// LDS-assisted block reduction; `lds` must hold >= blockDim.x/WAVE floats.
__device__ __forceinline__ float block_reduce_sum(float v, float* lds) {
  int lane = threadIdx.x & (WAVE - 1);
  int wid = threadIdx.x / WAVE;
  int nwaves = (blockDim.x + WAVE - 1) / WAVE;
  v = wave_reduce_sum(v);
  if (lane == 0) lds[wid] = v;
  __syncthreads();
  v = (threadIdx.x < nwaves) ? lds[threadIdx.x] : 0.f;
  if (wid == 0) v = wave_reduce_sum(v);
  if (threadIdx.x == 0) lds[0] = v;
  __syncthreads();
  v = lds[0];
  __syncthreads();
  return v;
}

__device__ __forceinline__ float block_reduce_max(float v, float* lds) {
  int lane = threadIdx.x & (WAVE - 1);
  int wid = threadIdx.x / WAVE;
  int nwaves = (blockDim.x + WAVE - 1) / WAVE;
  v = wave_reduce_max(v);
  if (lane == 0) lds[wid] = v;
  __syncthreads();
  v = (threadIdx.x < nwaves) ? lds[threadIdx.x] : -INFINITY;
  if (wid == 0) v = wave_reduce_max(v);
  if (threadIdx.x == 0) lds[0] = v;
  __syncthreads();
  v = lds[0];
  __syncthreads();
  return v;
}

// memory-bound grid sizing (Guideline 11): cap blocks, grid-stride the rest
__host__ __forceinline__ int rocps_grid_for(long work_items, int block,
                                            int cap = 2048) {
  long blocks = (work_items + block - 1) / block;
  return (int)(blocks < cap ? (blocks > 1 ? blocks : 1) : cap);
}

#define HIP_CHECK_LAST()                                             \
  do {                                                               \
    hipError_t e_ = hipGetLastError();                               \
    if (e_ != hipSuccess) {                                          \
      printf("HIP error %s at %s:%d\n", hipGetErrorString(e_),       \
             __FILE__, __LINE__);                                    \
    }                                                                \
  } while (0)

__device__ __forceinline__ float wave_allreduce_sum(float v) {
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1)
    v += __shfl_xor(v, off, WAVE);
  return v;
}
