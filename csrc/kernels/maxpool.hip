// MaxPool 3x3/stride-2/pad-1 fwd+bwd on NHWC — the ResNet stem pool
// (C9's pooling slice; reference: TF MaxPool/MaxPoolGrad kernels).
//
// torch's maxpool stores int64 argmax indices (8 B/elem) and its
// backward scatters; here the argmax is a 1-byte window code (0..8)
// and the backward is a gather: each INPUT pixel sums dy over the <=4
// windows whose stored code points at it — no atomics, deterministic.
// Column-resident threads (8 channels / 16 B per thread, Guideline 13).

#include "common.h"
#include "launchers.h"

namespace {

struct alignas(16) u16x8m { unsigned short v[8]; };
struct alignas(8) u8x8 { unsigned char v[8]; };

template <typename T>
__device__ __forceinline__ void load8m(const T* p, float* out);
template <>
__device__ __forceinline__ void load8m<bf16>(const bf16* p, float* out) {
  u16x8m raw = *reinterpret_cast<const u16x8m*>(p);
#pragma unroll
  for (int k = 0; k < 8; k++)
    out[k] = __uint_as_float((unsigned int)raw.v[k] << 16);
}
template <>
__device__ __forceinline__ void load8m<float>(const float* p, float* out) {
#pragma unroll
  for (int k = 0; k < 8; k++) out[k] = p[k];
}

template <typename T>
__device__ __forceinline__ void store8m(T* p, const float* in);
template <>
__device__ __forceinline__ void store8m<bf16>(bf16* p, const float* in) {
  u16x8m raw;
#pragma unroll
  for (int k = 0; k < 8; k++)
    raw.v[k] = __hip_bfloat16_raw(__float2bfloat16(in[k])).x;
  *reinterpret_cast<u16x8m*>(p) = raw;
}
template <>
__device__ __forceinline__ void store8m<float>(float* p, const float* in) {
#pragma unroll
  for (int k = 0; k < 8; k++) p[k] = in[k];
}

// fwd: one thread = 8 channels of one output pixel
template <typename T>
__global__ void maxpool332_fwd_kernel(const T* __restrict__ x,
                                      T* __restrict__ y,
                                      unsigned char* __restrict__ idx,
                                      int N, int C, int H, int W, int OH,
                                      int OW) {
  int cols = C >> 3;
  long total = (long)N * OH * OW * cols;
  long tid = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long t = tid; t < total; t += stride) {
    int cg = (int)(t % cols);
    long pix = t / cols;
    int ox = (int)(pix % OW);
    long rest = pix / OW;
    int oy = (int)(rest % OH);
    int n = (int)(rest / OH);
    float best[8];
    unsigned char code[8];
#pragma unroll
    for (int k = 0; k < 8; k++) {
      best[k] = -INFINITY;
      code[k] = 255;
    }
    int iy0 = oy * 2 - 1, ix0 = ox * 2 - 1;
#pragma unroll
    for (int ky = 0; ky < 3; ky++) {
      int iy = iy0 + ky;
      if (iy < 0 || iy >= H) continue;
#pragma unroll
      for (int kx = 0; kx < 3; kx++) {
        int ix = ix0 + kx;
        if (ix < 0 || ix >= W) continue;
        float v[8];
        load8m<T>(x + (((long)n * H + iy) * W + ix) * C + cg * 8, v);
        unsigned char cd = (unsigned char)(ky * 3 + kx);
#pragma unroll
        for (int k = 0; k < 8; k++) {
          if (v[k] > best[k]) {
            best[k] = v[k];
            code[k] = cd;
          }
        }
      }
    }
    long off = (((long)n * OH + oy) * OW + ox) * C + cg * 8;
    store8m<T>(y + off, best);
    *reinterpret_cast<u8x8*>(idx + off) = *reinterpret_cast<u8x8*>(code);
  }
}

// bwd: one thread = 8 channels of one INPUT pixel; gather from <=4 windows
template <typename T>
__global__ void maxpool332_bwd_kernel(const T* __restrict__ dy,
                                      const unsigned char* __restrict__ idx,
                                      T* __restrict__ dx, int N, int C,
                                      int H, int W, int OH, int OW) {
  int cols = C >> 3;
  long total = (long)N * H * W * cols;
  long tid = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (long t = tid; t < total; t += stride) {
    int cg = (int)(t % cols);
    long pix = t / cols;
    int ix = (int)(pix % W);
    long rest = pix / W;
    int iy = (int)(rest % H);
    int n = (int)(rest / H);
    float acc[8] = {0.f};
    // windows covering (iy, ix): o with 2o-1 <= i <= 2o+1
    // -> o in [ceil((i-1)/2), floor((i+1)/2)] = [i>>1, (i+1)>>1]
    int oy_lo = iy >> 1;
    int oy_hi = (iy + 1) >> 1;
#pragma unroll 2
    for (int oy = oy_lo; oy <= oy_hi; oy++) {
      if (oy >= OH) continue;
      int ky = iy - (oy * 2 - 1);
      int ox_lo = ix >> 1;
      int ox_hi = (ix + 1) >> 1;
#pragma unroll 2
      for (int ox = ox_lo; ox <= ox_hi; ox++) {
        if (ox >= OW) continue;
        int kx = ix - (ox * 2 - 1);
        unsigned char cd = (unsigned char)(ky * 3 + kx);
        long off = (((long)n * OH + oy) * OW + ox) * C + cg * 8;
        u8x8 codes = *reinterpret_cast<const u8x8*>(idx + off);
        float dv[8];
        load8m<T>(dy + off, dv);
#pragma unroll
        for (int k = 0; k < 8; k++)
          if (codes.v[k] == cd) acc[k] += dv[k];
      }
    }
    store8m<T>(dx + (((long)n * H + iy) * W + ix) * C + cg * 8, acc);
  }
}

}  // namespace

template <typename T>
static void _mp_fwd(const T* x, T* y, unsigned char* idx, int N, int C,
                    int H, int W, int OH, int OW, hipStream_t st) {
  long work = (long)N * OH * OW * (C >> 3);
  int grid = rocps_grid_for(work, 256, 4096);
  hipLaunchKernelGGL((maxpool332_fwd_kernel<T>), dim3(grid), dim3(256), 0, st,
                     x, y, idx, N, C, H, W, OH, OW);
}
void launch_maxpool332_fwd_f32(const float* x, float* y, unsigned char* idx,
                               int N, int C, int H, int W, int OH, int OW,
                               hipStream_t st) {
  _mp_fwd(x, y, idx, N, C, H, W, OH, OW, st);
}
void launch_maxpool332_fwd_bf16(const rbf16* x, rbf16* y, unsigned char* idx,
                                int N, int C, int H, int W, int OH, int OW,
                                hipStream_t st) {
  _mp_fwd(reinterpret_cast<const bf16*>(x), reinterpret_cast<bf16*>(y), idx,
          N, C, H, W, OH, OW, st);
}

template <typename T>
static void _mp_bwd(const T* dy, const unsigned char* idx, T* dx, int N,
                    int C, int H, int W, int OH, int OW, hipStream_t st) {
  long work = (long)N * H * W * (C >> 3);
  int grid = rocps_grid_for(work, 256, 4096);
  hipLaunchKernelGGL((maxpool332_bwd_kernel<T>), dim3(grid), dim3(256), 0, st,
                     dy, idx, dx, N, C, H, W, OH, OW);
}
void launch_maxpool332_bwd_f32(const float* dy, const unsigned char* idx,
                               float* dx, int N, int C, int H, int W, int OH,
                               int OW, hipStream_t st) {
  _mp_bwd(dy, idx, dx, N, C, H, W, OH, OW, st);
}
void launch_maxpool332_bwd_bf16(const rbf16* dy, const unsigned char* idx,
                                rbf16* dx, int N, int C, int H, int W,
                                int OH, int OW, hipStream_t st) {
  _mp_bwd(reinterpret_cast<const bf16*>(dy), idx,
          reinterpret_cast<bf16*>(dx), N, C, H, W, OH, OW, st);
}
