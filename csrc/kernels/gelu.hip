// Fused tanh-GELU backward: g = dy * dgelu(pre) in ONE pass.
//
// The BERT MLP training backward (rocps/ops/gemm.py
// _LinearGeluTrainFn) needs dy scaled by the GELU derivative at the
// saved pre-activation.  Composing it from torch ops materializes
// several fp32 intermediates of (B*S, 3072) per layer; this kernel
// reads pre and dy once (bf16x8 vectors) and writes g once — pure
// HBM-bound, 3 tensor passes total.

#include "common.h"
#include "launchers.h"

namespace {

using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;

__device__ __forceinline__ float dgelu_tanh(float x) {
  const float c = 0.7978845608028654f;  // sqrt(2/pi)
  float u = c * (x + 0.044715f * x * x * x);
  float t = tanhf(u);
  return 0.5f * (1.f + t)
       + 0.5f * x * (1.f - t * t) * c * (1.f + 3.f * 0.044715f * x * x);
}

__global__ __launch_bounds__(256)
void gelu_bwd_kernel(const bf16* __restrict__ pre,
                     const bf16* __restrict__ dy, bf16* __restrict__ g,
                     long n) {
  long i0 = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  long stride = (long)gridDim.x * blockDim.x * 8;
  for (long i = i0; i < n; i += stride) {
    if (i + 8 <= n) {
      bf16x8 p = *(const bf16x8*)(pre + i);
      bf16x8 d = *(const bf16x8*)(dy + i);
      bf16x8 o;
#pragma unroll
      for (int k = 0; k < 8; k++)
        o[k] = (__bf16)((float)d[k] * dgelu_tanh((float)p[k]));
      *(bf16x8*)(g + i) = o;
    } else {
      for (long j = i; j < n; j++)
        g[j] = from_f32<bf16>(to_f32(dy[j]) * dgelu_tanh(to_f32(pre[j])));
    }
  }
}

}  // namespace

void launch_gelu_bwd(const rbf16* pre, const rbf16* dy, rbf16* g, long n,
                     hipStream_t st) {
  long blocks = (n / 8 + 255) / 256;
  if (blocks > 4096) blocks = 4096;  // grid-stride; >>256 WGs fills 8 XCDs
  if (blocks < 1) blocks = 1;
  hipLaunchKernelGGL(gelu_bwd_kernel, dim3((unsigned)blocks), dim3(256), 0,
                     st, reinterpret_cast<const bf16*>(pre),
                     reinterpret_cast<const bf16*>(dy),
                     reinterpret_cast<bf16*>(g), n);
}
