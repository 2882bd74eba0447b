// Hand-written MFMA bf16 GEMM — capability C10.
//
// Computes D[M,N] = A[M,K] · W[N,K]^T (+bias[N]) (+GELU) in bf16 with
// fp32 accumulation — the torch.nn.Linear contract (weights stored
// (N,K)), which maps both operand fragments onto contiguous 16-byte
// K-runs.
//
// Structure (cdna_hip_programming.md §5 "step-3" ladder):
// - 128x128 macro-tile, BK=64, 4 waves (256 threads), each wave owns a
//   64x64 sub-tile as 4x4 fragments of v_mfma_f32_16x16x32_bf16.
// - Double-buffered LDS (2 x 16 KB per operand), staged by
//   __builtin_amdgcn_global_load_lds dwordx4 (16 B/lane — the width
//   that took the guide's ladder 546→911 TF).
// - XOR swizzle ((row&7)<<4 on the byte offset) applied to BOTH the
//   glds SOURCE address and the ds_read offset (rule 21: the LDS image
//   is lane-linear, so the swizzle rides on the source permutation),
//   killing the 8-way ds_read_b128 bank conflict of 128-B rows.
// - Epilogue: fp32 accum (+bias) (+tanh-GELU) → bf16 stores.
//
// Shape contract: M%128 == 0, N%128 == 0, K%64 == 0 (the BERT/MLP GEMM
// shapes; the Python wrapper falls back to rocBLAS outside it).

#include "common.h"
#include "launchers.h"

namespace {

using f32x4 = __attribute__((ext_vector_type(4))) float;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;

constexpr int BM = 128, BN = 128, BK = 64;
constexpr int TILE_BYTES = BM * BK * 2;  // 16 KB per operand

__device__ __forceinline__ int swz(int row, int byte_in_row) {
  // spread a 128-B row over 8 16-B slots (T2: byte ^= (row&7)<<4)
  return row * (BK * 2) + (byte_in_row ^ ((row & 7) << 4));
}

__device__ __forceinline__ float gelu_tanh(float x) {
  const float c = 0.7978845608028654f;  // sqrt(2/pi)
  float t = tanhf(c * (x + 0.044715f * x * x * x));
  return 0.5f * x * (1.f + t);
}

template <int EPI>  // 0 = none, 1 = +bias, 2 = +bias +gelu,
                    // 3 = +bias +gelu, pre-activation ALSO stored to P
                    //     (training: backward reuses pre instead of the
                    //      fp32 recompute chain — ROADMAP §Perf 5)
__global__ __launch_bounds__(256)
void gemm_bf16_kernel(const bf16* __restrict__ A, const bf16* __restrict__ W,
                      const float* __restrict__ bias, bf16* __restrict__ D,
                      bf16* __restrict__ P, int M, int N, int K) {
  __shared__ char smem[4 * TILE_BYTES];  // [buf][A|B]
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;

  // Flat grid; on TALL grids (few n-tiles — the Linear/conv1x1
  // shapes) the bijective XCD remap (T1) clusters each XCD on a
  // contiguous run of ids sharing the A row-panel, keeping the panel
  // in ONE L2 (+18% measured on 8192x768x3072,
  // profiles/gemm_micro_r2e.txt).  On wide/square grids the default
  // round-robin is faster (-14% at 8192^3 with the remap), so it is
  // gated on the n-tile count.
  const int ntx = N / BN;
  int bid = blockIdx.x;
  if (ntx <= 8) {
    int nwg = (int)gridDim.x;
    int q = nwg >> 3, r = nwg & 7, xcd = bid & 7, pos = bid >> 3;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
  }
  const int n0 = (bid % ntx) * BN;
  const int m0 = (bid / ntx) * BM;

  // wave sub-tile: 2x2 waves -> each 64x64
  const int wm = (wid >> 1) * 64;
  const int wn = (wid & 1) * 64;

  // ---- staging: thread t fills 16 B at lds offset t*16 + i*4096 ------
  // lane-linear dest (glds constraint); the matching source element is
  // row r = off/128, byte b = off%128 of the tile, with the XOR swizzle
  // applied to the SOURCE byte (rule 21): src_byte = b ^ ((r&7)<<4).
  auto a_src = [&](int off) -> const bf16* {
    int r = off >> 7, b = off & 127;
    int sb = b ^ ((r & 7) << 4);
    return A + (long)(m0 + r) * K + (sb >> 1);
  };
  auto w_src = [&](int off) -> const bf16* {
    int r = off >> 7, b = off & 127;
    int sb = b ^ ((r & 7) << 4);
    return W + (long)(n0 + r) * K + (sb >> 1);
  };

  auto stage = [&](int buf, int kt) {
    // 4 glds per operand: 256 threads x 16 B = 4 KB each
    char* base = smem + buf * 2 * TILE_BYTES;
#pragma unroll
    for (int i = 0; i < 4; i++) {
      int off = tid * 16 + i * 4096;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)
              (a_src(off) + kt * BK),
          (__attribute__((address_space(3))) unsigned int*)(base + off),
          16, 0, 0);
    }
    char* baseB = base + TILE_BYTES;
#pragma unroll
    for (int i = 0; i < 4; i++) {
      int off = tid * 16 + i * 4096;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)
              (w_src(off) + kt * BK),
          (__attribute__((address_space(3))) unsigned int*)(baseB + off),
          16, 0, 0);
    }
  };

  f32x4 acc[4][4] = {};

  const int nkt = K / BK;
  stage(0, 0);
  asm volatile("s_waitcnt vmcnt(0)");
  __syncthreads();

  for (int kt = 0; kt < nkt; kt++) {
    if (kt + 1 < nkt) stage((kt + 1) & 1, kt + 1);
    const char* base = smem + (kt & 1) * 2 * TILE_BYTES;
    const char* baseB = base + TILE_BYTES;
    // lane fragment addresses: row = (frag row) + (lane&15),
    // k-block = lane>>4 (16 B each); swizzled byte offset.
#pragma unroll
    for (int kk = 0; kk < BK / 32; kk++) {
      bf16x8 afrag[4], bfrag[4];
#pragma unroll
      for (int i = 0; i < 4; i++) {
        int ar = wm + i * 16 + (lane & 15);
        int ab = kk * 64 + (lane >> 4) * 16;
        afrag[i] = *(const bf16x8*)(base + swz(ar, ab));
        int br = wn + i * 16 + (lane & 15);
        bfrag[i] = *(const bf16x8*)(baseB + swz(br, ab));
      }
#pragma unroll
      for (int i = 0; i < 4; i++)
#pragma unroll
        for (int j = 0; j < 4; j++)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[i], bfrag[j], acc[i][j], 0, 0, 0);
    }
    asm volatile("s_waitcnt vmcnt(0)");
    __syncthreads();
  }

  // ---- epilogue: C/D map for 16x16x32: col = lane&15,
  // row = (lane>>4)*4 + reg (dtype-independent on gfx950, guide §3).
#pragma unroll
  for (int i = 0; i < 4; i++) {
#pragma unroll
    for (int j = 0; j < 4; j++) {
      int col = n0 + wn + j * 16 + (lane & 15);
      float bv = (EPI >= 1) ? bias[col] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; r++) {
        int row = m0 + wm + i * 16 + (lane >> 4) * 4 + r;
        float v = acc[i][j][r] + bv;
        if (EPI == 3) P[(long)row * N + col] = from_f32<bf16>(v);
        if (EPI >= 2) v = gelu_tanh(v);
        D[(long)row * N + col] = from_f32<bf16>(v);
      }
    }
  }
}

}  // namespace

void launch_gemm_bf16(const rbf16* A, const rbf16* W, const float* bias,
                      rbf16* D, rbf16* PRE, int M, int N, int K,
                      int epilogue, hipStream_t st) {
  dim3 grid((unsigned)((N / BN) * (M / BM)));
  const bf16* a = reinterpret_cast<const bf16*>(A);
  const bf16* w = reinterpret_cast<const bf16*>(W);
  bf16* d = reinterpret_cast<bf16*>(D);
  bf16* p = reinterpret_cast<bf16*>(PRE);
  if (epilogue == 3)
    hipLaunchKernelGGL((gemm_bf16_kernel<3>), grid, dim3(256), 0, st, a, w,
                       bias, d, p, M, N, K);
  else if (epilogue == 2)
    hipLaunchKernelGGL((gemm_bf16_kernel<2>), grid, dim3(256), 0, st, a, w,
                       bias, d, nullptr, M, N, K);
  else if (epilogue == 1)
    hipLaunchKernelGGL((gemm_bf16_kernel<1>), grid, dim3(256), 0, st, a, w,
                       bias, d, nullptr, M, N, K);
  else
    hipLaunchKernelGGL((gemm_bf16_kernel<0>), grid, dim3(256), 0, st, a, w,
                       bias, d, nullptr, M, N, K);
}
