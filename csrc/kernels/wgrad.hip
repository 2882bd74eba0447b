// conv1x1 weight gradient — the tall-skinny reduction GEMM
// dW[n][k] = sum_m dY[m][n] * X[m][k], M up to ~800k (capability C9's
// wgrad half; replaces MIOpen's wrw kernel on shapes where it wins).
//
// Both MFMA operands are CONSUMED transposed (contraction m is the
// row dimension of both inputs): tiles are staged in their natural
// [m][c] row-major form and fragments are read back with gfx950's
// ds_read_b64_tr_b16 hardware transpose (semantics probed on-device,
// profiles/tr16_probe_semantics.txt: a 16-lane group addressing
// consecutive 8-byte slots receives the covered [4 m][16 c] block
// column-per-lane).
//
// Layout: each operand tile is stored as 16-column PANELS —
// lds[panel][m][16] — so a fragment's two transpose reads cover rows
// (g*8..g*8+3) and (+4) of one panel contiguously (g = lane>>4 picks
// the fragment's 8-m chunk, matching the MFMA A/B operand maps).
//
// Split-M: grid.z blocks each own a contiguous M-range and atomicAdd
// their fp32 partial tile into a caller-zeroed fp32 dW workspace (the
// caller casts to bf16) — correct for any block placement, no
// cross-block ordering (cdna guide §6 Guideline 16: the accumulator
// alternative is acceptable here because the tile is tiny and each
// block adds once).

#include "common.h"
#include "launchers.h"

namespace {

using bf16x4 = __attribute__((ext_vector_type(4))) __bf16;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;
using f32x4 = __attribute__((ext_vector_type(4))) float;

constexpr int NT = 64;   // n-tile (dW rows)
constexpr int KT = 64;   // k-tile (dW cols)
constexpr int MT = 128;  // m rows staged per iteration
// one operand tile: 4 panels x [MT][16] bf16 = 16 KB; two operands,
// double-buffered: 64 KB LDS
constexpr int PANEL = MT * 16;          // elements per panel
constexpr int TILE = 4 * PANEL;         // elements per operand tile
constexpr int PASSES = MT * 64 / (256 * 8);  // 16-B moves per thread

// T14 split staging: LOAD (issue early, before the MFMA phase hides
// the HBM latency) and WRITE (after the compute, just before the
// barrier).  The full-tile path carries no per-load predicate (a
// runtime select around loads makes hipcc branch + drain per element
// — guide trap 4c); only the M-tail takes the predicated path.
template <int PASSES>
__device__ __forceinline__ void stage_load(
    const bf16* __restrict__ src, long row0, long rows, int ld, int c0,
    int tid, bf16x8* v) {
#pragma unroll
  for (int pass = 0; pass < PASSES; pass++) {
    int idx = tid + pass * 256;
    int m = idx >> 3;
    int half = idx & 7;
    int panel = half >> 1;
    int off8 = (half & 1) * 8;
    v[pass] = *(const bf16x8*)(src + (row0 + m) * (long)ld + c0 +
                               panel * 16 + off8);
  }
}

template <int PASSES>
__device__ __forceinline__ void stage_load_tail(
    const bf16* __restrict__ src, long row0, long rows, int ld, int c0,
    int tid, bf16x8* v) {
#pragma unroll
  for (int pass = 0; pass < PASSES; pass++) {
    int idx = tid + pass * 256;
    int m = idx >> 3;
    bf16x8 z = {};
    v[pass] = z;
    if (m < rows) {
      int half = idx & 7;
      int panel = half >> 1;
      int off8 = (half & 1) * 8;
      v[pass] = *(const bf16x8*)(src + (row0 + m) * (long)ld + c0 +
                                 panel * 16 + off8);
    }
  }
}

template <int PASSES>
__device__ __forceinline__ void stage_write(
    __attribute__((address_space(3))) bf16* dst, int tid, const bf16x8* v) {
#pragma unroll
  for (int pass = 0; pass < PASSES; pass++) {
    int idx = tid + pass * 256;
    int m = idx >> 3;
    int half = idx & 7;
    int panel = half >> 1;
    int off8 = (half & 1) * 8;
    *(__attribute__((address_space(3))) bf16x8*)
        (dst + panel * PANEL + m * 16 + off8) = v[pass];
  }
}

// fragment read: 8 m-contiguous bf16 for this lane's column, rows
// g*8..g*8+7 of panel p — two hardware-transpose reads of [4][16]
__device__ __forceinline__ bf16x8 frag_tr(
    const __attribute__((address_space(3))) bf16* tile, int panel,
    int mbase, int lane) {
  const __attribute__((address_space(3))) bf16* base =
      tile + panel * PANEL + mbase * 16 + (lane & 15) * 4;
  bf16x4 lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (__attribute__((address_space(3))) bf16x4*)base);
  bf16x4 hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (__attribute__((address_space(3))) bf16x4*)(base + 4 * 16));
  bf16x8 out;
#pragma unroll
  for (int i = 0; i < 4; i++) {
    out[i] = lo[i];
    out[4 + i] = hi[i];
  }
  return out;
}

__global__ __launch_bounds__(256)
void wgrad_kernel(const bf16* __restrict__ dY, const bf16* __restrict__ X,
                  float* __restrict__ dW, long M, int N, int K,
                  long m_per_block) {
  __shared__ bf16 smem[2 * 2 * TILE];  // [buf][dY|X][tile]
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wn = (wid >> 1) * 32;      // wave's n offset in the tile
  const int wk = (wid & 1) * 32;       // wave's k offset
  const int n0 = blockIdx.y * NT;
  const int k0 = blockIdx.x * KT;
  long row0 = (long)blockIdx.z * m_per_block;
  long row1 = row0 + m_per_block;
  if (row1 > M) row1 = M;
  if (row0 >= row1) return;

  auto dyt = [&](int buf) {
    return (__attribute__((address_space(3))) bf16*)smem + buf * 2 * TILE;
  };
  auto xt = [&](int buf) {
    return (__attribute__((address_space(3))) bf16*)smem + buf * 2 * TILE +
           TILE;
  };

  f32x4 acc[2][2] = {};
  const int g = lane >> 4;             // 8-m chunk selector

  bf16x8 vy[PASSES], vx[PASSES];
  auto load_chunk = [&](long r) {
    long rows = row1 - r;
    if (rows >= MT) {
      stage_load<PASSES>(dY, r, rows, N, n0, tid, vy);
      stage_load<PASSES>(X, r, rows, K, k0, tid, vx);
    } else {
      stage_load_tail<PASSES>(dY, r, rows, N, n0, tid, vy);
      stage_load_tail<PASSES>(X, r, rows, K, k0, tid, vx);
    }
  };

  int buf = 0;
  load_chunk(row0);
  stage_write<PASSES>(dyt(buf), tid, vy);
  stage_write<PASSES>(xt(buf), tid, vx);
  if (row0 + MT < row1) load_chunk(row0 + MT);  // issue next loads early
  __syncthreads();

  for (long r = row0; r < row1; r += MT) {
    // MFMA steps over the resident tile; the NEXT chunk's global
    // loads are already in flight (issued before the barrier)
#pragma unroll
    for (int ms = 0; ms < MT / 32; ms++) {
      int mbase = ms * 32 + g * 8;
      bf16x8 a[2], b[2];
#pragma unroll
      for (int i = 0; i < 2; i++) {
        a[i] = frag_tr(dyt(buf), (wn >> 4) + i, mbase, lane);  // dY^T
        b[i] = frag_tr(xt(buf), (wk >> 4) + i, mbase, lane);   // X
      }
#pragma unroll
      for (int i = 0; i < 2; i++)
#pragma unroll
        for (int j = 0; j < 2; j++)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a[i], b[j], acc[i][j], 0, 0, 0);
    }
    long rn = r + MT;
    if (rn < row1) {
      __syncthreads();  // everyone done reading buf^1's OLD content
      stage_write<PASSES>(dyt(buf ^ 1), tid, vy);
      stage_write<PASSES>(xt(buf ^ 1), tid, vx);
      if (rn + MT < row1) load_chunk(rn + MT);  // re-issue immediately
      __syncthreads();
      buf ^= 1;
    }
  }

  // C-map: col = lane&15 (k), row = (lane>>4)*4 + r (n)
#pragma unroll
  for (int i = 0; i < 2; i++) {
#pragma unroll
    for (int j = 0; j < 2; j++) {
      int col = k0 + wk + j * 16 + (lane & 15);
#pragma unroll
      for (int r = 0; r < 4; r++) {
        int row = n0 + wn + i * 16 + (lane >> 4) * 4 + r;
        atomicAdd(&dW[(long)row * K + col], acc[i][j][r]);
      }
    }
  }
}

}  // namespace

bool wgrad_supported(long M, int N, int K) {
  return N % NT == 0 && K % KT == 0;
}

void launch_conv1x1_wgrad(const rbf16* dY, const rbf16* X, float* dW,
                          long M, int N, int K, hipStream_t st) {
  int nt = N / NT, kt = K / KT;
  // fill the chip: ~2 blocks per CU across the split dimension
  long want = (512 + nt * kt - 1) / (nt * kt);
  long chunks = (M + MT - 1) / MT;
  long split = want < 1 ? 1 : (want > chunks ? chunks : want);
  long m_per_block = ((chunks + split - 1) / split) * MT;
  split = (M + m_per_block - 1) / m_per_block;
  dim3 grid((unsigned)kt, (unsigned)nt, (unsigned)split);
  hipLaunchKernelGGL(wgrad_kernel, grid, dim3(256), 0, st,
                     reinterpret_cast<const bf16*>(dY),
                     reinterpret_cast<const bf16*>(X), dW, M, N, K,
                     m_per_block);
}
