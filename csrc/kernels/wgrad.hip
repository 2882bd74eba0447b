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

// Tile geometry is templated: the 64x64 tile (4 panels each op, 128 m
// rows) wins when N or K is 64; the 128x128 tile (8 panels, 64 m rows
// — same 64 KB LDS) halves the cross-tile re-read traffic when both
// dims are >=128.  Waves are fixed 2x2; per-wave fragments = NP/2 x
// KP/2 of 16x16.

// T14 split staging: LOAD (issue early, before the MFMA phase hides
// the HBM latency) and WRITE (after the compute, just before the
// barrier).  The full-tile path carries no per-load predicate (a
// runtime select around loads makes hipcc branch + drain per element
// — guide trap 4c); only the M-tail takes the predicated path.
// NC = panels-per-tile (columns/16); index decomposition: each 16-B
// move covers half a panel row.
template <int PASSES, int NC>
__device__ __forceinline__ void stage_load(
    const bf16* __restrict__ src, long row0, long rows, int ld, int c0,
    int tid, bf16x8* v) {
#pragma unroll
  for (int pass = 0; pass < PASSES; pass++) {
    int idx = tid + pass * 256;
    int m = idx / (NC * 2);
    int half = idx % (NC * 2);
    int panel = half >> 1;
    int off8 = (half & 1) * 8;
    v[pass] = *(const bf16x8*)(src + (row0 + m) * (long)ld + c0 +
                               panel * 16 + off8);
  }
}

template <int PASSES, int NC>
__device__ __forceinline__ void stage_load_tail(
    const bf16* __restrict__ src, long row0, long rows, int ld, int c0,
    int tid, bf16x8* v) {
#pragma unroll
  for (int pass = 0; pass < PASSES; pass++) {
    int idx = tid + pass * 256;
    int m = idx / (NC * 2);
    bf16x8 z = {};
    v[pass] = z;
    if (m < rows) {
      int half = idx % (NC * 2);
      int panel = half >> 1;
      int off8 = (half & 1) * 8;
      v[pass] = *(const bf16x8*)(src + (row0 + m) * (long)ld + c0 +
                                 panel * 16 + off8);
    }
  }
}

template <int PASSES, int NC, int PANEL>
__device__ __forceinline__ void stage_write(
    __attribute__((address_space(3))) bf16* dst, int tid, const bf16x8* v) {
#pragma unroll
  for (int pass = 0; pass < PASSES; pass++) {
    int idx = tid + pass * 256;
    int m = idx / (NC * 2);
    int half = idx % (NC * 2);
    int panel = half >> 1;
    int off8 = (half & 1) * 8;
    *(__attribute__((address_space(3))) bf16x8*)
        (dst + panel * PANEL + m * 16 + off8) = v[pass];
  }
}

// fragment read: 8 m-contiguous bf16 for this lane's column, rows
// g*8..g*8+7 of panel p — two hardware-transpose reads of [4][16]
template <int PANEL>
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

// NP/KP: 16-col panels per dY/X tile (tile dims = NP*16 x KP*16);
// MT_: m rows staged per iteration.  64x64 tile: NP=KP=4, MT_=128;
// 128x128: NP=KP=8, MT_=64 (both 64 KB LDS double-buffered).
template <int NP, int KP, int MT_>
__global__ __launch_bounds__(256)
void wgrad_kernel(const bf16* __restrict__ dY, const bf16* __restrict__ X,
                  float* __restrict__ dW, long M, int N, int K,
                  long m_per_block, int kt, int nt) {
  constexpr int PANEL = MT_ * 16;
  constexpr int TILEY = NP * PANEL;
  constexpr int TILEX = KP * PANEL;
  constexpr int PY = NP * 16 * MT_ / (256 * 8);
  constexpr int PX = KP * 16 * MT_ / (256 * 8);
  constexpr int FN = NP / 2;           // per-wave n fragments
  constexpr int FK = KP / 2;           // per-wave k fragments
  __shared__ bf16 smem[2 * (TILEY + TILEX)];  // [buf][dY|X]
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wn = (wid >> 1) * (NP * 8);  // wave's n offset in the tile
  const int wk = (wid & 1) * (KP * 8);   // wave's k offset
  // 1-D grid with a bijective XCD remap (T1): an XCD receives
  // contiguous runs of (n-tile, m-split) groups, so the k-tiles that
  // re-read the SAME dY slice land on the same L2 instead of being
  // round-robined across all eight.
  int bid = blockIdx.x;
  {
    int nwg = (int)gridDim.x;
    int q = nwg >> 3, rm = nwg & 7, xcd = bid & 7, pos = bid >> 3;
    bid = (xcd < rm ? xcd * (q + 1) : rm * (q + 1) + (xcd - rm) * q) + pos;
  }
  const int k0 = (bid % kt) * KP * 16;
  const int n0 = ((bid / kt) % nt) * NP * 16;
  const int bz = bid / (kt * nt);
  long row0 = (long)bz * m_per_block;
  long row1 = row0 + m_per_block;
  if (row1 > M) row1 = M;
  if (row0 >= row1) return;

  auto dyt = [&](int buf) {
    return (__attribute__((address_space(3))) bf16*)smem +
           buf * (TILEY + TILEX);
  };
  auto xt = [&](int buf) {
    return (__attribute__((address_space(3))) bf16*)smem +
           buf * (TILEY + TILEX) + TILEY;
  };

  f32x4 acc[FN][FK] = {};
  const int g = lane >> 4;             // 8-m chunk selector

  bf16x8 vy[PY], vx[PX];
  auto load_chunk = [&](long r) {
    long rows = row1 - r;
    if (rows >= MT_) {
      stage_load<PY, NP>(dY, r, rows, N, n0, tid, vy);
      stage_load<PX, KP>(X, r, rows, K, k0, tid, vx);
    } else {
      stage_load_tail<PY, NP>(dY, r, rows, N, n0, tid, vy);
      stage_load_tail<PX, KP>(X, r, rows, K, k0, tid, vx);
    }
  };

  int buf = 0;
  load_chunk(row0);
  stage_write<PY, NP, PANEL>(dyt(buf), tid, vy);
  stage_write<PX, KP, PANEL>(xt(buf), tid, vx);
  if (row0 + MT_ < row1) load_chunk(row0 + MT_);  // issue next loads early
  __syncthreads();

  for (long r = row0; r < row1; r += MT_) {
    // MFMA steps over the resident tile; the NEXT chunk's global
    // loads are already in flight (issued before the barrier)
#pragma unroll
    for (int ms = 0; ms < MT_ / 32; ms++) {
      int mbase = ms * 32 + g * 8;
      bf16x8 a[FN], b[FK];
#pragma unroll
      for (int i = 0; i < FN; i++)
        a[i] = frag_tr<PANEL>(dyt(buf), (wn >> 4) + i, mbase, lane);  // dY^T
#pragma unroll
      for (int j = 0; j < FK; j++)
        b[j] = frag_tr<PANEL>(xt(buf), (wk >> 4) + j, mbase, lane);   // X
#pragma unroll
      for (int i = 0; i < FN; i++)
#pragma unroll
        for (int j = 0; j < FK; j++)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a[i], b[j], acc[i][j], 0, 0, 0);
    }
    long rn = r + MT_;
    if (rn < row1) {
      __syncthreads();  // everyone done reading buf^1's OLD content
      stage_write<PY, NP, PANEL>(dyt(buf ^ 1), tid, vy);
      stage_write<PX, KP, PANEL>(xt(buf ^ 1), tid, vx);
      if (rn + MT_ < row1) load_chunk(rn + MT_);  // re-issue immediately
      __syncthreads();
      buf ^= 1;
    }
  }

  // C-map: col = lane&15 (k), row = (lane>>4)*4 + r (n)
#pragma unroll
  for (int i = 0; i < FN; i++) {
#pragma unroll
    for (int j = 0; j < FK; j++) {
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
  return N % 64 == 0 && K % 64 == 0;
}

void launch_conv1x1_wgrad(const rbf16* dY, const rbf16* X, float* dW,
                          long M, int N, int K, hipStream_t st) {
  // 128x128 tile halves cross-tile re-reads when both dims allow it
  const bool big = (N % 128 == 0) && (K % 128 == 0);
  const int tile = big ? 128 : 64;
  const int mt = big ? 64 : 128;
  int nt = N / tile, kt = K / tile;
  // fill the chip: ~2 blocks per CU across the split dimension
  long want = (512 + nt * kt - 1) / (nt * kt);
  long chunks = (M + mt - 1) / mt;
  long split = want < 1 ? 1 : (want > chunks ? chunks : want);
  long m_per_block = ((chunks + split - 1) / split) * mt;
  split = (M + m_per_block - 1) / m_per_block;
  dim3 grid((unsigned)(kt * nt * split));
  if (big)
    hipLaunchKernelGGL((wgrad_kernel<8, 8, 64>), grid, dim3(256), 0, st,
                       reinterpret_cast<const bf16*>(dY),
                       reinterpret_cast<const bf16*>(X), dW, M, N, K,
                       m_per_block, kt, nt);
  else
    hipLaunchKernelGGL((wgrad_kernel<4, 4, 128>), grid, dim3(256), 0, st,
                       reinterpret_cast<const bf16*>(dY),
                       reinterpret_cast<const bf16*>(X), dW, M, N, K,
                       m_per_block, kt, nt);
}
