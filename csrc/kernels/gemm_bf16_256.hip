// 256x256 8-phase MFMA bf16 GEMM — the deep-pipelined upgrade of
// gemm_bf16.hip (capability C10, ROADMAP round-2 §Perf 2).
//
// Same contract as the 128² kernel: D[M,N] = A[M,K] · W[N,K]^T (+bias)
// (+GELU) in bf16, fp32 accumulation.  Structure per the CDNA4 guide's
// 256² 8-phase template (cdna_hip_programming.md §5): one workgroup of
// 8 waves (2 M x 4 N) owns a 256x256 output tile; each wave computes a
// 128x64 sub-tile as 8x4 fragments of v_mfma_f32_16x16x32_bf16.
//
// Pipeline: the K loop runs in phases of {issue 4-or-8 ds_read_b128 ||
// stage one 16 KB half-tile with 2 global_load_lds_dwordx4 || counted
// s_waitcnt vmcnt(6) || raw s_barrier || 16 MFMA wrapped
// in s_setprio(1)}.  A half-tile staged at phase p is guaranteed by the
// vmcnt at p+3 and read at p+4, so 3-4 halves (6-8 glds) stay in
// flight ACROSS the barriers — the counted-vmcnt pipelining that the
// 128² kernel's per-tile vmcnt(0) drain cannot do.  LDS is one 128 KiB
// array (2 K-tile buffers x 4 halves x 16 KB); the XOR swizzle rides
// on the per-lane glds SOURCE address (lane-linear LDS image), same as
// the 128² kernel.
//
// Half-tiles are CONSUMPTION-shaped, not row-order-shaped: e.g. the
// "A-q0" half holds global rows {0..63, 128..191} — each wave-row's
// first 4 M-fragments — so every wave's phase-p ds_reads hit the same
// half regardless of its wave id, keeping the per-wave vmcnt guarantee
// uniform across the workgroup.
//
// Shape contract: M%256 == 0, N%256 == 0, K%128 == 0 (an even number
// of BK=64 K-tiles; the pair-unrolled loop gives each tile a static
// LDS buffer parity).  The Python wrapper falls back to the 128²
// kernel / rocBLAS outside it.

#include "common.h"
#include "launchers.h"

namespace {

using f32x4 = __attribute__((ext_vector_type(4))) float;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;

constexpr int BM = 256, BN = 256, BK = 64;
constexpr int HALF = 16384;        // one half-tile: 128 rows x 64 k x 2 B
constexpr int BUF = 4 * HALF;      // A-q0 | B-first | B-second | A-q1
// half indices within a buffer
constexpr int H_A0 = 0, H_B0 = 1, H_B1 = 2, H_A1 = 3;

__device__ __forceinline__ float gelu_tanh(float x) {
  const float c = 0.7978845608028654f;  // sqrt(2/pi)
  float t = tanhf(c * (x + 0.044715f * x * x * x));
  return 0.5f * x * (1.f + t);
}

// packed row -> global tile row for each half's row set
__device__ __forceinline__ int map_row(int h, int pr) {
  switch (h) {
    case H_A0: return pr + (pr & 64);         // {0..63, 128..191}
    case H_A1: return pr + 64 + (pr & 64);    // {64..127, 192..255}
    case H_B0: return ((pr >> 5) << 6) + (pr & 31);        // wc*64+[0,32)
    default:   return ((pr >> 5) << 6) + 32 + (pr & 31);   // wc*64+[32,64)
  }
}

template <int EPI, bool INPH = false, int SWZ = 0>
// EPI: 0 none, 1 +bias, 2 +bias+gelu, 3 = 2 + store pre.
// INPH: phase reads its OWN MFMA operands (the guide template's
// literal shape — reads gated by the phase's barrier + the compiler's
// counted lgkm waits) instead of reading one phase ahead.
// SWZ: LDS bank swizzle — 0: byte ^= (pr&7)<<4 (spread over 8 16-B
// slots); 1: the guide's st_16x32 (byte bit5 ^= pr bit2, 4-way
// spread, and the kk step stays a plain +64 immediate).
__global__ __launch_bounds__(512)
void gemm256_kernel(const bf16* __restrict__ A, const bf16* __restrict__ W,
                    const float* __restrict__ bias, bf16* __restrict__ D,
                    bf16* __restrict__ P, int M, int N, int K) {
  __shared__ char smem[2 * BUF];  // 128 KiB — ONE shared object (trap 4a)
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid >> 2;   // wave row (2)
  const int wc = wid & 3;    // wave col (4)

  // ---- bijective XCD-aware block remap (T1) -------------------------
  const int mt = M / BM, ntl = N / BN;
  const int nwg = mt * ntl;
  int bid = blockIdx.x;
  {
    int q = nwg >> 3, r = nwg & 7, xcd = bid & 7, pos = bid >> 3;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
  }
  // consecutive remapped ids share the B (weight) panel
  const int m0 = (bid % mt) * BM;
  const int n0 = (bid / mt) * BN;

  const int nkt = K / BK;

  // ---- staging: one half-tile = 2 glds/thread, lane-linear LDS ------
  // source row/byte from the packed offset, XOR swizzle on the SOURCE
  // byte (rule 21) so ds_read_b128 is bank-conflict-free.
  auto stage = [&](int buf, int h, int kt) {
    if (kt >= nkt) kt = nkt - 1;  // tail: stage valid memory, never read
    const bf16* base = (h == H_A0 || h == H_A1) ? A + (long)m0 * K
                                                : W + (long)n0 * K;
    char* lds = smem + buf * BUF + h * HALF;
#pragma unroll
    for (int i = 0; i < 2; i++) {
      int off = i * 8192 + tid * 16;
      int pr = off >> 7, b = off & 127;
      int sb = SWZ ? (b ^ ((pr & 4) << 3)) : (b ^ ((pr & 7) << 4));
      const bf16* src = base + (long)map_row(h, pr) * K + kt * BK + (sb >> 1);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)src,
          (__attribute__((address_space(3))) unsigned int*)(lds + off), 16,
          0, 0);
    }
  };

  // ---- fragment reads (ds_read_b128) --------------------------------
  // Per phase ONE base address VGPR: fragment steps (f*2048) are
  // instruction immediates, and the kk=1 address is the kk=0 address
  // with byte-bit 6 flipped (kk*64 lives in bit 6, the XOR swizzle in
  // bits 4-6, the row field above bit 7 — no carries), so it costs one
  // v_xor instead of a second live-range (a spilled address VGPR here
  // forces a compiler vmcnt(0) that drains the whole glds pipeline).
  const int xorv = SWZ ? ((lane & 4) << 3) : ((lane & 7) << 4);
  const int rdb = (((lane >> 4) << 4) ^ xorv);
  // a-frags: pr = wr*64 + f*16 + (lane&15)   (f = 0..3)
  // b-frags: pr = wc*32 + j*16 + (lane&15)   (j = 0..1 within the half)
  const int apr = wr * 64 + (lane & 15);
  const int bpr = wc * 32 + (lane & 15);
  const int abase = apr * 128 + rdb;  // + buf*BUF + h*HALF + f*2048 (^64 kk)
  const int bbase = bpr * 128 + rdb;

  bf16x8 a0[4][2], a1[4][2], b0[2][2], b1[2][2];
  f32x4 acc[8][4] = {};

#define VMCNT(n) asm volatile("s_waitcnt vmcnt(" #n ")" ::: "memory")
#define BAR() __builtin_amdgcn_s_barrier()

  if constexpr (INPH) {
    // ---- in-phase schedule -----------------------------------------
    // phase X.P0: read a0(X)+b0(X) [12], MFMA q00, stage A0(X+1)
    //       X.P1: read b1(X)       [4],  MFMA q01, stage B0(X+1)
    //       X.P2: read a1(X)       [8],  MFMA q10, stage B1(X+1)
    //       X.P3: (cached)         [0],  MFMA q11, stage A1(X+1)
    // vmcnt(4) at P0/P1/P3 guarantees the halves the NEXT phase reads
    // (staged 3-4 phases earlier); reads are covered by the previous
    // phase's vmcnt + barrier, and hipcc's counted lgkm waits order
    // each phase's reads before its own MFMAs.
#define RD_A(dst, buf, h)                                                  {                                                                          const char* p0 = smem + (buf) * BUF + (h) * HALF + abase;                const char* p1 = smem + (((buf) * BUF + (h) * HALF + abase) ^ 64);       _Pragma("unroll") for (int f = 0; f < 4; f++) {                            dst[f][0] = *(const bf16x8*)(p0 + f * 2048);                             dst[f][1] = *(const bf16x8*)(p1 + f * 2048);                           }                                                                      }
#define RD_B(dst, buf, h)                                                  {                                                                          const char* p0 = smem + (buf) * BUF + (h) * HALF + bbase;                const char* p1 = smem + (((buf) * BUF + (h) * HALF + bbase) ^ 64);       _Pragma("unroll") for (int f = 0; f < 2; f++) {                            dst[f][0] = *(const bf16x8*)(p0 + f * 2048);                             dst[f][1] = *(const bf16x8*)(p1 + f * 2048);                           }                                                                      }
#define MM16(am, bm, mb, nb)                                               __builtin_amdgcn_s_setprio(1);                                           _Pragma("unroll") for (int i = 0; i < 4; i++)                                _Pragma("unroll") for (int j = 0; j < 2; j++)                                _Pragma("unroll") for (int kk = 0; kk < 2; kk++)                             acc[(mb) + i][(nb) + j] =                                                    __builtin_amdgcn_mfma_f32_16x16x32_bf16(                                     am[i][kk], bm[j][kk], acc[(mb) + i][(nb) + j],                           0, 0, 0);                                            __builtin_amdgcn_s_setprio(0)
#define VM4 asm volatile("s_waitcnt vmcnt(4)" ::: "memory")
    stage(0, H_A0, 0); stage(0, H_B0, 0); stage(0, H_B1, 0);
    stage(0, H_A1, 0);
    VM4;  // A0(0), B0(0) landed
    BAR();
    for (int t = 0; t < nkt; t += 2) {
      // tile t (buffer 0)
      RD_A(a0, 0, H_A0); RD_B(b0, 0, H_B0);
      stage(1, H_A0, t + 1); VM4; MM16(a0, b0, 0, 0); BAR();  // P0 q00
      RD_B(b1, 0, H_B1);
      stage(1, H_B0, t + 1); VM4; MM16(a0, b1, 0, 2); BAR();  // P1 q01
      RD_A(a1, 0, H_A1);
      stage(1, H_B1, t + 1);      MM16(a1, b0, 4, 0); BAR();  // P2 q10
      stage(1, H_A1, t + 1); VM4; MM16(a1, b1, 4, 2); BAR();  // P3 q11
      // tile t+1 (buffer 1)
      RD_A(a0, 1, H_A0); RD_B(b0, 1, H_B0);
      stage(0, H_A0, t + 2); VM4; MM16(a0, b0, 0, 0); BAR();
      RD_B(b1, 1, H_B1);
      stage(0, H_B0, t + 2); VM4; MM16(a0, b1, 0, 2); BAR();
      RD_A(a1, 1, H_A1);
      stage(0, H_B1, t + 2);      MM16(a1, b0, 4, 0); BAR();
      stage(0, H_A1, t + 2); VM4; MM16(a1, b1, 4, 2); BAR();
    }
#undef RD_A
#undef RD_B
#undef MM16
#undef VM4
  } else {
  // ---- prologue: 6 halves (tiles 0 and 1), first reads --------------
  stage(0, H_A0, 0); stage(0, H_B0, 0); stage(0, H_B1, 0); stage(0, H_A1, 0);
  stage(1, H_A0, 1); stage(1, H_B0, 1);
  VMCNT(8);   // A0(0), B0(0) landed
  BAR();
#pragma unroll
  for (int f = 0; f < 4; f++) {
    a0[f][0] = *(const bf16x8*)(smem + H_A0 * HALF + abase + f * 2048);
    a0[f][1] = *(const bf16x8*)(smem + ((H_A0 * HALF + abase) ^ 64)
                                + f * 2048);
  }
#pragma unroll
  for (int j = 0; j < 2; j++) {
    b0[j][0] = *(const bf16x8*)(smem + H_B0 * HALF + bbase + j * 2048);
    b0[j][1] = *(const bf16x8*)(smem + ((H_B0 * HALF + bbase) ^ 64)
                                + j * 2048);
  }
  VMCNT(6);   // B1(0) landed
  BAR();

  // ---- main loop: 2 K-tiles (8 phases) per iteration ----------------
  // phase X.P0: read b1(X)   [h B1, buf X]   stage B1(X+1) -> buf X+1
  //       X.P1: read a1(X)   [h A1, buf X]   stage A1(X+1) -> buf X+1
  //       X.P2: read a0(X+1) [h A0, buf X+1] stage A0(X+2) -> buf X
  //       X.P3: read b0(X+1) [h B0, buf X+1] stage B0(X+2) -> buf X
  // each phase: vmcnt(6) guarantees the half read NEXT phase (staged 4
  // phases earlier) across all waves once the barrier is passed.
  // Macro, not a lambda: pointer-passing the fragment arrays would
  // spill them to scratch; every index below is a compile-time
  // constant after unrolling.
#define READ_A(dst, buf, h)                                              \
  {                                                                      \
    const char* p0 = smem + (buf) * BUF + (h) * HALF + abase;            \
    const char* p1 = smem + (((buf) * BUF + (h) * HALF + abase) ^ 64);   \
    _Pragma("unroll") for (int f = 0; f < 4; f++) {                      \
      dst[f][0] = *(const bf16x8*)(p0 + f * 2048);                       \
      dst[f][1] = *(const bf16x8*)(p1 + f * 2048);                       \
    }                                                                    \
  }
#define READ_B(dst, buf, h)                                              \
  {                                                                      \
    const char* p0 = smem + (buf) * BUF + (h) * HALF + bbase;            \
    const char* p1 = smem + (((buf) * BUF + (h) * HALF + bbase) ^ 64);   \
    _Pragma("unroll") for (int f = 0; f < 2; f++) {                      \
      dst[f][0] = *(const bf16x8*)(p0 + f * 2048);                       \
      dst[f][1] = *(const bf16x8*)(p1 + f * 2048);                       \
    }                                                                    \
  }
#define MFMA16(am, bm, mbase, nbase)                                    \
  __builtin_amdgcn_s_setprio(1);                                        \
  _Pragma("unroll") for (int i = 0; i < 4; i++)                         \
      _Pragma("unroll") for (int j = 0; j < 2; j++)                     \
          _Pragma("unroll") for (int kk = 0; kk < 2; kk++)              \
              acc[(mbase) + i][(nbase) + j] =                           \
                  __builtin_amdgcn_mfma_f32_16x16x32_bf16(              \
                      am[i][kk], bm[j][kk], acc[(mbase) + i][(nbase) + j], \
                      0, 0, 0);                                         \
  __builtin_amdgcn_s_setprio(0)
// No explicit lgkmcnt here: each phase's ds_reads feed the NEXT
// phase's MFMAs, and hipcc already inserts counted lgkmcnt waits for
// its own ds_read->MFMA data deps — an explicit lgkmcnt(0) would drain
// the just-issued reads instead of hiding them under this phase's MFMA.
#define PHASE_TAIL() \
  VMCNT(6);          \
  BAR()

  for (int t = 0; t < nkt; t += 2) {
    // tile t (buffer 0)
    READ_B(b1, 0, H_B1); stage(1, H_B1, t + 1); PHASE_TAIL();  // P0
    MFMA16(a0, b0, 0, 0); BAR();                               //   q00
    READ_A(a1, 0, H_A1); stage(1, H_A1, t + 1); PHASE_TAIL();  // P1
    MFMA16(a0, b1, 0, 2); BAR();                               //   q01
    READ_A(a0, 1, H_A0); stage(0, H_A0, t + 2); PHASE_TAIL();  // P2
    MFMA16(a1, b0, 4, 0); BAR();                               //   q10
    READ_B(b0, 1, H_B0); stage(0, H_B0, t + 2); PHASE_TAIL();  // P3
    MFMA16(a1, b1, 4, 2); BAR();                               //   q11
    // tile t+1 (buffer 1)
    READ_B(b1, 1, H_B1); stage(0, H_B1, t + 2); PHASE_TAIL();
    MFMA16(a0, b0, 0, 0); BAR();
    READ_A(a1, 1, H_A1); stage(0, H_A1, t + 2); PHASE_TAIL();
    MFMA16(a0, b1, 0, 2); BAR();
    READ_A(a0, 0, H_A0); stage(1, H_A0, t + 3); PHASE_TAIL();
    MFMA16(a1, b0, 4, 0); BAR();
    READ_B(b0, 0, H_B0); stage(1, H_B0, t + 3); PHASE_TAIL();
    MFMA16(a1, b1, 4, 2); BAR();
  }
#undef READ_A
#undef READ_B
#undef MFMA16
#undef PHASE_TAIL
  }
#undef VMCNT
#undef BAR

  // ---- epilogue: C/D map col = lane&15, row = (lane>>4)*4 + r -------
#pragma unroll
  for (int i = 0; i < 8; i++) {
#pragma unroll
    for (int j = 0; j < 4; j++) {
      int col = n0 + wc * 64 + j * 16 + (lane & 15);
      float bv = (EPI >= 1) ? bias[col] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; r++) {
        int row = m0 + wr * 128 + i * 16 + ((lane >> 4) << 2) + r;
        float v = acc[i][j][r] + bv;
        if (EPI == 3) P[(long)row * N + col] = from_f32<bf16>(v);
        if (EPI >= 2) v = gelu_tanh(v);
        D[(long)row * N + col] = from_f32<bf16>(v);
      }
    }
  }
}

}  // namespace

bool gemm256_supported(int M, int N, int K) {
  return M % 256 == 0 && N % 256 == 0 && K % 128 == 0;
}

void launch_gemm_bf16_256(const rbf16* A, const rbf16* W, const float* bias,
                          rbf16* D, rbf16* PRE, int M, int N, int K,
                          int epilogue, hipStream_t st, int sched) {
  dim3 grid((M / BM) * (N / BN));
  const bf16* a = reinterpret_cast<const bf16*>(A);
  const bf16* w = reinterpret_cast<const bf16*>(W);
  bf16* d = reinterpret_cast<bf16*>(D);
  bf16* p = reinterpret_cast<bf16*>(PRE);
  if (sched == 1) {  // legacy look-ahead schedule (A/B benching)
    hipLaunchKernelGGL((gemm256_kernel<0, false>), grid, dim3(512), 0, st,
                       a, w, bias, d, nullptr, M, N, K);
    return;
  }
  if (sched == 2) {  // in-phase + st_16x32 swizzle (A/B benching)
    hipLaunchKernelGGL((gemm256_kernel<0, true, 1>), grid, dim3(512), 0, st,
                       a, w, bias, d, nullptr, M, N, K);
    return;
  }
  // in-phase schedule is the production 256-tile pipeline (measured
  // +9% over the look-ahead schedule on every dispatched shape)
  if (epilogue == 3)
    hipLaunchKernelGGL((gemm256_kernel<3, true>), grid, dim3(512), 0, st, a,
                       w, bias, d, p, M, N, K);
  else if (epilogue == 2)
    hipLaunchKernelGGL((gemm256_kernel<2, true>), grid, dim3(512), 0, st, a,
                       w, bias, d, nullptr, M, N, K);
  else if (epilogue == 1)
    hipLaunchKernelGGL((gemm256_kernel<1, true>), grid, dim3(512), 0, st, a,
                       w, bias, d, nullptr, M, N, K);
  else
    hipLaunchKernelGGL((gemm256_kernel<0, true>), grid, dim3(512), 0, st, a,
                       w, bias, d, nullptr, M, N, K);
}
