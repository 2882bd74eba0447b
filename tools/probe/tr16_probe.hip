// Empirical semantics probe for ds_read_b64_tr_b16 on gfx950 (the
// cdna4 ISA doc is not shipped here): fill LDS with element-index
// markers, tr-read at a per-lane address, dump each lane's two dwords
// so the host can reconstruct lane/element -> register mapping.
#include <hip/hip_runtime.h>
#include <cstdio>

typedef __attribute__((ext_vector_type(2))) unsigned int u32x2;

__global__ void tr_probe(const unsigned short* __restrict__ pattern,
                         unsigned int* __restrict__ out, int addr_mode) {
  __shared__ unsigned short lds[4096];
  int tid = threadIdx.x;
  for (int i = tid; i < 4096; i += blockDim.x) lds[i] = pattern[i];
  __syncthreads();
  int lane = tid & 63;
  unsigned int byte_addr;
  switch (addr_mode) {
    // mode 0: each lane at its own 8-byte slot, linear
    case 0: byte_addr = lane * 8; break;
    // mode 1: the guide's V-subtile recipe lds[(l&15) + j*16 + (l>>4)*64]
    // (elements, j=0) -> bytes
    case 1: byte_addr = ((lane & 15) + ((lane >> 4) * 64)) * 2; break;
    // mode 2: like 1 with j=1 (next 16-element group)
    case 2: byte_addr = ((lane & 15) + 16 + ((lane >> 4) * 64)) * 2; break;
    default: byte_addr = lane * 8; break;
  }
  typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;
  bf16x4 d = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (__attribute__((address_space(3))) bf16x4*)
          ((__attribute__((address_space(3))) char*)lds + byte_addr));
  union { bf16x4 v; unsigned int u[2]; } cv;
  cv.v = d;
  out[lane * 2] = cv.u[0];
  out[lane * 2 + 1] = cv.u[1];
}

#define CK(x) do { hipError_t e = (x); if (e != hipSuccess) { \
    printf("HIP error %s at %d\n", hipGetErrorString(e), __LINE__); \
    return 1; } } while (0)

int main() {
  unsigned short hpat[4096];
  unsigned int hout[128];
  unsigned short* pat;
  unsigned int* out;
  CK(hipMalloc(&pat, sizeof(hpat)));
  CK(hipMalloc(&out, sizeof(hout)));
  for (int i = 0; i < 4096; i++) hpat[i] = (unsigned short)i;
  CK(hipMemcpy(pat, hpat, sizeof(hpat), hipMemcpyHostToDevice));
  for (int mode = 0; mode < 3; mode++) {
    CK(hipMemset(out, 0, sizeof(hout)));
    hipLaunchKernelGGL(tr_probe, dim3(1), dim3(64), 0, 0, pat, out, mode);
    CK(hipGetLastError());
    CK(hipDeviceSynchronize());
    CK(hipMemcpy(hout, out, sizeof(hout), hipMemcpyDeviceToHost));
    printf("== mode %d: lane -> [elem ids of reg pair (4x bf16)]\n", mode);
    for (int l = 0; l < 64; l++) {
      unsigned int a = hout[l * 2], b = hout[l * 2 + 1];
      printf("lane %2d: %4u %4u %4u %4u\n", l, a & 0xffff, a >> 16,
             b & 0xffff, b >> 16);
    }
  }
  hipFree(pat); hipFree(out);
  return 0;
}
