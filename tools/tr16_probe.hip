// Empirical semantics probe for gfx950 ds_read_b64_tr_b16.
// LDS filled with identity pattern lds[i] = i; each lane reads 8 bytes at
// lane*8 (and a second test at a row-strided address); the returned 4 u16
// values reveal the (lane, reg) -> lds-index transpose mapping.
#include <hip/hip_runtime.h>
#include <stdio.h>
typedef __attribute__((ext_vector_type(4))) short s16x4;
typedef __attribute__((address_space(3))) s16x4* las4;
extern "C" __global__ void trprobe(short* out, int mode) {
  __shared__ short lds[4096];
  for (int i = threadIdx.x; i < 4096; i += blockDim.x) lds[i] = (short)i;
  __syncthreads();
  const int l = threadIdx.x;
  int addr;
  if (mode == 0) addr = l * 4;            // contiguous 8B per lane
  else if (mode == 1) addr = l * 16;      // 32B stride
  else addr = (l & 15) * 4 + (l >> 4) * 1024;  // 16-lane groups
  s16x4 v = __builtin_amdgcn_ds_read_tr16_b64_v4i16((las4)&lds[addr]);
  for (int j = 0; j < 4; ++j) out[l * 4 + j] = v[j];
}
int main() {
  short* out;
  hipMalloc(&out, 64 * 4 * sizeof(short));
  short h[256];
  for (int mode = 0; mode < 3; ++mode) {
    hipLaunchKernelGGL(trprobe, dim3(1), dim3(64), 0, 0, out, mode);
    hipMemcpy(h, out, sizeof(h), hipMemcpyDeviceToHost);
    printf("mode %d:\n", mode);
    for (int l = 0; l < 64; ++l) {
      printf("lane %2d: %5d %5d %5d %5d\n", l, h[l*4], h[l*4+1], h[l*4+2], h[l*4+3]);
    }
  }
  return 0;
}
