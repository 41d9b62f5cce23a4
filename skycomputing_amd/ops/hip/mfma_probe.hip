// MFMA fragment-layout probe for gfx950: computes one 16x16 tile
// D = A(16x32) @ B(32x16) with v_mfma_f32_16x16x32_bf16 using the assumed
// lane->element mappings, so the GPU test can verify them against torch.
//
// Assumed mappings (verified by tests/test_ops_gpu.py::test_mfma_layout):
//   A[i][k]: lane l holds i = l%16, k = 8*(l/16) + j  (j = 0..7)
//   B[k][n]: lane l holds n = l%16, k = 8*(l/16) + j
//   D[i][n]: lane l holds n = l%16, i = 4*(l/16) + r  (r = 0..3)

#include "common.h"

typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

__global__ __launch_bounds__(64) void mfma_probe_kernel(
    const ushort_t* __restrict__ A, const ushort_t* __restrict__ B,
    float* __restrict__ D) {
  const int l = threadIdx.x;
  bf16x8 a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    int i = l % 16, k = 8 * (l / 16) + j;
    a[j] = (short)A[i * 32 + k];
    int n = l % 16;
    b[j] = (short)B[k * 16 + n];
  }
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int i = 4 * (l / 16) + r, n = l % 16;
    D[i * 16 + n] = acc[r];
  }
}

SKY_EXPORT int sky_mfma_probe(uint64_t stream, uint64_t A, uint64_t B,
                              uint64_t D) {
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0,
                     (hipStream_t)stream, (const ushort_t*)A,
                     (const ushort_t*)B, (float*)D);
  LAUNCH_CHECK();
  return 0;
}

// glds semantics probe: copy 4 KB global -> LDS via global_load_lds with a
// per-lane (mode 0) or wave-uniform (mode 1) LDS pointer, then write the
// LDS back out so the host can check which addressing lands correctly.
typedef const __attribute__((address_space(1))) unsigned int* gp_u32;
typedef __attribute__((address_space(3))) unsigned int* lp_u32;

__global__ __launch_bounds__(256) void glds_probe_kernel(
    const ushort_t* __restrict__ src, ushort_t* __restrict__ dst, int mode) {
  __shared__ __attribute__((aligned(16))) char lbuf[4096];
  const int tid = threadIdx.x;
  const int w = tid >> 6;
  const ushort_t* g = src + tid * 8;
  if (mode == 0) {
    __builtin_amdgcn_global_load_lds((gp_u32)g, (lp_u32)(lbuf + tid * 16), 16, 0, 0);
  } else {
    __builtin_amdgcn_global_load_lds((gp_u32)g, (lp_u32)(lbuf + w * 64 * 16), 16, 0, 0);
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  *(ushort8_t*)(dst + tid * 8) = *(const ushort8_t*)(lbuf + tid * 16);
}

SKY_EXPORT int sky_glds_probe(uint64_t stream, uint64_t src, uint64_t dst,
                              int mode) {
  hipLaunchKernelGGL(glds_probe_kernel, dim3(1), dim3(256), 0,
                     (hipStream_t)stream, (const ushort_t*)src,
                     (ushort_t*)dst, mode);
  LAUNCH_CHECK();
  return 0;
}
