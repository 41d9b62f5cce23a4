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
