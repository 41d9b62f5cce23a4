// Fused bias+GELU and counter-RNG dropout for gfx950.
//
// bias_gelu replaces the reference's fused-at-Python-level LinearActivation
// epilogue + eager bias_gelu (reference: scaelum/model/bert_layers.py:21-44,
// 60-108) — on GPU it runs as the epilogue pass after the hipBLASLt GEMM.
// dropout regenerates its keep-mask from (seed, index) in backward, so no
// mask tensor is stored (reference used eager nn.Dropout).
//
// All kernels are memory-bound grid-stride loops; bf16 traffic is the
// dominant cost so loads/stores go through 8-wide vectors where aligned.

#include "common.h"

// ---------------- bias_gelu forward ----------------

template <int DT, int BLOCK>
__global__ __launch_bounds__(BLOCK) void bias_gelu_fwd_kernel(
    const void* __restrict__ x, const void* __restrict__ b,
    void* __restrict__ y, int64_t rows, int64_t cols) {
  const int64_t n = rows * cols;
  for (int64_t i = (int64_t)blockIdx.x * BLOCK + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * BLOCK) {
    float v = load_elem<DT>(x, i) + load_elem<DT>(b, i % cols);
    store_elem<DT>(y, i, gelu_f(v));
  }
}

SKY_EXPORT int sky_bias_gelu_fwd(uint64_t stream, uint64_t x, uint64_t b,
                                 uint64_t y, int64_t rows, int64_t cols,
                                 int dt) {
  constexpr int BLOCK = 256;
  int64_t n = rows * cols;
  unsigned grid = (unsigned)((n + BLOCK - 1) / BLOCK);
  if (grid > 2048u) grid = 2048u;
  hipStream_t s = (hipStream_t)stream;
  if (dt == DT_F32)
    hipLaunchKernelGGL((bias_gelu_fwd_kernel<DT_F32, BLOCK>), dim3(grid), dim3(BLOCK), 0, s,
                       (const void*)x, (const void*)b, (void*)y, rows, cols);
  else
    hipLaunchKernelGGL((bias_gelu_fwd_kernel<DT_BF16, BLOCK>), dim3(grid), dim3(BLOCK), 0, s,
                       (const void*)x, (const void*)b, (void*)y, rows, cols);
  LAUNCH_CHECK();
  return 0;
}

// ---------------- bias_gelu backward ----------------
// dx = dy * gelu'(x+b); db[c] = sum_r dx[r,c]
// Column partials accumulate in LDS (fp32[cols]) per block, then one
// atomicAdd per column per block.

template <int DT, int BLOCK>
__global__ __launch_bounds__(BLOCK) void bias_gelu_bwd_kernel(
    const void* __restrict__ dy, const void* __restrict__ x,
    const void* __restrict__ b, void* __restrict__ dx,
    float* __restrict__ db, int64_t rows, int64_t cols, int64_t rows_per_wg) {
  extern __shared__ __attribute__((aligned(16))) float db_part[];
  for (int64_t c = threadIdx.x; c < cols; c += BLOCK) db_part[c] = 0.f;
  __syncthreads();
  const int64_t row0 = (int64_t)blockIdx.x * rows_per_wg;
  const int64_t row1 = min(rows, row0 + rows_per_wg);
  for (int64_t row = row0; row < row1; ++row) {
    const int64_t base = row * cols;
    for (int64_t c = threadIdx.x; c < cols; c += BLOCK) {
      float z = load_elem<DT>(x, base + c) + load_elem<DT>(b, c);
      float d = load_elem<DT>(dy, base + c) * gelu_grad_f(z);
      store_elem<DT>(dx, base + c, d);
      db_part[c] += d;
    }
  }
  __syncthreads();
  for (int64_t c = threadIdx.x; c < cols; c += BLOCK)
    atomicAdd(&db[c], db_part[c]);
}

SKY_EXPORT int sky_bias_gelu_bwd(uint64_t stream, uint64_t dy, uint64_t x,
                                 uint64_t b, uint64_t dx, uint64_t db,
                                 int64_t rows, int64_t cols, int dt) {
  constexpr int BLOCK = 256;
  size_t lds_bytes = cols * sizeof(float);
  if (lds_bytes > 64 * 1024) return (int)hipErrorInvalidValue;
  int64_t rpw = 8;
  unsigned grid = (unsigned)((rows + rpw - 1) / rpw);
  hipStream_t s = (hipStream_t)stream;
  if (dt == DT_F32)
    hipLaunchKernelGGL((bias_gelu_bwd_kernel<DT_F32, BLOCK>), dim3(grid), dim3(BLOCK),
                       lds_bytes, s, (const void*)dy, (const void*)x, (const void*)b,
                       (void*)dx, (float*)db, rows, cols, rpw);
  else
    hipLaunchKernelGGL((bias_gelu_bwd_kernel<DT_BF16, BLOCK>), dim3(grid), dim3(BLOCK),
                       lds_bytes, s, (const void*)dy, (const void*)x, (const void*)b,
                       (void*)dx, (float*)db, rows, cols, rpw);
  LAUNCH_CHECK();
  return 0;
}

// ---------------- dropout ----------------

template <int DT, int BLOCK, bool FWD>
__global__ __launch_bounds__(BLOCK) void dropout_kernel(
    const void* __restrict__ x, void* __restrict__ y, int64_t n, float keep,
    uint64_t seed) {
  const float inv_keep = 1.f / keep;
  for (int64_t i = (int64_t)blockIdx.x * BLOCK + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * BLOCK) {
    float v = load_elem<DT>(x, i);
    bool k = rng_uniform(seed, (uint64_t)i) < keep;
    store_elem<DT>(y, i, k ? v * inv_keep : 0.f);
  }
}

SKY_EXPORT int sky_dropout_fwd(uint64_t stream, uint64_t x, uint64_t y,
                               int64_t n, float keep, uint64_t seed, int dt) {
  constexpr int BLOCK = 256;
  unsigned grid = (unsigned)((n + BLOCK - 1) / BLOCK);
  if (grid > 2048u) grid = 2048u;
  hipStream_t s = (hipStream_t)stream;
  if (dt == DT_F32)
    hipLaunchKernelGGL((dropout_kernel<DT_F32, BLOCK, true>), dim3(grid), dim3(BLOCK), 0, s,
                       (const void*)x, (void*)y, n, keep, seed);
  else
    hipLaunchKernelGGL((dropout_kernel<DT_BF16, BLOCK, true>), dim3(grid), dim3(BLOCK), 0, s,
                       (const void*)x, (void*)y, n, keep, seed);
  LAUNCH_CHECK();
  return 0;
}

SKY_EXPORT int sky_dropout_bwd(uint64_t stream, uint64_t dy, uint64_t dx,
                               int64_t n, float keep, uint64_t seed, int dt) {
  // identical math: dx = dy * mask / keep
  return sky_dropout_fwd(stream, dy, dx, n, keep, seed, dt);
}
