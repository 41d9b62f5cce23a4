// Fused (residual-add +) LayerNorm forward/backward for gfx950.
//
// Replaces the reference's optional apex FusedLayerNormAffineFunction
// (reference: scaelum/model/bert_layers.py:128-168) and fuses the
// preceding residual add + dropout-output add pattern
// (bert_layers.py:283-288,321-326) into the same pass.
//
// Forward: one 256-thread block per row-group; per-row mean/var accumulated
// in fp32 via wave shuffles + LDS cross-wave reduce; saves mean/rstd (fp32)
// for backward. Memory-bound: bf16 loads vectorized 8-wide.
// Backward: one block per ROWS_PER_WG rows; dweight/dbias accumulated in
// registers across the block's rows, then one fp32 atomicAdd per column per
// block (guide Guideline 12: partial-reduce before atomics).

#include "common.h"

// ---------------- forward ----------------

template <int DT, int BLOCK, bool HAS_RES>
__global__ __launch_bounds__(BLOCK) void ln_fwd_kernel(
    const void* __restrict__ x, const void* __restrict__ res,
    const void* __restrict__ w, const void* __restrict__ b,
    void* __restrict__ y, float* __restrict__ mean_out,
    float* __restrict__ rstd_out, int64_t rows, int64_t cols, float eps) {
  __shared__ float lds[BLOCK / WAVE > 2 ? BLOCK / WAVE : 2];
  for (int64_t row = blockIdx.x; row < rows; row += gridDim.x) {
    const int64_t base = row * cols;
    float s = 0.f, s2 = 0.f;
    for (int64_t c = threadIdx.x; c < cols; c += BLOCK) {
      float v = load_elem<DT>(x, base + c);
      if (HAS_RES) v += load_elem<DT>(res, base + c);
      s += v;
      s2 += v * v;
    }
    s = block_sum<BLOCK>(s, lds);
    s2 = block_sum<BLOCK>(s2, lds);
    const float mean = s / (float)cols;
    float var = s2 / (float)cols - mean * mean;
    const float rstd = rsqrtf(fmaxf(var, 0.f) + eps);
    if (threadIdx.x == 0) {
      mean_out[row] = mean;
      rstd_out[row] = rstd;
    }
    for (int64_t c = threadIdx.x; c < cols; c += BLOCK) {
      float v = load_elem<DT>(x, base + c);
      if (HAS_RES) v += load_elem<DT>(res, base + c);
      float wv = load_elem<DT>(w, c), bv = load_elem<DT>(b, c);
      store_elem<DT>(y, base + c, (v - mean) * rstd * wv + bv);
    }
  }
}

SKY_EXPORT int sky_layernorm_fwd(uint64_t stream, uint64_t x, uint64_t res,
                                 uint64_t w, uint64_t b, uint64_t y,
                                 uint64_t mean, uint64_t rstd, int64_t rows,
                                 int64_t cols, float eps, int dt) {
  constexpr int BLOCK = 256;
  dim3 grid((unsigned)(rows < 4096 ? rows : 4096));
  hipStream_t s = (hipStream_t)stream;
  bool has_res = res != 0;
#define LNF(DT, HR)                                                         \
  hipLaunchKernelGGL((ln_fwd_kernel<DT, BLOCK, HR>), grid, dim3(BLOCK), 0, s, \
                     (const void*)x, (const void*)res, (const void*)w,      \
                     (const void*)b, (void*)y, (float*)mean, (float*)rstd,  \
                     rows, cols, eps)
  if (dt == DT_F32) { if (has_res) LNF(DT_F32, true); else LNF(DT_F32, false); }
  else              { if (has_res) LNF(DT_BF16, true); else LNF(DT_BF16, false); }
#undef LNF
  LAUNCH_CHECK();
  return 0;
}

// ---------------- backward ----------------
// dxhat = dy * w
// dx = rstd * (dxhat - mean_c(dxhat) - xhat * mean_c(dxhat * xhat))
// dw[c] += sum_r dy * xhat ; db[c] += sum_r dy

template <int DT, int BLOCK, bool HAS_RES, int ROWS_PER_WG>
__global__ __launch_bounds__(BLOCK) void ln_bwd_kernel(
    const void* __restrict__ dy, const void* __restrict__ x,
    const void* __restrict__ res, const void* __restrict__ w,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    void* __restrict__ dx, float* __restrict__ dw, float* __restrict__ db,
    int64_t rows, int64_t cols) {
  // single dynamic LDS region: [cols] dw partials, [cols] db partials,
  // [BLOCK/WAVE] reduce scratch (one __shared__ object, 16B-aligned base —
  // guide Guideline 17)
  extern __shared__ __attribute__((aligned(16))) float dwdb[];
  float* dw_part = dwdb;
  float* db_part = dwdb + cols;
  float* lds = dwdb + 2 * cols;
  for (int64_t c = threadIdx.x; c < cols; c += BLOCK) {
    dw_part[c] = 0.f;
    db_part[c] = 0.f;
  }
  __syncthreads();

  const int64_t group = blockIdx.x;
  const int64_t row0 = group * ROWS_PER_WG;
  for (int64_t row = row0; row < rows && row < row0 + ROWS_PER_WG; ++row) {
    const int64_t base = row * cols;
    const float mu = mean[row], rs = rstd[row];
    float s1 = 0.f, s2 = 0.f;
    for (int64_t c = threadIdx.x; c < cols; c += BLOCK) {
      float xv = load_elem<DT>(x, base + c);
      if (HAS_RES) xv += load_elem<DT>(res, base + c);
      float xhat = (xv - mu) * rs;
      float dyv = load_elem<DT>(dy, base + c);
      float wv = load_elem<DT>(w, c);
      float dxhat = dyv * wv;
      s1 += dxhat;
      s2 += dxhat * xhat;
      dw_part[c] += dyv * xhat;
      db_part[c] += dyv;
    }
    s1 = block_sum<BLOCK>(s1, lds) / (float)cols;
    s2 = block_sum<BLOCK>(s2, lds) / (float)cols;
    for (int64_t c = threadIdx.x; c < cols; c += BLOCK) {
      float xv = load_elem<DT>(x, base + c);
      if (HAS_RES) xv += load_elem<DT>(res, base + c);
      float xhat = (xv - mu) * rs;
      float dxhat = load_elem<DT>(dy, base + c) * load_elem<DT>(w, c);
      store_elem<DT>(dx, base + c, rs * (dxhat - s1 - xhat * s2));
    }
    __syncthreads();
  }
  for (int64_t c = threadIdx.x; c < cols; c += BLOCK) {
    atomicAdd(&dw[c], dw_part[c]);
    atomicAdd(&db[c], db_part[c]);
  }
}

SKY_EXPORT int sky_layernorm_bwd(uint64_t stream, uint64_t dy, uint64_t x,
                                 uint64_t res, uint64_t w, uint64_t mean,
                                 uint64_t rstd, uint64_t dx, uint64_t dw,
                                 uint64_t db, int64_t rows, int64_t cols,
                                 int dt) {
  constexpr int BLOCK = 256;
  constexpr int RPW = 8;
  size_t lds_bytes = (2 * cols + BLOCK / WAVE) * sizeof(float);
  if (lds_bytes > 64 * 1024) return (int)hipErrorInvalidValue;
  dim3 grid((unsigned)((rows + RPW - 1) / RPW));
  hipStream_t s = (hipStream_t)stream;
  bool has_res = res != 0;
#define LNB(DT, HR)                                                          \
  hipLaunchKernelGGL((ln_bwd_kernel<DT, BLOCK, HR, RPW>), grid, dim3(BLOCK), \
                     lds_bytes, s, (const void*)dy, (const void*)x,          \
                     (const void*)res, (const void*)w, (const float*)mean,   \
                     (const float*)rstd, (void*)dx, (float*)dw, (float*)db,  \
                     rows, cols)
  if (dt == DT_F32) { if (has_res) LNB(DT_F32, true); else LNB(DT_F32, false); }
  else              { if (has_res) LNB(DT_BF16, true); else LNB(DT_BF16, false); }
#undef LNB
  LAUNCH_CHECK();
  return 0;
}
