// Fused word+position+type embedding gather + 3-way add + LayerNorm for
// gfx950 (fwd/bwd).
//
// Replaces the reference's eager gather/add/LN sequence
// (reference: scaelum/model/bert_layers.py:191-212). One block per row
// group; LN statistics saved for backward; backward recomputes the pre-LN
// sum by re-gathering and scatter-adds fp32 grads into the three tables.

#include "common.h"

template <int DT, int BLOCK>
__global__ __launch_bounds__(BLOCK) void emb_fwd_kernel(
    const int64_t* __restrict__ ids, const int64_t* __restrict__ tids,
    const int64_t* __restrict__ pids, const void* __restrict__ wemb,
    const void* __restrict__ temb, const void* __restrict__ pemb,
    const void* __restrict__ lnw, const void* __restrict__ lnb,
    void* __restrict__ y, float* __restrict__ mean_out,
    float* __restrict__ rstd_out, int64_t rows, int64_t cols, float eps) {
  __shared__ float lds[BLOCK / WAVE > 2 ? BLOCK / WAVE : 2];
  for (int64_t row = blockIdx.x; row < rows; row += gridDim.x) {
    const int64_t wb = ids[row] * cols;
    const int64_t tb = tids[row] * cols;
    const int64_t pb = pids[row] * cols;
    const int64_t base = row * cols;
    float s = 0.f, s2 = 0.f;
    for (int64_t c = threadIdx.x; c < cols; c += BLOCK) {
      float v = load_elem<DT>(wemb, wb + c) + load_elem<DT>(temb, tb + c) +
                load_elem<DT>(pemb, pb + c);
      s += v;
      s2 += v * v;
    }
    s = block_sum<BLOCK>(s, lds);
    s2 = block_sum<BLOCK>(s2, lds);
    const float mean = s / (float)cols;
    const float rstd = rsqrtf(fmaxf(s2 / (float)cols - mean * mean, 0.f) + eps);
    if (threadIdx.x == 0) {
      mean_out[row] = mean;
      rstd_out[row] = rstd;
    }
    for (int64_t c = threadIdx.x; c < cols; c += BLOCK) {
      float v = load_elem<DT>(wemb, wb + c) + load_elem<DT>(temb, tb + c) +
                load_elem<DT>(pemb, pb + c);
      float wv = load_elem<DT>(lnw, c), bv = load_elem<DT>(lnb, c);
      store_elem<DT>(y, base + c, (v - mean) * rstd * wv + bv);
    }
  }
}

SKY_EXPORT int sky_embedding_fwd(uint64_t stream, uint64_t ids, uint64_t tids,
                                 uint64_t pids, uint64_t wemb, uint64_t temb,
                                 uint64_t pemb, uint64_t lnw, uint64_t lnb,
                                 uint64_t y, uint64_t mean, uint64_t rstd,
                                 int64_t rows, int64_t cols, int64_t vocab,
                                 float eps, int dt) {
  (void)vocab;
  constexpr int BLOCK = 256;
  dim3 grid((unsigned)(rows < 4096 ? rows : 4096));
  hipStream_t s = (hipStream_t)stream;
#define EMF(DT)                                                                \
  hipLaunchKernelGGL((emb_fwd_kernel<DT, BLOCK>), grid, dim3(BLOCK), 0, s,     \
                     (const int64_t*)ids, (const int64_t*)tids,                \
                     (const int64_t*)pids, (const void*)wemb,                  \
                     (const void*)temb, (const void*)pemb, (const void*)lnw,   \
                     (const void*)lnb, (void*)y, (float*)mean, (float*)rstd,   \
                     rows, cols, eps)
  if (dt == DT_F32) EMF(DT_F32); else EMF(DT_BF16);
#undef EMF
  LAUNCH_CHECK();
  return 0;
}

// backward: LN backward per row -> dxs; scatter-add dxs into the three
// tables; accumulate dlnw/dlnb.

template <int DT, int BLOCK>
__global__ __launch_bounds__(BLOCK) void emb_bwd_kernel(
    const void* __restrict__ dy, const int64_t* __restrict__ ids,
    const int64_t* __restrict__ tids, const int64_t* __restrict__ pids,
    const void* __restrict__ wemb, const void* __restrict__ temb,
    const void* __restrict__ pemb, const void* __restrict__ lnw,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    float* __restrict__ dwe, float* __restrict__ dte, float* __restrict__ dpe,
    float* __restrict__ dlnw, float* __restrict__ dlnb, int64_t rows,
    int64_t cols) {
  extern __shared__ __attribute__((aligned(16))) float smem[];
  float* dlnw_part = smem;            // [cols]
  float* dlnb_part = smem + cols;     // [cols]
  float* lds = smem + 2 * cols;       // [BLOCK/WAVE]
  for (int64_t c = threadIdx.x; c < cols; c += BLOCK) {
    dlnw_part[c] = 0.f;
    dlnb_part[c] = 0.f;
  }
  __syncthreads();
  const int64_t RPW = 8;
  const int64_t row0 = (int64_t)blockIdx.x * RPW;
  const int64_t row1 = min(rows, row0 + RPW);
  for (int64_t row = row0; row < row1; ++row) {
    const int64_t wb = ids[row] * cols;
    const int64_t tb = tids[row] * cols;
    const int64_t pb = pids[row] * cols;
    const int64_t base = row * cols;
    const float mu = mean[row], rs = rstd[row];
    float s1 = 0.f, s2 = 0.f;
    for (int64_t c = threadIdx.x; c < cols; c += BLOCK) {
      float xv = load_elem<DT>(wemb, wb + c) + load_elem<DT>(temb, tb + c) +
                 load_elem<DT>(pemb, pb + c);
      float xhat = (xv - mu) * rs;
      float dyv = load_elem<DT>(dy, base + c);
      float dxhat = dyv * load_elem<DT>(lnw, c);
      s1 += dxhat;
      s2 += dxhat * xhat;
      dlnw_part[c] += dyv * xhat;
      dlnb_part[c] += dyv;
    }
    s1 = block_sum<BLOCK>(s1, lds) / (float)cols;
    s2 = block_sum<BLOCK>(s2, lds) / (float)cols;
    for (int64_t c = threadIdx.x; c < cols; c += BLOCK) {
      float xv = load_elem<DT>(wemb, wb + c) + load_elem<DT>(temb, tb + c) +
                 load_elem<DT>(pemb, pb + c);
      float xhat = (xv - mu) * rs;
      float dxhat = load_elem<DT>(dy, base + c) * load_elem<DT>(lnw, c);
      float dxs = rs * (dxhat - s1 - xhat * s2);
      atomicAdd(&dwe[wb + c], dxs);
      atomicAdd(&dte[tb + c], dxs);
      atomicAdd(&dpe[pb + c], dxs);
    }
    __syncthreads();
  }
  for (int64_t c = threadIdx.x; c < cols; c += BLOCK) {
    atomicAdd(&dlnw[c], dlnw_part[c]);
    atomicAdd(&dlnb[c], dlnb_part[c]);
  }
}

SKY_EXPORT int sky_embedding_bwd(uint64_t stream, uint64_t dy, uint64_t ids,
                                 uint64_t tids, uint64_t pids, uint64_t wemb,
                                 uint64_t temb, uint64_t pemb, uint64_t lnw,
                                 uint64_t mean, uint64_t rstd, uint64_t dwe,
                                 uint64_t dte, uint64_t dpe, uint64_t dlnw,
                                 uint64_t dlnb, int64_t rows, int64_t cols,
                                 int dt) {
  constexpr int BLOCK = 256;
  size_t lds_bytes = (2 * cols + BLOCK / WAVE) * sizeof(float);
  if (lds_bytes > 64 * 1024) return (int)hipErrorInvalidValue;
  unsigned grid = (unsigned)((rows + 7) / 8);
  hipStream_t s = (hipStream_t)stream;
#define EMB(DT)                                                                \
  hipLaunchKernelGGL((emb_bwd_kernel<DT, BLOCK>), dim3(grid), dim3(BLOCK),     \
                     lds_bytes, s, (const void*)dy, (const int64_t*)ids,       \
                     (const int64_t*)tids, (const int64_t*)pids,               \
                     (const void*)wemb, (const void*)temb, (const void*)pemb,  \
                     (const void*)lnw, (const float*)mean, (const float*)rstd, \
                     (float*)dwe, (float*)dte, (float*)dpe, (float*)dlnw,      \
                     (float*)dlnb, rows, cols)
  if (dt == DT_F32) EMB(DT_F32); else EMB(DT_BF16);
#undef EMB
  LAUNCH_CHECK();
  return 0;
}
