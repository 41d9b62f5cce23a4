// Fused (residual-add +) LayerNorm forward/backward for gfx950.
//
// Replaces the reference's optional apex FusedLayerNormAffineFunction
// (reference: scaelum/model/bert_layers.py:128-168) and fuses the
// preceding residual add (bert_layers.py:283-288,321-326) into the pass.
//
// Fast path (cols % 8 == 0, cols <= 4096): WAVE-PER-ROW, 16 B/lane
// vectorized loads, values register-cached across the stats/normalize
// passes, reductions via 64-lane shuffles only — no LDS, no barriers.
// Backward splits into a wave-per-row dx kernel plus a column-parallel
// dweight/dbias reduction kernel (coalesced down-column traversal, few
// atomics) instead of row-serial LDS accumulation.
// Generic fallback (odd cols): block-per-row scalar kernels.

#include "common.h"

#define LN_MAXCH 8  // max chunks of 512 elements -> cols <= 4096 fast path

// ---------------- forward (vectorized, wave per row) ----------------

// DROP: apply dropout to x BEFORE the residual add (the BERT pattern
// LN(dropout(dense(x)) + residual), reference bert_layers.py:283-288);
// the mask is regenerated from (salt,state,element index) in backward.
template <int DT, bool HAS_RES, bool DROP, int CH>
__global__ __launch_bounds__(256) void ln_fwd_vec_kernel(
    const void* __restrict__ x, const void* __restrict__ res,
    const void* __restrict__ w, const void* __restrict__ b,
    void* __restrict__ y, float* __restrict__ mean_out,
    float* __restrict__ rstd_out, int64_t rows, int64_t cols, float eps,
    float keep, uint64_t salt, const unsigned long long* __restrict__ state) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int64_t cols8 = cols / 8;
  const uint64_t seed =
      DROP ? salt + (state ? *state : 0ull) * 0xD1B54A32D192ED03ull : 0;
  const unsigned keep16 = keep_to_16(keep);
  const float inv_keep = 1.f / keep;
  float v[CH][8];
  for (int64_t row = (int64_t)blockIdx.x * 4 + wid; row < rows;
       row += (int64_t)gridDim.x * 4) {
    const int64_t base8 = row * cols8;
    float s = 0.f, s2 = 0.f;
#pragma unroll
    for (int c = 0; c < CH; ++c) {
      const int64_t i8 = (int64_t)c * WAVE + lane;
      if (i8 < cols8) {
        Vec8<DT>::load(x, base8 + i8, v[c]);
        if (DROP) {
          const uint64_t e8 = (uint64_t)(base8 + i8);
          const uint64_t z0 = rng_hash(seed, e8 * 2);
          const uint64_t z1 = rng_hash(seed, e8 * 2 + 1);
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            const uint64_t zz = j < 4 ? z0 : z1;
            bool kb = (unsigned)((zz >> (16 * (j & 3))) & 0xFFFFu) < keep16;
            v[c][j] = kb ? v[c][j] * inv_keep : 0.f;
          }
        }
        if (HAS_RES) {
          float r[8];
          Vec8<DT>::load(res, base8 + i8, r);
#pragma unroll
          for (int j = 0; j < 8; ++j) v[c][j] += r[j];
        }
#pragma unroll
        for (int j = 0; j < 8; ++j) { s += v[c][j]; s2 += v[c][j] * v[c][j]; }
      }
    }
    s = wave_sum(s);
    s2 = wave_sum(s2);
    const float mean = s / (float)cols;
    const float rstd = rsqrtf(fmaxf(s2 / (float)cols - mean * mean, 0.f) + eps);
    if (lane == 0) { mean_out[row] = mean; rstd_out[row] = rstd; }
#pragma unroll
    for (int c = 0; c < CH; ++c) {
      const int64_t i8 = (int64_t)c * WAVE + lane;
      if (i8 < cols8) {
        float wv[8], bv[8], o[8];
        Vec8<DT>::load(w, i8, wv);
        Vec8<DT>::load(b, i8, bv);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          o[j] = (v[c][j] - mean) * rstd * wv[j] + bv[j];
        Vec8<DT>::store(y, base8 + i8, o);
      }
    }
  }
}

// generic scalar fallback (block per row)
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
    const float rstd = rsqrtf(fmaxf(s2 / (float)cols - mean * mean, 0.f) + eps);
    if (threadIdx.x == 0) { mean_out[row] = mean; rstd_out[row] = rstd; }
    for (int64_t c = threadIdx.x; c < cols; c += BLOCK) {
      float v = load_elem<DT>(x, base + c);
      if (HAS_RES) v += load_elem<DT>(res, base + c);
      store_elem<DT>(y, base + c,
                     (v - mean) * rstd * load_elem<DT>(w, c) + load_elem<DT>(b, c));
    }
  }
}

static inline bool ln_fast_ok(int64_t cols) {
  return (cols % 8 == 0) && cols <= 64 * 8 * LN_MAXCH;
}

SKY_EXPORT int sky_layernorm_fwd(uint64_t stream, uint64_t x, uint64_t res,
                                 uint64_t w, uint64_t b, uint64_t y,
                                 uint64_t mean, uint64_t rstd, int64_t rows,
                                 int64_t cols, float eps, int dt, float keep,
                                 uint64_t salt, uint64_t state) {
  hipStream_t s = (hipStream_t)stream;
  bool has_res = res != 0;
  bool drop = keep < 1.f;
  if (drop && !ln_fast_ok(cols)) return (int)hipErrorInvalidValue;
  if (ln_fast_ok(cols)) {
    unsigned grid = (unsigned)((rows + 3) / 4);
    if (grid > 8192u) grid = 8192u;
    const int ch = (int)((cols / 8 + WAVE - 1) / WAVE);
#define LNFV(DT, HR, DR, CH)                                                  \
  hipLaunchKernelGGL((ln_fwd_vec_kernel<DT, HR, DR, CH>), dim3(grid),         \
                     dim3(256), 0, s, (const void*)x, (const void*)res,       \
                     (const void*)w, (const void*)b, (void*)y, (float*)mean,  \
                     (float*)rstd, rows, cols, eps, keep, salt,               \
                     (const unsigned long long*)state)
#define LNFV_CH(DT, HR, DR)                                                   \
  do {                                                                        \
    if (ch <= 2) LNFV(DT, HR, DR, 2);                                         \
    else if (ch <= 4) LNFV(DT, HR, DR, 4);                                    \
    else LNFV(DT, HR, DR, 8);                                                 \
  } while (0)
#define LNFV_D(DT, HR)                                                        \
  do { if (drop) LNFV_CH(DT, HR, true); else LNFV_CH(DT, HR, false); } while (0)
    if (dt == DT_F32) { if (has_res) LNFV_D(DT_F32, true); else LNFV_D(DT_F32, false); }
    else              { if (has_res) LNFV_D(DT_BF16, true); else LNFV_D(DT_BF16, false); }
#undef LNFV_D
#undef LNFV_CH
#undef LNFV
  } else {
    dim3 grid((unsigned)(rows < 4096 ? rows : 4096));
#define LNF(DT, HR)                                                           \
  hipLaunchKernelGGL((ln_fwd_kernel<DT, 256, HR>), grid, dim3(256), 0, s,     \
                     (const void*)x, (const void*)res, (const void*)w,        \
                     (const void*)b, (void*)y, (float*)mean, (float*)rstd,    \
                     rows, cols, eps)
    if (dt == DT_F32) { if (has_res) LNF(DT_F32, true); else LNF(DT_F32, false); }
    else              { if (has_res) LNF(DT_BF16, true); else LNF(DT_BF16, false); }
#undef LNF
  }
  LAUNCH_CHECK();
  return 0;
}

// ---------------- backward ----------------
// dxhat = dy * w
// dx = rstd * (dxhat - mean_c(dxhat) - xhat * mean_c(dxhat * xhat))
// dw[c] += sum_r dy * xhat ; db[c] += sum_r dy

// DROP: x is the PRE-dropout tensor; the mask is regenerated to rebuild
// xs = dropout(x)+res for the statistics, dx gets the mask/keep factor,
// and (when dres != null) the residual grad d_xs is written separately.
template <int DT, bool HAS_RES, bool DROP, int CH>
__global__ __launch_bounds__(256) void ln_bwd_dx_vec_kernel(
    const void* __restrict__ dy, const void* __restrict__ x,
    const void* __restrict__ res, const void* __restrict__ w,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    void* __restrict__ dx, void* __restrict__ dres, int64_t rows,
    int64_t cols, float keep, uint64_t salt,
    const unsigned long long* __restrict__ state) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int64_t cols8 = cols / 8;
  const uint64_t seed =
      DROP ? salt + (state ? *state : 0ull) * 0xD1B54A32D192ED03ull : 0;
  const unsigned keep16 = keep_to_16(keep);
  const float inv_keep = 1.f / keep;
  float xh[CH][8], dxh[CH][8];
  unsigned kbits[CH];
  for (int64_t row = (int64_t)blockIdx.x * 4 + wid; row < rows;
       row += (int64_t)gridDim.x * 4) {
    const int64_t base8 = row * cols8;
    const float mu = mean[row], rs = rstd[row];
    float s1 = 0.f, s2 = 0.f;
#pragma unroll
    for (int c = 0; c < CH; ++c) {
      const int64_t i8 = (int64_t)c * WAVE + lane;
      if (i8 < cols8) {
        float xv[8], dyv[8], wv[8];
        Vec8<DT>::load(x, base8 + i8, xv);
        if (DROP) {
          const uint64_t e8 = (uint64_t)(base8 + i8);
          const uint64_t z0 = rng_hash(seed, e8 * 2);
          const uint64_t z1 = rng_hash(seed, e8 * 2 + 1);
          kbits[c] = 0;
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            const uint64_t zz = j < 4 ? z0 : z1;
            bool kb = (unsigned)((zz >> (16 * (j & 3))) & 0xFFFFu) < keep16;
            kbits[c] |= (unsigned)kb << j;
            xv[j] = kb ? xv[j] * inv_keep : 0.f;
          }
        }
        if (HAS_RES) {
          float r[8];
          Vec8<DT>::load(res, base8 + i8, r);
#pragma unroll
          for (int j = 0; j < 8; ++j) xv[j] += r[j];
        }
        Vec8<DT>::load(dy, base8 + i8, dyv);
        Vec8<DT>::load(w, i8, wv);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          xh[c][j] = (xv[j] - mu) * rs;
          dxh[c][j] = dyv[j] * wv[j];
          s1 += dxh[c][j];
          s2 += dxh[c][j] * xh[c][j];
        }
      }
    }
    s1 = wave_sum(s1) / (float)cols;
    s2 = wave_sum(s2) / (float)cols;
#pragma unroll
    for (int c = 0; c < CH; ++c) {
      const int64_t i8 = (int64_t)c * WAVE + lane;
      if (i8 < cols8) {
        float o[8];
#pragma unroll
        for (int j = 0; j < 8; ++j)
          o[j] = rs * (dxh[c][j] - s1 - xh[c][j] * s2);
        if (DROP && dres != nullptr) Vec8<DT>::store(dres, base8 + i8, o);
        if (DROP) {
#pragma unroll
          for (int j = 0; j < 8; ++j)
            o[j] = ((kbits[c] >> j) & 1u) ? o[j] * inv_keep : 0.f;
        }
        Vec8<DT>::store(dx, base8 + i8, o);
      }
    }
  }
}

// Fused dx + dw/db column partials: the dx pass already holds dy and xhat
// in registers for every element, and a wave's lane covers the SAME
// column slices on every row it processes — so the dw/db partial sums
// accumulate in registers for free and wb_part's 16-24 MB re-read of
// dy/x disappears. Each WAVE is one scratch slab (slab = blockIdx*4+wid,
// no LDS merge); the grid is capped so nslabs <= LN_SLABS.
template <int DT, bool HAS_RES, bool DROP, int CH>
__global__ __launch_bounds__(256) void ln_bwd_dx_cs_kernel(
    const void* __restrict__ dy, const void* __restrict__ x,
    const void* __restrict__ res, const void* __restrict__ w,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    void* __restrict__ dx, void* __restrict__ dres,
    float* __restrict__ scratch, int64_t rows, int64_t cols, float keep,
    uint64_t salt, const unsigned long long* __restrict__ state) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int64_t cols8 = cols / 8;
  const uint64_t seed =
      DROP ? salt + (state ? *state : 0ull) * 0xD1B54A32D192ED03ull : 0;
  const unsigned keep16 = keep_to_16(keep);
  const float inv_keep = 1.f / keep;
  float xh[CH][8], dxh[CH][8];
  float sw[CH][8], sb[CH][8];
#pragma unroll
  for (int c = 0; c < CH; ++c)
#pragma unroll
    for (int j = 0; j < 8; ++j) { sw[c][j] = 0.f; sb[c][j] = 0.f; }
  unsigned kbits[CH];
  for (int64_t row = (int64_t)blockIdx.x * 4 + wid; row < rows;
       row += (int64_t)gridDim.x * 4) {
    const int64_t base8 = row * cols8;
    const float mu = mean[row], rs = rstd[row];
    float s1 = 0.f, s2 = 0.f;
#pragma unroll
    for (int c = 0; c < CH; ++c) {
      const int64_t i8 = (int64_t)c * WAVE + lane;
      if (i8 < cols8) {
        float xv[8], dyv[8], wv[8];
        Vec8<DT>::load(x, base8 + i8, xv);
        if (DROP) {
          const uint64_t e8 = (uint64_t)(base8 + i8);
          const uint64_t z0 = rng_hash(seed, e8 * 2);
          const uint64_t z1 = rng_hash(seed, e8 * 2 + 1);
          kbits[c] = 0;
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            const uint64_t zz = j < 4 ? z0 : z1;
            bool kb = (unsigned)((zz >> (16 * (j & 3))) & 0xFFFFu) < keep16;
            kbits[c] |= (unsigned)kb << j;
            xv[j] = kb ? xv[j] * inv_keep : 0.f;
          }
        }
        if (HAS_RES) {
          float r[8];
          Vec8<DT>::load(res, base8 + i8, r);
#pragma unroll
          for (int j = 0; j < 8; ++j) xv[j] += r[j];
        }
        Vec8<DT>::load(dy, base8 + i8, dyv);
        Vec8<DT>::load(w, i8, wv);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          xh[c][j] = (xv[j] - mu) * rs;
          dxh[c][j] = dyv[j] * wv[j];
          s1 += dxh[c][j];
          s2 += dxh[c][j] * xh[c][j];
          sw[c][j] += dyv[j] * xh[c][j];
          sb[c][j] += dyv[j];
        }
      }
    }
    s1 = wave_sum(s1) / (float)cols;
    s2 = wave_sum(s2) / (float)cols;
#pragma unroll
    for (int c = 0; c < CH; ++c) {
      const int64_t i8 = (int64_t)c * WAVE + lane;
      if (i8 < cols8) {
        float o[8];
#pragma unroll
        for (int j = 0; j < 8; ++j)
          o[j] = rs * (dxh[c][j] - s1 - xh[c][j] * s2);
        if (DROP && dres != nullptr) Vec8<DT>::store(dres, base8 + i8, o);
        if (DROP) {
#pragma unroll
          for (int j = 0; j < 8; ++j)
            o[j] = ((kbits[c] >> j) & 1u) ? o[j] * inv_keep : 0.f;
        }
        Vec8<DT>::store(dx, base8 + i8, o);
      }
    }
  }
  // merge the 4 waves' partials through LDS (halves the slab count vs
  // per-wave slabs -> half the scratch the final stage must re-read),
  // then ONE slab write per block.
  __shared__ float mrg[2][CH * 512];
  for (int i = threadIdx.x; i < 2 * CH * 512; i += 256)
    (&mrg[0][0])[i] = 0.f;
  __syncthreads();
  for (int k = 0; k < 4; ++k) {
    if (wid == k) {
#pragma unroll
      for (int c = 0; c < CH; ++c) {
        const int64_t i8 = (int64_t)c * WAVE + lane;
        if (i8 < cols8) {
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            mrg[0][i8 * 8 + j] += sw[c][j];
            mrg[1][i8 * 8 + j] += sb[c][j];
          }
        }
      }
    }
    __syncthreads();
  }
  float* base = scratch + (int64_t)blockIdx.x * 2 * cols8 * 8;
  for (int i = threadIdx.x; i < cols8 * 8; i += 256) {
    base[i] = mrg[0][i];
    base[cols8 * 8 + i] = mrg[1][i];
  }
}

// column-parallel dw/db, two-stage, no atomics, outputs need no zero-init:
// stage 1 writes per-slab partials to scratch [nslabs][2*cols] fp32
// (dw partial at [y][c], db partial at [y][cols+c]); stage 2 reduces.
#define LN_SLABS 1024

template <int DT, bool HAS_RES, bool DROP, int BLOCK>
__global__ __launch_bounds__(BLOCK) void ln_bwd_wb_part_kernel(
    const void* __restrict__ dy, const void* __restrict__ x,
    const void* __restrict__ res, const float* __restrict__ mean,
    const float* __restrict__ rstd, float* __restrict__ scratch,
    int64_t rows, int64_t cols8, int64_t rows_per_slab, float keep,
    uint64_t salt, const unsigned long long* __restrict__ state) {
  const int64_t c8 = (int64_t)blockIdx.x * BLOCK + threadIdx.x;
  if (c8 >= cols8) return;
  const int64_t slab = blockIdx.y;
  const int64_t r0 = slab * rows_per_slab;
  const int64_t r1 = min(rows, r0 + rows_per_slab);
  const uint64_t seed =
      DROP ? salt + (state ? *state : 0ull) * 0xD1B54A32D192ED03ull : 0;
  const unsigned keep16 = keep_to_16(keep);
  const float inv_keep = 1.f / keep;
  float sw[8] = {0.f}, sb[8] = {0.f};
  for (int64_t r = r0; r < r1; ++r) {
    const float mu = mean[r], rs = rstd[r];
    float xv[8], dyv[8];
    Vec8<DT>::load(x, r * cols8 + c8, xv);
    Vec8<DT>::load(dy, r * cols8 + c8, dyv);
    if (DROP) {
      const uint64_t e8 = (uint64_t)(r * cols8 + c8);
      const uint64_t z0 = rng_hash(seed, e8 * 2);
      const uint64_t z1 = rng_hash(seed, e8 * 2 + 1);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const uint64_t zz = j < 4 ? z0 : z1;
        bool kb = (unsigned)((zz >> (16 * (j & 3))) & 0xFFFFu) < keep16;
        xv[j] = kb ? xv[j] * inv_keep : 0.f;
      }
    }
    if (HAS_RES) {
      float rv[8];
      Vec8<DT>::load(res, r * cols8 + c8, rv);
#pragma unroll
      for (int j = 0; j < 8; ++j) xv[j] += rv[j];
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      sw[j] += dyv[j] * (xv[j] - mu) * rs;
      sb[j] += dyv[j];
    }
  }
  float* base = scratch + slab * 2 * cols8 * 8;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    base[c8 * 8 + j] = sw[j];
    base[cols8 * 8 + c8 * 8 + j] = sb[j];
  }
}

// two-level: a 1024-thread block covers 16 columns x 64 slab-groups,
// parallel LDS tree, one write per column (see colsum_final_kernel).
template <int BLOCK, int DTOUT>
__global__ __launch_bounds__(1024) void ln_bwd_wb_final_kernel(
    const float* __restrict__ scratch, void* __restrict__ dw,
    void* __restrict__ db, int64_t cols, int64_t nslabs) {
  __shared__ float ldsw[64][17];
  __shared__ float ldsb[64][17];
  const int c = threadIdx.x & 15;
  const int g = threadIdx.x >> 4;  // 0..63
  const int64_t col = (int64_t)blockIdx.x * 16 + c;
  const int64_t per = (nslabs + 63) / 64;
  float sw = 0.f, sb = 0.f;
  if (col < cols) {
    const int64_t y1 = min(nslabs, (int64_t)(g + 1) * per);
    for (int64_t y = (int64_t)g * per; y < y1; ++y) {
      sw += scratch[y * 2 * cols + col];
      sb += scratch[y * 2 * cols + cols + col];
    }
  }
  ldsw[g][c] = sw;
  ldsb[g][c] = sb;
  __syncthreads();
  for (int st = 32; st >= 1; st >>= 1) {
    if (g < st) { ldsw[g][c] += ldsw[g + st][c]; ldsb[g][c] += ldsb[g + st][c]; }
    __syncthreads();
  }
  if (g == 0 && col < cols) {
    store_elem<DTOUT>(dw, col, ldsw[0][c]);
    store_elem<DTOUT>(db, col, ldsb[0][c]);
  }
}

// final reduction, CPB=4 columns per 1024-thread block (256 slab-groups):
// 4x the block count and 4x the per-thread memory-level parallelism of the
// 16-col variant (that one ran ~1 TB/s on an 8 MB scratch — latency-bound
// at 64 blocks). Treats scratch as one [nslabs][2*cols] matrix; col >= cols
// lands in db.
template <int DTOUT>
__global__ __launch_bounds__(1024) void ln_bwd_wb_final4_kernel(
    const float* __restrict__ scratch, void* __restrict__ dw,
    void* __restrict__ db, int64_t cols, int64_t nslabs) {
  __shared__ float ldsv[256][5];
  const int c = threadIdx.x & 3;
  const int g = threadIdx.x >> 2;  // 0..255
  const int64_t col = (int64_t)blockIdx.x * 4 + c;
  const int64_t per = (nslabs + 255) / 256;
  float s = 0.f;
  if (col < 2 * cols) {
    const int64_t y1 = min(nslabs, (int64_t)(g + 1) * per);
    for (int64_t y = (int64_t)g * per; y < y1; ++y)
      s += scratch[y * 2 * cols + col];
  }
  ldsv[g][c] = s;
  __syncthreads();
  for (int st = 128; st >= 1; st >>= 1) {
    if (g < st) ldsv[g][c] += ldsv[g + st][c];
    __syncthreads();
  }
  if (g == 0 && col < 2 * cols) {
    if (col < cols) store_elem<DTOUT>(dw, col, ldsv[0][c]);
    else store_elem<DTOUT>(db, col - cols, ldsv[0][c]);
  }
}

template <int DT, bool HAS_RES, int BLOCK>
__global__ __launch_bounds__(BLOCK) void ln_bwd_wb_kernel(
    const void* __restrict__ dy, const void* __restrict__ x,
    const void* __restrict__ res, const float* __restrict__ mean,
    const float* __restrict__ rstd, float* __restrict__ dw,
    float* __restrict__ db, int64_t rows, int64_t cols, int64_t rows_per_slab) {
  const int64_t col = (int64_t)blockIdx.x * BLOCK + threadIdx.x;
  if (col >= cols) return;
  const int64_t r0 = (int64_t)blockIdx.y * rows_per_slab;
  const int64_t r1 = min(rows, r0 + rows_per_slab);
  float sw = 0.f, sb = 0.f;
  for (int64_t r = r0; r < r1; ++r) {
    float xv = load_elem<DT>(x, r * cols + col);
    if (HAS_RES) xv += load_elem<DT>(res, r * cols + col);
    float dyv = load_elem<DT>(dy, r * cols + col);
    sw += dyv * (xv - mean[r]) * rstd[r];
    sb += dyv;
  }
  atomicAdd(&dw[col], sw);
  atomicAdd(&db[col], sb);
}

// generic scalar fallback dx (block per row)
template <int DT, int BLOCK, bool HAS_RES>
__global__ __launch_bounds__(BLOCK) void ln_bwd_dx_kernel(
    const void* __restrict__ dy, const void* __restrict__ x,
    const void* __restrict__ res, const void* __restrict__ w,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    void* __restrict__ dx, int64_t rows, int64_t cols) {
  __shared__ float lds[BLOCK / WAVE > 2 ? BLOCK / WAVE : 2];
  for (int64_t row = blockIdx.x; row < rows; row += gridDim.x) {
    const int64_t base = row * cols;
    const float mu = mean[row], rs = rstd[row];
    float s1 = 0.f, s2 = 0.f;
    for (int64_t c = threadIdx.x; c < cols; c += BLOCK) {
      float xv = load_elem<DT>(x, base + c);
      if (HAS_RES) xv += load_elem<DT>(res, base + c);
      float dxhat = load_elem<DT>(dy, base + c) * load_elem<DT>(w, c);
      s1 += dxhat;
      s2 += dxhat * (xv - mu) * rs;
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
  }
}

SKY_EXPORT int sky_layernorm_bwd(uint64_t stream, uint64_t dy, uint64_t x,
                                 uint64_t res, uint64_t w, uint64_t mean,
                                 uint64_t rstd, uint64_t dx, uint64_t dw,
                                 uint64_t db, uint64_t scratch, int64_t rows,
                                 int64_t cols, int dt, float keep,
                                 uint64_t salt, uint64_t state,
                                 uint64_t dres) {
  hipStream_t s = (hipStream_t)stream;
  bool has_res = res != 0;
  bool drop = keep < 1.f;
  if (drop && !ln_fast_ok(cols)) return (int)hipErrorInvalidValue;
  // fused path: the dx kernel also produces the dw/db per-wave column
  // partials (kills wb_part's re-read of dy/x); SKY_LN_SPLIT_WB=1 restores
  // the separate 3-kernel chain for A/B.
  const bool fused_wb =
      ln_fast_ok(cols) && scratch != 0 && !getenv("SKY_LN_SPLIT_WB");
  if (fused_wb) {
    // one slab per BLOCK (LDS-merged): grid 512 keeps the dx streaming fed
    // (256 starved it: 16.9 vs 9.9 us) while the final re-reads only
    // 512 slabs (SKY_LN_CS_GRID overrides for sweeps)
    unsigned gmax = 512;
    if (const char* e = getenv("SKY_LN_CS_GRID")) gmax = (unsigned)atoi(e);
    if (gmax > LN_SLABS) gmax = LN_SLABS;
    unsigned grid = (unsigned)((rows + 3) / 4);
    if (grid > gmax) grid = gmax;
    const int ch = (int)((cols / 8 + WAVE - 1) / WAVE);
#define LNCS(DT, HR, DR, CH)                                                   \
  hipLaunchKernelGGL((ln_bwd_dx_cs_kernel<DT, HR, DR, CH>), dim3(grid),        \
                     dim3(256), 0, s, (const void*)dy, (const void*)x,         \
                     (const void*)res, (const void*)w, (const float*)mean,     \
                     (const float*)rstd, (void*)dx, (void*)dres,               \
                     (float*)scratch, rows, cols, keep, salt,                  \
                     (const unsigned long long*)state)
#define LNCS_CH(DT, HR, DR)                                                    \
  do {                                                                         \
    if (ch <= 2) LNCS(DT, HR, DR, 2);                                          \
    else if (ch <= 4) LNCS(DT, HR, DR, 4);                                     \
    else LNCS(DT, HR, DR, 8);                                                  \
  } while (0)
#define LNCS_D(DT, HR)                                                         \
  do { if (drop) LNCS_CH(DT, HR, true); else LNCS_CH(DT, HR, false); } while (0)
    if (dt == DT_F32) { if (has_res) LNCS_D(DT_F32, true); else LNCS_D(DT_F32, false); }
    else              { if (has_res) LNCS_D(DT_BF16, true); else LNCS_D(DT_BF16, false); }
#undef LNCS_D
#undef LNCS_CH
#undef LNCS
    const int64_t nslabs = (int64_t)grid;  // one slab per block
    dim3 g2((unsigned)((cols + 15) / 16));
    if (dt == DT_BF16)
      hipLaunchKernelGGL((ln_bwd_wb_final_kernel<256, DT_BF16>), g2, dim3(1024),
                         0, s, (const float*)scratch, (void*)dw, (void*)db,
                         cols, nslabs);
    else
      hipLaunchKernelGGL((ln_bwd_wb_final_kernel<256, DT_F32>), g2, dim3(1024),
                         0, s, (const float*)scratch, (void*)dw, (void*)db,
                         cols, nslabs);
    LAUNCH_CHECK();
    return 0;
  }
  if (ln_fast_ok(cols)) {
    unsigned grid = (unsigned)((rows + 3) / 4);
    if (grid > 8192u) grid = 8192u;
    const int ch = (int)((cols / 8 + WAVE - 1) / WAVE);
#define LNBV(DT, HR, DR, CH)                                                   \
  hipLaunchKernelGGL((ln_bwd_dx_vec_kernel<DT, HR, DR, CH>), dim3(grid),       \
                     dim3(256), 0, s, (const void*)dy, (const void*)x,         \
                     (const void*)res, (const void*)w, (const float*)mean,     \
                     (const float*)rstd, (void*)dx, (void*)dres, rows, cols,   \
                     keep, salt, (const unsigned long long*)state)
#define LNBV_CH(DT, HR, DR)                                                    \
  do {                                                                         \
    if (ch <= 2) LNBV(DT, HR, DR, 2);                                          \
    else if (ch <= 4) LNBV(DT, HR, DR, 4);                                     \
    else LNBV(DT, HR, DR, 8);                                                  \
  } while (0)
#define LNBV_D(DT, HR)                                                         \
  do { if (drop) LNBV_CH(DT, HR, true); else LNBV_CH(DT, HR, false); } while (0)
    if (dt == DT_F32) { if (has_res) LNBV_D(DT_F32, true); else LNBV_D(DT_F32, false); }
    else              { if (has_res) LNBV_D(DT_BF16, true); else LNBV_D(DT_BF16, false); }
#undef LNBV_D
#undef LNBV_CH
#undef LNBV
  } else {
    dim3 grid((unsigned)(rows < 4096 ? rows : 4096));
#define LNBD(DT, HR)                                                           \
  hipLaunchKernelGGL((ln_bwd_dx_kernel<DT, 256, HR>), grid, dim3(256), 0, s,   \
                     (const void*)dy, (const void*)x, (const void*)res,        \
                     (const void*)w, (const float*)mean, (const float*)rstd,   \
                     (void*)dx, rows, cols)
    if (dt == DT_F32) { if (has_res) LNBD(DT_F32, true); else LNBD(DT_F32, false); }
    else              { if (has_res) LNBD(DT_BF16, true); else LNBD(DT_BF16, false); }
#undef LNBD
  }
  if (cols % 8 == 0 && scratch != 0) {
    constexpr int BLOCK = 128;
    const int64_t cols8 = cols / 8;
    // ~1024 workgroups total (see launch_colsum): more slabs only where
    // the column dimension is too narrow to fill the chip
    const int64_t gx = (cols8 + BLOCK - 1) / BLOCK;
    int64_t nslabs = 1024 / gx;
    if (nslabs < 128) nslabs = 128;
    if (nslabs > LN_SLABS) nslabs = LN_SLABS;
    if (nslabs > rows) nslabs = rows;
    const int64_t slab = (rows + nslabs - 1) / nslabs;
    dim3 grid((unsigned)gx, (unsigned)nslabs);
#define LNWBP(DT, HR, DR)                                                      \
  hipLaunchKernelGGL((ln_bwd_wb_part_kernel<DT, HR, DR, BLOCK>), grid,         \
                     dim3(BLOCK), 0, s, (const void*)dy, (const void*)x,       \
                     (const void*)res, (const float*)mean,                     \
                     (const float*)rstd, (float*)scratch, rows, cols8, slab,   \
                     keep, salt, (const unsigned long long*)state)
#define LNWBP_D(DT, HR)                                                        \
  do { if (drop) LNWBP(DT, HR, true); else LNWBP(DT, HR, false); } while (0)
    if (dt == DT_F32) { if (has_res) LNWBP_D(DT_F32, true); else LNWBP_D(DT_F32, false); }
    else              { if (has_res) LNWBP_D(DT_BF16, true); else LNWBP_D(DT_BF16, false); }
#undef LNWBP_D
#undef LNWBP
    dim3 g2((unsigned)((cols + 15) / 16));
    if (dt == DT_BF16)
      hipLaunchKernelGGL((ln_bwd_wb_final_kernel<256, DT_BF16>), g2, dim3(1024),
                         0, s, (const float*)scratch, (void*)dw, (void*)db,
                         cols, nslabs);
    else
      hipLaunchKernelGGL((ln_bwd_wb_final_kernel<256, DT_F32>), g2, dim3(1024),
                         0, s, (const float*)scratch, (void*)dw, (void*)db,
                         cols, nslabs);
  } else {
    constexpr int BLOCK = 256;
    // slab sized so the grid fills 256 CUs (cols/256 col-blocks * row-slabs)
    int64_t slab = 64;
    while ((cols + BLOCK - 1) / BLOCK * ((rows + slab - 1) / slab) > 2048 && slab < rows)
      slab *= 2;
    dim3 grid((unsigned)((cols + BLOCK - 1) / BLOCK),
              (unsigned)((rows + slab - 1) / slab));
#define LNWB(DT, HR)                                                           \
  hipLaunchKernelGGL((ln_bwd_wb_kernel<DT, HR, BLOCK>), grid, dim3(BLOCK), 0,  \
                     s, (const void*)dy, (const void*)x, (const void*)res,     \
                     (const float*)mean, (const float*)rstd, (float*)dw,       \
                     (float*)db, rows, cols, slab)
    if (dt == DT_F32) { if (has_res) LNWB(DT_F32, true); else LNWB(DT_F32, false); }
    else              { if (has_res) LNWB(DT_BF16, true); else LNWB(DT_BF16, false); }
#undef LNWB
  }
  LAUNCH_CHECK();
  return 0;
}
