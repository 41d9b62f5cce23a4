// Fused bias+GELU and counter-RNG dropout for gfx950.
//
// bias_gelu replaces the reference's eager bias_gelu / LinearActivation
// epilogue (reference: scaelum/model/bert_layers.py:21-44,60-108) — on GPU
// it runs as the epilogue pass after the hipBLASLt GEMM. dropout
// regenerates its keep-mask from (seed, index) in backward (no stored
// mask). All memory-bound: 16 B/lane vectorized fast paths with scalar
// fallbacks for odd shapes; dbias is a separate column-parallel reduction
// (coalesced down-column walk, few atomics).

#include "common.h"

// ---------------- bias_gelu forward ----------------

template <int DT>
__global__ __launch_bounds__(256) void bias_gelu_fwd_vec_kernel(
    const void* __restrict__ x, const void* __restrict__ b,
    void* __restrict__ y, int64_t n8, int64_t cols8) {
  for (int64_t i8 = (int64_t)blockIdx.x * 256 + threadIdx.x; i8 < n8;
       i8 += (int64_t)gridDim.x * 256) {
    float v[8], bv[8];
    Vec8<DT>::load(x, i8, v);
    Vec8<DT>::load(b, i8 % cols8, bv);
#pragma unroll
    for (int j = 0; j < 8; ++j) v[j] = gelu_f(v[j] + bv[j]);
    Vec8<DT>::store(y, i8, v);
  }
}

template <int DT>
__global__ __launch_bounds__(256) void bias_gelu_fwd_kernel(
    const void* __restrict__ x, const void* __restrict__ b,
    void* __restrict__ y, int64_t n, int64_t cols) {
  for (int64_t i = (int64_t)blockIdx.x * 256 + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * 256) {
    store_elem<DT>(y, i, gelu_f(load_elem<DT>(x, i) + load_elem<DT>(b, i % cols)));
  }
}

SKY_EXPORT int sky_bias_gelu_fwd(uint64_t stream, uint64_t x, uint64_t b,
                                 uint64_t y, int64_t rows, int64_t cols,
                                 int dt) {
  hipStream_t s = (hipStream_t)stream;
  int64_t n = rows * cols;
  if (cols % 8 == 0) {
    int64_t n8 = n / 8;
    unsigned grid = (unsigned)((n8 + 255) / 256);
    if (grid > 2048u) grid = 2048u;
    if (dt == DT_F32)
      hipLaunchKernelGGL((bias_gelu_fwd_vec_kernel<DT_F32>), dim3(grid), dim3(256), 0, s,
                         (const void*)x, (const void*)b, (void*)y, n8, cols / 8);
    else
      hipLaunchKernelGGL((bias_gelu_fwd_vec_kernel<DT_BF16>), dim3(grid), dim3(256), 0, s,
                         (const void*)x, (const void*)b, (void*)y, n8, cols / 8);
  } else {
    unsigned grid = (unsigned)((n + 255) / 256);
    if (grid > 2048u) grid = 2048u;
    if (dt == DT_F32)
      hipLaunchKernelGGL((bias_gelu_fwd_kernel<DT_F32>), dim3(grid), dim3(256), 0, s,
                         (const void*)x, (const void*)b, (void*)y, n, cols);
    else
      hipLaunchKernelGGL((bias_gelu_fwd_kernel<DT_BF16>), dim3(grid), dim3(256), 0, s,
                         (const void*)x, (const void*)b, (void*)y, n, cols);
  }
  LAUNCH_CHECK();
  return 0;
}

// ---------------- bias_gelu backward ----------------
// dx = dy * gelu'(x+b), then db[c] = sum_r dx[r,c] via column reduction.
// With b == nullptr, x is already the pre-activation (the hipBLASLt
// GELU_AUX_BIAS epilogue saved it) and the add is skipped.

template <int DT, bool HAS_B>
__global__ __launch_bounds__(256) void bias_gelu_bwd_dx_vec_kernel(
    const void* __restrict__ dy, const void* __restrict__ x,
    const void* __restrict__ b, void* __restrict__ dx, int64_t n8,
    int64_t cols8) {
  for (int64_t i8 = (int64_t)blockIdx.x * 256 + threadIdx.x; i8 < n8;
       i8 += (int64_t)gridDim.x * 256) {
    float v[8], bv[8], d[8];
    Vec8<DT>::load(x, i8, v);
    if (HAS_B) Vec8<DT>::load(b, i8 % cols8, bv);
    Vec8<DT>::load(dy, i8, d);
#pragma unroll
    for (int j = 0; j < 8; ++j) d[j] *= gelu_grad_f(HAS_B ? v[j] + bv[j] : v[j]);
    Vec8<DT>::store(dx, i8, d);
  }
}

template <int DT, bool HAS_B>
__global__ __launch_bounds__(256) void bias_gelu_bwd_dx_kernel(
    const void* __restrict__ dy, const void* __restrict__ x,
    const void* __restrict__ b, void* __restrict__ dx, int64_t n,
    int64_t cols) {
  for (int64_t i = (int64_t)blockIdx.x * 256 + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * 256) {
    float z = load_elem<DT>(x, i) + (HAS_B ? load_elem<DT>(b, i % cols) : 0.f);
    store_elem<DT>(dx, i, load_elem<DT>(dy, i) * gelu_grad_f(z));
  }
}

// fused dx + db column partials: computes dx = dy * gelu'(pre) AND
// accumulates its per-slab column sums in the same pass, so db needs no
// re-read of the 32 MB dx tensor (stage 2 = colsum_final over scratch).
// Same (column-block x row-slab) grid as colsum_part_kernel.
template <int DT, bool HAS_B, int BLOCK>
__global__ __launch_bounds__(BLOCK) void bias_gelu_bwd_part_kernel(
    const void* __restrict__ dy, const void* __restrict__ x,
    const void* __restrict__ b, void* __restrict__ dx,
    float* __restrict__ scratch, int64_t rows, int64_t cols8,
    int64_t rows_per_slab) {
  const int64_t c8 = (int64_t)blockIdx.x * BLOCK + threadIdx.x;
  if (c8 >= cols8) return;
  const int64_t slab = blockIdx.y;
  const int64_t r0 = slab * rows_per_slab;
  const int64_t r1 = min(rows, r0 + rows_per_slab);
  float bv[8];
  if (HAS_B) Vec8<DT>::load(b, c8, bv);
  float s[8] = {0.f};
  for (int64_t r = r0; r < r1; ++r) {
    float v[8], d[8];
    Vec8<DT>::load(x, r * cols8 + c8, v);
    Vec8<DT>::load(dy, r * cols8 + c8, d);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      d[j] *= gelu_grad_f(HAS_B ? v[j] + bv[j] : v[j]);
      s[j] += d[j];
    }
    Vec8<DT>::store(dx, r * cols8 + c8, d);
  }
#pragma unroll
  for (int j = 0; j < 8; ++j) scratch[(slab * cols8 + c8) * 8 + j] = s[j];
}

// Streaming fused dx + db partials (v2): unlike bias_gelu_bwd_part_kernel
// (column-parallel, 2 KB-strided main traffic — measured SLOWER in-app,
// r01), this keeps dx_vec's large sequential reads: each block owns a
// (row-slab x 256-c8 column window) tile, reads/writes 4 KB strips per
// row, and every THREAD keeps its fixed column-slice partial in registers
// — db costs one fp32 slab write instead of a 32 MB re-read of dx.
// Requires cols8 % 256 == 0 (cols % 2048 == 0; the FFN width).
template <int DT, bool HAS_B>
__global__ __launch_bounds__(256) void bias_gelu_bwd_cs_kernel(
    const void* __restrict__ dy, const void* __restrict__ x,
    const void* __restrict__ b, void* __restrict__ dx,
    float* __restrict__ scratch, int64_t rows, int64_t cols8,
    int64_t rows_per_slab) {
  const int64_t nwin = cols8 >> 8;  // 256-c8 windows per row
  const int64_t win = blockIdx.x % nwin;
  const int64_t slab = blockIdx.x / nwin;
  const int64_t c8 = win * 256 + threadIdx.x;
  const int64_t r0 = slab * rows_per_slab;
  const int64_t r1 = min(rows, r0 + rows_per_slab);
  float bv[8];
  if (HAS_B) Vec8<DT>::load(b, c8, bv);
  float s[8] = {0.f};
  for (int64_t r = r0; r < r1; ++r) {
    float v[8], d[8];
    Vec8<DT>::load(x, r * cols8 + c8, v);
    Vec8<DT>::load(dy, r * cols8 + c8, d);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      d[j] *= gelu_grad_f(HAS_B ? v[j] + bv[j] : v[j]);
      s[j] += d[j];
    }
    Vec8<DT>::store(dx, r * cols8 + c8, d);
  }
#pragma unroll
  for (int j = 0; j < 8; ++j) scratch[(slab * cols8 + c8) * 8 + j] = s[j];
}

// db[c] = sum over rows of dx[r, c] — two-stage, no atomics, outputs need
// no zero-init: stage 1 writes per-slab partials [nslabs][cols] fp32 to a
// scratch buffer (thread owns 8 consecutive columns, 16 B loads); stage 2
// reduces the slab axis and WRITES the result.
#define CS_SLABS 1024

template <int DT, int BLOCK>
__global__ __launch_bounds__(BLOCK) void colsum_part_kernel(
    const void* __restrict__ src, float* __restrict__ scratch, int64_t rows,
    int64_t cols8, int64_t rows_per_slab) {
  const int64_t c8 = (int64_t)blockIdx.x * BLOCK + threadIdx.x;
  if (c8 >= cols8) return;
  const int64_t slab = blockIdx.y;
  const int64_t r0 = slab * rows_per_slab;
  const int64_t r1 = min(rows, r0 + rows_per_slab);
  float s[8] = {0.f};
  for (int64_t r = r0; r < r1; ++r) {
    float v[8];
    Vec8<DT>::load(src, r * cols8 + c8, v);
#pragma unroll
    for (int j = 0; j < 8; ++j) s[j] += v[j];
  }
#pragma unroll
  for (int j = 0; j < 8; ++j) scratch[(slab * cols8 + c8) * 8 + j] = s[j];
}

// two-level: a 1024-thread block covers 16 columns x 64 slab-groups (each
// thread sums nslabs/64 slabs of one column), parallel LDS tree, one write
// per column. 16 waves/block keeps the chip's XCDs busy even though cols
// only yields cols/16 blocks.
template <int BLOCK, int DTOUT>
__global__ __launch_bounds__(1024) void colsum_final_kernel(
    const float* __restrict__ scratch, void* __restrict__ out, int64_t cols,
    int64_t nslabs) {
  __shared__ float lds[64][17];
  const int c = threadIdx.x & 15;
  const int g = threadIdx.x >> 4;  // 0..63
  const int64_t col = (int64_t)blockIdx.x * 16 + c;
  const int64_t per = (nslabs + 63) / 64;
  float s = 0.f;
  if (col < cols) {
    const int64_t y1 = min(nslabs, (int64_t)(g + 1) * per);
    for (int64_t y = (int64_t)g * per; y < y1; ++y)
      s += scratch[y * cols + col];
  }
  lds[g][c] = s;
  __syncthreads();
  for (int st = 32; st >= 1; st >>= 1) {
    if (g < st) lds[g][c] += lds[g + st][c];
    __syncthreads();
  }
  if (g == 0 && col < cols) store_elem<DTOUT>(out, col, lds[0][c]);
}

template <int DT, int BLOCK>
__global__ __launch_bounds__(BLOCK) void colsum_kernel(
    const void* __restrict__ src, float* __restrict__ out, int64_t rows,
    int64_t cols, int64_t rows_per_slab) {
  const int64_t col = (int64_t)blockIdx.x * BLOCK + threadIdx.x;
  if (col >= cols) return;
  const int64_t r0 = (int64_t)blockIdx.y * rows_per_slab;
  const int64_t r1 = min(rows, r0 + rows_per_slab);
  float s = 0.f;
  for (int64_t r = r0; r < r1; ++r) s += load_elem<DT>(src, r * cols + col);
  atomicAdd(&out[col], s);
}

// scratch must hold CS_SLABS * cols floats when the fast path is taken
// (cols % 8 == 0); pass scratch = nullptr to force the atomic fallback
// (out must then be pre-zeroed).
template <int DT>
static void launch_colsum(hipStream_t s, const void* src, void* out,
                          float* scratch, int64_t rows, int64_t cols,
                          int dtout) {
  if (cols % 8 == 0 && scratch != nullptr) {
    constexpr int BLOCK = 128;
    const int64_t cols8 = cols / 8;
    // adapt the slab count to the column-block count: ~1024 workgroups
    // total fills the chip without inflating scratch traffic on wide
    // matrices (which already parallelize across columns)
    const int64_t gx = (cols8 + BLOCK - 1) / BLOCK;
    int64_t nslabs = 1024 / gx;
    if (nslabs < 128) nslabs = 128;
    if (nslabs > CS_SLABS) nslabs = CS_SLABS;
    if (nslabs > rows) nslabs = rows;
    const int64_t slab = (rows + nslabs - 1) / nslabs;
    dim3 grid((unsigned)gx, (unsigned)nslabs);
    hipLaunchKernelGGL((colsum_part_kernel<DT, BLOCK>), grid, dim3(BLOCK), 0,
                       s, src, scratch, rows, cols8, slab);
    dim3 g2((unsigned)((cols + 15) / 16));
    if (dtout == DT_BF16)
      hipLaunchKernelGGL((colsum_final_kernel<256, DT_BF16>), g2, dim3(1024), 0,
                         s, scratch, out, cols, nslabs);
    else
      hipLaunchKernelGGL((colsum_final_kernel<256, DT_F32>), g2, dim3(1024), 0,
                         s, scratch, out, cols, nslabs);
    return;
  }
  constexpr int BLOCK = 256;
  int64_t slab = 64;
  while ((cols + BLOCK - 1) / BLOCK * ((rows + slab - 1) / slab) > 2048 && slab < rows)
    slab *= 2;
  dim3 grid((unsigned)((cols + BLOCK - 1) / BLOCK),
            (unsigned)((rows + slab - 1) / slab));
  (void)dtout;  // atomic fallback accumulates fp32 into `out`
  hipLaunchKernelGGL((colsum_kernel<DT, BLOCK>), grid, dim3(BLOCK), 0, s, src,
                     (float*)out, rows, cols, slab);
}

SKY_EXPORT int sky_colsum(uint64_t stream, uint64_t src, uint64_t out,
                          uint64_t scratch, int64_t rows, int64_t cols,
                          int dt, int dtout) {
  hipStream_t s = (hipStream_t)stream;
  if (dt == DT_F32)
    launch_colsum<DT_F32>(s, (const void*)src, (void*)out, (float*)scratch, rows, cols, dtout);
  else
    launch_colsum<DT_BF16>(s, (const void*)src, (void*)out, (float*)scratch, rows, cols, dtout);
  LAUNCH_CHECK();
  return 0;
}

SKY_EXPORT int sky_bias_gelu_bwd(uint64_t stream, uint64_t dy, uint64_t x,
                                 uint64_t b, uint64_t dx, uint64_t db,
                                 uint64_t scratch, int64_t rows, int64_t cols,
                                 int dt) {
  hipStream_t s = (hipStream_t)stream;
  int64_t n = rows * cols;
  const bool hb = b != 0;
  // The fused one-pass variant measures FASTER standalone but SLOWER
  // in-app (+4.9 us/layer vs the dx + two-stage-colsum chain at the bench
  // shape — profiles r01_bench160 diff); the proven chain is the default
  // and the fused kernel stays selectable for re-measurement.
  // v2 streaming fused dx+db (see bias_gelu_bwd_cs_kernel); default where
  // the shape allows, SKY_GELU_SPLIT_DB=1 restores the dx + colsum chain.
  if (cols % 2048 == 0 && scratch != 0 && !getenv("SKY_GELU_SPLIT_DB")) {
    const int64_t cols8 = cols / 8;
    const int64_t nwin = cols8 >> 8;
    // ~1024 blocks feeds the streaming; fewer slabs shrink the final's
    // scratch re-read (A/B'd: 1024 slabs put 16 MB on the final)
    int64_t nslabs = 1024 / nwin;
    if (const char* e = getenv("SKY_GELU_CS_SLABS")) nslabs = atoll(e);
    if (nslabs > CS_SLABS) nslabs = CS_SLABS;
    if (nslabs > rows) nslabs = rows;
    const int64_t slab = (rows + nslabs - 1) / nslabs;
    dim3 grid((unsigned)(nwin * nslabs));
    auto launch_cs = [&](auto kern) {
      hipLaunchKernelGGL(kern, grid, dim3(256), 0, s, (const void*)dy,
                         (const void*)x, (const void*)b, (void*)dx,
                         (float*)scratch, rows, cols8, slab);
    };
    if (dt == DT_F32)
      hb ? launch_cs(bias_gelu_bwd_cs_kernel<DT_F32, true>)
         : launch_cs(bias_gelu_bwd_cs_kernel<DT_F32, false>);
    else
      hb ? launch_cs(bias_gelu_bwd_cs_kernel<DT_BF16, true>)
         : launch_cs(bias_gelu_bwd_cs_kernel<DT_BF16, false>);
    dim3 g2((unsigned)((cols + 15) / 16));
    if (dt == DT_BF16)
      hipLaunchKernelGGL((colsum_final_kernel<256, DT_BF16>), g2, dim3(1024),
                         0, s, (float*)scratch, (void*)db, cols, nslabs);
    else
      hipLaunchKernelGGL((colsum_final_kernel<256, DT_F32>), g2, dim3(1024),
                         0, s, (float*)scratch, (void*)db, cols, nslabs);
    LAUNCH_CHECK();
    return 0;
  }
  if (cols % 8 == 0 && scratch != 0 && getenv("SKY_GELU_FUSED_BWD")) {
    // fused one-pass dx + db partials (see bias_gelu_bwd_part_kernel).
    // This kernel moves 3 tensors (dy, x read; dx write), so unlike the
    // read-only colsum part it wants FULL thread count (~4096 waves, short
    // row chains); the extra scratch volume is small against 96 MB of
    // main traffic.
    constexpr int BLOCK = 128;
    const int64_t cols8 = cols / 8;
    const int64_t gx = (cols8 + BLOCK - 1) / BLOCK;
    int64_t nslabs = 4096 / gx;
    if (nslabs < 128) nslabs = 128;
    if (nslabs > CS_SLABS) nslabs = CS_SLABS;
    if (nslabs > rows) nslabs = rows;
    const int64_t slab = (rows + nslabs - 1) / nslabs;
    dim3 grid((unsigned)gx, (unsigned)nslabs);
    auto launch_part = [&](auto kern) {
      hipLaunchKernelGGL(kern, grid, dim3(BLOCK), 0, s, (const void*)dy,
                         (const void*)x, (const void*)b, (void*)dx,
                         (float*)scratch, rows, cols8, slab);
    };
    if (dt == DT_F32)
      hb ? launch_part(bias_gelu_bwd_part_kernel<DT_F32, true, BLOCK>)
         : launch_part(bias_gelu_bwd_part_kernel<DT_F32, false, BLOCK>);
    else
      hb ? launch_part(bias_gelu_bwd_part_kernel<DT_BF16, true, BLOCK>)
         : launch_part(bias_gelu_bwd_part_kernel<DT_BF16, false, BLOCK>);
    dim3 g2((unsigned)((cols + 15) / 16));
    if (dt == DT_BF16)
      hipLaunchKernelGGL((colsum_final_kernel<256, DT_BF16>), g2, dim3(1024),
                         0, s, (float*)scratch, (void*)db, cols, nslabs);
    else
      hipLaunchKernelGGL((colsum_final_kernel<256, DT_F32>), g2, dim3(1024),
                         0, s, (float*)scratch, (void*)db, cols, nslabs);
    LAUNCH_CHECK();
    return 0;
  }
  if (cols % 8 == 0) {
    int64_t n8 = n / 8;
    unsigned grid = (unsigned)((n8 + 255) / 256);
    if (grid > 2048u) grid = 2048u;
    auto launch_vec = [&](auto kern) {
      hipLaunchKernelGGL(kern, dim3(grid), dim3(256), 0, s, (const void*)dy,
                         (const void*)x, (const void*)b, (void*)dx, n8, cols / 8);
    };
    if (dt == DT_F32)
      hb ? launch_vec(bias_gelu_bwd_dx_vec_kernel<DT_F32, true>)
         : launch_vec(bias_gelu_bwd_dx_vec_kernel<DT_F32, false>);
    else
      hb ? launch_vec(bias_gelu_bwd_dx_vec_kernel<DT_BF16, true>)
         : launch_vec(bias_gelu_bwd_dx_vec_kernel<DT_BF16, false>);
  } else {
    unsigned grid = (unsigned)((n + 255) / 256);
    if (grid > 2048u) grid = 2048u;
    auto launch_sc = [&](auto kern) {
      hipLaunchKernelGGL(kern, dim3(grid), dim3(256), 0, s, (const void*)dy,
                         (const void*)x, (const void*)b, (void*)dx, n, cols);
    };
    if (dt == DT_F32)
      hb ? launch_sc(bias_gelu_bwd_dx_kernel<DT_F32, true>)
         : launch_sc(bias_gelu_bwd_dx_kernel<DT_F32, false>);
    else
      hb ? launch_sc(bias_gelu_bwd_dx_kernel<DT_BF16, true>)
         : launch_sc(bias_gelu_bwd_dx_kernel<DT_BF16, false>);
  }
  if (dt == DT_F32)
    launch_colsum<DT_F32>(s, (const void*)dx, (void*)db, (float*)scratch, rows, cols, dt);
  else
    launch_colsum<DT_BF16>(s, (const void*)dx, (void*)db, (float*)scratch, rows, cols, dt);
  LAUNCH_CHECK();
  return 0;
}

// ---------------- dropout ----------------
// RNG indexed by ABSOLUTE element index so forward and backward agree for
// any vectorization.

// The effective RNG stream is (salt + device step counter): the counter is
// bumped ON DEVICE by sky_rng_tick at the start of every training
// iteration, so dropout masks vary across steps even when the launch
// parameters are frozen inside a captured hipGraph (salt alone would be
// constant under replay).

template <int DT>
__global__ __launch_bounds__(256) void dropout_vec_kernel(
    const void* __restrict__ x, void* __restrict__ y, int64_t n8, float keep,
    uint64_t salt, const unsigned long long* __restrict__ state) {
  const float inv_keep = 1.f / keep;
  const unsigned keep16 = keep_to_16(keep);
  const uint64_t seed = salt + (state ? *state : 0ull) * 0xD1B54A32D192ED03ull;
  for (int64_t i8 = (int64_t)blockIdx.x * 256 + threadIdx.x; i8 < n8;
       i8 += (int64_t)gridDim.x * 256) {
    float v[8];
    Vec8<DT>::load(x, i8, v);
    // 8 consecutive elements = exactly 2 hash quads
    const uint64_t z0 = rng_hash(seed, (uint64_t)i8 * 2);
    const uint64_t z1 = rng_hash(seed, (uint64_t)i8 * 2 + 1);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const uint64_t z = j < 4 ? z0 : z1;
      bool k = (unsigned)((z >> (16 * (j & 3))) & 0xFFFFu) < keep16;
      v[j] = k ? v[j] * inv_keep : 0.f;
    }
    Vec8<DT>::store(y, i8, v);
  }
}

template <int DT>
__global__ __launch_bounds__(256) void dropout_kernel(
    const void* __restrict__ x, void* __restrict__ y, int64_t n, float keep,
    uint64_t salt, const unsigned long long* __restrict__ state) {
  const float inv_keep = 1.f / keep;
  const uint64_t seed = salt + (state ? *state : 0ull) * 0xD1B54A32D192ED03ull;
  for (int64_t i = (int64_t)blockIdx.x * 256 + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * 256) {
    float v = load_elem<DT>(x, i);
    bool k = rng_keep16(seed, (uint64_t)i, keep_to_16(keep));
    store_elem<DT>(y, i, k ? v * inv_keep : 0.f);
  }
}

__global__ void rng_tick_kernel(unsigned long long* state) {
  if (threadIdx.x == 0 && blockIdx.x == 0) state[0] += 1;
}

SKY_EXPORT int sky_rng_tick(uint64_t stream, uint64_t state) {
  hipLaunchKernelGGL(rng_tick_kernel, dim3(1), dim3(64), 0,
                     (hipStream_t)stream, (unsigned long long*)state);
  LAUNCH_CHECK();
  return 0;
}

SKY_EXPORT int sky_dropout_fwd(uint64_t stream, uint64_t x, uint64_t y,
                               int64_t n, float keep, uint64_t salt,
                               uint64_t state, int dt) {
  hipStream_t s = (hipStream_t)stream;
  if (n % 8 == 0) {
    int64_t n8 = n / 8;
    unsigned grid = (unsigned)((n8 + 255) / 256);
    if (grid > 2048u) grid = 2048u;
    if (dt == DT_F32)
      hipLaunchKernelGGL((dropout_vec_kernel<DT_F32>), dim3(grid), dim3(256), 0, s,
                         (const void*)x, (void*)y, n8, keep, salt,
                         (const unsigned long long*)state);
    else
      hipLaunchKernelGGL((dropout_vec_kernel<DT_BF16>), dim3(grid), dim3(256), 0, s,
                         (const void*)x, (void*)y, n8, keep, salt,
                         (const unsigned long long*)state);
  } else {
    unsigned grid = (unsigned)((n + 255) / 256);
    if (grid > 2048u) grid = 2048u;
    if (dt == DT_F32)
      hipLaunchKernelGGL((dropout_kernel<DT_F32>), dim3(grid), dim3(256), 0, s,
                         (const void*)x, (void*)y, n, keep, salt,
                         (const unsigned long long*)state);
    else
      hipLaunchKernelGGL((dropout_kernel<DT_BF16>), dim3(grid), dim3(256), 0, s,
                         (const void*)x, (void*)y, n, keep, salt,
                         (const unsigned long long*)state);
  }
  LAUNCH_CHECK();
  return 0;
}

SKY_EXPORT int sky_dropout_bwd(uint64_t stream, uint64_t dy, uint64_t dx,
                               int64_t n, float keep, uint64_t salt,
                               uint64_t state, int dt) {
  // identical math: dx = dy * mask / keep
  return sky_dropout_fwd(stream, dy, dx, n, keep, salt, state, dt);
}
