// Fused scale + additive-mask + row softmax (fwd/bwd) for gfx950.
//
// Replaces the reference's eager mask-add + nn.Softmax over attention
// scores (reference: scaelum/model/bert_layers.py:259-269). Scores
// [B, h, Sq, Sk]; mask [B, 1, 1, Sk] additive (built by BertEmbeddings).
//
// Fast path (Sk % 8 == 0, Sk <= 512): SUBWAVE-PER-ROW — a row is owned by
// 16/32/64 lanes each holding one 16 B vector (8 elems) in registers, so a
// wave processes up to 4 rows with full lane utilization at Sk=128 (the
// BERT-MNLI shape). Reductions are subwave shuffles; one global read.
// Fallback: wave-per-row strided re-read (rows are L2-resident).

#include "common.h"

template <int SUBW>
DEV float subwave_max(float v) {
#pragma unroll
  for (int off = SUBW / 2; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}
template <int SUBW>
DEV float subwave_sum(float v) {
#pragma unroll
  for (int off = SUBW / 2; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

template <int DT, bool HAS_MASK, int SUBW>
__global__ __launch_bounds__(256) void softmax_fwd_sub_kernel(
    const void* __restrict__ scores, const void* __restrict__ mask,
    void* __restrict__ probs, int64_t nrows, int64_t hSq, int64_t cols8,
    float scale) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int sl = lane & (SUBW - 1);          // lane within subgroup
  const int sub = lane / SUBW;               // subgroup within wave
  const int wid = threadIdx.x / WAVE;
  constexpr int RPW = WAVE / SUBW;           // rows per wave
  const int64_t row = ((int64_t)blockIdx.x * 4 + wid) * RPW + sub;
  if (row >= nrows) return;
  const int64_t base8 = row * cols8;
  const int64_t mbase8 = (row / hSq) * cols8;
  float z[8];
  float m = -3.4e38f;
  const bool active = sl < cols8;
  if (active) {
    Vec8<DT>::load(scores, base8 + sl, z);
    if (HAS_MASK) {
      float mv[8];
      Vec8<DT>::load(mask, mbase8 + sl, mv);
#pragma unroll
      for (int j = 0; j < 8; ++j) z[j] = z[j] * scale + mv[j];
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) z[j] *= scale;
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) m = fmaxf(m, z[j]);
  }
  m = subwave_max<SUBW>(m);
  float s = 0.f;
  if (active) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      z[j] = __expf(z[j] - m);
      s += z[j];
    }
  }
  s = subwave_sum<SUBW>(s);
  if (active) {
    const float inv = 1.f / s;
#pragma unroll
    for (int j = 0; j < 8; ++j) z[j] *= inv;
    Vec8<DT>::store(probs, base8 + sl, z);
  }
}

template <int DT, bool HAS_MASK>
__global__ __launch_bounds__(256) void softmax_fwd_kernel(
    const void* __restrict__ scores, const void* __restrict__ mask,
    void* __restrict__ probs, int64_t nrows, int64_t hSq, int64_t Sk,
    float scale) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int64_t row = (int64_t)blockIdx.x * 4 + wid;
  if (row >= nrows) return;
  const int64_t base = row * Sk;
  const int64_t mbase = (row / hSq) * Sk;
  float m = -3.4e38f;
  for (int64_t k = lane; k < Sk; k += WAVE) {
    float v = load_elem<DT>(scores, base + k) * scale;
    if (HAS_MASK) v += load_elem<DT>(mask, mbase + k);
    m = fmaxf(m, v);
  }
  m = wave_max(m);
  float s = 0.f;
  for (int64_t k = lane; k < Sk; k += WAVE) {
    float v = load_elem<DT>(scores, base + k) * scale;
    if (HAS_MASK) v += load_elem<DT>(mask, mbase + k);
    s += __expf(v - m);
  }
  s = wave_sum(s);
  const float inv = 1.f / s;
  for (int64_t k = lane; k < Sk; k += WAVE) {
    float v = load_elem<DT>(scores, base + k) * scale;
    if (HAS_MASK) v += load_elem<DT>(mask, mbase + k);
    store_elem<DT>(probs, base + k, __expf(v - m) * inv);
  }
}

SKY_EXPORT int sky_masked_softmax_fwd(uint64_t stream, uint64_t scores,
                                      uint64_t mask, uint64_t probs, int64_t B,
                                      int64_t h, int64_t Sq, int64_t Sk,
                                      float scale, float keep, uint64_t seed,
                                      int dt) {
  (void)keep; (void)seed;  // dropout handled by sky_dropout_* (separate op)
  hipStream_t s = (hipStream_t)stream;
  const int64_t nrows = B * h * Sq;
  const int64_t hSq = h * Sq;
  bool hm = mask != 0;
  if (Sk % 8 == 0 && Sk <= 512) {
    const int64_t cols8 = Sk / 8;
    int subw = cols8 <= 16 ? 16 : (cols8 <= 32 ? 32 : 64);
    const int rpw = WAVE / subw;
    unsigned grid = (unsigned)((nrows + 4 * rpw - 1) / (4 * rpw));
#define SMS(DT, HM, SW)                                                        \
  hipLaunchKernelGGL((softmax_fwd_sub_kernel<DT, HM, SW>), dim3(grid),         \
                     dim3(256), 0, s, (const void*)scores, (const void*)mask,  \
                     (void*)probs, nrows, hSq, cols8, scale)
#define SMS_SW(DT, HM)                                                         \
  do {                                                                         \
    if (subw == 16) SMS(DT, HM, 16);                                           \
    else if (subw == 32) SMS(DT, HM, 32);                                      \
    else SMS(DT, HM, 64);                                                      \
  } while (0)
    if (dt == DT_F32) { if (hm) SMS_SW(DT_F32, true); else SMS_SW(DT_F32, false); }
    else              { if (hm) SMS_SW(DT_BF16, true); else SMS_SW(DT_BF16, false); }
#undef SMS_SW
#undef SMS
  } else {
    unsigned grid = (unsigned)((nrows + 3) / 4);
#define SMF(DT, HM)                                                            \
  hipLaunchKernelGGL((softmax_fwd_kernel<DT, HM>), dim3(grid), dim3(256), 0,   \
                     s, (const void*)scores, (const void*)mask, (void*)probs,  \
                     nrows, hSq, Sk, scale)
    if (dt == DT_F32) { if (hm) SMF(DT_F32, true); else SMF(DT_F32, false); }
    else              { if (hm) SMF(DT_BF16, true); else SMF(DT_BF16, false); }
#undef SMF
  }
  LAUNCH_CHECK();
  return 0;
}

// backward: ds = scale * p * (dp - sum_k dp*p)

template <int DT, int SUBW>
__global__ __launch_bounds__(256) void softmax_bwd_sub_kernel(
    const void* __restrict__ dp, const void* __restrict__ probs,
    void* __restrict__ ds, int64_t nrows, int64_t cols8, float scale) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int sl = lane & (SUBW - 1);
  const int sub = lane / SUBW;
  const int wid = threadIdx.x / WAVE;
  constexpr int RPW = WAVE / SUBW;
  const int64_t row = ((int64_t)blockIdx.x * 4 + wid) * RPW + sub;
  if (row >= nrows) return;
  const int64_t base8 = row * cols8;
  const bool active = sl < cols8;
  float p[8], d[8];
  float dot = 0.f;
  if (active) {
    Vec8<DT>::load(probs, base8 + sl, p);
    Vec8<DT>::load(dp, base8 + sl, d);
#pragma unroll
    for (int j = 0; j < 8; ++j) dot += p[j] * d[j];
  }
  dot = subwave_sum<SUBW>(dot);
  if (active) {
#pragma unroll
    for (int j = 0; j < 8; ++j) d[j] = scale * p[j] * (d[j] - dot);
    Vec8<DT>::store(ds, base8 + sl, d);
  }
}

template <int DT>
__global__ __launch_bounds__(256) void softmax_bwd_kernel(
    const void* __restrict__ dp, const void* __restrict__ probs,
    void* __restrict__ ds, int64_t nrows, int64_t Sk, float scale) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int64_t row = (int64_t)blockIdx.x * 4 + wid;
  if (row >= nrows) return;
  const int64_t base = row * Sk;
  float dot = 0.f;
  for (int64_t k = lane; k < Sk; k += WAVE)
    dot += load_elem<DT>(dp, base + k) * load_elem<DT>(probs, base + k);
  dot = wave_sum(dot);
  for (int64_t k = lane; k < Sk; k += WAVE) {
    float p = load_elem<DT>(probs, base + k);
    float d = load_elem<DT>(dp, base + k);
    store_elem<DT>(ds, base + k, scale * p * (d - dot));
  }
}

SKY_EXPORT int sky_masked_softmax_bwd(uint64_t stream, uint64_t dp,
                                      uint64_t probs, uint64_t ds, int64_t B,
                                      int64_t h, int64_t Sq, int64_t Sk,
                                      float scale, float keep, uint64_t seed,
                                      int dt) {
  (void)keep; (void)seed;
  hipStream_t s = (hipStream_t)stream;
  const int64_t nrows = B * h * Sq;
  if (Sk % 8 == 0 && Sk <= 512) {
    const int64_t cols8 = Sk / 8;
    int subw = cols8 <= 16 ? 16 : (cols8 <= 32 ? 32 : 64);
    const int rpw = WAVE / subw;
    unsigned grid = (unsigned)((nrows + 4 * rpw - 1) / (4 * rpw));
#define SMB(DT, SW)                                                            \
  hipLaunchKernelGGL((softmax_bwd_sub_kernel<DT, SW>), dim3(grid), dim3(256),  \
                     0, s, (const void*)dp, (const void*)probs, (void*)ds,     \
                     nrows, cols8, scale)
#define SMB_SW(DT)                                                             \
  do {                                                                         \
    if (subw == 16) SMB(DT, 16);                                               \
    else if (subw == 32) SMB(DT, 32);                                          \
    else SMB(DT, 64);                                                          \
  } while (0)
    if (dt == DT_F32) SMB_SW(DT_F32); else SMB_SW(DT_BF16);
#undef SMB_SW
#undef SMB
  } else {
    unsigned grid = (unsigned)((nrows + 3) / 4);
    if (dt == DT_F32)
      hipLaunchKernelGGL((softmax_bwd_kernel<DT_F32>), dim3(grid), dim3(256), 0, s,
                         (const void*)dp, (const void*)probs, (void*)ds, nrows, Sk, scale);
    else
      hipLaunchKernelGGL((softmax_bwd_kernel<DT_BF16>), dim3(grid), dim3(256), 0, s,
                         (const void*)dp, (const void*)probs, (void*)ds, nrows, Sk, scale);
  }
  LAUNCH_CHECK();
  return 0;
}
