// Fused scale + additive-mask + row softmax (fwd/bwd) for gfx950.
//
// Replaces the reference's eager mask-add + nn.Softmax over attention
// scores (reference: scaelum/model/bert_layers.py:259-269). Scores
// [B, h, Sq, Sk]; mask [B, 1, 1, Sk] additive (built by BertEmbeddings).
//
// One wave per row (4 rows per 256-thread block): row max / row sum via
// 64-lane shuffles in fp32. For Sk <= REG_MAX*64 each lane caches its
// elements in registers (one global read); longer rows re-read (L2-hot).

#include "common.h"

#define SM_REG_MAX 8  // register-cached path covers Sk <= 512

template <int DT, bool HAS_MASK>
__global__ __launch_bounds__(256) void softmax_fwd_kernel(
    const void* __restrict__ scores, const void* __restrict__ mask,
    void* __restrict__ probs, int64_t B, int64_t h, int64_t Sq, int64_t Sk,
    float scale) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int64_t nrows = B * h * Sq;
  const int64_t row = (int64_t)blockIdx.x * 4 + wid;
  if (row >= nrows) return;
  const int64_t b = row / (h * Sq);
  const int64_t base = row * Sk;
  const int64_t mbase = b * Sk;

  float reg[SM_REG_MAX];
  const int nchunk = (int)((Sk + WAVE - 1) / WAVE);
  float m = -3.4e38f;
  if (nchunk <= SM_REG_MAX) {
#pragma unroll
    for (int c = 0; c < SM_REG_MAX; ++c) {
      int64_t k = (int64_t)c * WAVE + lane;
      if (c < nchunk && k < Sk) {
        float v = load_elem<DT>(scores, base + k) * scale;
        if (HAS_MASK) v += load_elem<DT>(mask, mbase + k);
        reg[c] = v;
        m = fmaxf(m, v);
      } else {
        reg[c] = -3.4e38f;
      }
    }
    m = wave_max(m);
    float s = 0.f;
#pragma unroll
    for (int c = 0; c < SM_REG_MAX; ++c) {
      if (c < nchunk) {
        reg[c] = __expf(reg[c] - m);
        s += reg[c];
      }
    }
    s = wave_sum(s);
    const float inv = 1.f / s;
#pragma unroll
    for (int c = 0; c < SM_REG_MAX; ++c) {
      int64_t k = (int64_t)c * WAVE + lane;
      if (c < nchunk && k < Sk) store_elem<DT>(probs, base + k, reg[c] * inv);
    }
  } else {
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
}

SKY_EXPORT int sky_masked_softmax_fwd(uint64_t stream, uint64_t scores,
                                      uint64_t mask, uint64_t probs, int64_t B,
                                      int64_t h, int64_t Sq, int64_t Sk,
                                      float scale, float keep, uint64_t seed,
                                      int dt) {
  (void)keep; (void)seed;  // dropout handled by sky_dropout_* (separate op)
  int64_t nrows = B * h * Sq;
  unsigned grid = (unsigned)((nrows + 3) / 4);
  hipStream_t s = (hipStream_t)stream;
  bool hm = mask != 0;
#define SMF(DT, HM)                                                           \
  hipLaunchKernelGGL((softmax_fwd_kernel<DT, HM>), dim3(grid), dim3(256), 0, s, \
                     (const void*)scores, (const void*)mask, (void*)probs, B, \
                     h, Sq, Sk, scale)
  if (dt == DT_F32) { if (hm) SMF(DT_F32, true); else SMF(DT_F32, false); }
  else              { if (hm) SMF(DT_BF16, true); else SMF(DT_BF16, false); }
#undef SMF
  LAUNCH_CHECK();
  return 0;
}

// backward: ds = scale * p * (dp - sum_k dp*p)

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
  int64_t nrows = B * h * Sq;
  unsigned grid = (unsigned)((nrows + 3) / 4);
  hipStream_t s = (hipStream_t)stream;
  if (dt == DT_F32)
    hipLaunchKernelGGL((softmax_bwd_kernel<DT_F32>), dim3(grid), dim3(256), 0, s,
                       (const void*)dp, (const void*)probs, (void*)ds, nrows, Sk, scale);
  else
    hipLaunchKernelGGL((softmax_bwd_kernel<DT_BF16>), dim3(grid), dim3(256), 0, s,
                       (const void*)dp, (const void*)probs, (void*)ds, nrows, Sk, scale);
  LAUNCH_CHECK();
  return 0;
}
