// Common helpers for skycomputing_amd gfx950 (CDNA4) kernels.
//
// Conventions (per /opt/skills/guides/cdna_hip_programming.md):
//   * wave = 64 lanes, hard-coded;
//   * 256-thread blocks (multiple of 64);
//   * bf16 I/O is vectorized as ushort4/ushort8 reinterprets (scalar bf16
//     loads are ~2-2.5x slower, guide G13);
//   * fp32 accumulation everywhere.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <stdint.h>

#define WAVE 64
#define DEV __device__ __forceinline__

typedef unsigned short ushort_t;
typedef __attribute__((ext_vector_type(2))) unsigned short ushort2_t;
typedef __attribute__((ext_vector_type(4))) unsigned short ushort4_t;
typedef __attribute__((ext_vector_type(8))) unsigned short ushort8_t;
typedef __attribute__((ext_vector_type(4))) float float4_t;

DEV float bf16_to_f32(ushort_t u) {
  union { unsigned int i; float f; } v;
  v.i = ((unsigned int)u) << 16;
  return v.f;
}

DEV ushort_t f32_to_bf16(float f) {
  union { unsigned int i; float f; } v;
  v.f = f;
  // round-to-nearest-even
  unsigned int lsb = (v.i >> 16) & 1u;
  v.i += 0x7fffu + lsb;
  return (ushort_t)(v.i >> 16);
}

// dtype tags matching hiplib._DT
enum { DT_F32 = 0, DT_BF16 = 1 };

// 8-element vector I/O (16 B/lane for bf16 — the coalescing sweet spot,
// guide G13; fp32 goes through two float4). Index i8 is in units of 8
// elements; pointers must be 16-byte aligned at those offsets.
template <int DT> struct Vec8;
template <> struct Vec8<DT_BF16> {
  static DEV void load(const void* p, int64_t i8, float f[8]) {
    ushort8_t v = ((const ushort8_t*)p)[i8];
#pragma unroll
    for (int j = 0; j < 8; ++j) f[j] = bf16_to_f32(v[j]);
  }
  static DEV void store(void* p, int64_t i8, const float f[8]) {
    ushort8_t v;
#pragma unroll
    for (int j = 0; j < 8; ++j) v[j] = f32_to_bf16(f[j]);
    ((ushort8_t*)p)[i8] = v;
  }
};
template <> struct Vec8<DT_F32> {
  static DEV void load(const void* p, int64_t i8, float f[8]) {
    float4_t a = ((const float4_t*)p)[i8 * 2];
    float4_t b = ((const float4_t*)p)[i8 * 2 + 1];
#pragma unroll
    for (int j = 0; j < 4; ++j) { f[j] = a[j]; f[4 + j] = b[j]; }
  }
  static DEV void store(void* p, int64_t i8, const float f[8]) {
    float4_t a, b;
#pragma unroll
    for (int j = 0; j < 4; ++j) { a[j] = f[j]; b[j] = f[4 + j]; }
    ((float4_t*)p)[i8 * 2] = a;
    ((float4_t*)p)[i8 * 2 + 1] = b;
  }
};

// generic element load/store through a dtype tag (scalar path)
template <int DT> DEV float load_elem(const void* p, int64_t i);
template <> DEV float load_elem<DT_F32>(const void* p, int64_t i) {
  return ((const float*)p)[i];
}
template <> DEV float load_elem<DT_BF16>(const void* p, int64_t i) {
  return bf16_to_f32(((const ushort_t*)p)[i]);
}
template <int DT> DEV void store_elem(void* p, int64_t i, float v);
template <> DEV void store_elem<DT_F32>(void* p, int64_t i, float v) {
  ((float*)p)[i] = v;
}
template <> DEV void store_elem<DT_BF16>(void* p, int64_t i, float v) {
  ((ushort_t*)p)[i] = f32_to_bf16(v);
}

// wave-wide reductions (64 lanes)
DEV float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}
DEV float wave_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}

// block-wide reductions through LDS (BLOCK threads = BLOCK/64 waves)
template <int BLOCK>
DEV float block_sum(float v, float* lds /* >= BLOCK/64 floats */) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  v = wave_sum(v);
  if (lane == 0) lds[wid] = v;
  __syncthreads();
  float r = 0.f;
  if (wid == 0) {
    r = (lane < BLOCK / WAVE) ? lds[lane] : 0.f;
    r = wave_sum(r);
    if (lane == 0) lds[0] = r;
  }
  __syncthreads();
  r = lds[0];
  __syncthreads();
  return r;
}

// counter-based RNG: splitmix64 -> uniform [0,1)
DEV uint64_t rng_hash(uint64_t seed, uint64_t q) {
  uint64_t z = seed + q * 0x9E3779B97F4A7C15ull;
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ull;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBull;
  return z ^ (z >> 31);
}

DEV float rng_uniform(uint64_t seed, uint64_t idx) {
  return (float)(rng_hash(seed, idx) >> 40) * 0x1.0p-24f;
}

// dropout keep decision: one splitmix hash feeds FOUR consecutive indices
// (16-bit thresholds) — 4x fewer hashes when a lane owns consecutive
// elements. keep16 = keep * 65536 (keep == 1 -> always true).
DEV bool rng_keep16(uint64_t seed, uint64_t idx, unsigned keep16) {
  uint64_t z = rng_hash(seed, idx >> 2);
  return (unsigned)((z >> (16 * (idx & 3))) & 0xFFFFu) < keep16;
}

DEV unsigned keep_to_16(float keep) {
  return keep >= 1.f ? 0x20000u : (unsigned)(keep * 65536.f + 0.5f);
}

// erf-formula GELU and its derivative (fp32)
// Branch-free erf (Abramowitz & Stegun 7.1.26, |err| <= 1.5e-7 absolute —
// far below bf16 resolution): ~12 VALU ops vs OCML erff's ~30 with range
// branches. Feeds the exact (erf) GELU, not the tanh approximation.
DEV float erf_fast(float x) {
  const float ax = fabsf(x);
  const float t = 1.f / fmaf(0.3275911f, ax, 1.f);
  const float p = t * fmaf(t, fmaf(t, fmaf(t, fmaf(t, 1.061405429f, -1.453152027f),
                                           1.421413741f), -0.284496736f),
                           0.254829592f);
  const float r = 1.f - p * __expf(-ax * ax);
  return copysignf(r, x);
}
DEV float gelu_f(float x) { return 0.5f * x * (1.f + erf_fast(x * 0.70710678118654752f)); }
DEV float gelu_grad_f(float x) {
  const float k = 0.70710678118654752f;       // 1/sqrt(2)
  const float c = 0.3989422804014327f;        // 1/sqrt(2*pi)
  float cdf = 0.5f * (1.f + erf_fast(x * k));
  float pdf = c * __expf(-0.5f * x * x);
  return cdf + x * pdf;
}

#define LAUNCH_CHECK() do { hipError_t e_ = hipGetLastError(); if (e_ != hipSuccess) return (int)e_; } while (0)

#define SKY_EXPORT extern "C" __attribute__((visibility("default")))
