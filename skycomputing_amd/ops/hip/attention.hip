// Fused multi-head attention forward for gfx950 (MFMA 16x16x32 bf16).
//
// Computes out = dropout(softmax(scale * Q K^T + mask)) V for one (batch,
// head) per workgroup, S <= 128, d = 64 (BERT-large heads) — replacing the
// reference's QK^T GEMM + mask-add + softmax + dropout + PV GEMM + head
// permutes (reference: scaelum/model/bert_layers.py:249-275) with ONE
// kernel. The S x S score matrix never touches HBM; row max / row sum are
// saved (fp32) so backward recomputes probabilities with sky_attn_probs.
//
// Layouts (chosen so no transpose copies happen anywhere in the layer):
//   qkv  [B, S, 3, h, d]  — straight out of the fused QKV GEMM
//   out  [B, S, h, d]     — views as [B, S, h*d] for the output projection
//
// Structure per workgroup (4 waves, 256 threads):
//   * K tile staged to LDS [S][64] (XOR-swizzled), V staged TRANSPOSED to
//     LDS [64][S] so PV's B-fragments are contiguous ds_read_b128;
//   * each wave owns 32 query rows: QK^T via mfma_f32_16x16x32_bf16
//     (A = Q fragments straight from global, B = K from LDS), softmax in
//     registers (subwave shuffles), dropout via the counter RNG, P written
//     to a per-wave LDS tile, PV via MFMA (A = P, B = V^T from LDS).
//
// Fragment mappings verified on hardware by sky_mfma_probe
// (tests/test_ops_gpu.py::test_mfma_layout).

#include "common.h"

typedef const __attribute__((address_space(1))) unsigned int* att_gas;
typedef __attribute__((address_space(3))) unsigned int* att_las;

typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define ATT_SMAX 128
#define ATT_D 64

// LDS layout (bytes): K [128][64]bf16 at 0 (16K); P per wave [32][128]
// at 16K + w*8K (32K). After QK^T, V^T [64][128] ALIASES K's region
// (K is dead) — total 48K, so 3 workgroups fit per CU (160K LDS) instead
// of 2, +50% latency-hiding concurrency.
#define K_OFF 0
#define VT_OFF 0
#define P_OFF (16 * 1024)
#define ATT_LDS_BYTES (48 * 1024)

DEV void* lds_at(char* base, int byte) { return (void*)(base + byte); }

DEV int swz(int byte_addr, int row, int rmask) {
  return byte_addr ^ ((row & rmask) << 4);
}

typedef __attribute__((ext_vector_type(4))) short s16x4_a;
typedef __attribute__((address_space(3))) s16x4_a* abf_las4;

// tr16 read of an MFMA fragment from a row-major [row][64] sw7 image:
// lane lm indexes the fragment's 16 output columns (cquad base cq0),
// regs j = 4 consecutive image rows starting at r0 + (per-lane s>>2).
DEV bf16x8 abf_tr_rows64(const char* lds_base, int off, int r0, int cq0,
                         int s4) {
  s16x4_a v[2];
#pragma unroll
  for (int half = 0; half < 2; ++half) {
    const int row = r0 + half * 4 + (s4 >> 2);
    const int byte = (off + row * 128 + (cq0 + (s4 & 3)) * 8) ^ ((row & 7) << 4);
    v[half] = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
        (abf_las4)(lds_base + byte));
  }
  return (bf16x8){v[0][0], v[0][1], v[0][2], v[0][3],
                  v[1][0], v[1][1], v[1][2], v[1][3]};
}

// same for the [64 q][128 k] sw15 tiles (256-B rows)
DEV bf16x8 abf_tr_rows128(const char* lds_base, int off, int r0, int cq0,
                          int s4) {
  s16x4_a v[2];
#pragma unroll
  for (int half = 0; half < 2; ++half) {
    const int row = r0 + half * 4 + (s4 >> 2);
    const int byte = (off + (row & 63) * 256 + (cq0 + (s4 & 3)) * 8) ^
                     ((row & 15) << 4);
    v[half] = __builtin_amdgcn_ds_read_tr16_b64_v4i16(
        (abf_las4)(lds_base + byte));
  }
  return (bf16x8){v[0][0], v[0][1], v[0][2], v[0][3],
                  v[1][0], v[1][1], v[1][2], v[1][3]};
}


template <bool HAS_MASK, bool SAVE_ML, bool PROBS_MODE, bool TRV = false>
__global__ __launch_bounds__(256, 2) void attn_fwd_kernel(
    const ushort_t* __restrict__ qkv, const ushort_t* __restrict__ mask,
    ushort_t* __restrict__ out, float* __restrict__ m_io,
    float* __restrict__ l_io, ushort_t* __restrict__ p_out,
    ushort_t* __restrict__ pd_out, int B, int S, int h, float scale,
    float keep, uint64_t salt, const unsigned long long* __restrict__ state) {
  extern __shared__ __attribute__((aligned(16))) char lds[];
  const int tid = threadIdx.x;
  const int l = tid & 63;
  const int w = tid >> 6;
  const int bh = blockIdx.x;
  const int b = bh / h;
  const int hh = bh % h;
  const int ts = 3 * h * ATT_D;  // token stride in qkv
  const int NT = (S + 15) / 16;  // 16-token k tiles
  const uint64_t seed =
      salt + (state ? *state : 0ull) * 0xD1B54A32D192ED03ull;
  const float inv_keep = 1.f / keep;
  const unsigned keep16 = keep_to_16(keep);

  const ushort_t* qbase = qkv + (size_t)b * S * ts + (size_t)hh * ATT_D;
  const ushort_t* kbase = qbase + (size_t)h * ATT_D;
  const ushort_t* vbase = qbase + (size_t)2 * h * ATT_D;

  // ---- zero LDS when S < full tile coverage (avoid NaN poisoning) ----
  if (S < ATT_SMAX) {
    for (int u = tid; u < ATT_LDS_BYTES / 16; u += 256)
      *(ushort8_t*)lds_at(lds, u * 16) = (ushort8_t)(ushort_t)0;
    __syncthreads();
  }

  // ---- stage K into LDS [S][64] via global_load_lds: the XOR swizzle
  // moves to the SOURCE address (lane-linear LDS image == swizzled layout;
  // guide rule 21) ----
  for (int u = tid; u < S * 8; u += 256) {
    const int tok = u >> 3;
    const int c16s = (u & 7) ^ (tok & 7);  // pre-swizzled source column
    __builtin_amdgcn_global_load_lds(
        (att_gas)(kbase + (size_t)tok * ts + c16s * 8),
        (att_las)lds_at(lds, K_OFF + u * 16), 16, 0, 0);
  }
  // ---- per-wave constants (issued BEFORE the barrier so the global
  // loads land under the staging latency) ----
  const int qt0 = 2 * w;
  const int lm = l & 15;
  const int lg = l >> 4;  // lane group 0..3

  // additive mask values for this lane's k columns (col = kt*16 + lm)
  float mval[8];
#pragma unroll
  for (int kt = 0; kt < 8; ++kt) {
    const int col = kt * 16 + lm;
    float mv = 0.f;
    if (HAS_MASK && col < S) mv = bf16_to_f32(mask[(size_t)b * S + col]);
    mval[kt] = (col < S) ? mv : -3.0e38f;
  }
  // Q fragments for both q tiles (independent of LDS)
  bf16x8 aq_all[2][2];
#pragma unroll
  for (int qi = 0; qi < 2; ++qi) {
    int qtok = (qt0 + qi) * 16 + lm;
    if (qtok >= S) qtok = S - 1;
#pragma unroll
    for (int ks = 0; ks < 2; ++ks)
      aq_all[qi][ks] = *(const bf16x8*)(qbase + (size_t)qtok * ts + ks * 32 + lg * 8);
  }
  __syncthreads();

#pragma clang loop unroll(disable)
  for (int qi = 0; qi < 2; ++qi) {
    const int qtile = qt0 + qi;
    const int qtok_base = qtile * 16;
    if (qtok_base >= S) break;
    // A fragments of Q (preloaded before the barrier)
    bf16x8 aq[2] = {aq_all[qi][0], aq_all[qi][1]};
    f32x4 sacc[8];
#pragma unroll
    for (int kt = 0; kt < 8; ++kt) sacc[kt] = (f32x4){0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int kt = 0; kt < 8; ++kt) {
      if (kt >= NT) continue;
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        const int ktok = kt * 16 + lm;
        bf16x8 bk = *(const bf16x8*)lds_at(
            lds, swz(K_OFF + ktok * 128 + (ks * 32 + lg * 8) * 2, ktok, 7));
        sacc[kt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(aq[ks], bk, sacc[kt], 0, 0, 0);
      }
    }
    // softmax over this wave's 4 rows per (lg, r); the transformed
    // scores overwrite sacc in place (register budget)
    float mx[4], sm[4], inv[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) mx[r] = -3.0e38f;
#pragma unroll
    for (int kt = 0; kt < 8; ++kt) {
      if (kt >= NT) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float zz = sacc[kt][r] * scale + mval[kt];
        sacc[kt][r] = zz;
        mx[r] = fmaxf(mx[r], zz);
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
#pragma unroll
      for (int off = 8; off > 0; off >>= 1)
        mx[r] = fmaxf(mx[r], __shfl_xor(mx[r], off, 64));
    }
    if (SAVE_ML) {
      // nothing: m loaded below in PROBS_MODE only
    }
    float mrow[4], lrow[4];
    if (PROBS_MODE) {
      // use saved statistics for exact recomputation
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = qtok_base + lg * 4 + r;
        const int rr = row < S ? row : S - 1;
        mrow[r] = m_io[((size_t)bh) * S + rr];
        lrow[r] = l_io[((size_t)bh) * S + rr];
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) sm[r] = 0.f;
#pragma unroll
    for (int kt = 0; kt < 8; ++kt) {
      if (kt >= NT) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float p = __expf(sacc[kt][r] - (PROBS_MODE ? mrow[r] : mx[r]));
        sacc[kt][r] = p;
        sm[r] += p;
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
#pragma unroll
      for (int off = 8; off > 0; off >>= 1) sm[r] += __shfl_xor(sm[r], off, 64);
      inv[r] = 1.f / (PROBS_MODE ? lrow[r] : sm[r]);
    }
    if (SAVE_ML && lm == 0) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = qtok_base + lg * 4 + r;
        if (row < S) {
          m_io[((size_t)bh) * S + row] = mx[r];
          l_io[((size_t)bh) * S + row] = sm[r];
        }
      }
    }
    // normalize, dropout, write P (LDS in fwd mode; global in probs mode).
    // Dropout indices are lane-local (base*8 + kt), so one hash covers 4
    // consecutive kt tiles; backward reconstructs the mask from Pd != 0
    // (FusedAttentionFn.backward), so the index scheme is kernel-private.
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = qtok_base + lg * 4 + r;
      uint64_t z0 = 0, z1 = 0;
      if (keep < 1.f) {
        const uint64_t base = ((uint64_t)bh * S + row) * 16 + lm;
        z0 = rng_hash(seed, base * 2);
        z1 = rng_hash(seed, base * 2 + 1);
      }
#pragma unroll
      for (int kt = 0; kt < 8; ++kt) {
        if (kt >= NT) continue;
        const int col = kt * 16 + lm;
        float p = sacc[kt][r] * inv[r];
        float pd = p;
        if (keep < 1.f) {
          const uint64_t zz = kt < 4 ? z0 : z1;
          bool kbit = (unsigned)((zz >> (16 * (kt & 3))) & 0xFFFFu) < keep16;
          pd = kbit ? p * inv_keep : 0.f;
        }
        if (PROBS_MODE) {
          if (row < S && col < S) {
            p_out[((size_t)bh * S + row) * S + col] = f32_to_bf16(p);
            pd_out[((size_t)bh * S + row) * S + col] = f32_to_bf16(pd);
          }
        } else {
          const int rl = qi * 16 + lg * 4 + r;  // local row 0..31
          *(ushort_t*)lds_at(
              lds, swz(P_OFF + w * 8192 + rl * 256 + col * 2, rl, 15)) =
              f32_to_bf16((row < S && col < S) ? pd : 0.f);
        }
      }
    }
  }

  if (PROBS_MODE) return;

  // PV accumulators live only from here (declaring them earlier keeps 32
  // VGPRs hot through the scores/softmax phase and forces spills)
  f32x4 oacc[2][4];
#pragma unroll
  for (int qi = 0; qi < 2; ++qi)
#pragma unroll
    for (int dv = 0; dv < 4; ++dv) oacc[qi][dv] = (f32x4){0.f, 0.f, 0.f, 0.f};

  // ---- stage V over K's (now dead) region ----
  __syncthreads();  // all waves done reading K and writing their P tiles
  if (TRV) {
    // DIRECT [tok][64] sw7 row image via glds; PV reads it TRANSPOSED
    // with ds_read_b64_tr_b16 (garbage rows tok>=S multiply P zeros)
    for (int u = tid; u < 128 * 8; u += 256) {
      const int tok = u >> 3;
      const int gtok = tok < S ? tok : S - 1;
      const int c16s = (u & 7) ^ (tok & 7);
      __builtin_amdgcn_global_load_lds(
          (att_gas)(vbase + (size_t)gtok * ts + c16s * 8),
          (att_las)lds_at(lds, VT_OFF + u * 16), 16, 0, 0);
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  } else {
    if (S < ATT_SMAX) {
      for (int u = tid; u < (16 * 1024) / 16; u += 256)
        *(ushort8_t*)lds_at(lds, VT_OFF + u * 16) = (ushort8_t)(ushort_t)0;
      __syncthreads();  // zero before scattered staging writes land
    }
    for (int u = tid; u < S * 8; u += 256) {
      const int tok = u >> 3;
      const int c16 = u & 7;
      ushort8_t v = *(const ushort8_t*)(vbase + (size_t)tok * ts + c16 * 8);
#pragma unroll
      for (int jj = 0; jj < 8; ++jj) {
        const int j = (jj + tok) & 7;  // bank-spread write order (see bwd2)
        const int c = c16 * 8 + j;
        *(ushort_t*)lds_at(lds, swz(VT_OFF + c * 256 + tok * 2, c, 15)) = v[j];
      }
    }
  }
  __syncthreads();

  // ---- PV: O[32 rows][64] per wave ----
  // (same-wave LDS write->read; compiler inserts the lgkm waits)
  for (int qi = 0; qi < 2; ++qi) {
    if ((qt0 + qi) * 16 >= S) break;
#pragma unroll
    for (int ks = 0; ks < 4; ++ks) {
      if (ks * 32 >= S) break;
      // A fragment: P rows
      const int rl = qi * 16 + lm;
      bf16x8 ap = *(const bf16x8*)lds_at(
          lds, swz(P_OFF + w * 8192 + rl * 256 + (ks * 32 + lg * 8) * 2, rl, 15));
#pragma unroll
      for (int dvt = 0; dvt < 4; ++dvt) {
        bf16x8 bv;
        if (TRV) {
          bv = abf_tr_rows64(lds, VT_OFF, ks * 32 + lg * 8, dvt * 4, lm);
        } else {
          const int dv = dvt * 16 + lm;
          bv = *(const bf16x8*)lds_at(
              lds, swz(VT_OFF + dv * 256 + (ks * 32 + lg * 8) * 2, dv, 15));
        }
        oacc[qi][dvt] =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(ap, bv, oacc[qi][dvt], 0, 0, 0);
      }
    }
  }
  // write out [B,S,h,d]
#pragma unroll
  for (int qi = 0; qi < 2; ++qi) {
#pragma unroll
    for (int dvt = 0; dvt < 4; ++dvt) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = (qt0 + qi) * 16 + lg * 4 + r;
        const int dv = dvt * 16 + lm;
        if (row < S)
          out[(((size_t)b * S + row) * h + hh) * ATT_D + dv] =
              f32_to_bf16(oacc[qi][dvt][r]);
      }
    }
  }
}

// pack dQ/dK/dV (each [B,h,S,d] contiguous) into dqkv [B,S,3,h,d] in one
// pass — replaces three permute-copies in the attention backward.
template <int DT>
__global__ __launch_bounds__(256) void pack3_kernel(
    const void* __restrict__ s0, const void* __restrict__ s1,
    const void* __restrict__ s2, void* __restrict__ dst, int64_t B,
    int64_t S, int64_t h, int64_t d8) {
  const int64_t n = B * h * S * d8;
  for (int64_t u = (int64_t)blockIdx.x * 256 + threadIdx.x; u < n;
       u += (int64_t)gridDim.x * 256) {
    int64_t t = u;
    const int64_t di = t % d8; t /= d8;
    const int64_t tok = t % S; t /= S;
    const int64_t hh = t % h;
    const int64_t b = t / h;
    const int64_t src_off = u;  // [B,h,S,d] contiguous
    const int64_t dst_base = (((b * S + tok) * 3) * h + hh) * d8 + di;
    float v[8];
    Vec8<DT>::load(s0, src_off, v);
    Vec8<DT>::store(dst, dst_base, v);
    Vec8<DT>::load(s1, src_off, v);
    Vec8<DT>::store(dst, dst_base + h * d8, v);
    Vec8<DT>::load(s2, src_off, v);
    Vec8<DT>::store(dst, dst_base + 2 * h * d8, v);
  }
}

SKY_EXPORT int sky_pack3(uint64_t stream, uint64_t s0, uint64_t s1,
                         uint64_t s2, uint64_t dst, int64_t B, int64_t S,
                         int64_t h, int64_t d, int dt) {
  if (d % 8 != 0) return (int)hipErrorInvalidValue;
  const int64_t n = B * h * S * (d / 8);
  unsigned grid = (unsigned)((n + 255) / 256);
  if (grid > 2048u) grid = 2048u;
  hipStream_t s = (hipStream_t)stream;
  if (dt == DT_F32)
    hipLaunchKernelGGL((pack3_kernel<DT_F32>), dim3(grid), dim3(256), 0, s,
                       (const void*)s0, (const void*)s1, (const void*)s2,
                       (void*)dst, B, S, h, d / 8);
  else
    hipLaunchKernelGGL((pack3_kernel<DT_BF16>), dim3(grid), dim3(256), 0, s,
                       (const void*)s0, (const void*)s1, (const void*)s2,
                       (void*)dst, B, S, h, d / 8);
  LAUNCH_CHECK();
  return 0;
}

static int attn_launch(uint64_t stream, uint64_t qkv, uint64_t mask,
                       uint64_t out, uint64_t m, uint64_t lsum, uint64_t p,
                       uint64_t pd, int64_t B, int64_t S, int64_t h,
                       int64_t d, float scale, float keep, uint64_t salt,
                       uint64_t state, bool probs_mode) {
  if (d != ATT_D || S > ATT_SMAX) return (int)hipErrorInvalidValue;
  hipStream_t s = (hipStream_t)stream;
  dim3 grid((unsigned)(B * h));
  size_t lds_bytes = ATT_LDS_BYTES;
  bool hm = mask != 0;
  // (a split-q forward variant — 2 blocks per (b,h), 32K LDS — measured
  // SLOWER than this one-block kernel: 29.2 vs 27.2 us; the duplicated K
  // staging doesn't pay without an occupancy gain, and the fwd register
  // budget pins 2 waves/SIMD either way. See profiles/r01_notes.md.)
  // Default: stage V DIRECT via glds and read PV fragments with
  // ds_read_b64_tr_b16 — measured 27.0 vs 30.3 us standalone and
  // -0.55 ms/step in-app over the register-transpose V^T staging.
  // SKY_ATTN_TRV=0 restores the old staging for A/B.
  const char* trve = getenv("SKY_ATTN_TRV");
  const bool trv = !(trve && trve[0] == '0');
#define ATT(HM, SML, PM)                                                      \
  do {                                                                        \
    if (trv)                                                                  \
      hipLaunchKernelGGL((attn_fwd_kernel<HM, SML, PM, true>), grid,          \
                         dim3(256), lds_bytes, s, (const ushort_t*)qkv,       \
                         (const ushort_t*)mask, (ushort_t*)out, (float*)m,    \
                         (float*)lsum, (ushort_t*)p, (ushort_t*)pd, (int)B,   \
                         (int)S, (int)h, scale, keep, salt,                   \
                         (const unsigned long long*)state);                   \
    else                                                                      \
      hipLaunchKernelGGL((attn_fwd_kernel<HM, SML, PM, false>), grid,         \
                         dim3(256), lds_bytes, s, (const ushort_t*)qkv,       \
                         (const ushort_t*)mask, (ushort_t*)out, (float*)m,    \
                         (float*)lsum, (ushort_t*)p, (ushort_t*)pd, (int)B,   \
                         (int)S, (int)h, scale, keep, salt,                   \
                         (const unsigned long long*)state);                   \
  } while (0)
  if (probs_mode) { if (hm) ATT(true, false, true); else ATT(false, false, true); }
  else            { if (hm) ATT(true, true, false); else ATT(false, true, false); }
#undef ATT
  LAUNCH_CHECK();
  return 0;
}

SKY_EXPORT int sky_attn_fwd(uint64_t stream, uint64_t qkv, uint64_t mask,
                            uint64_t out, uint64_t m, uint64_t lsum,
                            int64_t B, int64_t S, int64_t h, int64_t d,
                            float scale, float keep, uint64_t salt,
                            uint64_t state) {
  return attn_launch(stream, qkv, mask, out, m, lsum, 0, 0, B, S, h, d,
                     scale, keep, salt, state, false);
}

SKY_EXPORT int sky_attn_probs(uint64_t stream, uint64_t qkv, uint64_t mask,
                              uint64_t m, uint64_t lsum, uint64_t p,
                              uint64_t pd, int64_t B, int64_t S, int64_t h,
                              int64_t d, float scale, float keep,
                              uint64_t salt, uint64_t state) {
  return attn_launch(stream, qkv, mask, 0, m, lsum, p, pd, B, S, h, d, scale,
                     keep, salt, state, true);
}

// ============================================================================
// Fully-fused attention backward (two kernels).
//
// B1, per (b,h): recompute scores -> P (saved m/l) + dropout mask; compute
//   dPd = dO V^T (MFMA, A = dO straight from global), dP, the softmax
//   row-dot, dS = scale*P*(dP - rowdot); write dS row-major to per-wave LDS
//   and dS^T / Pd^T TRANSPOSED to global scratch; compute
//   dQ = dS K (A = dS from LDS, B = K^T from LDS) -> dqkv[:,:,0].
// B2, per (b,h): dV = Pd^T dO and dK = dS^T Q, reading the transposed
//   scratch directly as MFMA A-fragments (contiguous q per lane) and
//   staging Q^T / dO^T in LDS -> dqkv[:,:,1:3].
//
// Replaces: sky_attn_probs + 4 hipBLASLt bmms + softmax-bwd + dropout-bwd
// + permute copies + the dqkv pack (FusedAttentionFn.backward fast path).
// ============================================================================

template <bool HAS_MASK>
__global__ __launch_bounds__(256, 2) void attn_bwd1_kernel(
    const ushort_t* __restrict__ qkv, const ushort_t* __restrict__ dout,
    const ushort_t* __restrict__ mask, const float* __restrict__ m_io,
    const float* __restrict__ l_io, ushort_t* __restrict__ pdT,
    ushort_t* __restrict__ dsT, ushort_t* __restrict__ dqkv, int B, int S,
    int h, float scale, float keep, uint64_t salt,
    const unsigned long long* __restrict__ state) {
  // LDS: V [128][64] sw7 @0 (16K), K [128][64] sw7 @16K, Kt [64][128] sw15
  // @32K, per-wave dSrow [32][128] sw15 @48K+w*8K. Total 80K.
  extern __shared__ __attribute__((aligned(16))) char lds[];
  const int tid = threadIdx.x;
  const int l = tid & 63;
  const int w = tid >> 6;
  const int lm = l & 15;
  const int lg = l >> 4;
  const int bh = blockIdx.x;
  const int b = bh / h;
  const int hh = bh % h;
  const int ts = 3 * h * ATT_D;
  const int NT = (S + 15) / 16;
  const uint64_t seed = salt + (state ? *state : 0ull) * 0xD1B54A32D192ED03ull;
  const float inv_keep = 1.f / keep;
  const unsigned keep16 = keep_to_16(keep);

  const ushort_t* qbase = qkv + (size_t)b * S * ts + (size_t)hh * ATT_D;
  const ushort_t* kbase = qbase + (size_t)h * ATT_D;
  const ushort_t* vbase = qbase + (size_t)2 * h * ATT_D;
  const ushort_t* dobase = dout + ((size_t)b * S * h + hh) * ATT_D;
  const int dots = h * ATT_D;  // dout token stride

  if (S < ATT_SMAX) {
    for (int u = tid; u < (80 * 1024) / 16; u += 256)
      *(ushort8_t*)lds_at(lds, u * 16) = (ushort8_t)(ushort_t)0;
    __syncthreads();
  }
  // stage V and K [tok][64] sw7, Kt [c][tok] sw15
  for (int u = tid; u < S * 8; u += 256) {
    const int tok = u >> 3;
    const int c16 = u & 7;
    ushort8_t v = *(const ushort8_t*)(vbase + (size_t)tok * ts + c16 * 8);
    *(ushort8_t*)lds_at(lds, swz(0 + tok * 128 + c16 * 16, tok, 7)) = v;
    ushort8_t kv = *(const ushort8_t*)(kbase + (size_t)tok * ts + c16 * 8);
    *(ushort8_t*)lds_at(lds, swz(16384 + tok * 128 + c16 * 16, tok, 7)) = kv;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int c = c16 * 8 + j;
      *(ushort_t*)lds_at(lds, swz(32768 + c * 256 + tok * 2, c, 15)) = kv[j];
    }
  }
  const int qt0 = 2 * w;
  // preload Q and dO fragments + mask before the barrier
  float mval[8];
#pragma unroll
  for (int kt = 0; kt < 8; ++kt) {
    const int col = kt * 16 + lm;
    float mv = 0.f;
    if (HAS_MASK && col < S) mv = bf16_to_f32(mask[(size_t)b * S + col]);
    mval[kt] = (col < S) ? mv : -3.0e38f;
  }
  bf16x8 aq_all[2][2], ado_all[2][2];
#pragma unroll
  for (int qi = 0; qi < 2; ++qi) {
    int qtok = (qt0 + qi) * 16 + lm;
    if (qtok >= S) qtok = S - 1;
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      aq_all[qi][ks] = *(const bf16x8*)(qbase + (size_t)qtok * ts + ks * 32 + lg * 8);
      ado_all[qi][ks] = *(const bf16x8*)(dobase + (size_t)qtok * dots + ks * 32 + lg * 8);
    }
  }
  __syncthreads();

#pragma clang loop unroll(disable)
  for (int qi = 0; qi < 2; ++qi) {
    const int qtok_base = (qt0 + qi) * 16;
    if (qtok_base >= S) break;
    f32x4 sacc[8], dacc[8];
#pragma unroll
    for (int kt = 0; kt < 8; ++kt) {
      sacc[kt] = (f32x4){0.f, 0.f, 0.f, 0.f};
      dacc[kt] = (f32x4){0.f, 0.f, 0.f, 0.f};
    }
#pragma unroll
    for (int kt = 0; kt < 8; ++kt) {
      if (kt >= NT) continue;
      const int ktok = kt * 16 + lm;
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        bf16x8 bk = *(const bf16x8*)lds_at(
            lds, swz(16384 + ktok * 128 + (ks * 32 + lg * 8) * 2, ktok, 7));
        sacc[kt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(aq_all[qi][ks], bk, sacc[kt], 0, 0, 0);
        bf16x8 bv = *(const bf16x8*)lds_at(
            lds, swz(0 + ktok * 128 + (ks * 32 + lg * 8) * 2, ktok, 7));
        dacc[kt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ado_all[qi][ks], bv, dacc[kt], 0, 0, 0);
      }
    }
    // P, Pd, dP, rowdot, dS. Register budget: p/pd/dp are NOT kept in
    // arrays (3x32 VGPRs would spill); the store loop recomputes them
    // from the still-live sacc/dacc and the cached dropout hashes.
    float mrow[4], lrow[4], dot[4];
    uint64_t zs[4][2];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = qtok_base + lg * 4 + r;
      const int rr = row < S ? row : S - 1;
      mrow[r] = m_io[(size_t)bh * S + rr];
      lrow[r] = 1.f / l_io[(size_t)bh * S + rr];
      dot[r] = 0.f;
      zs[r][0] = zs[r][1] = 0;
      if (keep < 1.f) {
        const uint64_t base = ((uint64_t)bh * S + row) * 16 + lm;
        zs[r][0] = rng_hash(seed, base * 2);
        zs[r][1] = rng_hash(seed, base * 2 + 1);
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = qtok_base + lg * 4 + r;
#pragma unroll
      for (int kt = 0; kt < 8; ++kt) {
        if (kt >= NT) continue;
        const int col = kt * 16 + lm;
        float p = __expf(sacc[kt][r] * scale + mval[kt] - mrow[r]) * lrow[r];
        if (!(row < S && col < S)) p = 0.f;
        float dp = dacc[kt][r];
        if (keep < 1.f) {
          const uint64_t zz = kt < 4 ? zs[r][0] : zs[r][1];
          bool kbit = (unsigned)((zz >> (16 * (kt & 3))) & 0xFFFFu) < keep16;
          dp = kbit ? dp * inv_keep : 0.f;
        }
        dot[r] += dp * p;
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
#pragma unroll
      for (int off = 8; off > 0; off >>= 1) dot[r] += __shfl_xor(dot[r], off, 64);
    }
    // dS = scale * P * (dP - dot); write dSrow (LDS) + dS^T, Pd^T (global,
    // 4 consecutive rows packed per 8-byte store — issue-bound tail, T21)
#pragma unroll
    for (int kt = 0; kt < 8; ++kt) {
      if (kt >= NT) continue;
      const int col = kt * 16 + lm;
      ushort4_t ds4, pd4;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = qtok_base + lg * 4 + r;
        float p = __expf(sacc[kt][r] * scale + mval[kt] - mrow[r]) * lrow[r];
        if (!(row < S && col < S)) p = 0.f;
        float pd = p, dp = dacc[kt][r];
        if (keep < 1.f) {
          const uint64_t zz = kt < 4 ? zs[r][0] : zs[r][1];
          bool kbit = (unsigned)((zz >> (16 * (kt & 3))) & 0xFFFFu) < keep16;
          pd = kbit ? p * inv_keep : 0.f;
          dp = kbit ? dp * inv_keep : 0.f;
        }
        const float ds = scale * p * (dp - dot[r]);
        const int rl = qi * 16 + lg * 4 + r;
        *(ushort_t*)lds_at(lds, swz(49152 + w * 8192 + rl * 256 + col * 2, rl, 15)) =
            f32_to_bf16(ds);
        ds4[r] = f32_to_bf16(ds);
        pd4[r] = f32_to_bf16(pd);
      }
      if (col < S) {
        const size_t o = ((size_t)bh * S + col) * S + qtok_base + lg * 4;
        if (qtok_base + lg * 4 + 3 < S) {  // whole 4-pack in range
          *(ushort4_t*)(dsT + o) = ds4;
          *(ushort4_t*)(pdT + o) = pd4;
        } else {  // edge tile of a non-multiple-of-4 S: scalar tail
#pragma unroll
          for (int r = 0; r < 4; ++r)
            if (qtok_base + lg * 4 + r < S) {
              dsT[o + r] = ds4[r];
              pdT[o + r] = pd4[r];
            }
        }
      }
    }
    // dQ = dS K : A = dSrow (this wave's LDS), B = Kt
    f32x4 qacc[4];
#pragma unroll
    for (int ct = 0; ct < 4; ++ct) qacc[ct] = (f32x4){0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int ks = 0; ks < 4; ++ks) {
      if (ks * 32 >= S) break;
      const int rl = qi * 16 + lm;
      bf16x8 asr = *(const bf16x8*)lds_at(
          lds, swz(49152 + w * 8192 + rl * 256 + (ks * 32 + lg * 8) * 2, rl, 15));
#pragma unroll
      for (int ct = 0; ct < 4; ++ct) {
        const int c = ct * 16 + lm;
        bf16x8 bkt = *(const bf16x8*)lds_at(
            lds, swz(32768 + c * 256 + (ks * 32 + lg * 8) * 2, c, 15));
        qacc[ct] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(asr, bkt, qacc[ct], 0, 0, 0);
      }
    }
    ushort_t* dq = dqkv + (size_t)b * S * ts + (size_t)hh * ATT_D;
#pragma unroll
    for (int ct = 0; ct < 4; ++ct) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = qtok_base + lg * 4 + r;
        const int c = ct * 16 + lm;
        if (row < S) dq[(size_t)row * ts + c] = f32_to_bf16(qacc[ct][r]);
      }
    }
  }
}

// Split-q variant of B1: (S+63)/64 blocks per (b,h), each covering 64 query
// rows (one 16-row tile per wave, no qi loop). Exists because the one-block
// B1 is latency-bound at 2 waves/SIMD — its 80 KB LDS caps residency at
// 2 blocks/CU and its 201 VGPRs + 144 B/lane spill cap it at 2 waves/SIMD.
// Here K^T is rebuilt IN PLACE over the dead V region after the score/dPd
// MFMAs (one extra LDS round trip of K) and the per-wave dS buffer drops to
// 16 rows, so the whole block fits 48 KB LDS with a smaller register
// footprint -> 3 blocks/CU at 3+ waves/SIMD and 2x the workgroups.
template <bool HAS_MASK>
__global__ __launch_bounds__(256, 3) void attn_bwd1s_kernel(
    const ushort_t* __restrict__ qkv, const ushort_t* __restrict__ dout,
    const ushort_t* __restrict__ mask, const float* __restrict__ m_io,
    const float* __restrict__ l_io, ushort_t* __restrict__ pdT,
    ushort_t* __restrict__ dsT, ushort_t* __restrict__ dqkv, int B, int S,
    int h, float scale, float keep, uint64_t salt,
    const unsigned long long* __restrict__ state) {
  // LDS: V [128][64] sw7 @0 (16K, becomes Kt [64][128] sw15 after phase A),
  // K [128][64] sw7 @16K, per-wave dS [16][128] sw15 @32K + w*4K. Total 48K.
  extern __shared__ __attribute__((aligned(16))) char lds[];
  const int tid = threadIdx.x;
  const int l = tid & 63;
  const int w = tid >> 6;
  const int lm = l & 15;
  const int lg = l >> 4;
  const int QB = (S + 63) / 64;
  const int bh = blockIdx.x / QB;
  const int qb = blockIdx.x % QB;
  const int b = bh / h;
  const int hh = bh % h;
  const int ts = 3 * h * ATT_D;
  const int NT = (S + 15) / 16;
  const uint64_t seed = salt + (state ? *state : 0ull) * 0xD1B54A32D192ED03ull;
  const float inv_keep = 1.f / keep;
  const unsigned keep16 = keep_to_16(keep);

  const ushort_t* qbase = qkv + (size_t)b * S * ts + (size_t)hh * ATT_D;
  const ushort_t* kbase = qbase + (size_t)h * ATT_D;
  const ushort_t* vbase = qbase + (size_t)2 * h * ATT_D;
  const ushort_t* dobase = dout + ((size_t)b * S * h + hh) * ATT_D;
  const int dots = h * ATT_D;

  const int qtok_base = qb * 64 + w * 16;
  const bool active = qtok_base < S;

  if (S < ATT_SMAX) {
    for (int u = tid; u < (48 * 1024) / 16; u += 256)
      *(ushort8_t*)lds_at(lds, u * 16) = (ushort8_t)(ushort_t)0;
    __syncthreads();
  }
  // stage V [tok][64] sw7 @0 and K [tok][64] sw7 @16K
  for (int u = tid; u < S * 8; u += 256) {
    const int tok = u >> 3;
    const int c16 = u & 7;
    *(ushort8_t*)lds_at(lds, swz(0 + tok * 128 + c16 * 16, tok, 7)) =
        *(const ushort8_t*)(vbase + (size_t)tok * ts + c16 * 8);
    *(ushort8_t*)lds_at(lds, swz(16384 + tok * 128 + c16 * 16, tok, 7)) =
        *(const ushort8_t*)(kbase + (size_t)tok * ts + c16 * 8);
  }
  // preload this wave's Q / dO fragments + mask column values
  float mval[8];
#pragma unroll
  for (int kt = 0; kt < 8; ++kt) {
    const int col = kt * 16 + lm;
    float mv = 0.f;
    if (HAS_MASK && col < S) mv = bf16_to_f32(mask[(size_t)b * S + col]);
    mval[kt] = (col < S) ? mv : -3.0e38f;
  }
  int qtok = qtok_base + lm;
  if (qtok >= S) qtok = S - 1;
  bf16x8 aq[2], ado[2];
#pragma unroll
  for (int ks = 0; ks < 2; ++ks) {
    aq[ks] = *(const bf16x8*)(qbase + (size_t)qtok * ts + ks * 32 + lg * 8);
    ado[ks] = *(const bf16x8*)(dobase + (size_t)qtok * dots + ks * 32 + lg * 8);
  }
  __syncthreads();

  // phase A: scores + dPd
  f32x4 sacc[8], dacc[8];
#pragma unroll
  for (int kt = 0; kt < 8; ++kt) {
    sacc[kt] = (f32x4){0.f, 0.f, 0.f, 0.f};
    dacc[kt] = (f32x4){0.f, 0.f, 0.f, 0.f};
  }
  if (active) {
#pragma unroll
    for (int kt = 0; kt < 8; ++kt) {
      if (kt >= NT) continue;
      const int ktok = kt * 16 + lm;
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        bf16x8 bk = *(const bf16x8*)lds_at(
            lds, swz(16384 + ktok * 128 + (ks * 32 + lg * 8) * 2, ktok, 7));
        sacc[kt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(aq[ks], bk, sacc[kt], 0, 0, 0);
        bf16x8 bv = *(const bf16x8*)lds_at(
            lds, swz(0 + ktok * 128 + (ks * 32 + lg * 8) * 2, ktok, 7));
        dacc[kt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ado[ks], bv, dacc[kt], 0, 0, 0);
      }
    }
  }
  // (v2: no Kt rebuild — dQ reads the K image transposed via
  // ds_read_b64_tr_b16, same recipe as the forward's V path)

  if (active) {
    // P, dP, rowdot, dS (p/pd recomputed in the store loop from live
    // sacc/dacc + cached dropout hashes — same register discipline as B1)
    float mrow[4], lrow[4], dot[4];
    uint64_t zs[4][2];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = qtok_base + lg * 4 + r;
      const int rr = row < S ? row : S - 1;
      mrow[r] = m_io[(size_t)bh * S + rr];
      lrow[r] = 1.f / l_io[(size_t)bh * S + rr];
      dot[r] = 0.f;
      zs[r][0] = zs[r][1] = 0;
      if (keep < 1.f) {
        const uint64_t base = ((uint64_t)bh * S + row) * 16 + lm;
        zs[r][0] = rng_hash(seed, base * 2);
        zs[r][1] = rng_hash(seed, base * 2 + 1);
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = qtok_base + lg * 4 + r;
#pragma unroll
      for (int kt = 0; kt < 8; ++kt) {
        if (kt >= NT) continue;
        const int col = kt * 16 + lm;
        float p = __expf(sacc[kt][r] * scale + mval[kt] - mrow[r]) * lrow[r];
        if (!(row < S && col < S)) p = 0.f;
        float dp = dacc[kt][r];
        if (keep < 1.f) {
          const uint64_t zz = kt < 4 ? zs[r][0] : zs[r][1];
          bool kbit = (unsigned)((zz >> (16 * (kt & 3))) & 0xFFFFu) < keep16;
          dp = kbit ? dp * inv_keep : 0.f;
        }
        dot[r] += dp * p;
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
#pragma unroll
      for (int off = 8; off > 0; off >>= 1) dot[r] += __shfl_xor(dot[r], off, 64);
    }
#pragma unroll
    for (int kt = 0; kt < 8; ++kt) {
      if (kt >= NT) continue;
      const int col = kt * 16 + lm;
      ushort4_t ds4, pd4;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = qtok_base + lg * 4 + r;
        float p = __expf(sacc[kt][r] * scale + mval[kt] - mrow[r]) * lrow[r];
        if (!(row < S && col < S)) p = 0.f;
        float pd = p, dp = dacc[kt][r];
        if (keep < 1.f) {
          const uint64_t zz = kt < 4 ? zs[r][0] : zs[r][1];
          bool kbit = (unsigned)((zz >> (16 * (kt & 3))) & 0xFFFFu) < keep16;
          pd = kbit ? p * inv_keep : 0.f;
          dp = kbit ? dp * inv_keep : 0.f;
        }
        const float ds = scale * p * (dp - dot[r]);
        const int rl = lg * 4 + r;
        *(ushort_t*)lds_at(lds, swz(32768 + w * 4096 + rl * 256 + col * 2, rl, 15)) =
            f32_to_bf16(ds);
        ds4[r] = f32_to_bf16(ds);
        pd4[r] = f32_to_bf16(pd);
      }
      if (col < S) {
        const size_t o = ((size_t)bh * S + col) * S + qtok_base + lg * 4;
        if (qtok_base + lg * 4 + 3 < S) {
          *(ushort4_t*)(dsT + o) = ds4;
          *(ushort4_t*)(pdT + o) = pd4;
        } else {
#pragma unroll
          for (int r = 0; r < 4; ++r)
            if (qtok_base + lg * 4 + r < S) {
              dsT[o + r] = ds4[r];
              pdT[o + r] = pd4[r];
            }
        }
      }
    }
  }
  __syncthreads();  // dS tiles complete before the dQ MFMAs read them

  if (active) {
    // dQ = dS K : A = this wave's dS rows (LDS), B = K^T via tr16
    f32x4 qacc[4];
#pragma unroll
    for (int ct = 0; ct < 4; ++ct) qacc[ct] = (f32x4){0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int ks = 0; ks < 4; ++ks) {
      if (ks * 32 >= S) break;
      bf16x8 asr = *(const bf16x8*)lds_at(
          lds, swz(32768 + w * 4096 + lm * 256 + (ks * 32 + lg * 8) * 2, lm, 15));
#pragma unroll
      for (int ct = 0; ct < 4; ++ct) {
        bf16x8 bkt = abf_tr_rows64(lds, 16384, ks * 32 + lg * 8, ct * 4, lm);
        qacc[ct] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(asr, bkt, qacc[ct], 0, 0, 0);
      }
    }
    ushort_t* dq = dqkv + (size_t)b * S * ts + (size_t)hh * ATT_D;
#pragma unroll
    for (int ct = 0; ct < 4; ++ct) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = qtok_base + lg * 4 + r;
        const int c = ct * 16 + lm;
        if (row < S) dq[(size_t)row * ts + c] = f32_to_bf16(qacc[ct][r]);
      }
    }
  }
}

__global__ __launch_bounds__(256) void attn_bwd2_kernel(
    const ushort_t* __restrict__ qkv, const ushort_t* __restrict__ dout,
    const ushort_t* __restrict__ pdT, const ushort_t* __restrict__ dsT,
    ushort_t* __restrict__ dqkv, int B, int S, int h) {
  // LDS: Qt [64][128] sw15 @0 (16K), dOt [64][128] sw15 @16K. Total 32K.
  extern __shared__ __attribute__((aligned(16))) char lds[];
  const int tid = threadIdx.x;
  const int l = tid & 63;
  const int w = tid >> 6;
  const int lm = l & 15;
  const int lg = l >> 4;
  const int bh = blockIdx.x;
  const int b = bh / h;
  const int hh = bh % h;
  const int ts = 3 * h * ATT_D;
  const int dots = h * ATT_D;

  const ushort_t* qbase = qkv + (size_t)b * S * ts + (size_t)hh * ATT_D;
  const ushort_t* dobase = dout + ((size_t)b * S * h + hh) * ATT_D;

  // Q and dO staged DIRECT [tok][64] sw7 via glds; the dV/dK B-fragments
  // read them transposed with ds_read_b64_tr_b16 (v2 — replaces the
  // register-scatter Qt/dOt build and its residual bank conflicts).
  // A-side zeros (pdT/dsT rows beyond S) null any garbage tail rows.
  for (int u = tid; u < S * 8; u += 256) {
    const int tok = u >> 3;
    const int c16s = (u & 7) ^ (tok & 7);
    __builtin_amdgcn_global_load_lds(
        (att_gas)(qbase + (size_t)tok * ts + c16s * 8),
        (att_las)lds_at(lds, 0 + u * 16), 16, 0, 0);
    __builtin_amdgcn_global_load_lds(
        (att_gas)(dobase + (size_t)tok * dots + c16s * 8),
        (att_las)lds_at(lds, 16384 + u * 16), 16, 0, 0);
  }
  // rows >= S must be ZERO: the pdT/dsT A-fragment reads of a partial
  // 32-q block run past row S into the next k-row's data (row stride S),
  // and only a zero B side nulls that contribution.
  for (int u = S * 8 + tid; u < 128 * 8; u += 256) {
    *(ushort8_t*)lds_at(lds, 0 + u * 16) = (ushort8_t)(ushort_t)0;
    *(ushort8_t*)lds_at(lds, 16384 + u * 16) = (ushort8_t)(ushort_t)0;
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  const int kt0 = 2 * w;  // this wave's two 16-row k tiles
  for (int ki = 0; ki < 2; ++ki) {
    const int ktok_base = (kt0 + ki) * 16;
    if (ktok_base >= S) break;
    int ktok = ktok_base + lm;
    if (ktok >= S) ktok = S - 1;
    f32x4 vacc[4], kacc[4];
#pragma unroll
    for (int ct = 0; ct < 4; ++ct) {
      vacc[ct] = (f32x4){0.f, 0.f, 0.f, 0.f};
      kacc[ct] = (f32x4){0.f, 0.f, 0.f, 0.f};
    }
#pragma unroll
    for (int ks = 0; ks < 4; ++ks) {
      if (ks * 32 >= S) break;
      // A fragments from the transposed global scratch (contiguous q)
      const size_t abase = ((size_t)bh * S + ktok) * S + ks * 32 + lg * 8;
      bf16x8 apd = *(const bf16x8*)(pdT + abase);
      bf16x8 ads = *(const bf16x8*)(dsT + abase);
#pragma unroll
      for (int ct = 0; ct < 4; ++ct) {
        bf16x8 bdo = abf_tr_rows64(lds, 16384, ks * 32 + lg * 8, ct * 4, lm);
        vacc[ct] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(apd, bdo, vacc[ct], 0, 0, 0);
        bf16x8 bq = abf_tr_rows64(lds, 0, ks * 32 + lg * 8, ct * 4, lm);
        kacc[ct] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ads, bq, kacc[ct], 0, 0, 0);
      }
    }
    ushort_t* dk = dqkv + (size_t)b * S * ts + (size_t)(h + hh) * ATT_D;
    ushort_t* dv = dqkv + (size_t)b * S * ts + (size_t)(2 * h + hh) * ATT_D;
#pragma unroll
    for (int ct = 0; ct < 4; ++ct) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = ktok_base + lg * 4 + r;
        const int c = ct * 16 + lm;
        if (row < S) {
          dk[(size_t)row * ts + c] = f32_to_bf16(kacc[ct][r]);
          dv[(size_t)row * ts + c] = f32_to_bf16(vacc[ct][r]);
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Fully-fused attention backward (S <= 128): ONE kernel per (b,h) computes
// dQ, dK, dV with P and dS living ONLY in LDS — the bwd1s/bwd2 pair's
// transposed pdT/dsT scratch (2 x S x S bf16 per (b,h), written scattered
// and re-read next kernel) never touches HBM, and Q/dO are staged once
// instead of twice. Transposed operand fragments (Kt for dQ; pd^T/dS^T,
// Q^T, dO^T for dK/dV) are read with ds_read_b64_tr_b16 (lane mapping
// derived on HW, tools/tr16_probe.hip) straight from row-major images.
//
// LDS (80 KB -> 2 WGs/CU): K[128][64]sw7 @0, Q @16K, dO @32K (all glds-
// staged), dS[64 q][128 k]sw15 @48K, pd @64K (per-pass tiles; q rows are
// processed in two 64-row passes, dK/dV accumulate in registers across
// passes; each wave owns one 16-row q tile per pass and k-tiles 2w,2w+1).
#define ABF_K 0
#define ABF_Q (16 * 1024)
#define ABF_DO (32 * 1024)
#define ABF_DS (48 * 1024)
#define ABF_PD (64 * 1024)
#define ABF_LDS (80 * 1024)

template <bool HAS_MASK>
__global__ __launch_bounds__(256, 1) void attn_bwd_fused_kernel(
    const ushort_t* __restrict__ qkv, const ushort_t* __restrict__ dout,
    const ushort_t* __restrict__ mask, const float* __restrict__ m_io,
    const float* __restrict__ l_io, ushort_t* __restrict__ dqkv, int B,
    int S, int h, float scale, float keep, uint64_t salt,
    const unsigned long long* __restrict__ state) {
  extern __shared__ __attribute__((aligned(16))) char lds[];
  const int tid = threadIdx.x;
  const int l = tid & 63;
  const int w = tid >> 6;
  const int lm = l & 15;
  const int lg = l >> 4;
  const int bh = blockIdx.x;
  const int b = bh / h;
  const int hh = bh % h;
  const int ts = 3 * h * ATT_D;
  const int dots = h * ATT_D;
  const int NT = (S + 15) / 16;
  const uint64_t seed = salt + (state ? *state : 0ull) * 0xD1B54A32D192ED03ull;
  const float inv_keep = 1.f / keep;
  const unsigned keep16 = keep_to_16(keep);

  const ushort_t* qbase = qkv + (size_t)b * S * ts + (size_t)hh * ATT_D;
  const ushort_t* kbase = qbase + (size_t)h * ATT_D;
  const ushort_t* vbase = qbase + (size_t)2 * h * ATT_D;
  const ushort_t* dobase = dout + ((size_t)b * S * h + hh) * ATT_D;

  // ---- stage K, Q, dO row images via glds (source-side swizzle) ----
  for (int u = tid; u < 128 * 8; u += 256) {
    const int tok = u >> 3;
    const int gtok = tok < S ? tok : S - 1;
    const int c16s = (u & 7) ^ (tok & 7);
    __builtin_amdgcn_global_load_lds(
        (att_gas)(kbase + (size_t)gtok * ts + c16s * 8),
        (att_las)lds_at(lds, ABF_K + u * 16), 16, 0, 0);
    __builtin_amdgcn_global_load_lds(
        (att_gas)(qbase + (size_t)gtok * ts + c16s * 8),
        (att_las)lds_at(lds, ABF_Q + u * 16), 16, 0, 0);
    __builtin_amdgcn_global_load_lds(
        (att_gas)(dobase + (size_t)gtok * dots + c16s * 8),
        (att_las)lds_at(lds, ABF_DO + u * 16), 16, 0, 0);
  }
  // mask column values (pass-independent)
  float mval[8];
#pragma unroll
  for (int kt = 0; kt < 8; ++kt) {
    const int col = kt * 16 + lm;
    float mv = 0.f;
    if (HAS_MASK && col < S) mv = bf16_to_f32(mask[(size_t)b * S + col]);
    mval[kt] = (col < S) ? mv : -3.0e38f;
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  // persistent dK/dV accumulators: wave w owns k-tiles 2w and 2w+1
  f32x4 kacc[2][4], vacc[2][4];
#pragma unroll
  for (int ki = 0; ki < 2; ++ki)
#pragma unroll
    for (int ct = 0; ct < 4; ++ct) {
      kacc[ki][ct] = (f32x4){0.f, 0.f, 0.f, 0.f};
      vacc[ki][ct] = (f32x4){0.f, 0.f, 0.f, 0.f};
    }

  const int npass = (S + 63) / 64;
  for (int pass = 0; pass < npass; ++pass) {
    const int qtok_base = pass * 64 + w * 16;
    const bool active = qtok_base < S;
    // ---- phase A: scores (K from LDS) + dPd (V fragments from global);
    // A fragments (this wave's q rows of Q / dO) read per-use from the
    // staged LDS row images — no held registers ----
    const int qrow = qtok_base + lm;
    const int qrl = qrow < S ? qrow : S - 1;
    f32x4 sacc[8], dacc[8];
#pragma unroll
    for (int kt = 0; kt < 8; ++kt) {
      sacc[kt] = (f32x4){0.f, 0.f, 0.f, 0.f};
      dacc[kt] = (f32x4){0.f, 0.f, 0.f, 0.f};
    }
    if (active) {
      bf16x8 aq[2], ado[2];
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        aq[ks] = *(const bf16x8*)lds_at(
            lds, swz(ABF_Q + qrl * 128 + (ks * 32 + lg * 8) * 2, qrl, 7));
        ado[ks] = *(const bf16x8*)lds_at(
            lds, swz(ABF_DO + qrl * 128 + (ks * 32 + lg * 8) * 2, qrl, 7));
      }
#pragma unroll
      for (int kt = 0; kt < 8; ++kt) {
        if (kt >= NT) continue;
        const int ktok = kt * 16 + lm;
        const int vtok = ktok < S ? ktok : S - 1;
#pragma unroll
        for (int ks = 0; ks < 2; ++ks) {
          bf16x8 bk = *(const bf16x8*)lds_at(
              lds, swz(ABF_K + ktok * 128 + (ks * 32 + lg * 8) * 2, ktok, 7));
          sacc[kt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(aq[ks], bk, sacc[kt], 0, 0, 0);
          bf16x8 bv = *(const bf16x8*)(vbase + (size_t)vtok * ts + ks * 32 + lg * 8);
          dacc[kt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ado[ks], bv, dacc[kt], 0, 0, 0);
        }
      }
    }
    // ---- P, dP, rowdot, dS/pd tiles (EVERY wave writes its 16 rows) ----
    float mrow[4], lrow[4], dot[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = qtok_base + lg * 4 + r;
      const int rr = row < S ? row : S - 1;
      mrow[r] = m_io[(size_t)bh * S + rr];
      lrow[r] = 1.f / l_io[(size_t)bh * S + rr];
      dot[r] = 0.f;
    }
    // dot pass: p and the dropout-masked dp are PARKED in the pd/dS tile
    // slots (bf16) so sacc/dacc die here — the register peak would
    // otherwise spill (sacc+dacc 64 + kacc/vacc 64 persistent).
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = qtok_base + lg * 4 + r;
      const int rl = lg * 4 + r;
      uint64_t z0 = 0, z1 = 0;
      if (keep < 1.f) {
        const uint64_t zbase = ((uint64_t)bh * S + row) * 16 + lm;
        z0 = rng_hash(seed, zbase * 2);
        z1 = rng_hash(seed, zbase * 2 + 1);
      }
#pragma unroll
      for (int kt = 0; kt < 8; ++kt) {
        const int col = kt * 16 + lm;
        float p = (kt < NT)
                      ? __expf(sacc[kt][r] * scale + mval[kt] - mrow[r]) * lrow[r]
                      : 0.f;
        if (!(row < S && col < S)) p = 0.f;
        float dp = dacc[kt][r];
        if (keep < 1.f) {
          const uint64_t zz = kt < 4 ? z0 : z1;
          bool kbit = (unsigned)((zz >> (16 * (kt & 3))) & 0xFFFFu) < keep16;
          dp = kbit ? dp * inv_keep : 0.f;
        }
        dot[r] += dp * p;
        *(ushort_t*)lds_at(lds, swz(ABF_PD + (w * 16 + rl) * 256 + col * 2, rl, 15)) =
            f32_to_bf16(p);
        *(ushort_t*)lds_at(lds, swz(ABF_DS + (w * 16 + rl) * 256 + col * 2, rl, 15)) =
            f32_to_bf16(dp);
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
#pragma unroll
      for (int off = 8; off > 0; off >>= 1) dot[r] += __shfl_xor(dot[r], off, 64);
    }
    // finish pass: read the parked p/dp back, apply dropout to p -> pd,
    // ds = scale * p * (dp - dot); overwrite the same slots in place.
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = qtok_base + lg * 4 + r;
      const int rl = lg * 4 + r;
      uint64_t z0 = 0, z1 = 0;
      if (keep < 1.f) {
        const uint64_t zbase = ((uint64_t)bh * S + row) * 16 + lm;
        z0 = rng_hash(seed, zbase * 2);
        z1 = rng_hash(seed, zbase * 2 + 1);
      }
#pragma unroll
      for (int kt = 0; kt < 8; ++kt) {
        const int col = kt * 16 + lm;
        const float p = bf16_to_f32(*(const ushort_t*)lds_at(
            lds, swz(ABF_PD + (w * 16 + rl) * 256 + col * 2, rl, 15)));
        const float dp = bf16_to_f32(*(const ushort_t*)lds_at(
            lds, swz(ABF_DS + (w * 16 + rl) * 256 + col * 2, rl, 15)));
        float pd = p;
        if (keep < 1.f) {
          const uint64_t zz = kt < 4 ? z0 : z1;
          bool kbit = (unsigned)((zz >> (16 * (kt & 3))) & 0xFFFFu) < keep16;
          pd = kbit ? p * inv_keep : 0.f;
        }
        const float ds = scale * p * (dp - dot[r]);
        *(ushort_t*)lds_at(lds, swz(ABF_DS + (w * 16 + rl) * 256 + col * 2, rl, 15)) =
            f32_to_bf16(ds);
        *(ushort_t*)lds_at(lds, swz(ABF_PD + (w * 16 + rl) * 256 + col * 2, rl, 15)) =
            f32_to_bf16(pd);
      }
    }
    __syncthreads();  // tiles complete for cross-wave reads

    // ---- dQ = dS K (A = own tile rows, direct b128; B = Kt via tr16) ----
    if (active) {
      f32x4 qacc[4];
#pragma unroll
      for (int ct = 0; ct < 4; ++ct) qacc[ct] = (f32x4){0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int ks = 0; ks < 4; ++ks) {
        if (ks * 32 >= S) break;
        bf16x8 asr = *(const bf16x8*)lds_at(
            lds, swz(ABF_DS + (w * 16 + lm) * 256 + (ks * 32 + lg * 8) * 2, lm, 15));
#pragma unroll
        for (int ct = 0; ct < 4; ++ct) {
          bf16x8 bkt = abf_tr_rows64(lds, ABF_K, ks * 32 + lg * 8, ct * 4, lm);
          qacc[ct] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(asr, bkt, qacc[ct], 0, 0, 0);
        }
      }
      ushort_t* dq = dqkv + (size_t)b * S * ts + (size_t)hh * ATT_D;
#pragma unroll
      for (int ct = 0; ct < 4; ++ct) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = qtok_base + lg * 4 + r;
          const int c = ct * 16 + lm;
          if (row < S) dq[(size_t)row * ts + c] = f32_to_bf16(qacc[ct][r]);
        }
      }
    }

    // ---- dK += dS^T Q, dV += pd^T dO over this pass's 64 q rows ----
    // wave's k-tiles: 2w, 2w+1; A frags via tr16 on the tiles, B frags
    // via tr16 on the Q/dO row images (k-dim = global q row).
#pragma unroll
    for (int ks2 = 0; ks2 < 2; ++ks2) {
      const int q0t = ks2 * 32 + lg * 8;             // tile-local q base
      bf16x8 apd[2], ads[2];
#pragma unroll
      for (int ki = 0; ki < 2; ++ki) {
        const int kt = 2 * w + ki;
        apd[ki] = abf_tr_rows128(lds, ABF_PD, q0t, kt * 4, lm);
        ads[ki] = abf_tr_rows128(lds, ABF_DS, q0t, kt * 4, lm);
      }
#pragma unroll
      for (int ct = 0; ct < 4; ++ct) {
        bf16x8 bq = abf_tr_rows64(lds, ABF_Q, pass * 64 + q0t, ct * 4, lm);
        bf16x8 bdo = abf_tr_rows64(lds, ABF_DO, pass * 64 + q0t, ct * 4, lm);
#pragma unroll
        for (int ki = 0; ki < 2; ++ki) {
          vacc[ki][ct] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(apd[ki], bdo, vacc[ki][ct], 0, 0, 0);
          kacc[ki][ct] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ads[ki], bq, kacc[ki][ct], 0, 0, 0);
        }
      }
    }
    __syncthreads();  // tiles free for the next pass
  }

  // ---- write dK/dV (wave's 32 k rows) ----
  ushort_t* dk = dqkv + (size_t)b * S * ts + (size_t)(h + hh) * ATT_D;
  ushort_t* dv = dqkv + (size_t)b * S * ts + (size_t)(2 * h + hh) * ATT_D;
#pragma unroll
  for (int ki = 0; ki < 2; ++ki) {
#pragma unroll
    for (int ct = 0; ct < 4; ++ct) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = (2 * w + ki) * 16 + lg * 4 + r;
        const int c = ct * 16 + lm;
        if (row < S) {
          dk[(size_t)row * ts + c] = f32_to_bf16(kacc[ki][ct][r]);
          dv[(size_t)row * ts + c] = f32_to_bf16(vacc[ki][ct][r]);
        }
      }
    }
  }
}

SKY_EXPORT int sky_attn_bwd_fused(uint64_t stream, uint64_t qkv,
                                  uint64_t dout, uint64_t mask, uint64_t m,
                                  uint64_t lsum, uint64_t dqkv, int64_t B,
                                  int64_t S, int64_t h, int64_t d,
                                  float scale, float keep, uint64_t salt,
                                  uint64_t state) {
  if (d != ATT_D || S > ATT_SMAX) return (int)hipErrorInvalidValue;
  hipStream_t s = (hipStream_t)stream;
  dim3 grid((unsigned)(B * h));
  if (mask)
    hipLaunchKernelGGL((attn_bwd_fused_kernel<true>), grid, dim3(256),
                       ABF_LDS, s, (const ushort_t*)qkv,
                       (const ushort_t*)dout, (const ushort_t*)mask,
                       (const float*)m, (const float*)lsum, (ushort_t*)dqkv,
                       (int)B, (int)S, (int)h, scale, keep, salt,
                       (const unsigned long long*)state);
  else
    hipLaunchKernelGGL((attn_bwd_fused_kernel<false>), grid, dim3(256),
                       ABF_LDS, s, (const ushort_t*)qkv,
                       (const ushort_t*)dout, (const ushort_t*)mask,
                       (const float*)m, (const float*)lsum, (ushort_t*)dqkv,
                       (int)B, (int)S, (int)h, scale, keep, salt,
                       (const unsigned long long*)state);
  LAUNCH_CHECK();
  return 0;
}

SKY_EXPORT int sky_attn_bwd(uint64_t stream, uint64_t qkv, uint64_t dout,
                            uint64_t mask, uint64_t m, uint64_t lsum,
                            uint64_t pdT, uint64_t dsT, uint64_t dqkv,
                            int64_t B, int64_t S, int64_t h, int64_t d,
                            float scale, float keep, uint64_t salt,
                            uint64_t state) {
  if (d != ATT_D || S > ATT_SMAX) return (int)hipErrorInvalidValue;
  hipStream_t s = (hipStream_t)stream;
  bool hm = mask != 0;
  // B1 variant: split-q (2 blocks per (b,h), 48K LDS, 3 waves/SIMD) is the
  // measured default; SKY_ATTN_BWD1=wide selects the one-block variant.
  const char* e = getenv("SKY_ATTN_BWD1");
  const bool wide = e && e[0] == 'w';
  if (wide) {
    dim3 grid((unsigned)(B * h));
    if (hm)
      hipLaunchKernelGGL((attn_bwd1_kernel<true>), grid, dim3(256), 80 * 1024, s,
                         (const ushort_t*)qkv, (const ushort_t*)dout,
                         (const ushort_t*)mask, (const float*)m,
                         (const float*)lsum, (ushort_t*)pdT, (ushort_t*)dsT,
                         (ushort_t*)dqkv, (int)B, (int)S, (int)h, scale, keep,
                         salt, (const unsigned long long*)state);
    else
      hipLaunchKernelGGL((attn_bwd1_kernel<false>), grid, dim3(256), 80 * 1024, s,
                         (const ushort_t*)qkv, (const ushort_t*)dout,
                         (const ushort_t*)mask, (const float*)m,
                         (const float*)lsum, (ushort_t*)pdT, (ushort_t*)dsT,
                         (ushort_t*)dqkv, (int)B, (int)S, (int)h, scale, keep,
                         salt, (const unsigned long long*)state);
  } else {
    dim3 grid((unsigned)(B * h * ((S + 63) / 64)));
    if (hm)
      hipLaunchKernelGGL((attn_bwd1s_kernel<true>), grid, dim3(256), 48 * 1024, s,
                         (const ushort_t*)qkv, (const ushort_t*)dout,
                         (const ushort_t*)mask, (const float*)m,
                         (const float*)lsum, (ushort_t*)pdT, (ushort_t*)dsT,
                         (ushort_t*)dqkv, (int)B, (int)S, (int)h, scale, keep,
                         salt, (const unsigned long long*)state);
    else
      hipLaunchKernelGGL((attn_bwd1s_kernel<false>), grid, dim3(256), 48 * 1024, s,
                         (const ushort_t*)qkv, (const ushort_t*)dout,
                         (const ushort_t*)mask, (const float*)m,
                         (const float*)lsum, (ushort_t*)pdT, (ushort_t*)dsT,
                         (ushort_t*)dqkv, (int)B, (int)S, (int)h, scale, keep,
                         salt, (const unsigned long long*)state);
  }
  hipLaunchKernelGGL(attn_bwd2_kernel, dim3((unsigned)(B * h)), dim3(256),
                     32 * 1024, s, (const ushort_t*)qkv,
                     (const ushort_t*)dout, (const ushort_t*)pdT,
                     (const ushort_t*)dsT, (ushort_t*)dqkv, (int)B, (int)S,
                     (int)h);
  LAUNCH_CHECK();
  return 0;
}

// ============================================================================
// Flash-style attention forward for ARBITRARY sequence length (d = 64).
//
// Online-softmax over 128-key kv tiles: one workgroup per
// (batch, head, 128-query block); running row max m / row sum l with a
// rescale of the O accumulator at every tile (plain rescale, no defer —
// the guide's T13 hazard does not apply). The S x S matrix never exists;
// row stats are saved so the Python backward can recompute probabilities.
// Dropout uses LINEAR element indices ((bh*S+row)*S+col) so the generic
// dropout kernels regenerate the same mask in the decomposed backward.
//
// LDS reuses the S<=128 kernel's 48 KB layout per kv tile: K staged via
// global_load_lds with a source-side XOR swizzle, V^T aliased over K after
// QK^T (two barriers per tile), per-wave P tile.
// ============================================================================

// Flash forward v2: one workgroup covers QPW 128-query blocks of one
// (batch, head) — K and V^T live in SEPARATE LDS regions staged ONCE per
// kv tile and reused across the q blocks (the v1 kernel aliased V over K
// and restaged both for every q block: S/128x K/V read amplification,
// 176 vs 107 us at S=512 against the decomposed path). Consecutive
// blockIdx values = same (b,h), so with the XCD-contiguous remap a head's
// K/V streams from its XCD's L2 across its q-block workgroups.
// LDS: K [128][64] @0 (16K), V^T [64][128] @16K, P per wave @32K+w*8K.
#define F2_K_OFF 0
#define F2_VT_OFF (16 * 1024)
#define F2_P_OFF (32 * 1024)
#define F2_LDS_BYTES (64 * 1024)

template <int QPW>
__global__ __launch_bounds__(256, 1) void attn_flash_fwd_kernel(
    const ushort_t* __restrict__ qkv, const ushort_t* __restrict__ mask,
    ushort_t* __restrict__ out, float* __restrict__ m_io,
    float* __restrict__ l_io, int B, int S, int h, float scale, float keep,
    uint64_t salt, const unsigned long long* __restrict__ state) {
  extern __shared__ __attribute__((aligned(16))) char lds[];
  const int tid = threadIdx.x;
  const int l = tid & 63;
  const int w = tid >> 6;
  const int nqb = (S + 127) / 128;
  const int nchunk = (nqb + QPW - 1) / QPW;
  int bid = blockIdx.x;
  {  // XCD-contiguous remap (bijective): consecutive ids share an XCD
    const int nwg = gridDim.x;
    const int q = nwg / 8, rr = nwg % 8;
    const int xcd = bid % 8, idx = bid / 8;
    bid = (xcd < rr ? xcd * (q + 1) : rr * (q + 1) + (xcd - rr) * q) + idx;
  }
  const int bh = bid / nchunk;
  const int qc = bid % nchunk;
  const int b = bh / h;
  const int hh = bh % h;
  const int ts = 3 * h * ATT_D;
  const uint64_t seed = salt + (state ? *state : 0ull) * 0xD1B54A32D192ED03ull;
  const float inv_keep = 1.f / keep;
  const unsigned keep16 = keep_to_16(keep);

  const ushort_t* qbase = qkv + (size_t)b * S * ts + (size_t)hh * ATT_D;
  const ushort_t* kbase = qbase + (size_t)h * ATT_D;
  const ushort_t* vbase = qbase + (size_t)2 * h * ATT_D;

  const int lm = l & 15;
  const int lg = l >> 4;
  const int q0 = qc * QPW * 128;    // first query row of this chunk
  const int qt0 = 2 * w;            // wave's first q tile within a block

  // Q fragments for the wave's 2 tiles in each of the QPW blocks
  bf16x8 aq_all[QPW][2][2];
#pragma unroll
  for (int blk = 0; blk < QPW; ++blk)
#pragma unroll
    for (int qi = 0; qi < 2; ++qi) {
      int qtok = q0 + blk * 128 + (qt0 + qi) * 16 + lm;
      if (qtok >= S) qtok = S - 1;
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
        aq_all[blk][qi][ks] =
            *(const bf16x8*)(qbase + (size_t)qtok * ts + ks * 32 + lg * 8);
    }

  float mx[QPW][2][4], sm[QPW][2][4];
  f32x4 oacc[QPW][2][4];
#pragma unroll
  for (int blk = 0; blk < QPW; ++blk)
#pragma unroll
    for (int qi = 0; qi < 2; ++qi)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        mx[blk][qi][r] = -3.0e38f;
        sm[blk][qi][r] = 0.f;
        oacc[blk][qi][r] = (f32x4){0.f, 0.f, 0.f, 0.f};
      }

  const int nkv = (S + 127) / 128;
  for (int kv = 0; kv < nkv; ++kv) {
    const int k0 = kv * 128;
    const int ktiles = min(8, (S - k0 + 15) / 16);
    __syncthreads();  // previous tile's K/Vt reads complete before overwrite
    // ---- stage K tile [128][64] via glds (source-side swizzle) ----
    for (int u = tid; u < 128 * 8; u += 256) {
      const int tok = u >> 3;
      int gtok = k0 + tok;
      if (gtok >= S) gtok = S - 1;  // clamped; masked off via mval
      const int c16s = (u & 7) ^ (tok & 7);
      __builtin_amdgcn_global_load_lds(
          (att_gas)(kbase + (size_t)gtok * ts + c16s * 8),
          (att_las)lds_at(lds, F2_K_OFF + u * 16), 16, 0, 0);
    }
    // ---- stage V^T [64 d][128 tok] (register transpose) ----
    for (int u = tid; u < 128 * 8; u += 256) {
      const int tok = u >> 3;
      const int gtok = k0 + tok;
      const int c16 = u & 7;
      ushort8_t v;
      if (gtok < S)
        v = *(const ushort8_t*)(vbase + (size_t)gtok * ts + c16 * 8);
      else
        v = (ushort8_t)(ushort_t)0;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int c = c16 * 8 + j;
        *(ushort_t*)lds_at(lds, swz(F2_VT_OFF + c * 256 + tok * 2, c, 15)) = v[j];
      }
    }
    // additive mask values for this tile's columns
    float mval[8];
#pragma unroll
    for (int kt = 0; kt < 8; ++kt) {
      const int col = k0 + kt * 16 + lm;
      float mv = 0.f;
      if (mask && col < S) mv = bf16_to_f32(mask[(size_t)b * S + col]);
      mval[kt] = (col < S) ? mv : -3.0e38f;
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();

#pragma clang loop unroll(disable)
    for (int blk = 0; blk < QPW; ++blk) {
      if (q0 + blk * 128 >= S) break;
#pragma clang loop unroll(disable)
      for (int qi = 0; qi < 2; ++qi) {
        const int qrow_base = q0 + blk * 128 + (qt0 + qi) * 16;
        if (qrow_base >= S) break;
        f32x4 sacc[8];
#pragma unroll
        for (int kt = 0; kt < 8; ++kt) sacc[kt] = (f32x4){0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int kt = 0; kt < 8; ++kt) {
          if (kt >= ktiles) continue;
          const int ktok = kt * 16 + lm;
#pragma unroll
          for (int ks = 0; ks < 2; ++ks) {
            bf16x8 bk = *(const bf16x8*)lds_at(
                lds, swz(F2_K_OFF + ktok * 128 + (ks * 32 + lg * 8) * 2, ktok, 7));
            sacc[kt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                aq_all[blk][qi][ks], bk, sacc[kt], 0, 0, 0);
          }
        }
        // online softmax update for this tile
        float tmx[4];
#pragma unroll
        for (int r = 0; r < 4; ++r) tmx[r] = -3.0e38f;
#pragma unroll
        for (int kt = 0; kt < 8; ++kt) {
          if (kt >= ktiles) continue;
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            float zz = sacc[kt][r] * scale + mval[kt];
            sacc[kt][r] = zz;
            tmx[r] = fmaxf(tmx[r], zz);
          }
        }
#pragma unroll
        for (int r = 0; r < 4; ++r) {
#pragma unroll
          for (int off = 8; off > 0; off >>= 1)
            tmx[r] = fmaxf(tmx[r], __shfl_xor(tmx[r], off, 64));
          const float mn = fmaxf(mx[blk][qi][r], tmx[r]);
          const float f = __expf(mx[blk][qi][r] - mn);
          mx[blk][qi][r] = mn;
          sm[blk][qi][r] *= f;
#pragma unroll
          for (int dv = 0; dv < 4; ++dv) oacc[blk][qi][dv][r] *= f;
        }
        float tsum[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int kt = 0; kt < 8; ++kt) {
          if (kt >= ktiles) continue;
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            float p = __expf(sacc[kt][r] - mx[blk][qi][r]);
            sacc[kt][r] = p;
            tsum[r] += p;
          }
        }
#pragma unroll
        for (int r = 0; r < 4; ++r) {
#pragma unroll
          for (int off = 8; off > 0; off >>= 1)
            tsum[r] += __shfl_xor(tsum[r], off, 64);
          sm[blk][qi][r] += tsum[r];
        }
        // dropout + UNNORMALIZED P tile to per-wave LDS (linear-index RNG)
#pragma unroll
        for (int kt = 0; kt < 8; ++kt) {
          const int col = k0 + kt * 16 + lm;
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int row = qrow_base + lg * 4 + r;
            float p = (kt < ktiles) ? sacc[kt][r] : 0.f;
            if (keep < 1.f && p != 0.f && row < S && col < S) {
              const uint64_t idx = ((uint64_t)bh * S + row) * S + col;
              p = rng_keep16(seed, idx, keep16) ? p * inv_keep : 0.f;
            }
            const int rl = qi * 16 + lg * 4 + r;
            *(ushort_t*)lds_at(
                lds,
                swz(F2_P_OFF + w * 8192 + rl * 256 + (kt * 16 + lm) * 2, rl, 15)) =
                f32_to_bf16((row < S && col < S) ? p : 0.f);
          }
        }
        // ---- PV for this qi straight away (V^T resident, P per-wave) ----
        // same-wave LDS RAW on the P tile: wait the ds_writes explicitly
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        {
          const int rl = qi * 16 + lm;
#pragma unroll
          for (int ks = 0; ks < 4; ++ks) {
            bf16x8 ap = *(const bf16x8*)lds_at(
                lds, swz(F2_P_OFF + w * 8192 + rl * 256 + (ks * 32 + lg * 8) * 2,
                         rl, 15));
#pragma unroll
            for (int dvt = 0; dvt < 4; ++dvt) {
              const int dv = dvt * 16 + lm;
              bf16x8 bv = *(const bf16x8*)lds_at(
                  lds, swz(F2_VT_OFF + dv * 256 + (ks * 32 + lg * 8) * 2, dv, 15));
              oacc[blk][qi][dvt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                  ap, bv, oacc[blk][qi][dvt], 0, 0, 0);
            }
          }
        }
      }
    }
  }

  // ---- epilogue: normalize by l, write out + stats ----
#pragma unroll
  for (int blk = 0; blk < QPW; ++blk) {
#pragma unroll
    for (int qi = 0; qi < 2; ++qi) {
      const int qrow_base = q0 + blk * 128 + (qt0 + qi) * 16;
      if (qrow_base >= S) break;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = qrow_base + lg * 4 + r;
        const float inv = 1.f / sm[blk][qi][r];
        if (lm == 0 && row < S) {
          m_io[(size_t)bh * S + row] = mx[blk][qi][r];
          l_io[(size_t)bh * S + row] = sm[blk][qi][r];
        }
#pragma unroll
        for (int dvt = 0; dvt < 4; ++dvt) {
          const int dv = dvt * 16 + lm;
          if (row < S)
            out[(((size_t)b * S + row) * h + hh) * ATT_D + dv] =
                f32_to_bf16(oacc[blk][qi][dvt][r] * inv);
        }
      }
    }
  }
}

SKY_EXPORT int sky_attn_flash_fwd(uint64_t stream, uint64_t qkv,
                                  uint64_t mask, uint64_t out, uint64_t m,
                                  uint64_t lsum, int64_t B, int64_t S,
                                  int64_t h, int64_t d, float scale,
                                  float keep, uint64_t salt, uint64_t state) {
  if (d != ATT_D) return (int)hipErrorInvalidValue;
  hipStream_t s = (hipStream_t)stream;
  const int nqb = (int)((S + 127) / 128);
  // 2 query blocks per workgroup where the grid stays >= 256 WGs
  int qpw = (nqb >= 2 && B * h * ((nqb + 1) / 2) >= 512) ? 2 : 1;
  const int nchunk = (nqb + qpw - 1) / qpw;
  dim3 grid((unsigned)(B * h * nchunk));
#define F2L(QPW)                                                               \
  hipLaunchKernelGGL((attn_flash_fwd_kernel<QPW>), grid, dim3(256),            \
                     F2_LDS_BYTES, s, (const ushort_t*)qkv,                    \
                     (const ushort_t*)mask, (ushort_t*)out, (float*)m,         \
                     (float*)lsum, (int)B, (int)S, (int)h, scale, keep, salt,  \
                     (const unsigned long long*)state)
  if (qpw == 2) F2L(2);
  else F2L(1);
#undef F2L

  LAUNCH_CHECK();
  return 0;
}
