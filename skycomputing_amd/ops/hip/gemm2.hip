// 256x256x64 8-phase bf16 MFMA GEMM ("v2") for gfx950.
//
// Replaces hipBLASLt for the BERT linear-layer forwards (SURVEY.md §2c rows
// 1/5/6; reference call sites scaelum/model/bert_layers.py:227-229,281,319,
// 60-108 use nn.Linear -> cuBLAS). The v1 kernel (gemm.hip) is the guide's
// "step-3" 128^2 structure whose __syncthreads-drained staging ceilings at
// ~880 TF; this kernel is the guide's 256^2 8-phase template
// (cdna_hip_programming.md §5 "The 256² 8-phase template"): raw s_barrier +
// counted vmcnt so global_load_lds prefetches stay in flight across
// barriers, 8 waves (2Mx4N) each owning a 128x64 accumulator, MFMA issued
// one accumulator QUADRANT per phase while one 16 KB half-tile stages.
//
// Geometry: BM=BN=256, BK=64, 512 threads. LDS = 2 buffers x (A 256x64 +
// B 256x64) bf16 = 128 KiB. A-half h = tile rows with bit6==h (the rows the
// phase-pair mh==h reads); B-half h = tile cols with bit5==h. Quadrant
// order per K-tile: (mh,nh) = (0,0),(0,1),(1,1),(1,0) — consecutive phases
// share one operand's fragments, so phases issue 12/4/8/4 ds_read_b128.
//
// Staging schedule (phase φ = 4t+p while computing K-tile t):
//   p0: issue A1[t+1], B0[t+1]   (into buffer (t+1)&1)
//   p1: —
//   p2: issue A0[t+2]            (into buffer t&1, slot freed at p1)
//   p3: issue B1[t+2]; s_waitcnt vmcnt(4) before the closing barrier
// Every half is issued >=1 full phase after the last ds_read of the slot's
// previous content and >=4 phases before its first read; the vmcnt(4) at
// each tile boundary retires everything except the last two halves
// (2 glds16/thread each). Prologue stages A0[0],B1[0],A1[0],B0[0],A0[1],
// B1[1] then waits vmcnt(4) (tile 0 fully resident).
//
// Split-K (gsu>1): grid = ntiles*gsu, slice g computes K-range
// [g*K/gsu, ...) and writes fp32 partials to Wk[g][M][N] (non-temporal);
// sky_gemm2_reduce sums slices and applies the epilogue. Used where the
// tile grid alone cannot fill 256 CUs (N=1024 outputs: 64 tiles).

#include "common.h"

typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define G2_BM 256
#define G2_BN 256
#define G2_BK 64
#define G2_BLOCK 512
#define G2_LDS (128 * 1024)

typedef const __attribute__((address_space(1))) unsigned int* g2_gas;
typedef __attribute__((address_space(3))) unsigned int* g2_las;

DEV void g2_glds16(const void* g, void* l) {
  __builtin_amdgcn_global_load_lds((g2_gas)g, (g2_las)l, 16, 0, 0);
}

// LDS images are lane-linear under glds; the XOR bank swizzle
// (byte ^= ((row&7)<<4)) is applied on the SOURCE column (guide rule 21).
DEV int g2_swz(int byte_addr, int row) { return byte_addr ^ ((row & 7) << 4); }

enum { G2_EPI_NONE = 0, G2_EPI_BIAS = 1, G2_EPI_BIAS_GELU = 2 };

// operand storage modes: DIRECT = stored [out][red] (reduction contiguous,
// staged as [out][64k] images read by ds_read_b128); KMAJOR = stored
// [red][out] (out contiguous — dgrad's W, wgrad's dY/X), staged as
// [half][64k][128out] images (glds along out) and read TRANSPOSED with
// ds_read_b64_tr_b16 (semantics measured by tools/tr16_probe.hip:
// result(lane 4q+r, reg j) = elem at lane(4j+q).addr + r).
enum { G2_DIRECT = 0, G2_KMAJOR = 1 };

typedef __attribute__((ext_vector_type(4))) short s16x4;
typedef __attribute__((address_space(3))) s16x4* g2_las4;

// k-dependent XOR on within-half column quads (bank spread for the tr16
// reads; bits 2-4 of the quad index, preserves 16-B staging granularity)
DEV int g2_kswz(int k) { return 4 * ((((k >> 3) & 1) << 2) | (k & 3)); }

// ---- half-tile staging (16 KB, 2 x glds16 per thread) ----
// A image: [256][64] bf16 row-major at abuf; half h = rows {h*64..h*64+63,
// 128+h*64..}; chunk c covers rows c*128+h*64+rl.
DEV void g2_stageA(const ushort_t* __restrict__ A, int lda, int m0, int k0,
                   char* abuf, int h, int tid) {
  const int rl = tid >> 3, c16 = tid & 7;
#pragma unroll
  for (int c = 0; c < 2; ++c) {
    const int row = c * 128 + h * 64 + rl;
    const int c16s = c16 ^ (row & 7);
    const ushort_t* g = A + (size_t)(m0 + row) * lda + k0 + c16s * 8;
    g2_glds16(g, abuf + row * 128 + c16 * 16);
  }
}

// B image: [256][64] bf16 row-major at bbuf (row = output col n); half h =
// cols n with bit5==h: n = chunk*64 + h*32 + rl32, 4 chunks of 32 rows.
DEV void g2_stageB(const ushort_t* __restrict__ B, int ldb, int n0, int k0,
                   char* bbuf, int h, int tid) {
  const int c16 = tid & 7;
#pragma unroll
  for (int c = 0; c < 2; ++c) {
    const int u = tid + c * 512;
    const int chunk = u >> 8;
    const int n = chunk * 64 + h * 32 + ((u & 255) >> 3);
    const int c16s = c16 ^ (n & 7);
    const ushort_t* g = B + (size_t)(n0 + n) * ldb + k0 + c16s * 8;
    g2_glds16(g, bbuf + n * 128 + c16 * 16);
  }
}

// KMAJOR half staging: operand stored [red][out] (ld = out-row stride);
// image = [half 16KB][k 0..63][16 x 16B slots]; slot v16 of row k holds the
// 8 source columns starting at out-offset of quad Q = (2*v16) ^ g2_kswz(k).
// outsel(h, Q) maps a within-half quad to the tile-local out coordinate:
//   A halves split on out bit6: out = (Q>>4)*128 + h*64 + (Q&15)*4
//   B halves split on out bit5: out = (Q>>3)*64  + h*32 + (Q&7)*4
template <bool BHALF>
DEV void g2_stage_kmaj(const ushort_t* __restrict__ X, int ld, int out0,
                       int k0, char* img, int h, int tid) {
#pragma unroll
  for (int c = 0; c < 2; ++c) {
    const int u = tid + c * 512;
    const int k = u >> 4, v16 = u & 15;
    const int Q = (2 * v16) ^ g2_kswz(k);
    const int out = BHALF ? (Q >> 3) * 64 + h * 32 + (Q & 7) * 4
                          : (Q >> 4) * 128 + h * 64 + (Q & 15) * 4;
    const ushort_t* g = X + (size_t)(k0 + k) * ld + out0 + out;
    g2_glds16(g, img + h * 16384 + k * 256 + v16 * 16);
  }
}

template <int AM, int BM, int EPI, bool STORE_Z, bool SPLIT>
__global__ __launch_bounds__(G2_BLOCK, 1) void gemm2_kernel(
    const ushort_t* __restrict__ A, const ushort_t* __restrict__ B,
    ushort_t* __restrict__ C, const ushort_t* __restrict__ bias,
    ushort_t* __restrict__ Z, float* __restrict__ Wk, int M, int N, int K,
    int lda, int ldb, int ldc, int gsu) {
  extern __shared__ __attribute__((aligned(16))) char lds[];
  const int tid = threadIdx.x;
  const int l = tid & 63;
  const int w = tid >> 6;
  const int lm = l & 15, lg = l >> 4;
  const int wr = w >> 2, wc = w & 3;  // 8 waves as 2M x 4N

  // XCD-aware bijective remap (guide T1), then a supertile decode: within
  // an XCD's contiguous logical range, tiles cover a 4(M)x*(N) rectangle
  // instead of a full M-row — the 32-tile group's A panels (2 MB) AND a
  // few B panels co-reside in the XCD's 4 MB L2, cutting HBM re-reads
  // (row-major order re-reads every B panel per XCD).
  int bid = blockIdx.x;
  {
    const int nwg = gridDim.x;
    const int q = nwg / 8, rr = nwg % 8;
    const int xcd = bid % 8, idx = bid / 8;
    bid = (xcd < rr ? xcd * (q + 1) : rr * (q + 1) + (xcd - rr) * q) + idx;
  }
  const int ntn = N / G2_BN;
  const int ntm = M / G2_BM;
  const int ntiles = ntm * ntn;
  const int g = SPLIT ? bid / ntiles : 0;
  int tile = SPLIT ? bid % ntiles : bid;
  {
    const int SM = (ntm >= 4) ? 4 : ntm;  // supertile height (m tiles)
    const int stripe = SM * ntn;
    const int mb = tile / stripe, r = tile % stripe;
    tile = (mb * SM + r % SM) * ntn + r / SM;
  }
  const int m0 = (tile / ntn) * G2_BM;
  const int n0 = (tile % ntn) * G2_BN;
  const int kpg = SPLIT ? K / gsu : K;
  const int kbase = g * kpg;
  const int NT = kpg / G2_BK;

  auto issueA = [&](int t, int h) {
    char* abuf = lds + (t & 1) * 65536;
    const int tt = t < NT ? t : NT - 1;  // clamp SOURCE only; slot stays t&1
    if (AM == G2_DIRECT)
      g2_stageA(A, lda, m0, kbase + tt * G2_BK, abuf, h, tid);
    else
      g2_stage_kmaj<false>(A, lda, m0, kbase + tt * G2_BK, abuf, h, tid);
  };
  auto issueB = [&](int t, int h) {
    char* bbuf = lds + (t & 1) * 65536 + 32768;
    const int tt = t < NT ? t : NT - 1;
    if (BM == G2_DIRECT)
      g2_stageB(B, ldb, n0, kbase + tt * G2_BK, bbuf, h, tid);
    else
      g2_stage_kmaj<true>(B, ldb, n0, kbase + tt * G2_BK, bbuf, h, tid);
  };

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4){0.f, 0.f, 0.f, 0.f};

  // prologue: both LDS buffers staged (tiles 0 and 1), in steady-state
  // issue order; the vmcnt(8) retires tile 0, leaving tile 1's 8 ops in
  // flight — the same depth the post-MFMA issue schedule maintains.
  issueA(0, 0);
  issueB(0, 1);
  issueA(0, 1);
  issueB(0, 0);
  issueA(1, 0);
  issueB(1, 1);
  issueA(1, 1);
  issueB(1, 0);
  asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  bf16x8 af[4][2], bfr[2][2];
  const int swz_x = (lm & 7) << 4;  // row&7 == lm&7 for every fragment row

  const int s4 = l & 15;  // tr16 source-lane index within the 16-lane group

  // transposed fragment read from a KMAJOR image: one tr16 pair yields the
  // 8 k-values of out-row (frag base + lm); lane s supplies the address of
  // (k-row k0 + s>>2, within-half quad bq + (s&3)) per the probed mapping.
  auto read_kmaj = [&](char* img, int h, int bq, int ks) -> bf16x8 {
    s16x4 v[2];
#pragma unroll
    for (int half = 0; half < 2; ++half) {
      const int kl = ks * 32 + lg * 8 + half * 4 + (s4 >> 2);
      const int byte = h * 16384 + kl * 256 +
                       (((bq + (s4 & 3)) ^ g2_kswz(kl)) << 3);
      v[half] = __builtin_amdgcn_ds_read_tr16_b64_v4i16((g2_las4)(img + byte));
    }
    return (bf16x8){v[0][0], v[0][1], v[0][2], v[0][3],
                    v[1][0], v[1][1], v[1][2], v[1][3]};
  };

  auto readA = [&](char* abuf, int mh) {
#pragma unroll
    for (int i = 0; i < 4; ++i) {
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        if (AM == G2_DIRECT) {
          const int row = wr * 128 + mh * 64 + i * 16 + lm;
          af[i][ks] = *(const bf16x8*)(abuf +
                                       ((row * 128 + (ks * 32 + lg * 8) * 2) ^ swz_x));
        } else {
          af[i][ks] = read_kmaj(abuf, mh, wr * 16 + i * 4, ks);
        }
      }
    }
  };
  auto readB = [&](char* bbuf, int nh) {
#pragma unroll
    for (int j = 0; j < 2; ++j) {
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        if (BM == G2_DIRECT) {
          const int n = wc * 64 + nh * 32 + j * 16 + lm;
          bfr[j][ks] = *(const bf16x8*)(bbuf +
                                        ((n * 128 + (ks * 32 + lg * 8) * 2) ^ swz_x));
        } else {
          bfr[j][ks] = read_kmaj(bbuf, nh, wc * 8 + j * 4, ks);
        }
      }
    }
  };
  auto mma_span = [&](int mh, int nh, int i0, int i1) {
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      if (i < i0 || i >= i1) continue;
#pragma unroll
      for (int j = 0; j < 2; ++j)
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
          acc[mh * 4 + i][nh * 2 + j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[i][ks], bfr[j][ks], acc[mh * 4 + i][nh * 2 + j], 0, 0, 0);
    }
    __builtin_amdgcn_s_setprio(0);
  };
  auto mma = [&](int mh, int nh) { mma_span(mh, nh, 0, 4); };

  // nounroll: a x2-unrolled body gets per-copy acc register assignments and
  // the compiler bridges them with ~256 v_mov per iteration (measured in
  // the asm); a single body keeps acc pinned.
#pragma clang loop unroll(disable)
  // Post-MFMA prefetch issue: each phase's glds targets the slot whose
  // last ds_read RETIRED at this phase's own lgkmcnt(0) — so all four of
  // tile t+2's halves issue during tile t (one phase after the slot
  // frees), keeping ~2 full tiles (8-16 glds) in flight instead of 2
  // halves. The boundary wait relaxes to vmcnt(8): tile t+1's issues may
  // stay outstanding; everything older (tile t+2's data... consumed next)
  // has retired.
  for (int t = 0; t < NT; ++t) {
    char* abuf = lds + (t & 1) * 65536;
    char* bbuf = abuf + 32768;
    // phase 0: quadrant (0,0)
    readA(abuf, 0);
    readB(bbuf, 0);
    __builtin_amdgcn_s_barrier();
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    mma(0, 0);
    __builtin_amdgcn_s_barrier();
    // phase 1: quadrant (0,1); af reused; A0[t]'s slot freed at p0
    readB(bbuf, 1);
    __builtin_amdgcn_s_barrier();
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    mma_span(0, 1, 0, 2);
    issueA(t + 2, 0);   // DMA launches under the second MFMA half
    mma_span(0, 1, 2, 4);
    __builtin_amdgcn_s_barrier();
    // phase 2: quadrant (1,1); bfr reused; B1[t]'s slot freed at p1
    readA(abuf, 1);
    __builtin_amdgcn_s_barrier();
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    mma_span(1, 1, 0, 2);
    issueB(t + 2, 1);
    mma_span(1, 1, 2, 4);
    __builtin_amdgcn_s_barrier();
    // phase 3: quadrant (1,0); A1/B0[t]'s slots free at this phase's lgkm
    readB(bbuf, 0);
    __builtin_amdgcn_s_barrier();
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    mma_span(1, 0, 0, 2);
    issueA(t + 2, 1);
    mma_span(1, 0, 2, 4);
    issueB(t + 2, 0);
    asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }

  if (SPLIT) {
    // fp32 partials, non-temporal (touch-once until the reduce kernel)
    float* out = Wk + (size_t)g * M * N;
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      const int m = m0 + wr * 128 + i * 16 + lg * 4;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int n = n0 + wc * 64 + j * 16 + lm;
#pragma unroll
        for (int r = 0; r < 4; ++r)
          __builtin_nontemporal_store(acc[i][j][r],
                                      out + (size_t)(m + r) * N + n);
      }
    }
    return;
  }

  float bv[4];
  if (EPI != G2_EPI_NONE) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int n = n0 + wc * 64 + j * 16 + lm;
      bv[j] = bias ? bf16_to_f32(bias[n]) : 0.f;
    }
  }
#pragma unroll
  for (int i = 0; i < 8; ++i) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int m = m0 + wr * 128 + i * 16 + lg * 4 + r;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int n = n0 + wc * 64 + j * 16 + lm;
        float v = acc[i][j][r];
        if (EPI != G2_EPI_NONE) v += bv[j];
        if (EPI == G2_EPI_BIAS_GELU) {
          if (STORE_Z) Z[(size_t)m * ldc + n] = f32_to_bf16(v);
          v = gelu_f(v);
        }
        C[(size_t)m * ldc + n] = f32_to_bf16(v);
      }
    }
  }
}

// ---- split-K reduce + epilogue ----
template <int EPI, bool STORE_Z>
__global__ __launch_bounds__(256) void gemm2_reduce_kernel(
    const float* __restrict__ Wk, ushort_t* __restrict__ C,
    const ushort_t* __restrict__ bias, ushort_t* __restrict__ Z, long MN,
    int N, int ldc, int gsu) {
  const long i4 = ((long)blockIdx.x * 256 + threadIdx.x) * 4;
  if (i4 >= MN) return;
  f32x4 s = *(const f32x4*)(Wk + i4);
  for (int g = 1; g < gsu; ++g) {
    f32x4 p = *(const f32x4*)(Wk + (size_t)g * MN + i4);
#pragma unroll
    for (int r = 0; r < 4; ++r) s[r] += p[r];
  }
  const int m = (int)(i4 / N), n = (int)(i4 % N);
  ushort4_t o, zo;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    float v = s[r];
    if (EPI != G2_EPI_NONE) v += bf16_to_f32(bias[n + r]);
    if (EPI == G2_EPI_BIAS_GELU) {
      if (STORE_Z) zo[r] = f32_to_bf16(v);
      v = gelu_f(v);
    }
    o[r] = f32_to_bf16(v);
  }
  *(ushort4_t*)(C + (size_t)m * ldc + n) = o;
  if (EPI == G2_EPI_BIAS_GELU && STORE_Z) *(ushort4_t*)(Z + (size_t)m * ldc + n) = zo;
}

static int g2_set_lds(const void* f) {
  // dynamic LDS >64 KiB needs the opt-in attribute, once per kernel
  static const void* done[32];
  static int ndone = 0;
  for (int i = 0; i < ndone; ++i)
    if (done[i] == f) return 0;
  hipError_t e = hipFuncSetAttribute(
      f, hipFuncAttributeMaxDynamicSharedMemorySize, G2_LDS);
  if (e != hipSuccess) return (int)e;
  if (ndone < 32) done[ndone++] = f;
  return 0;
}

SKY_EXPORT int sky_gemm2(uint64_t stream, uint64_t A, uint64_t B, uint64_t C,
                         uint64_t bias, uint64_t Z, uint64_t Wk, int64_t M,
                         int64_t N, int64_t K, int64_t lda, int64_t ldb,
                         int64_t ldc, int transA, int transB, int epi,
                         int gsu) {
  if (M % G2_BM || N % G2_BN || K % G2_BK) return (int)hipErrorInvalidValue;
  if (gsu > 1 && (K % (gsu * G2_BK) || !Wk)) return (int)hipErrorInvalidValue;
  // 16-B vector I/O: every row stride must keep 8-element alignment
  if (lda % 8 || ldb % 8 || ldc % 8) return (int)hipErrorInvalidValue;
  hipStream_t s = (hipStream_t)stream;
  const long ntiles = (M / G2_BM) * (N / G2_BN);
  dim3 grid((unsigned)(ntiles * (gsu > 1 ? gsu : 1)));
  const bool split = gsu > 1;
  const bool sz = Z != 0;
  int rc = 0;
#define G2K(AM, BM, EP, SZ, SP)                                                \
  do {                                                                         \
    rc = g2_set_lds((const void*)&gemm2_kernel<AM, BM, EP, SZ, SP>);           \
    if (rc) return rc;                                                         \
    hipLaunchKernelGGL((gemm2_kernel<AM, BM, EP, SZ, SP>), grid,               \
                       dim3(G2_BLOCK), G2_LDS, s, (const ushort_t*)A,          \
                       (const ushort_t*)B, (ushort_t*)C,                       \
                       (const ushort_t*)bias, (ushort_t*)Z, (float*)Wk,        \
                       (int)M, (int)N, (int)K, (int)lda, (int)ldb, (int)ldc,   \
                       gsu);                                                   \
  } while (0)
  // supported orientations: NT fwd (direct,direct) with full epilogues;
  // dgrad NN (direct,kmajor) and wgrad TN (kmajor,kmajor) epilogue-free
  if (transA == 0 && transB == 0) {
    // split kernels ignore the epilogue (applied in the reduce)
    if (split) {
      G2K(G2_DIRECT, G2_DIRECT, G2_EPI_NONE, false, true);
    } else if (epi == G2_EPI_NONE) {
      G2K(G2_DIRECT, G2_DIRECT, G2_EPI_NONE, false, false);
    } else if (epi == G2_EPI_BIAS) {
      G2K(G2_DIRECT, G2_DIRECT, G2_EPI_BIAS, false, false);
    } else if (sz) {
      G2K(G2_DIRECT, G2_DIRECT, G2_EPI_BIAS_GELU, true, false);
    } else {
      G2K(G2_DIRECT, G2_DIRECT, G2_EPI_BIAS_GELU, false, false);
    }
  } else if (transA == 0 && transB == 1) {
    if (epi != G2_EPI_NONE && !split) return (int)hipErrorInvalidValue;
    if (split) G2K(G2_DIRECT, G2_KMAJOR, G2_EPI_NONE, false, true);
    else G2K(G2_DIRECT, G2_KMAJOR, G2_EPI_NONE, false, false);
  } else if (transA == 1 && transB == 1) {
    if (epi != G2_EPI_NONE && !split) return (int)hipErrorInvalidValue;
    if (split) G2K(G2_KMAJOR, G2_KMAJOR, G2_EPI_NONE, false, true);
    else G2K(G2_KMAJOR, G2_KMAJOR, G2_EPI_NONE, false, false);
  } else {
    return (int)hipErrorInvalidValue;
  }
#undef G2K
  LAUNCH_CHECK();
  if (split) {
    const long MN = M * N;
    dim3 rgrid((unsigned)((MN / 4 + 255) / 256));
#define G2R(EP, SZ)                                                            \
  hipLaunchKernelGGL((gemm2_reduce_kernel<EP, SZ>), rgrid, dim3(256), 0, s,    \
                     (const float*)Wk, (ushort_t*)C, (const ushort_t*)bias,    \
                     (ushort_t*)Z, MN, (int)N, (int)ldc, gsu)
    if (epi == G2_EPI_NONE) G2R(G2_EPI_NONE, false);
    else if (epi == G2_EPI_BIAS) G2R(G2_EPI_BIAS, false);
    else if (sz) G2R(G2_EPI_BIAS_GELU, true);
    else G2R(G2_EPI_BIAS_GELU, false);
#undef G2R
    LAUNCH_CHECK();
  }
  return 0;
}
