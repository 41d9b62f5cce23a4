// Hand-written bf16 MFMA GEMM for gfx950 with fused epilogues.
//
// Replaces hipBLASLt for the BERT linear layers (forward X@W^T, dgrad
// dY@W, wgrad dY^T@X — SURVEY.md §2c rows 1/5/6) where shapes are clean
// multiples of the tile. Structure = the guide's "step-3 ladder" shape
// (cdna_hip_programming.md §5): 128x128 output tile, BK=64, 4 waves each
// owning a 64x64 subtile as 4x4 fragments of mfma_f32_16x16x32_bf16,
// double-buffered LDS, direct global->LDS DMA (global_load_lds, 16 B) for
// operands whose reduction dim is contiguous in memory, register+scatter
// staging for transposed operands.
//
// C[M][N] = OpA(A) @ OpB(B) (+bias) (+erf-GELU, optionally storing the
// pre-activation Z for backward). A_DIRECT means A is stored [M][R]
// row-major (reduction contiguous); !A_DIRECT means [R][M]. B_DIRECT
// means B stored [N][R]; !B_DIRECT means [R][N].
//
// LDS images (both operands): [tile_row][BK] bf16, reduction contiguous —
// A fragments (i = lane%16, k = 8*(lane/16)+j) and B fragments
// (n = lane%16, same k) are then single ds_read_b128s.

#include "common.h"

typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define GM_BM 128
#define GM_BN 128
#define GM_BK 64
#define GM_BLOCK 256

typedef const __attribute__((address_space(1))) unsigned int* gas_u32;
typedef __attribute__((address_space(3))) unsigned int* las_u32;

DEV void glds16(const void* g, void* l) {
  __builtin_amdgcn_global_load_lds((gas_u32)g, (las_u32)l, 16, 0, 0);
}

// XOR swizzle for the LDS tile images (rows are 128 B = 8 x 16 B units):
// without it every lane of a fragment-read quad starts on the SAME bank
// (row stride 128 B), serializing ds_read_b128 16-way. The glds path
// applies the swizzle on the SOURCE column instead (lane-linear LDS image
// == swizzled layout, guide rule 21).
DEV int gswz(int byte_addr, int row) { return byte_addr ^ ((row & 7) << 4); }

// epilogue modes
enum { EPI_NONE = 0, EPI_BIAS = 1, EPI_BIAS_GELU = 2 };

template <bool A_DIRECT, bool B_DIRECT, int EPI, bool STORE_Z, bool USE_GLDS>
__global__ __launch_bounds__(GM_BLOCK) void gemm_kernel(
    const ushort_t* __restrict__ A, const ushort_t* __restrict__ B,
    ushort_t* __restrict__ C, const ushort_t* __restrict__ bias,
    ushort_t* __restrict__ Z, int M, int N, int K, int lda, int ldb,
    int ldc) {
  // two LDS buffers, each: A tile [128][64] + B tile [128][64] bf16
  extern __shared__ __attribute__((aligned(16))) char lds[];
  const int tid = threadIdx.x;
  const int l = tid & 63;
  const int w = tid >> 6;
  const int lm = l & 15;
  const int lg = l >> 4;

  // XCD-aware bijective block swizzle (guide T1): contiguous grid chunks
  // per XCD for L2 locality.
  int bid = blockIdx.x;
  {
    const int nwg = gridDim.x;
    const int q = nwg / 8, rr = nwg % 8;
    const int xcd = bid % 8, idx = bid / 8;
    bid = (xcd < rr ? xcd * (q + 1) : rr * (q + 1) + (xcd - rr) * q) + idx;
  }
  const int ntn = N / GM_BN;
  const int m0 = (bid / ntn) * GM_BM;
  const int n0 = (bid % ntn) * GM_BN;
  const int wr = w >> 1, wc = w & 1;  // wave's 64x64 subtile

  // staging: each tile is 1024 x 16B units; 256 threads x 4 units.
  // direct: unit u -> row = u/8, col16 = u%8 (row-major [128][64]).
  // transpose: unit u -> src row r = u/16, col8 = u%16; scatter 8 u16.
  auto stage_direct = [&](const ushort_t* src, int ld, int row0, int k0,
                          char* dst, int rows) {
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      const int u = it * 256 + tid;
      const int row = u >> 3, c16 = u & 7;
      if (USE_GLDS) {
        // source-side swizzle: fetch the column that belongs at this
        // lane-linear LDS slot under the XOR layout
        const int c16s = c16 ^ (row & 7);
        const ushort_t* g = src + (size_t)(row0 + (row < rows ? row : rows - 1)) * ld + k0 + c16s * 8;
        glds16(g, dst + u * 16);
      } else {
        const ushort_t* g = src + (size_t)(row0 + (row < rows ? row : rows - 1)) * ld + k0 + c16 * 8;
        *(ushort8_t*)(dst + gswz(row * 128 + c16 * 16, row)) = *(const ushort8_t*)g;
      }
    }
  };
  auto stage_trans = [&](const ushort_t* src, int ld, int k0, int col0,
                         char* dst, int cols) {
    // src stored [R][cols-dim]; fill dst_lds[col][r]
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      const int u = it * 256 + tid;
      const int r = u >> 4, c8 = u & 15;
      ushort8_t v = *(const ushort8_t*)(src + (size_t)(k0 + r) * ld + col0 + c8 * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int row = c8 * 8 + j;
        *(ushort_t*)(dst + gswz(row * 128 + r * 2, row)) = v[j];
      }
    }
  };

  auto stage = [&](int kt, int buf) {
    char* abuf = lds + buf * 32 * 1024;
    char* bbuf = abuf + 16 * 1024;
    const int k0 = kt * GM_BK;
    if (A_DIRECT)
      stage_direct(A, lda, m0, k0, abuf, M - m0 < GM_BM ? M - m0 : GM_BM);
    else
      stage_trans(A, lda, k0, m0, abuf, GM_BM);
    if (B_DIRECT)
      stage_direct(B, ldb, n0, k0, bbuf, N - n0 < GM_BN ? N - n0 : GM_BN);
    else
      stage_trans(B, ldb, k0, n0, bbuf, GM_BN);
  };

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4){0.f, 0.f, 0.f, 0.f};

  const int NT = K / GM_BK;
  stage(0, 0);
  if (USE_GLDS && (A_DIRECT || B_DIRECT))
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  for (int kt = 0; kt < NT; ++kt) {
    const int cur = kt & 1;
    if (kt + 1 < NT) stage(kt + 1, cur ^ 1);
    char* abuf = lds + cur * 32 * 1024;
    char* bbuf = abuf + 16 * 1024;
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8 af[4], bf[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int am = wr * 64 + i * 16 + lm;
        af[i] = *(const bf16x8*)(abuf + gswz(am * 128 + (ks * 32 + lg * 8) * 2, am));
        const int bn = wc * 64 + i * 16 + lm;
        bf[i] = *(const bf16x8*)(bbuf + gswz(bn * 128 + (ks * 32 + lg * 8) * 2, bn));
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af[i], bf[j], acc[i][j], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
    if (USE_GLDS && (A_DIRECT || B_DIRECT))
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
  }

  // epilogue
  float bv[4];
  if (EPI != EPI_NONE) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int n = n0 + wc * 64 + j * 16 + lm;
      bv[j] = bias ? bf16_to_f32(bias[n]) : 0.f;
    }
  }
#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int m = m0 + wr * 64 + i * 16 + lg * 4 + r;
      if (m >= M) continue;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int n = n0 + wc * 64 + j * 16 + lm;
        float v = acc[i][j][r];
        if (EPI != EPI_NONE) v += bv[j];
        if (EPI == EPI_BIAS_GELU) {
          if (STORE_Z) Z[(size_t)m * ldc + n] = f32_to_bf16(v);
          v = gelu_f(v);
        }
        C[(size_t)m * ldc + n] = f32_to_bf16(v);
      }
    }
  }
}

SKY_EXPORT int sky_gemm(uint64_t stream, uint64_t A, uint64_t B, uint64_t C,
                        uint64_t bias, uint64_t Z, int64_t M, int64_t N,
                        int64_t K, int64_t lda, int64_t ldb, int64_t ldc,
                        int transA, int transB, int epi, int dt,
                        int use_glds) {
  if (dt != DT_BF16) return (int)hipErrorInvalidValue;
  if (M % GM_BM || N % GM_BN || K % GM_BK) return (int)hipErrorInvalidValue;
  hipStream_t s = (hipStream_t)stream;
  dim3 grid((unsigned)((M / GM_BM) * (N / GM_BN)));
  size_t lds_bytes = 64 * 1024;
  const bool ad = transA == 0, bd = transB != 0;
  const bool sz = Z != 0;
#define GK(AD, BD, EP, SZ)                                                     \
  do {                                                                         \
    if (use_glds)                                                              \
      hipLaunchKernelGGL((gemm_kernel<AD, BD, EP, SZ, true>), grid,            \
                         dim3(GM_BLOCK), lds_bytes, s, (const ushort_t*)A,     \
                         (const ushort_t*)B, (ushort_t*)C,                     \
                         (const ushort_t*)bias, (ushort_t*)Z, (int)M, (int)N,  \
                         (int)K, (int)lda, (int)ldb, (int)ldc);                \
    else                                                                       \
      hipLaunchKernelGGL((gemm_kernel<AD, BD, EP, SZ, false>), grid,           \
                         dim3(GM_BLOCK), lds_bytes, s, (const ushort_t*)A,     \
                         (const ushort_t*)B, (ushort_t*)C,                     \
                         (const ushort_t*)bias, (ushort_t*)Z, (int)M, (int)N,  \
                         (int)K, (int)lda, (int)ldb, (int)ldc);                \
  } while (0)
#define GK_EPI(AD, BD)                                                         \
  do {                                                                         \
    if (epi == EPI_NONE) GK(AD, BD, EPI_NONE, false);                          \
    else if (epi == EPI_BIAS) GK(AD, BD, EPI_BIAS, false);                     \
    else if (sz) GK(AD, BD, EPI_BIAS_GELU, true);                              \
    else GK(AD, BD, EPI_BIAS_GELU, false);                                     \
  } while (0)
  if (ad) { if (bd) GK_EPI(true, true); else GK_EPI(true, false); }
  else    { if (bd) GK_EPI(false, true); else GK_EPI(false, false); }
#undef GK_EPI
#undef GK
  LAUNCH_CHECK();
  return 0;
}
