// Multi-tensor fused SGD for gfx950.
//
// Replaces per-parameter eager SGD (reference: DistributedOptimizer's local
// SGD steps, experiment/launch.py:152-156): ONE kernel launch updates every
// parameter of the stage. The host packs per-slab descriptors
// {param_ptr, grad_ptr, master_ptr, momentum_ptr, count} into a device
// int64 array (built once; pointers are stable after the first backward);
// each block processes one slab. With fp32 master weights the update runs
// on the master and the bf16 param is refreshed in the same pass.
//
// memutil: sky_detect_mem wraps hipMemGetInfo (replacing the reference's
// nvidia-smi subprocess probe, module_wrapper.py:210-219).

#include "common.h"

struct SgdDesc {
  uint64_t p, g, m, mom;
  int64_t n;
  int64_t pad;
};

// NOTE on non-temporal I/O: every byte in the SGD step is touch-once, so
// r01 used NT streaming loads/stores throughout. Measured in-app in r02,
// PLAIN write-back I/O is 1.1 ms/step faster on the full bench — the NT
// hints cost more at the memory controller than the avoided L2
// write-allocate saves. NT kept selectable (SKY_SGD_NONT=0 -> NT).
template <int DT>
DEV void nt_load8(const void* p, int64_t i8, float f[8]);
template <>
DEV void nt_load8<DT_BF16>(const void* p, int64_t i8, float f[8]) {
  ushort8_t v = __builtin_nontemporal_load((const ushort8_t*)p + i8);
#pragma unroll
  for (int j = 0; j < 8; ++j) f[j] = bf16_to_f32(v[j]);
}
template <>
DEV void nt_load8<DT_F32>(const void* p, int64_t i8, float f[8]) {
  float4_t a = __builtin_nontemporal_load((const float4_t*)p + i8 * 2);
  float4_t b = __builtin_nontemporal_load((const float4_t*)p + i8 * 2 + 1);
#pragma unroll
  for (int j = 0; j < 4; ++j) { f[j] = a[j]; f[4 + j] = b[j]; }
}
template <int DT>
DEV void nt_store8(void* p, int64_t i8, const float f[8]);
template <>
DEV void nt_store8<DT_BF16>(void* p, int64_t i8, const float f[8]) {
  ushort8_t v;
#pragma unroll
  for (int j = 0; j < 8; ++j) v[j] = f32_to_bf16(f[j]);
  __builtin_nontemporal_store(v, (ushort8_t*)p + i8);
}
template <>
DEV void nt_store8<DT_F32>(void* p, int64_t i8, const float f[8]) {
  float4_t a, b;
#pragma unroll
  for (int j = 0; j < 4; ++j) { a[j] = f[j]; b[j] = f[4 + j]; }
  __builtin_nontemporal_store(a, (float4_t*)p + i8 * 2);
  __builtin_nontemporal_store(b, (float4_t*)p + i8 * 2 + 1);
}

template <int DT, int BLOCK, bool MASTER, bool MOM, int UNROLL = 2,
          bool NT = true>
__global__ __launch_bounds__(BLOCK) void sgd_kernel(
    const SgdDesc* __restrict__ descs, float lr, float momentum, float wd) {
  const SgdDesc d = descs[blockIdx.x];
  void* p = (void*)d.p;
  const void* g = (const void*)d.g;
  float* master = (float*)d.m;
  float* mbuf = (float*)d.mom;
  const int64_t n8 = d.n / 8;
  // vectorized body (slab bases are 16B-aligned; tail handled below);
  // UNROLLx for memory-level parallelism
  for (int64_t i8 = threadIdx.x; i8 < n8; i8 += UNROLL * BLOCK) {
#pragma unroll
    for (int u = 0; u < UNROLL; ++u) {
      const int64_t k8 = i8 + u * BLOCK;
      if (k8 >= n8) break;
      float gv[8], pv[8];
      if (NT) {
        nt_load8<DT>(g, k8, gv);
        if (MASTER) nt_load8<DT_F32>(master, k8, pv);
        else nt_load8<DT>(p, k8, pv);
      } else {
        Vec8<DT>::load(g, k8, gv);
        if (MASTER) Vec8<DT_F32>::load(master, k8, pv);
        else Vec8<DT>::load(p, k8, pv);
      }
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float gg = gv[j];
        if (wd != 0.f) gg += wd * pv[j];
        if (MOM) {
          float b = mbuf[k8 * 8 + j] * momentum + gg;
          mbuf[k8 * 8 + j] = b;
          gg = b;
        }
        pv[j] -= lr * gg;
      }
      if (NT) {
        if (MASTER) nt_store8<DT_F32>(master, k8, pv);
        nt_store8<DT>(p, k8, pv);
      } else {
        if (MASTER) Vec8<DT_F32>::store(master, k8, pv);
        Vec8<DT>::store(p, k8, pv);
      }
    }
  }
  for (int64_t i = n8 * 8 + threadIdx.x; i < d.n; i += BLOCK) {
    float gv = load_elem<DT>(g, i);
    float pv = MASTER ? master[i] : load_elem<DT>(p, i);
    if (wd != 0.f) gv += wd * pv;
    if (MOM) {
      float b = mbuf[i] * momentum + gv;
      mbuf[i] = b;
      gv = b;
    }
    pv -= lr * gv;
    if (MASTER) master[i] = pv;
    store_elem<DT>(p, i, pv);
  }
}

SKY_EXPORT int sky_sgd_step(uint64_t stream, uint64_t descs, int64_t n_descs,
                            float lr, float momentum, float wd, int dt,
                            int flags) {
  constexpr int BLOCK = 256;
  const bool has_master = flags & 1;
  const bool has_mom = flags & 2;
  hipStream_t s = (hipStream_t)stream;
  dim3 grid((unsigned)n_descs);
  const char* ue = getenv("SKY_SGD_UNROLL");
  const int unroll = ue ? atoi(ue) : 2;
  // plain (write-back) I/O is the measured in-app default: 108.95 vs
  // 110.07 ms/step against the non-temporal variant (same box) — the r01
  // touch-once/NT reasoning loses to L2 write buffering in practice.
  // SKY_SGD_NONT=0 selects the NT variant for re-measurement.
  const char* nte = getenv("SKY_SGD_NONT");
  const bool plain = !(nte && nte[0] == '0');
#define SGD(DT, MA, MO)                                                      \
  do {                                                                       \
    if (plain)                                                               \
      hipLaunchKernelGGL((sgd_kernel<DT, BLOCK, MA, MO, 2, false>), grid,    \
                         dim3(BLOCK), 0, s, (const SgdDesc*)descs, lr,       \
                         momentum, wd);                                      \
    else if (unroll >= 8)                                                    \
      hipLaunchKernelGGL((sgd_kernel<DT, BLOCK, MA, MO, 8>), grid,           \
                         dim3(BLOCK), 0, s, (const SgdDesc*)descs, lr,       \
                         momentum, wd);                                      \
    else if (unroll >= 4)                                                    \
      hipLaunchKernelGGL((sgd_kernel<DT, BLOCK, MA, MO, 4>), grid,           \
                         dim3(BLOCK), 0, s, (const SgdDesc*)descs, lr,       \
                         momentum, wd);                                      \
    else                                                                     \
      hipLaunchKernelGGL((sgd_kernel<DT, BLOCK, MA, MO, 2>), grid,           \
                         dim3(BLOCK), 0, s, (const SgdDesc*)descs, lr,       \
                         momentum, wd);                                      \
  } while (0)
  if (dt == DT_F32) {
    if (has_master) { if (has_mom) SGD(DT_F32, true, true); else SGD(DT_F32, true, false); }
    else            { if (has_mom) SGD(DT_F32, false, true); else SGD(DT_F32, false, false); }
  } else {
    if (has_master) { if (has_mom) SGD(DT_BF16, true, true); else SGD(DT_BF16, true, false); }
    else            { if (has_mom) SGD(DT_BF16, false, true); else SGD(DT_BF16, false, false); }
  }
#undef SGD
  LAUNCH_CHECK();
  return 0;
}

SKY_EXPORT int sky_detect_mem(uint64_t free_out, uint64_t total_out) {
  size_t f = 0, t = 0;
  hipError_t e = hipMemGetInfo(&f, &t);
  if (e != hipSuccess) return (int)e;
  *(uint64_t*)free_out = (uint64_t)f;
  *(uint64_t*)total_out = (uint64_t)t;
  return 0;
}
