"""ctypes loader for the in-tree HIP kernel library (libskyhip.so).

The kernels are hand-written CDNA4 (gfx950) HIP — see ops/hip/*.hip — built
in-tree by ``skycomputing_amd.ops.build`` (driven from __graft_entry__.build())
so the .so travels with the repo snapshot to GPU boxes.

Design: a plain C ABI over raw device pointers + the current HIP stream,
loaded with ctypes. This avoids libtorch C++ ABI coupling entirely; autograd
integration lives in Python (ops/functions.py).
"""

from __future__ import annotations

import ctypes
import os

_LIB = None
_LIB_PATH = os.path.join(os.path.dirname(os.path.abspath(__file__)), "hip", "libskyhip.so")
_TRIED = False

# every kernel entry point returns int (0 = ok, nonzero = hipError_t)
_SIGNATURES: dict[str, list] = {}

u64 = ctypes.c_uint64
i64 = ctypes.c_int64
i32 = ctypes.c_int32
f32 = ctypes.c_float
f64 = ctypes.c_double


def _register_signatures(lib):
    sigs = {
        # layernorm: fwd saves mean/rstd (fp32 per row) for bwd
        "sky_layernorm_fwd": [u64, u64, u64, u64, u64, u64, u64, u64, i64, i64, f32, i32, f32, u64, u64],
        #        strm x res w b y mean rstd rows cols eps dt keep salt state
        "sky_layernorm_bwd": [u64, u64, u64, u64, u64, u64, u64, u64, u64, u64, u64, i64, i64, i32, f32, u64, u64, u64],
        #        strm dy x res w mean rstd dx dw db scratch | rows cols dt keep salt state dres
        "sky_bias_gelu_fwd": [u64, u64, u64, u64, i64, i64, i32],
        #                     strm  x    b    y    rows cols dt
        "sky_bias_gelu_bwd": [u64, u64, u64, u64, u64, u64, u64, i64, i64, i32],
        #                 strm dy x b dx db scratch | rows cols dt
        "sky_masked_softmax_fwd": [u64, u64, u64, u64, i64, i64, i64, i64, f32, f32, u64, i32],
        #                     strm scores mask probs B   h    Sq   Sk  scale keep seed dt
        "sky_masked_softmax_bwd": [u64, u64, u64, u64, i64, i64, i64, i64, f32, f32, u64, i32],
        #                     strm  dp  probs ds    B    h    Sq   Sk  scale keep seed dt
        "sky_embedding_fwd": [u64] * 12 + [i64, i64, i64, f32, i32],
        # strm ids tids pids wemb temb pemb lnw lnb y mean rstd | rows cols vocab eps dt
        "sky_embedding_bwd": [u64] * 16 + [i64, i64, i32],
        # strm dy ids tids pids wemb temb pemb lnw mean rstd dwe dte dpe dlnw dlnb | rows cols dt
        "sky_sgd_step": [u64, u64, i64, f32, f32, f32, i32, i32],
        #                strm descs n   lr   mom  wd   dt   flags
        "sky_detect_mem": [u64, u64],  # free_out, total_out (host ptrs)
        "sky_dropout_fwd": [u64, u64, u64, i64, f32, u64, u64, i32],
        #                   strm  x    y    n   keep salt state dt
        "sky_dropout_bwd": [u64, u64, u64, i64, f32, u64, u64, i32],
        "sky_rng_tick": [u64, u64],  # strm, state ptr
        "sky_colsum": [u64, u64, u64, u64, i64, i64, i32, i32],  # strm src out scratch rows cols dt dtout
        "sky_hblt_wgrad_bgrad": [u64, u64, u64, u64, u64, i64, i64, i64, i32],  # strm x dy dw db M N K dt
        "sky_hblt_linear_gelu_aux": [u64, u64, u64, u64, u64, u64, i64, i64, i64, i32, i32],  # strm x w bias y aux M N K dt aux_dt
        "sky_mfma_probe": [u64, u64, u64, u64],  # strm A B D
        "sky_glds_probe": [u64, u64, u64, i32],
        "sky_gemm": [u64, u64, u64, u64, u64, u64, i64, i64, i64, i64, i64, i64, i32, i32, i32, i32, i32],
        # strm A B C bias Z | M N K lda ldb ldc | transA transB epi dt use_glds
        "sky_gemm2": [u64, u64, u64, u64, u64, u64, u64, i64, i64, i64, i64, i64, i64, i32, i32, i32, i32],
        # strm A B C bias Z Wk | M N K lda ldb ldc | transA transB epi gsu
        # (256^2 8-phase; transA/B: 0=direct [out][red], 1=kmajor [red][out])
        "sky_pack3": [u64, u64, u64, u64, u64, i64, i64, i64, i64, i32],
        "sky_attn_fwd": [u64, u64, u64, u64, u64, u64, i64, i64, i64, i64, f32, f32, u64, u64],
        #                strm qkv  mask out  m    l    B    S    h    d  scale keep salt state
        "sky_attn_probs": [u64, u64, u64, u64, u64, u64, u64, i64, i64, i64, i64, f32, f32, u64, u64],
        #                  strm qkv mask m    l    p    pd   B    S    h    d  scale keep salt state
        "sky_attn_flash_fwd": [u64, u64, u64, u64, u64, u64, i64, i64, i64, i64, f32, f32, u64, u64],
        #                      strm qkv mask out  m    l    B    S    h    d scale keep salt state
        "sky_attn_bwd": [u64] * 9 + [i64, i64, i64, i64, f32, f32, u64, u64],
        # strm qkv dout mask m l pdT dsT dqkv | B S h d scale keep salt state
        "sky_attn_bwd_fused": [u64] * 7 + [i64, i64, i64, i64, f32, f32, u64, u64],
        # strm qkv dout mask m l dqkv | B S h d scale keep salt state
        # (ONE kernel: dQ/dK/dV, P+dS never leave LDS)
    }
    for name, argtypes in sigs.items():
        if hasattr(lib, name):
            fn = getattr(lib, name)
            fn.argtypes = argtypes
            fn.restype = ctypes.c_int
    return sigs


def lib():
    """Return the loaded CDLL or None if unavailable."""
    global _LIB, _TRIED
    if _LIB is None and not _TRIED:
        _TRIED = True
        if os.path.isfile(_LIB_PATH):
            _LIB = ctypes.CDLL(_LIB_PATH)
            _register_signatures(_LIB)
    return _LIB


def available() -> bool:
    return lib() is not None


def require():
    l = lib()
    if l is None:
        raise RuntimeError(
            f"skycomputing_amd HIP kernel library not found at {_LIB_PATH}. "
            "Build it with `python -m skycomputing_amd.ops.build` (or __graft_entry__.build()). "
            "Running CUDA tensors through the eager fallback is disabled so GPU runs "
            "cannot silently lose the native path; set SKY_ALLOW_EAGER_GPU=1 to override."
        )
    return l


def check(rc: int, name: str):
    if rc != 0:
        raise RuntimeError(f"{name} failed with hipError_t={rc}")


def ptr(t) -> int:
    """Device pointer of a tensor (0 for None)."""
    return 0 if t is None else t.data_ptr()
