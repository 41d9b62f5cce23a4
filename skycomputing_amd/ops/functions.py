"""torch.autograd.Function wrappers over the HIP kernel library.

Each Function's forward/backward launches hand-written gfx950 kernels on the
current HIP stream via the C ABI in hiplib. Numerics contract: bf16 (or fp32)
I/O, fp32 accumulation inside the kernels; tested against ops/eager.py in
fp32 (tests/test_ops_gpu.py).
"""

from __future__ import annotations

import os

import torch

from . import hiplib
from .hiplib import check, ptr

_DT = {torch.float32: 0, torch.bfloat16: 1}

_seed_gen = torch.Generator()
_seed_gen.manual_seed(0x5EED)


def set_dropout_seed(seed: int):
    _seed_gen.manual_seed(seed)


def _next_seed() -> int:
    return int(torch.randint(0, 2**62, (1,), generator=_seed_gen).item())


def _stream() -> int:
    return torch.cuda.current_stream().cuda_stream


_CS_SLABS = 1024  # must match CS_SLABS / LN_SLABS in the HIP sources


def _red_scratch(cols: int, pairs: int, device) -> torch.Tensor:
    """fp32 scratch for the two-stage column reductions:
    [_CS_SLABS slabs][pairs*cols]."""
    return torch.empty(_CS_SLABS * pairs * cols, dtype=torch.float32, device=device)


def _dt(t: torch.Tensor) -> int:
    try:
        return _DT[t.dtype]
    except KeyError:
        raise TypeError(f"unsupported dtype {t.dtype} (fp32/bf16 only)") from None


class LayerNormFn(torch.autograd.Function):
    """Fused (dropout + residual-add +) LayerNorm, fwd+bwd in single-pass
    HIP kernels (replaces the reference's optional apex FusedLayerNorm +
    preceding nn.Dropout, reference: scaelum/model/bert_layers.py:128-168,
    283-288). With dropout_p > 0 the op computes
    LN(dropout(x) + residual); the mask regenerates from (salt, device
    step counter) in backward."""

    @staticmethod
    def forward(ctx, x, weight, bias, eps, residual, dropout_p=0.0):
        lib = hiplib.require()
        x = x.contiguous()
        residual = residual.contiguous() if residual is not None else None
        cols = x.shape[-1]
        rows = x.numel() // cols
        y = torch.empty_like(x)
        mean = torch.empty(rows, dtype=torch.float32, device=x.device)
        rstd = torch.empty(rows, dtype=torch.float32, device=x.device)
        keep = 1.0 - dropout_p
        salt = _next_seed() if keep < 1.0 else 0
        state_ptr = rng_state().data_ptr() if keep < 1.0 else 0
        check(
            lib.sky_layernorm_fwd(
                _stream(), ptr(x), ptr(residual), ptr(weight), ptr(bias),
                ptr(y), ptr(mean), ptr(rstd), rows, cols, eps, _dt(x),
                keep, salt, state_ptr,
            ),
            "sky_layernorm_fwd",
        )
        ctx.save_for_backward(x, weight, mean, rstd)
        ctx.residual = residual
        ctx.has_residual = residual is not None
        ctx.keep, ctx.salt = keep, salt
        return y

    @staticmethod
    def backward(ctx, dy):
        lib = hiplib.require()
        x, weight, mean, rstd = ctx.saved_tensors
        dy = dy.contiguous()
        cols = x.shape[-1]
        rows = x.numel() // cols
        dx = torch.empty_like(x)
        fast = cols % 8 == 0
        if fast:
            # the final reduction stage writes the param dtype directly
            dw = torch.empty(cols, dtype=weight.dtype, device=x.device)
            db = torch.empty(cols, dtype=weight.dtype, device=x.device)
            scratch = _red_scratch(cols, 2, x.device)
        else:
            dw = torch.zeros(cols, dtype=torch.float32, device=x.device)
            db = torch.zeros(cols, dtype=torch.float32, device=x.device)
            scratch = None
        drop = ctx.keep < 1.0
        dres = torch.empty_like(x) if (drop and ctx.has_residual) else None
        state_ptr = rng_state().data_ptr() if drop else 0
        check(
            lib.sky_layernorm_bwd(
                _stream(), ptr(dy), ptr(x), ptr(ctx.residual), ptr(weight),
                ptr(mean), ptr(rstd), ptr(dx), ptr(dw), ptr(db),
                ptr(scratch), rows, cols, _dt(x), ctx.keep, ctx.salt,
                state_ptr, ptr(dres),
            ),
            "sky_layernorm_bwd",
        )
        if not fast:
            dw = dw.to(weight.dtype)
            db = db.to(weight.dtype)
        if not ctx.has_residual:
            dgrad_res = None
        elif drop:
            dgrad_res = dres
        else:
            dgrad_res = dx
        return dx, dw, db, None, dgrad_res, None


class BiasGeluFn(torch.autograd.Function):
    """Fused bias + erf-GELU (reference: scaelum/model/bert_layers.py:21-44)."""

    @staticmethod
    def forward(ctx, x, bias):
        lib = hiplib.require()
        x = x.contiguous()
        cols = x.shape[-1]
        rows = x.numel() // cols
        y = torch.empty_like(x)
        check(
            lib.sky_bias_gelu_fwd(_stream(), ptr(x), ptr(bias), ptr(y), rows, cols, _dt(x)),
            "sky_bias_gelu_fwd",
        )
        ctx.save_for_backward(x, bias)
        return y

    @staticmethod
    def backward(ctx, dy):
        lib = hiplib.require()
        x, bias = ctx.saved_tensors
        dy = dy.contiguous()
        cols = x.shape[-1]
        rows = x.numel() // cols
        dx = torch.empty_like(x)
        fast = cols % 8 == 0
        db = (torch.empty(cols, dtype=bias.dtype, device=x.device) if fast
              else torch.zeros(cols, dtype=torch.float32, device=x.device))
        scratch = _red_scratch(cols, 1, x.device) if fast else None
        check(
            lib.sky_bias_gelu_bwd(
                _stream(), ptr(dy), ptr(x), ptr(bias), ptr(dx), ptr(db),
                ptr(scratch), rows, cols, _dt(x)
            ),
            "sky_bias_gelu_bwd",
        )
        return dx, (db if fast else db.to(bias.dtype))


class MaskedSoftmaxFn(torch.autograd.Function):
    """Fused scale + additive-mask + row softmax over attention scores
    [B, h, Sq, Sk]; mask [B, 1, 1, Sk] additive, broadcast over (h, Sq).
    Replaces the reference's eager mask-add + softmax
    (reference: scaelum/model/bert_layers.py:259-269)."""

    @staticmethod
    def forward(ctx, scores, mask, scale):
        lib = hiplib.require()
        scores = scores.contiguous()
        B, h, Sq, Sk = scores.shape
        probs = torch.empty_like(scores)
        mask = mask.contiguous() if mask is not None else None
        check(
            lib.sky_masked_softmax_fwd(
                _stream(), ptr(scores), ptr(mask), ptr(probs),
                B, h, Sq, Sk, scale, 1.0, 0, _dt(scores),
            ),
            "sky_masked_softmax_fwd",
        )
        ctx.save_for_backward(probs)
        ctx.scale = scale
        return probs

    @staticmethod
    def backward(ctx, dp):
        lib = hiplib.require()
        (probs,) = ctx.saved_tensors
        dp = dp.contiguous()
        B, h, Sq, Sk = probs.shape
        ds = torch.empty_like(probs)
        check(
            lib.sky_masked_softmax_bwd(
                _stream(), ptr(dp), ptr(probs), ptr(ds),
                B, h, Sq, Sk, ctx.scale, 1.0, 0, _dt(probs),
            ),
            "sky_masked_softmax_bwd",
        )
        return ds, None, None


_RNG_STATE: torch.Tensor | None = None


def rng_state() -> torch.Tensor:
    """Device-side step counter mixed into every dropout seed. Bump it with
    rng_tick() at iteration start — as a captured hipGraph node this keeps
    dropout masks varying under graph replay (launch args are frozen)."""
    global _RNG_STATE
    if _RNG_STATE is None:
        _RNG_STATE = torch.zeros(1, dtype=torch.int64, device="cuda")
    return _RNG_STATE


def rng_tick():
    lib = hiplib.require()
    check(lib.sky_rng_tick(_stream(), rng_state().data_ptr()), "sky_rng_tick")


class DropoutFn(torch.autograd.Function):
    """Dropout with counter-based RNG: the keep mask is regenerated from
    (salt, device step counter) in backward, so no mask tensor is stored
    and the op is hipGraph-capture safe."""

    @staticmethod
    def forward(ctx, x, p):
        lib = hiplib.require()
        x = x.contiguous()
        y = torch.empty_like(x)
        salt = _next_seed()
        keep = 1.0 - p
        state = rng_state()
        check(
            lib.sky_dropout_fwd(
                _stream(), ptr(x), ptr(y), x.numel(), keep, salt,
                state.data_ptr(), _dt(x),
            ),
            "sky_dropout_fwd",
        )
        ctx.salt = salt
        ctx.keep = keep
        return y

    @staticmethod
    def backward(ctx, dy):
        lib = hiplib.require()
        dy = dy.contiguous()
        dx = torch.empty_like(dy)
        check(
            lib.sky_dropout_bwd(
                _stream(), ptr(dy), ptr(dx), dy.numel(), ctx.keep, ctx.salt,
                rng_state().data_ptr(), _dt(dy),
            ),
            "sky_dropout_bwd",
        )
        return dx, None


class FusedAttentionFn(torch.autograd.Function):
    """Fused attention: out = dropout(softmax(scale*QK^T + mask)) V in ONE
    MFMA kernel (ops/hip/attention.hip); S x S probabilities are never
    materialized in forward (row max/sum saved, flash-style).

    Backward default: TWO fused MFMA kernels (sky_attn_bwd) — recompute +
    dS + dQ, then dK/dV over transposed scratch — no torch ops at all.
    SKY_NO_FUSED_ATTN_BWD=1 selects the decomposed fallback (sky_attn_probs
    recompute + hipBLASLt batched GEMMs + HIP softmax-backward).

    qkv: [B, S, 3, h, d] (d = 64), mask: [B, 1, 1, S] additive or None.
    Returns [B, S, h, d].
    """

    @staticmethod
    def forward(ctx, qkv, mask, scale, dropout_p, training):
        lib = hiplib.require()
        B, S, three, h, d = qkv.shape
        assert three == 3 and d == 64 and S <= 128
        qkv = qkv.contiguous()
        mask = mask.contiguous() if mask is not None else None
        out = torch.empty(B, S, h, d, dtype=qkv.dtype, device=qkv.device)
        m = torch.empty(B, h, S, dtype=torch.float32, device=qkv.device)
        lsum = torch.empty_like(m)
        keep = 1.0 - dropout_p if (dropout_p > 0 and training) else 1.0
        salt = _next_seed() if keep < 1.0 else 0
        state = rng_state()
        check(
            lib.sky_attn_fwd(
                _stream(), ptr(qkv), ptr(mask), ptr(out), ptr(m), ptr(lsum),
                B, S, h, d, scale, keep, salt, state.data_ptr(),
            ),
            "sky_attn_fwd",
        )
        ctx.save_for_backward(qkv, mask, m, lsum)
        ctx.scale, ctx.keep, ctx.salt = scale, keep, salt
        return out

    @staticmethod
    def backward(ctx, dout):
        lib = hiplib.require()
        qkv, mask, m, lsum = ctx.saved_tensors
        B, S, _, h, d = qkv.shape
        dev = qkv.device
        if os.environ.get("SKY_NO_FUSED_ATTN_BWD") != "1":
            dout = dout.contiguous()
            dqkv = torch.empty_like(qkv)
            if os.environ.get("SKY_ATTN_FUSED_BWD") == "1":
                # single-kernel path: dQ/dK/dV in one launch, P and dS
                # never leave LDS (no pdT/dsT scratch tensors). Measured
                # in-app TIE vs the split pair (111.97 vs 111.73 ms step)
                # with lower memory churn; opt-in until it wins.
                check(
                    lib.sky_attn_bwd_fused(
                        _stream(), ptr(qkv), ptr(dout), ptr(mask), ptr(m),
                        ptr(lsum), ptr(dqkv), B, S, h, d, ctx.scale,
                        ctx.keep, ctx.salt, rng_state().data_ptr(),
                    ),
                    "sky_attn_bwd_fused",
                )
                return dqkv, None, None, None, None
            # two-kernel path (bwd1s + bwd2 over transposed scratch)
            alloc = torch.empty if S == 128 else torch.zeros
            pdT = alloc((B, h, S, S), dtype=qkv.dtype, device=dev)
            dsT = alloc((B, h, S, S), dtype=qkv.dtype, device=dev)
            check(
                lib.sky_attn_bwd(
                    _stream(), ptr(qkv), ptr(dout), ptr(mask), ptr(m),
                    ptr(lsum), ptr(pdT), ptr(dsT), ptr(dqkv),
                    B, S, h, d, ctx.scale, ctx.keep, ctx.salt,
                    rng_state().data_ptr(),
                ),
                "sky_attn_bwd",
            )
            return dqkv, None, None, None, None
        P = torch.empty(B, h, S, S, dtype=qkv.dtype, device=dev)
        Pd = torch.empty_like(P) if ctx.keep < 1.0 else P
        check(
            lib.sky_attn_probs(
                _stream(), ptr(qkv), ptr(mask), ptr(m), ptr(lsum), ptr(P),
                ptr(Pd), B, S, h, d, ctx.scale, ctx.keep, ctx.salt,
                rng_state().data_ptr(),
            ),
            "sky_attn_probs",
        )
        dO = dout.permute(0, 2, 1, 3).contiguous()  # one copy, used twice
        q = qkv[:, :, 0].permute(0, 2, 1, 3)
        k = qkv[:, :, 1].permute(0, 2, 1, 3)
        v = qkv[:, :, 2].permute(0, 2, 1, 3)
        dV = torch.matmul(Pd.transpose(-1, -2), dO)
        dPd = torch.matmul(dO, v.transpose(-1, -2)).contiguous()
        if ctx.keep < 1.0:
            # regenerate the dropout mask from Pd (zero where dropped; a
            # P-underflow zero also zeroes dS, so no false positives matter)
            dP = torch.where(
                Pd != 0, dPd * (1.0 / ctx.keep), torch.zeros((), dtype=dPd.dtype, device=dPd.device)
            )
        else:
            dP = dPd
        dS = torch.empty_like(dP)
        check(
            lib.sky_masked_softmax_bwd(
                _stream(), ptr(dP), ptr(P), ptr(dS), B, h, S, S, ctx.scale,
                1.0, 0, _dt(P),
            ),
            "sky_masked_softmax_bwd",
        )
        dQ = torch.matmul(dS, k)                      # [B,h,S,d]
        dK = torch.matmul(dS.transpose(-1, -2), q)    # [B,h,S,d]
        dqkv = torch.empty_like(qkv)
        check(
            lib.sky_pack3(
                _stream(), ptr(dQ.contiguous()), ptr(dK.contiguous()),
                ptr(dV.contiguous()), ptr(dqkv), B, S, h, d, _dt(qkv),
            ),
            "sky_pack3",
        )
        return dqkv, None, None, None, None


class FlashAttentionFn(torch.autograd.Function):
    """Flash-style attention for arbitrary S (d = 64, bf16): the forward is
    one online-softmax MFMA kernel per (batch, head, 128-query block); the
    S x S matrix never exists in forward. Backward recomputes probabilities
    (hipBLASLt scores GEMM + HIP masked-softmax) and regenerates the
    dropout mask from the saved salt (linear element indices match the
    generic dropout kernels), then runs the standard five-GEMM backward."""

    @staticmethod
    def forward(ctx, qkv, mask, scale, dropout_p, training):
        lib = hiplib.require()
        B, S, three, h, d = qkv.shape
        assert three == 3 and d == 64
        qkv = qkv.contiguous()
        mask = mask.contiguous() if mask is not None else None
        out = torch.empty(B, S, h, d, dtype=qkv.dtype, device=qkv.device)
        m = torch.empty(B, h, S, dtype=torch.float32, device=qkv.device)
        lsum = torch.empty_like(m)
        keep = 1.0 - dropout_p if (dropout_p > 0 and training) else 1.0
        salt = _next_seed() if keep < 1.0 else 0
        check(
            lib.sky_attn_flash_fwd(
                _stream(), ptr(qkv), ptr(mask), ptr(out), ptr(m), ptr(lsum),
                B, S, h, d, scale, keep, salt, rng_state().data_ptr(),
            ),
            "sky_attn_flash_fwd",
        )
        ctx.save_for_backward(qkv, mask)
        ctx.scale, ctx.keep, ctx.salt = scale, keep, salt
        return out

    @staticmethod
    def backward(ctx, dout):
        lib = hiplib.require()
        qkv, mask = ctx.saved_tensors
        B, S, _, h, d = qkv.shape
        dev = qkv.device
        q = qkv[:, :, 0].permute(0, 2, 1, 3)
        k = qkv[:, :, 1].permute(0, 2, 1, 3)
        v = qkv[:, :, 2].permute(0, 2, 1, 3)
        scores = torch.matmul(q, k.transpose(-1, -2)).contiguous()
        P = torch.empty_like(scores)
        check(
            lib.sky_masked_softmax_fwd(
                _stream(), ptr(scores), ptr(mask), ptr(P), B, h, S, S,
                ctx.scale, 1.0, 0, _dt(scores),
            ),
            "sky_masked_softmax_fwd",
        )
        if ctx.keep < 1.0:
            Pd = torch.empty_like(P)
            check(
                lib.sky_dropout_fwd(
                    _stream(), ptr(P), ptr(Pd), P.numel(), ctx.keep, ctx.salt,
                    rng_state().data_ptr(), _dt(P),
                ),
                "sky_dropout_fwd",
            )
        else:
            Pd = P
        dO = dout.permute(0, 2, 1, 3).contiguous()
        dV = torch.matmul(Pd.transpose(-1, -2), dO)
        dPd = torch.matmul(dO, v.transpose(-1, -2)).contiguous()
        if ctx.keep < 1.0:
            dP = torch.empty_like(dPd)
            check(
                lib.sky_dropout_bwd(
                    _stream(), ptr(dPd), ptr(dP), dPd.numel(), ctx.keep,
                    ctx.salt, rng_state().data_ptr(), _dt(dPd),
                ),
                "sky_dropout_bwd",
            )
        else:
            dP = dPd
        dS = torch.empty_like(dP)
        check(
            lib.sky_masked_softmax_bwd(
                _stream(), ptr(dP), ptr(P), ptr(dS), B, h, S, S, ctx.scale,
                1.0, 0, _dt(P),
            ),
            "sky_masked_softmax_bwd",
        )
        dQ = torch.matmul(dS, k)
        dK = torch.matmul(dS.transpose(-1, -2), q)
        dqkv = torch.empty_like(qkv)
        check(
            lib.sky_pack3(
                _stream(), ptr(dQ.contiguous()), ptr(dK.contiguous()),
                ptr(dV.contiguous()), ptr(dqkv), B, S, h, d, _dt(qkv),
            ),
            "sky_pack3",
        )
        return dqkv, None, None, None, None


def colsum(src: torch.Tensor, out_dtype=None) -> torch.Tensor:
    """Column sum over a 2D [rows, cols] tensor via the HIP column-parallel
    reduction (fp32 accumulation)."""
    lib = hiplib.require()
    src = src.contiguous()
    rows, cols = src.shape
    fast = cols % 8 == 0
    want = out_dtype if (out_dtype is not None and fast) else torch.float32
    out = (torch.empty if fast else torch.zeros)(cols, dtype=want, device=src.device)
    scratch = _red_scratch(cols, 1, src.device) if fast else None
    check(
        lib.sky_colsum(_stream(), ptr(src), ptr(out), ptr(scratch), rows, cols,
                       _dt(src), _dt(out)),
        "sky_colsum",
    )
    if out_dtype is not None and out.dtype != out_dtype:
        out = out.to(out_dtype)
    return out


def _use_hblt() -> bool:
    return os.environ.get("SKY_NO_HBLT") != "1"


# ---- v2 hand GEMM dispatch (ops/hip/gemm2.hip) ----
# SKY_GEMM2 unset -> per-site measured defaults (the ffn-up trio, which
# ties hipBLASLt in-app: gpurun_out/ab_*.json r02 matrix); "0" = off;
# "1" = all sites; or a comma list of {fwd,dgrad,wgrad}. Only perfect-fit
# shapes (M,N %256; K %64) route here; everything else stays on hipBLASLt.

_G2_SITES: set | None = None
_G2_SITES_SET = False
_G2_WK: dict = {}

# in-app A/B (160L bench, r02): the fused ffn-up forward (one v2 GEMM with
# bias+gelu+Z epilogue) ties the library chain (112.26 vs 112.3 ms);
# v2 dgrad/wgrad regress ~25-35 us/layer against hipBLASLt and stay off
# by default (they only dispatch at all when the fused fwd owns the op).
_G2_DEFAULT = {
    "fwd": {(4096, 4096, 1024)},     # ffn-up fwd, bias+gelu+Z fused
    "dgrad": set(),
    "wgrad": set(),
}


def _g2_sites() -> set | None:
    """None means: use the measured per-site default allowlist."""
    global _G2_SITES, _G2_SITES_SET
    if not _G2_SITES_SET:
        v = os.environ.get("SKY_GEMM2", "").strip()
        if v == "":
            _G2_SITES = None
        elif v == "0":
            _G2_SITES = set()
        elif v == "1":
            _G2_SITES = {"fwd", "dgrad", "wgrad"}
        else:
            _G2_SITES = {s.strip() for s in v.split(",") if s.strip()}
        _G2_SITES_SET = True
    return _G2_SITES


def _g2_enabled(site: str, M: int, N: int, K: int) -> bool:
    sites = _g2_sites()
    if sites is None:
        return (M, N, K) in _G2_DEFAULT[site]
    return site in sites and _g2_shape_ok(M, N, K)


_G2_SHAPES: set | None = None


def _g2_shape_ok(M: int, N: int, K: int) -> bool:
    """Optional per-shape allowlist: SKY_GEMM2_SHAPES="MxNxK,MxNxK"."""
    global _G2_SHAPES
    if _G2_SHAPES is None:
        v = os.environ.get("SKY_GEMM2_SHAPES", "").strip()
        _G2_SHAPES = ({tuple(int(d) for d in s.split("x")) for s in v.split(",") if s}
                      if v else set())
    return not _G2_SHAPES or (M, N, K) in _G2_SHAPES


def _g2_fit(M: int, N: int, K: int) -> bool:
    return (M % 256 == 0 and N % 256 == 0 and K % 64 == 0
            and _g2_shape_ok(M, N, K))


def _g2_gsu(M: int, N: int, K: int) -> int:
    # split-K only where the tile grid badly underfills 256 CUs AND K is
    # deep enough to amortize the fp32 partial round-trip (measured:
    # K=1024 shapes run best at gsu=1 despite 64-WG grids)
    ntiles = (M // 256) * (N // 256)
    if ntiles < 128 and K >= 2048 and K % 256 == 0:
        return 4
    return 1


def _g2_workspace(M: int, N: int, gsu: int, device) -> torch.Tensor | None:
    if gsu <= 1:
        return None
    key = (M, N, gsu, device)
    wk = _G2_WK.get(key)
    if wk is None:
        wk = torch.empty(gsu * M * N, dtype=torch.float32, device=device)
        _G2_WK[key] = wk
    return wk


def _g2_call(a, b, c, bias, z, M, N, K, lda, ldb, ldc, tA, tB, epi, gsu):
    lib = hiplib.require()
    wk = _g2_workspace(M, N, gsu, a.device)
    check(
        lib.sky_gemm2(
            _stream(), ptr(a), ptr(b), ptr(c), ptr(bias), ptr(z),
            ptr(wk) if wk is not None else 0, M, N, K, lda, ldb, ldc,
            tA, tB, epi, gsu,
        ),
        "sky_gemm2",
    )
    return c


def gemm2_fwd(x2, weight, bias=None, gelu=False, want_z=False):
    """NT forward y = x2 @ weight.T (+bias)(+gelu). Returns (y, z|None) or
    None when the shape doesn't fit or the site is disabled."""
    if x2.dtype != torch.bfloat16:
        return None
    M, K = x2.shape
    N = weight.shape[0]
    if not (_g2_enabled("fwd", M, N, K) and _g2_fit(M, N, K)):
        return None
    epi = 2 if gelu else (1 if bias is not None else 0)
    gsu = _g2_gsu(M, N, K)
    if gsu > 1 and epi == 2 and want_z:
        pass  # reduce kernel handles bias+gelu+z too
    y = torch.empty(M, N, dtype=x2.dtype, device=x2.device)
    z = torch.empty_like(y) if (gelu and want_z) else None
    _g2_call(x2, weight, y, bias, z, M, N, K, x2.stride(0), weight.stride(0),
             y.stride(0), 0, 0, epi, gsu)
    return y, z


def gemm2_dgrad(dy2, weight):
    """NN dgrad dx = dy2 @ weight (weight stored [N,K] = kmajor operand)."""
    if dy2.dtype != torch.bfloat16:
        return None
    M, Kred = dy2.shape
    N = weight.shape[1]
    if not (_g2_enabled("dgrad", M, N, Kred) and _g2_fit(M, N, Kred)):
        return None
    dx = torch.empty(M, N, dtype=dy2.dtype, device=dy2.device)
    _g2_call(dy2, weight, dx, None, None, M, N, Kred, dy2.stride(0),
             weight.stride(0), dx.stride(0), 0, 1, 0, _g2_gsu(M, N, Kred))
    return dx


def gemm2_wgrad(dy2, x2):
    """TN wgrad dw = dy2.T @ x2 (both stored [Mtok, out] = kmajor)."""
    if dy2.dtype != torch.bfloat16:
        return None
    Kred, M = dy2.shape
    N = x2.shape[1]
    if not (_g2_enabled("wgrad", M, N, Kred) and _g2_fit(M, N, Kred)):
        return None
    dw = torch.empty(M, N, dtype=dy2.dtype, device=dy2.device)
    _g2_call(dy2, x2, dw, None, None, M, N, Kred, dy2.stride(0),
             x2.stride(0), dw.stride(0), 1, 1, 0, _g2_gsu(M, N, Kred))
    return dw


class LinearBiasFn(torch.autograd.Function):
    """Linear + bias with a custom backward: dgrad stays a hipBLASLt GEMM;
    wgrad is a direct hipBLASLt call whose BGRADB epilogue produces dbias
    inside the GEMM (ops/hip/hblt.hip), removing the separate column
    reduction per linear (a measured hot spot — profiles/r01_notes.md).
    SKY_NO_HBLT=1 falls back to torch wgrad + HIP colsum."""

    @staticmethod
    def forward(ctx, x, weight, bias):
        xs = x.shape
        x2 = x.reshape(-1, xs[-1])
        g2 = gemm2_fwd(x2 if x2.is_contiguous() else x2.contiguous(),
                       weight, bias)
        if g2 is not None:
            y = g2[0]
        else:
            y = torch.nn.functional.linear(x2, weight, bias)
        ctx.save_for_backward(x2, weight)
        ctx.xshape = xs
        ctx.has_bias = bias is not None
        return y.view(*xs[:-1], weight.shape[0])

    @staticmethod
    def backward(ctx, dy):
        x2, weight = ctx.saved_tensors
        dy2 = dy.reshape(-1, weight.shape[0])
        if not dy2.is_contiguous():
            dy2 = dy2.contiguous()
        dx = gemm2_dgrad(dy2, weight)
        dx = dx.view(ctx.xshape) if dx is not None else dy2.mm(weight).view(ctx.xshape)
        dw2 = gemm2_wgrad(dy2, x2 if x2.is_contiguous() else x2.contiguous())
        if dw2 is not None:
            db = colsum(dy2, weight.dtype) if ctx.has_bias else None
            return dx, dw2, db
        if ctx.has_bias and _use_hblt():
            lib = hiplib.require()
            M, K = x2.shape
            N = weight.shape[0]
            dw = torch.empty_like(weight)
            db = torch.empty(N, dtype=weight.dtype, device=weight.device)
            check(
                lib.sky_hblt_wgrad_bgrad(
                    _stream(), ptr(x2), ptr(dy2), ptr(dw), ptr(db), M, N, K, _dt(x2)
                ),
                "sky_hblt_wgrad_bgrad",
            )
            return dx, dw, db
        dw = dy2.t().mm(x2)
        db = colsum(dy2, weight.dtype) if ctx.has_bias else None
        return dx, dw, db


class LinearGeluFn(torch.autograd.Function):
    """Fused linear + bias + GELU via the hipBLASLt GELU_AUX_BIAS epilogue
    (the reference's LinearActivation, scaelum/model/bert_layers.py:60-108):
    the forward is ONE GEMM that also writes the pre-activation `aux`; the
    backward reuses the bias_gelu kernel with aux as the pre-activation
    (dpre + fused dbias), then plain dgrad/wgrad GEMMs. hipBLASLt's GELU is
    the tanh approximation; the erf-derivative backward differs by less
    than bf16 rounding (tests/test_ops_gpu.py)."""

    @staticmethod
    def forward(ctx, x, weight, bias):
        xs = x.shape
        x2 = x.reshape(-1, xs[-1]).contiguous()
        M, K = x2.shape
        N = weight.shape[0]
        g2 = gemm2_fwd(x2, weight, bias, gelu=True, want_z=True)
        if g2 is not None:
            y, aux = g2
        else:
            lib = hiplib.require()
            y = torch.empty(M, N, dtype=x2.dtype, device=x2.device)
            aux = torch.empty(M, N, dtype=x2.dtype, device=x2.device)
            check(
                lib.sky_hblt_linear_gelu_aux(
                    _stream(), ptr(x2), ptr(weight), ptr(bias), ptr(y), ptr(aux),
                    M, N, K, _dt(x2), _dt(aux)
                ),
                "sky_hblt_linear_gelu_aux",
            )
        ctx.save_for_backward(x2, weight, aux)
        ctx.xshape = xs
        return y.view(*xs[:-1], N)

    @staticmethod
    def backward(ctx, dy):
        lib = hiplib.require()
        x2, weight, aux = ctx.saved_tensors
        N = weight.shape[0]
        dy2 = dy.reshape(-1, N)
        if not dy2.is_contiguous():
            dy2 = dy2.contiguous()
        rows = dy2.shape[0]
        dpre = torch.empty_like(dy2)
        db = torch.empty(N, dtype=weight.dtype, device=weight.device)
        scratch = _red_scratch(N, 1, dy2.device)
        check(
            lib.sky_bias_gelu_bwd(
                _stream(), ptr(dy2), ptr(aux), 0, ptr(dpre), ptr(db),
                ptr(scratch), rows, N, _dt(dy2)
            ),
            "sky_bias_gelu_bwd",
        )
        dx = gemm2_dgrad(dpre, weight)
        dx = dx.view(ctx.xshape) if dx is not None else dpre.mm(weight).view(ctx.xshape)
        dw = gemm2_wgrad(dpre, x2)
        if dw is None:
            dw = dpre.t().mm(x2)
        return dx, dw, db


class EmbeddingFusedFn(torch.autograd.Function):
    """Fused word+position+type gather, 3-way add and LayerNorm
    (reference eager sequence: scaelum/model/bert_layers.py:191-212)."""

    @staticmethod
    def forward(ctx, input_ids, token_type_ids, position_ids,
                word_emb, pos_emb, type_emb, ln_w, ln_b, eps):
        lib = hiplib.require()
        ids = input_ids.contiguous().view(-1).to(torch.int64)
        tids = token_type_ids.contiguous().view(-1).to(torch.int64)
        pids = position_ids.contiguous().view(-1).to(torch.int64)
        rows = ids.numel()
        cols = word_emb.shape[1]
        y = torch.empty(
            (*input_ids.shape, cols), dtype=word_emb.dtype, device=word_emb.device
        )
        mean = torch.empty(rows, dtype=torch.float32, device=y.device)
        rstd = torch.empty(rows, dtype=torch.float32, device=y.device)
        check(
            lib.sky_embedding_fwd(
                _stream(), ptr(ids), ptr(tids), ptr(pids),
                ptr(word_emb), ptr(type_emb), ptr(pos_emb),
                ptr(ln_w), ptr(ln_b), ptr(y), ptr(mean), ptr(rstd),
                rows, cols, word_emb.shape[0], eps, _dt(y),
            ),
            "sky_embedding_fwd",
        )
        ctx.save_for_backward(ids, tids, pids, word_emb, pos_emb, type_emb, ln_w, mean, rstd)
        ctx.eps = eps
        return y

    @staticmethod
    def backward(ctx, dy):
        lib = hiplib.require()
        ids, tids, pids, word_emb, pos_emb, type_emb, ln_w, mean, rstd = ctx.saved_tensors
        dy = dy.contiguous()
        cols = word_emb.shape[1]
        rows = ids.numel()
        dev = dy.device
        dwe = torch.zeros(word_emb.shape, dtype=torch.float32, device=dev)
        dte = torch.zeros(type_emb.shape, dtype=torch.float32, device=dev)
        dpe = torch.zeros(pos_emb.shape, dtype=torch.float32, device=dev)
        dlnw = torch.zeros(cols, dtype=torch.float32, device=dev)
        dlnb = torch.zeros(cols, dtype=torch.float32, device=dev)
        check(
            lib.sky_embedding_bwd(
                _stream(), ptr(dy), ptr(ids), ptr(tids), ptr(pids),
                ptr(word_emb), ptr(type_emb), ptr(pos_emb), ptr(ln_w),
                ptr(mean), ptr(rstd), ptr(dwe), ptr(dte), ptr(dpe),
                ptr(dlnw), ptr(dlnb),
                rows, cols, _dt(dy),
            ),
            "sky_embedding_bwd",
        )
        return (
            None, None, None,
            dwe.to(word_emb.dtype), dpe.to(pos_emb.dtype), dte.to(type_emb.dtype),
            dlnw.to(ln_w.dtype), dlnb.to(ln_w.dtype), None,
        )
