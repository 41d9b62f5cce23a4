"""HIP kernel numerics vs plain PyTorch fp32 reference (ops/eager.py).

Each op is tested in fp32 (tight tolerance) and bf16 (bf16-rounding
tolerance), forward and backward, on MI355X.
"""

from __future__ import annotations

import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from skycomputing_amd.ops import eager, hiplib
    from skycomputing_amd.ops.functions import (
        BiasGeluFn, DropoutFn, EmbeddingFusedFn, LayerNormFn, MaskedSoftmaxFn,
    )


@pytest.fixture(scope="module", autouse=True)
def _require_gpu_and_lib():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    assert hiplib.available(), "libskyhip.so must be built (fail loudly, no eager fallback)"


def _tols(dtype):
    return dict(atol=1e-4, rtol=1e-4) if dtype == torch.float32 else dict(atol=5e-2, rtol=5e-2)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("residual", [False, True])
def test_layernorm_fwd_bwd(dtype, residual):
    torch.manual_seed(0)
    rows, cols = 512, 1024
    x = torch.randn(rows, cols, dtype=dtype, device="cuda", requires_grad=True)
    res = torch.randn(rows, cols, dtype=dtype, device="cuda", requires_grad=True) if residual else None
    w = (torch.rand(cols, device="cuda") + 0.5).to(dtype).requires_grad_(True)
    b = torch.randn(cols, device="cuda", dtype=dtype, requires_grad=True)
    y = LayerNormFn.apply(x, w, b, 1e-12, res)
    dy = torch.randn_like(y)
    y.backward(dy)

    xf = x.detach().float().requires_grad_(True)
    rf = res.detach().float().requires_grad_(True) if residual else None
    wf = w.detach().float().requires_grad_(True)
    bf = b.detach().float().requires_grad_(True)
    yr = eager.layer_norm(xf, wf, bf, 1e-12, rf)
    yr.backward(dy.float())

    t = _tols(dtype)
    assert torch.allclose(y.float(), yr, **t), (y.float() - yr).abs().max()
    assert torch.allclose(x.grad.float(), xf.grad, **t)
    if residual:
        assert torch.allclose(res.grad.float(), rf.grad, **t)
    # param grads accumulate over 512 rows -> scale tolerance
    assert torch.allclose(w.grad.float(), wf.grad, atol=t["atol"] * 30, rtol=0.05)
    assert torch.allclose(b.grad.float(), bf.grad, atol=t["atol"] * 30, rtol=0.05)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_bias_gelu_fwd_bwd(dtype):
    torch.manual_seed(1)
    rows, cols = 512, 4096
    x = torch.randn(rows, cols, dtype=dtype, device="cuda", requires_grad=True)
    b = torch.randn(cols, dtype=dtype, device="cuda", requires_grad=True)
    y = BiasGeluFn.apply(x, b)
    dy = torch.randn_like(y)
    y.backward(dy)

    xf = x.detach().float().requires_grad_(True)
    bf = b.detach().float().requires_grad_(True)
    yr = eager.bias_gelu(xf, bf)
    yr.backward(dy.float())
    t = _tols(dtype)
    assert torch.allclose(y.float(), yr, **t)
    assert torch.allclose(x.grad.float(), xf.grad, **t)
    assert torch.allclose(b.grad.float(), bf.grad, atol=t["atol"] * 30, rtol=0.05)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("Sk", [128, 384, 1024])
def test_masked_softmax_fwd_bwd(dtype, Sk):
    torch.manual_seed(2)
    B, h, Sq = 4, 4, 32
    scale = 0.125
    s = torch.randn(B, h, Sq, Sk, dtype=dtype, device="cuda", requires_grad=True)
    mask = torch.zeros(B, 1, 1, Sk, dtype=dtype, device="cuda")
    mask[:, :, :, Sk // 2:] = -10000.0
    p = MaskedSoftmaxFn.apply(s, mask, scale)
    dp = torch.randn_like(p)
    p.backward(dp)

    sf = s.detach().float().requires_grad_(True)
    pr = eager.masked_softmax(sf * scale, mask.float())
    pr.backward(dp.float())
    t = _tols(dtype)
    assert torch.allclose(p.float(), pr, atol=t["atol"], rtol=0.1)
    assert torch.allclose(s.grad.float(), sf.grad, atol=t["atol"], rtol=0.1)
    assert torch.allclose(p.float().sum(-1), torch.ones(B, h, Sq, device="cuda"), atol=1e-2)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_dropout_fwd_bwd_mask_consistency(dtype):
    torch.manual_seed(3)
    x = torch.randn(1 << 16, dtype=dtype, device="cuda", requires_grad=True)
    p = 0.3
    y = DropoutFn.apply(x, p)
    keep_frac = (y != 0).float().mean().item()
    assert abs(keep_frac - 0.7) < 0.02
    # kept elements scaled by 1/keep
    kept = y[y != 0].float()
    ref = (x.detach()[y != 0].float()) / 0.7
    assert torch.allclose(kept, ref, atol=2e-2, rtol=2e-2)
    # backward regenerates the same mask
    dy = torch.ones_like(y)
    y.backward(dy)
    g = x.grad.float()
    assert torch.equal((g != 0), (y.detach() != 0))


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_embedding_fused_fwd_bwd(dtype):
    torch.manual_seed(4)
    V, H, B, S, P = 1000, 256, 8, 32, 64
    we = torch.randn(V, H, dtype=dtype, device="cuda", requires_grad=True)
    pe = torch.randn(P, H, dtype=dtype, device="cuda", requires_grad=True)
    te = torch.randn(2, H, dtype=dtype, device="cuda", requires_grad=True)
    w = (torch.rand(H, device="cuda") + 0.5).to(dtype).requires_grad_(True)
    b = torch.randn(H, dtype=dtype, device="cuda", requires_grad=True)
    ids = torch.randint(0, V, (B, S), device="cuda")
    tids = torch.randint(0, 2, (B, S), device="cuda")
    pids = torch.arange(S, device="cuda").unsqueeze(0).expand(B, S)
    y = EmbeddingFusedFn.apply(ids, tids, pids, we, pe, te, w, b, 1e-12)
    dy = torch.randn_like(y)
    y.backward(dy)

    wef = we.detach().float().requires_grad_(True)
    pef = pe.detach().float().requires_grad_(True)
    tef = te.detach().float().requires_grad_(True)
    wf = w.detach().float().requires_grad_(True)
    bf = b.detach().float().requires_grad_(True)
    yr = eager.embedding_fused(ids, tids, pids, wef, pef, tef, wf, bf, 1e-12)
    yr.backward(dy.float())
    t = _tols(dtype)
    assert torch.allclose(y.float(), yr, **t)
    assert torch.allclose(we.grad.float(), wef.grad, atol=t["atol"] * 10, rtol=0.05)
    assert torch.allclose(pe.grad.float(), pef.grad, atol=t["atol"] * 10, rtol=0.05)
    assert torch.allclose(te.grad.float(), tef.grad, atol=t["atol"] * 30, rtol=0.05)
    assert torch.allclose(w.grad.float(), wf.grad, atol=t["atol"] * 30, rtol=0.05)


@pytest.mark.parametrize("dtype,master", [(torch.float32, False), (torch.bfloat16, True)])
def test_multi_tensor_sgd(dtype, master):
    torch.manual_seed(5)
    from skycomputing_amd.optim import FusedSGD

    params = [
        torch.randn(n, dtype=dtype, device="cuda", requires_grad=True)
        for n in (1024, 3 << 20, 77)
    ]
    grads = [torch.randn_like(p) for p in params]
    refs = [p.detach().float().clone() for p in params]
    for p, g in zip(params, grads):
        p.grad = g
    opt = FusedSGD(params, lr=0.1, momentum=0.9, master_weights=master)
    opt.step()
    opt.step()  # second step exercises momentum + cached plan
    for p, g, r in zip(params, grads, refs):
        buf = g.float().clone()
        r1 = r - 0.1 * buf
        buf = 0.9 * buf + g.float()
        r2 = r1 - 0.1 * buf
        tol = 1e-5 if dtype == torch.float32 else 2e-2
        assert torch.allclose(p.detach().float(), r2, atol=tol, rtol=tol), (
            (p.detach().float() - r2).abs().max()
        )


def test_attention_context_vs_fp32():
    torch.manual_seed(6)
    from skycomputing_amd import ops

    q, k, v = (torch.randn(2, 16, 128, 64, dtype=torch.bfloat16, device="cuda") for _ in range(3))
    mask = torch.zeros(2, 1, 1, 128, dtype=torch.bfloat16, device="cuda")
    mask[:, :, :, 100:] = -10000.0
    out = ops.attention_context(q, k, v, mask)
    ref = eager.attention_context(q.float(), k.float(), v.float(), mask.float())
    assert torch.allclose(out.float(), ref, atol=6e-2, rtol=6e-2)


def test_mfma_layout():
    """Verify the 16x16x32 bf16 MFMA fragment mappings the attention kernel
    assumes (A: i=l%16,k=8*(l/16)+j; B: n=l%16 same k; D: n=l%16,i=4*(l/16)+r)."""
    torch.manual_seed(7)
    A = torch.randn(16, 32).bfloat16().cuda()
    # ASYMMETRIC B to catch transposes (guide: A=I-check trap)
    B = (torch.randn(32, 16) * torch.linspace(0.5, 2.0, 16)).bfloat16().cuda()
    D = torch.zeros(16, 16, dtype=torch.float32, device="cuda")
    stream = torch.cuda.current_stream().cuda_stream
    rc = hiplib.lib().sky_mfma_probe(
        stream, A.view(torch.uint16).data_ptr(), B.view(torch.uint16).data_ptr(),
        D.data_ptr())
    torch.cuda.synchronize()
    assert rc == 0
    ref = A.float() @ B.float()
    assert torch.allclose(D, ref, atol=5e-2, rtol=5e-2), (D - ref).abs().max()


@pytest.mark.parametrize("S", [128, 100, 64, 48])
@pytest.mark.parametrize("with_mask", [True, False])
@pytest.mark.parametrize("bwd", ["split", "fused", "oldv"])
def test_fused_attention_fwd_bwd(S, with_mask, bwd, monkeypatch):
    # bwd="fused" exercises the single-kernel dQ/dK/dV path
    # (sky_attn_bwd_fused); "split" the default bwd1s+bwd2 pair;
    # "oldv" the register-transpose V-staging forward (SKY_ATTN_TRV=0).
    if bwd == "fused":
        monkeypatch.setenv("SKY_ATTN_FUSED_BWD", "1")
    elif bwd == "oldv":
        monkeypatch.setenv("SKY_ATTN_TRV", "0")
    torch.manual_seed(8)
    from skycomputing_amd.ops.functions import FusedAttentionFn

    B, h, d = 3, 4, 64
    scale = 1.0 / d ** 0.5
    qkv = torch.randn(B, S, 3, h, d, dtype=torch.bfloat16, device="cuda",
                      requires_grad=True)
    if with_mask:
        mask = torch.zeros(B, 1, 1, S, dtype=torch.bfloat16, device="cuda")
        mask[:, :, :, S - S // 4:] = -10000.0
    else:
        mask = None
    out = FusedAttentionFn.apply(qkv, mask, scale, 0.0, False)
    dout = torch.randn_like(out)
    out.backward(dout)

    qf = qkv.detach().float().requires_grad_(True)
    q = qf[:, :, 0].permute(0, 2, 1, 3)
    k = qf[:, :, 1].permute(0, 2, 1, 3)
    v = qf[:, :, 2].permute(0, 2, 1, 3)
    ref = eager.attention_context(q, k, v, mask.float() if mask is not None else None)
    ref = ref.permute(0, 2, 1, 3)
    ref.backward(dout.float())
    assert torch.allclose(out.float(), ref, atol=6e-2, rtol=6e-2), (
        (out.float() - ref).abs().max()
    )
    assert torch.allclose(qkv.grad.float(), qf.grad, atol=8e-2, rtol=8e-2), (
        (qkv.grad.float() - qf.grad).abs().max()
    )


def test_fused_attention_dropout_consistency():
    """With dropout on, E[out] matches the no-dropout output and backward
    regenerates exactly the forward mask (zero positions align)."""
    torch.manual_seed(9)
    from skycomputing_amd.ops.functions import FusedAttentionFn
    from skycomputing_amd.ops.hiplib import check, ptr

    B, S, h, d = 2, 128, 4, 64
    scale = 1.0 / d ** 0.5
    qkv = torch.randn(B, S, 3, h, d, dtype=torch.bfloat16, device="cuda",
                      requires_grad=True)
    out = FusedAttentionFn.apply(qkv, None, scale, 0.5, True)
    assert torch.isfinite(out.float()).all()
    out.sum().backward()
    assert torch.isfinite(qkv.grad.float()).all()
    # mean over many heads/batches approximates the dropout-free output
    out0 = FusedAttentionFn.apply(qkv.detach(), None, scale, 0.0, False)
    rel = (out.float().mean() - out0.float().mean()).abs()
    assert rel < 0.05


def test_layernorm_fused_dropout_matches_composition():
    """The LN kernel's fused dropout uses the SAME (salt,index) hash scheme
    as the standalone dropout kernel, so with a fixed salt
    LN_fused(x, keep, salt) == eager_LN(dropout_kernel(x, keep, salt) + res)."""
    torch.manual_seed(11)
    from skycomputing_amd.ops.hiplib import check, ptr

    lib = hiplib.require()
    rows, cols = 256, 1024
    dt = torch.bfloat16
    x = torch.randn(rows, cols, dtype=dt, device="cuda")
    res = torch.randn(rows, cols, dtype=dt, device="cuda")
    w = (torch.rand(cols, device="cuda") + 0.5).to(dt)
    b = torch.randn(cols, dtype=dt, device="cuda")
    keep, salt = 0.7, 987654321
    stream = torch.cuda.current_stream().cuda_stream

    # reference: standalone dropout kernel with the same salt, then eager LN
    xd = torch.empty_like(x)
    check(lib.sky_dropout_fwd(stream, ptr(x), ptr(xd), x.numel(), keep, salt, 0, 1), "d")
    ref = eager.layer_norm(xd.float(), w.float(), b.float(), 1e-12, res.float())

    y = torch.empty_like(x)
    mean = torch.empty(rows, dtype=torch.float32, device="cuda")
    rstd = torch.empty_like(mean)
    check(
        lib.sky_layernorm_fwd(stream, ptr(x), ptr(res), ptr(w), ptr(b), ptr(y),
                              ptr(mean), ptr(rstd), rows, cols, 1e-12, 1,
                              keep, salt, 0),
        "lnf",
    )
    torch.cuda.synchronize()
    assert torch.allclose(y.float(), ref, atol=6e-2, rtol=6e-2), (
        (y.float() - ref).abs().max()
    )


def test_layernorm_dropout_autograd_vs_reference():
    """Full autograd path of LN(dropout(x)+res): gradients must match the
    fp32 composition computed with the mask RECOVERED from the fused op
    (mask = positions where fused dx is zero given res grad differs)."""
    torch.manual_seed(12)
    from skycomputing_amd.ops.functions import LayerNormFn

    rows, cols = 128, 512
    x = torch.randn(rows, cols, dtype=torch.bfloat16, device="cuda", requires_grad=True)
    res = torch.randn(rows, cols, dtype=torch.bfloat16, device="cuda", requires_grad=True)
    w = (torch.rand(cols, device="cuda") + 0.5).bfloat16().requires_grad_(True)
    b = torch.randn(cols, dtype=torch.bfloat16, device="cuda", requires_grad=True)
    p = 0.4
    y = LayerNormFn.apply(x, w, b, 1e-12, res, p)
    dy = torch.randn_like(y)
    y.backward(dy)
    # recover the keep mask from dx (zero exactly where dropped)
    mask = (x.grad != 0).float()
    keep_frac = mask.mean().item()
    assert abs(keep_frac - (1 - p)) < 0.03
    # fp32 composition with the recovered mask must reproduce y and grads
    xf = x.detach().float().requires_grad_(True)
    rf = res.detach().float().requires_grad_(True)
    wf = w.detach().float().requires_grad_(True)
    bf = b.detach().float().requires_grad_(True)
    xd = xf * mask / (1 - p)
    yr = eager.layer_norm(xd, wf, bf, 1e-12, rf)
    yr.backward(dy.float())
    assert torch.allclose(y.float(), yr, atol=6e-2, rtol=6e-2)
    assert torch.allclose(x.grad.float(), xf.grad, atol=6e-2, rtol=6e-2)
    assert torch.allclose(res.grad.float(), rf.grad, atol=6e-2, rtol=6e-2)
    assert torch.allclose(w.grad.float(), wf.grad, atol=0.3, rtol=0.05)


@pytest.mark.parametrize("transA,transB", [(0, 1), (0, 0), (1, 0)])
def test_mfma_gemm_orientations(transA, transB):
    """Hand MFMA GEMM (128x128x64 tiles): fwd NT, dgrad NN, wgrad TN."""
    torch.manual_seed(13)
    from skycomputing_amd.ops.hiplib import check, ptr

    lib = hiplib.require()
    M, N, K = 256, 384, 128
    if transA == 0:
        a = torch.randn(M, K, dtype=torch.bfloat16, device="cuda") * 0.2
        ar = a.float()
    else:
        a = torch.randn(K, M, dtype=torch.bfloat16, device="cuda") * 0.2
        ar = a.float().t()
    if transB == 1:
        b = torch.randn(N, K, dtype=torch.bfloat16, device="cuda") * 0.2
        br = b.float().t()
    else:
        b = torch.randn(K, N, dtype=torch.bfloat16, device="cuda") * 0.2
        br = b.float()
    c = torch.empty(M, N, dtype=torch.bfloat16, device="cuda")
    rc = lib.sky_gemm(
        torch.cuda.current_stream().cuda_stream, ptr(a), ptr(b), ptr(c), 0, 0,
        M, N, K, a.stride(0), b.stride(0), c.stride(0), transA, transB, 0, 1, 1,
    )
    torch.cuda.synchronize()
    assert rc == 0
    ref = ar @ br
    assert torch.allclose(c.float(), ref, atol=5e-2, rtol=5e-2), (
        (c.float() - ref).abs().max()
    )


def test_mfma_gemm_bias_gelu_epilogue():
    torch.manual_seed(14)
    from skycomputing_amd.ops.hiplib import check, ptr

    lib = hiplib.require()
    M, N, K = 128, 256, 64
    x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda") * 0.3
    w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda") * 0.3
    b = torch.randn(N, dtype=torch.bfloat16, device="cuda")
    c = torch.empty(M, N, dtype=torch.bfloat16, device="cuda")
    z = torch.empty_like(c)
    rc = lib.sky_gemm(
        torch.cuda.current_stream().cuda_stream, ptr(x), ptr(w), ptr(c),
        ptr(b), ptr(z), M, N, K, K, K, N, 0, 1, 2, 1, 1,
    )
    torch.cuda.synchronize()
    assert rc == 0
    zr = x.float() @ w.float().t() + b.float()
    ref = eager.gelu(zr)
    assert torch.allclose(z.float(), zr, atol=5e-2, rtol=5e-2)
    assert torch.allclose(c.float(), ref, atol=5e-2, rtol=5e-2)


def test_attention_fused_vs_decomposed_paths(monkeypatch):
    """ops.attention: the fused kernel and the decomposed
    (bmm + masked-softmax) fallback must agree (no dropout)."""
    torch.manual_seed(15)
    from skycomputing_amd import ops

    qkv = torch.randn(2, 128, 3, 4, 64, dtype=torch.bfloat16, device="cuda")
    mask = torch.zeros(2, 1, 1, 128, dtype=torch.bfloat16, device="cuda")
    mask[:, :, :, 100:] = -10000.0
    fused = ops.attention(qkv, mask, 0.0, False)
    monkeypatch.setenv("SKY_NO_FUSED_ATTN", "1")
    decomposed = ops.attention(qkv, mask, 0.0, False)
    assert torch.allclose(fused.float(), decomposed.float(), atol=6e-2, rtol=6e-2), (
        (fused.float() - decomposed.float()).abs().max()
    )


def test_attention_decomposed_backward(monkeypatch):
    """The env-gated decomposed backward (probs recompute + bmms) stays
    correct — it is the fallback if the fused backward is disabled."""
    torch.manual_seed(16)
    from skycomputing_amd.ops.functions import FusedAttentionFn

    monkeypatch.setenv("SKY_NO_FUSED_ATTN_BWD", "1")
    qkv = torch.randn(2, 64, 3, 4, 64, dtype=torch.bfloat16, device="cuda",
                      requires_grad=True)
    out = FusedAttentionFn.apply(qkv, None, 0.125, 0.0, False)
    dout = torch.randn_like(out)
    out.backward(dout)
    qf = qkv.detach().float().requires_grad_(True)
    q = qf[:, :, 0].permute(0, 2, 1, 3)
    k = qf[:, :, 1].permute(0, 2, 1, 3)
    v = qf[:, :, 2].permute(0, 2, 1, 3)
    ref = eager.attention_context(q, k, v, None).permute(0, 2, 1, 3)
    ref.backward(dout.float())
    assert torch.allclose(qkv.grad.float(), qf.grad, atol=8e-2, rtol=8e-2)


@pytest.mark.parametrize("cols", [2048, 4096, 1000, 72])
def test_layernorm_other_widths(cols):
    """Covers the CH=4/8 vectorized template instantiations (2048/4096) and
    the scalar fallback paths (1000 non-x8... 1000 is x8? no: 1000%8==0 ->
    vec CH=2; 72 -> vec CH=1..; use 1001 for scalar) — parametrized widths
    exercise every dispatch branch."""
    torch.manual_seed(17)
    rows = 64
    x = torch.randn(rows, cols, dtype=torch.bfloat16, device="cuda", requires_grad=True)
    res = torch.randn_like(x).requires_grad_(True)
    w = (torch.rand(cols, device="cuda") + 0.5).bfloat16().requires_grad_(True)
    b = torch.randn(cols, dtype=torch.bfloat16, device="cuda", requires_grad=True)
    y = LayerNormFn.apply(x, w, b, 1e-12, res, 0.0)
    dy = torch.randn_like(y)
    y.backward(dy)
    xf = x.detach().float().requires_grad_(True)
    rf = res.detach().float().requires_grad_(True)
    wf = w.detach().float().requires_grad_(True)
    bf = b.detach().float().requires_grad_(True)
    yr = eager.layer_norm(xf, wf, bf, 1e-12, rf)
    yr.backward(dy.float())
    assert torch.allclose(y.float(), yr, atol=5e-2, rtol=5e-2)
    assert torch.allclose(x.grad.float(), xf.grad, atol=5e-2, rtol=5e-2)
    assert torch.allclose(w.grad.float(), wf.grad, atol=0.2, rtol=0.05)


def test_layernorm_scalar_fallback_width():
    """Odd width (not %8): block-per-row scalar kernels + atomic wb path."""
    torch.manual_seed(18)
    rows, cols = 32, 1001
    x = torch.randn(rows, cols, dtype=torch.float32, device="cuda", requires_grad=True)
    w = (torch.rand(cols, device="cuda") + 0.5).requires_grad_(True)
    b = torch.randn(cols, device="cuda", requires_grad=True)
    y = LayerNormFn.apply(x, w, b, 1e-12, None, 0.0)
    dy = torch.randn_like(y)
    y.backward(dy)
    xf = x.detach().clone().requires_grad_(True)
    wf = w.detach().clone().requires_grad_(True)
    bf = b.detach().clone().requires_grad_(True)
    yr = eager.layer_norm(xf, wf, bf, 1e-12)
    yr.backward(dy)
    assert torch.allclose(y, yr, atol=1e-4, rtol=1e-4)
    assert torch.allclose(x.grad, xf.grad, atol=1e-4, rtol=1e-4)
    assert torch.allclose(w.grad, wf.grad, atol=1e-3, rtol=1e-3)


def test_bias_gelu_odd_width_fallback():
    torch.manual_seed(19)
    rows, cols = 64, 301
    x = torch.randn(rows, cols, dtype=torch.float32, device="cuda", requires_grad=True)
    b = torch.randn(cols, device="cuda", requires_grad=True)
    y = BiasGeluFn.apply(x, b)
    dy = torch.randn_like(y)
    y.backward(dy)
    xf = x.detach().clone().requires_grad_(True)
    bf = b.detach().clone().requires_grad_(True)
    yr = eager.bias_gelu(xf, bf)
    yr.backward(dy)
    assert torch.allclose(y, yr, atol=1e-5, rtol=1e-5)
    assert torch.allclose(x.grad, xf.grad, atol=1e-5, rtol=1e-5)
    assert torch.allclose(b.grad, bf.grad, atol=1e-3, rtol=1e-3)


@pytest.mark.parametrize("S", [192, 256, 512])
def test_flash_attention_long_seq(S):
    """Flash-style forward (online softmax over kv tiles) + decomposed
    backward for S > 128, vs the eager fp32 reference."""
    torch.manual_seed(20)
    from skycomputing_amd.ops.functions import FlashAttentionFn

    B, h, d = 2, 4, 64
    scale = 1.0 / d ** 0.5
    qkv = torch.randn(B, S, 3, h, d, dtype=torch.bfloat16, device="cuda",
                      requires_grad=True)
    mask = torch.zeros(B, 1, 1, S, dtype=torch.bfloat16, device="cuda")
    mask[:, :, :, S - S // 5:] = -10000.0
    out = FlashAttentionFn.apply(qkv, mask, scale, 0.0, False)
    dout = torch.randn_like(out)
    out.backward(dout)

    qf = qkv.detach().float().requires_grad_(True)
    q = qf[:, :, 0].permute(0, 2, 1, 3)
    k = qf[:, :, 1].permute(0, 2, 1, 3)
    v = qf[:, :, 2].permute(0, 2, 1, 3)
    ref = eager.attention_context(q, k, v, mask.float()).permute(0, 2, 1, 3)
    ref.backward(dout.float())
    assert torch.allclose(out.float(), ref, atol=6e-2, rtol=6e-2), (
        S, (out.float() - ref).abs().max()
    )
    assert torch.allclose(qkv.grad.float(), qf.grad, atol=8e-2, rtol=8e-2), (
        S, (qkv.grad.float() - qf.grad).abs().max()
    )


@pytest.mark.parametrize("S", [512, 2048])
def test_attention_sdpa_dispatch_long_seq(S):
    """The DEFAULT long-S attention path (torch flash sdpa) fwd+bwd vs the
    eager fp32 reference — the dispatch must have no S regression window
    and no S x S tensor in either direction."""
    torch.manual_seed(22)
    from skycomputing_amd import ops

    B, h, d = 2, 4, 64
    qkv = torch.randn(B, S, 3, h, d, dtype=torch.bfloat16, device="cuda",
                      requires_grad=True)
    mask = torch.zeros(B, 1, 1, S, dtype=torch.bfloat16, device="cuda")
    mask[:, :, :, S - S // 5:] = -10000.0
    out = ops.attention(qkv, mask, 0.0, True)
    assert out.shape == (B, S, h * d)
    dout = torch.randn_like(out)
    out.backward(dout)

    qf = qkv.detach().float().requires_grad_(True)
    q = qf[:, :, 0].permute(0, 2, 1, 3)
    k = qf[:, :, 1].permute(0, 2, 1, 3)
    v = qf[:, :, 2].permute(0, 2, 1, 3)
    ref = eager.attention_context(q, k, v, mask.float()).permute(0, 2, 1, 3)
    ref = ref.reshape(B, S, h * d)
    ref.backward(dout.float())
    assert torch.allclose(out.float(), ref, atol=6e-2, rtol=6e-2), (
        S, (out.float() - ref).abs().max())
    assert torch.allclose(qkv.grad.float(), qf.grad, atol=8e-2, rtol=8e-2), (
        S, (qkv.grad.float() - qf.grad).abs().max())


def test_flash_attention_dropout_trains():
    torch.manual_seed(21)
    from skycomputing_amd.ops.functions import FlashAttentionFn

    qkv = torch.randn(2, 256, 3, 4, 64, dtype=torch.bfloat16, device="cuda",
                      requires_grad=True)
    out = FlashAttentionFn.apply(qkv, None, 0.125, 0.3, True)
    assert torch.isfinite(out.float()).all()
    out.sum().backward()
    assert torch.isfinite(qkv.grad.float()).all()
    # E[dropout out] ~ no-dropout out
    out0 = FlashAttentionFn.apply(qkv.detach(), None, 0.125, 0.0, False)
    assert (out.float().mean() - out0.float().mean()).abs() < 0.05


def test_gemm2_linear_fwd_bwd_matches_reference():
    """v2 hand-GEMM dispatch (SKY_GEMM2=1): LinearBiasFn fwd via the 256^2
    8-phase NT kernel, dgrad via NN (tr16 kmajor W), wgrad via TN — all
    against the fp32 torch reference on a bench-shaped problem."""
    import skycomputing_amd.ops.functions as F

    from skycomputing_amd.ops.functions import LinearBiasFn, LinearGeluFn

    old = F._G2_SITES
    F._G2_SITES = {"fwd", "dgrad", "wgrad"}
    try:
        torch.manual_seed(31)
        M, K, N = 4096, 1024, 3072
        x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda", requires_grad=True)
        w = (torch.randn(N, K, dtype=torch.bfloat16, device="cuda") * 0.03).requires_grad_()
        b = torch.randn(N, dtype=torch.bfloat16, device="cuda", requires_grad=True)
        y = LinearBiasFn.apply(x, w, b)
        dy = torch.randn_like(y) * 0.05
        y.backward(dy)
        xf = x.detach().float().requires_grad_(True)
        wf = w.detach().float().requires_grad_(True)
        bf = b.detach().float().requires_grad_(True)
        yr = torch.nn.functional.linear(xf, wf, bf)
        yr.backward(dy.float())
        assert torch.allclose(y.float(), yr, atol=0.5, rtol=0.05), (
            (y.float() - yr).abs().max())
        assert torch.allclose(x.grad.float(), xf.grad, atol=0.5, rtol=0.05), (
            (x.grad.float() - xf.grad).abs().max())
        assert torch.allclose(w.grad.float(), wf.grad, atol=1.0, rtol=0.05), (
            (w.grad.float() - wf.grad).abs().max())
        assert torch.allclose(b.grad.float(), bf.grad, atol=2.0, rtol=0.05)

        # gelu-fused forward (bf16 path only exists through gemm2)
        M, K, N = 4096, 1024, 4096
        x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda", requires_grad=True)
        w = (torch.randn(N, K, dtype=torch.bfloat16, device="cuda") * 0.03).requires_grad_()
        b = torch.randn(N, dtype=torch.bfloat16, device="cuda", requires_grad=True)
        y = LinearGeluFn.apply(x, w, b)
        dy = torch.randn_like(y) * 0.05
        y.backward(dy)
        xf = x.detach().float().requires_grad_(True)
        wf = w.detach().float().requires_grad_(True)
        bf = b.detach().float().requires_grad_(True)
        pre = torch.nn.functional.linear(xf, wf, bf)
        yr = pre * 0.5 * (1 + torch.erf(pre / 2**0.5))
        yr.backward(dy.float())
        assert torch.allclose(y.float(), yr, atol=0.5, rtol=0.05)
        assert torch.allclose(x.grad.float(), xf.grad, atol=0.5, rtol=0.05)
        assert torch.allclose(w.grad.float(), wf.grad, atol=1.0, rtol=0.05)
    finally:
        F._G2_SITES = old


# ---------------- hipBLASLt epilogue-fused linears (ops/hip/hblt.hip) ----------------


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_hblt_linear_bias_backward(dtype):
    """LinearBiasFn with the BGRADB wgrad epilogue: dw and db come out of
    ONE hipBLASLt call; compare against the fp32 torch reference."""
    from skycomputing_amd.ops.functions import LinearBiasFn

    torch.manual_seed(30)
    M, K, N = 512, 1024, 3072
    x = torch.randn(M, K, dtype=dtype, device="cuda", requires_grad=True)
    w = torch.randn(N, K, dtype=dtype, device="cuda", requires_grad=True) * 0.03
    w.retain_grad()
    b = torch.randn(N, dtype=dtype, device="cuda", requires_grad=True)
    y = LinearBiasFn.apply(x, w, b)
    dy = torch.randn_like(y)
    y.backward(dy)

    xf = x.detach().float().requires_grad_(True)
    wf = w.detach().float().requires_grad_(True)
    bf = b.detach().float().requires_grad_(True)
    yr = torch.nn.functional.linear(xf, wf, bf)
    yr.backward(dy.float())
    t = _tols(dtype)
    assert torch.allclose(y.float(), yr, atol=t["atol"] * 30, rtol=0.05)
    assert torch.allclose(x.grad.float(), xf.grad, atol=t["atol"] * 30, rtol=0.05)
    # dw/db accumulate over 512 rows
    assert torch.allclose(w.grad.float(), wf.grad, atol=t["atol"] * 100, rtol=0.05), (
        (w.grad.float() - wf.grad).abs().max()
    )
    assert torch.allclose(b.grad.float(), bf.grad, atol=t["atol"] * 100, rtol=0.05), (
        (b.grad.float() - bf.grad).abs().max()
    )


def test_hblt_linear_gelu_bf16_unsupported_fails_loudly():
    """This hipBLASLt build rejects the GELU_AUX_BIAS epilogue for bf16 D
    (HIPBLAS_STATUS_NOT_SUPPORTED, probed by tools/hblt_probe.py) — the op
    must fail loudly, never fall back silently; the dispatch layer
    (ops.linear_act) therefore gates the fusion to fp32."""
    from skycomputing_amd.ops.functions import LinearGeluFn

    x = torch.randn(64, 128, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(256, 128, dtype=torch.bfloat16, device="cuda")
    b = torch.randn(256, dtype=torch.bfloat16, device="cuda")
    with pytest.raises(RuntimeError, match="sky_hblt_linear_gelu_aux"):
        LinearGeluFn.apply(x, w, b)


@pytest.mark.parametrize("dtype", [torch.float32])
def test_hblt_linear_gelu(dtype):
    """LinearGeluFn (GELU_AUX_BIAS epilogue, fp32-only on this hipBLASLt
    build). hipBLASLt's GELU is the tanh approximation (max |erf-tanh|
    difference ~1.5e-3), so the forward is compared at a tolerance above
    that; grads use the erf derivative."""
    from skycomputing_amd.ops.functions import LinearGeluFn

    torch.manual_seed(31)
    M, K, N = 512, 1024, 4096
    x = torch.randn(M, K, dtype=dtype, device="cuda", requires_grad=True)
    w = torch.randn(N, K, dtype=dtype, device="cuda", requires_grad=True) * 0.03
    w.retain_grad()
    b = torch.randn(N, dtype=dtype, device="cuda", requires_grad=True)
    y = LinearGeluFn.apply(x, w, b)
    dy = torch.randn_like(y)
    y.backward(dy)

    xf = x.detach().float().requires_grad_(True)
    wf = w.detach().float().requires_grad_(True)
    bf = b.detach().float().requires_grad_(True)
    yr = torch.nn.functional.gelu(torch.nn.functional.linear(xf, wf, bf))
    yr.backward(dy.float())
    atol = 5e-3 if dtype == torch.float32 else 5e-2
    assert torch.allclose(y.float(), yr, atol=atol, rtol=0.05), (
        (y.float() - yr).abs().max()
    )
    assert torch.allclose(x.grad.float(), xf.grad, atol=atol * 4, rtol=0.05)
    assert torch.allclose(w.grad.float(), wf.grad, atol=atol * 20, rtol=0.05)
    assert torch.allclose(b.grad.float(), bf.grad, atol=atol * 20, rtol=0.05)


def test_hblt_matches_fallback_path():
    """The hipBLASLt wgrad epilogue and the colsum fallback agree."""
    import os

    from skycomputing_amd.ops.functions import LinearBiasFn

    torch.manual_seed(32)
    M, K, N = 256, 512, 1024
    x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda") * 0.03
    b = torch.randn(N, dtype=torch.bfloat16, device="cuda")
    dy = torch.randn(M, N, dtype=torch.bfloat16, device="cuda")

    def run():
        xi = x.clone().requires_grad_(True)
        wi = w.clone().requires_grad_(True)
        bi = b.clone().requires_grad_(True)
        LinearBiasFn.apply(xi, wi, bi).backward(dy)
        return xi.grad.float(), wi.grad.float(), bi.grad.float()

    g1 = run()
    os.environ["SKY_NO_HBLT"] = "1"
    try:
        g2 = run()
    finally:
        del os.environ["SKY_NO_HBLT"]
    for a, c in zip(g1, g2):
        assert torch.allclose(a, c, atol=2e-2, rtol=2e-2), (a - c).abs().max()
