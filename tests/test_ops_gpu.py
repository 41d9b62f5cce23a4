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
