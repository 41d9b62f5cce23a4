"""Eager op reference implementations: shape/gradient sanity (CPU).

The GPU counterpart (tests/test_ops_gpu.py) compares the HIP kernels
against these in fp32.
"""

from __future__ import annotations

import torch

from skycomputing_amd import ops
from skycomputing_amd.ops import eager


def test_gelu_matches_torch():
    x = torch.randn(100)
    assert torch.allclose(eager.gelu(x), torch.nn.functional.gelu(x), atol=1e-6)


def test_layer_norm_matches_torch():
    x = torch.randn(8, 32)
    w, b = torch.rand(32) + 0.5, torch.randn(32)
    y = eager.layer_norm(x, w, b, 1e-12)
    ref = torch.nn.functional.layer_norm(x, (32,), w, b, 1e-12)
    assert torch.allclose(y, ref, atol=1e-6)


def test_layer_norm_residual_fusion():
    x, r = torch.randn(4, 16), torch.randn(4, 16)
    w, b = torch.ones(16), torch.zeros(16)
    y = eager.layer_norm(x, w, b, 1e-12, residual=r)
    ref = torch.nn.functional.layer_norm(x + r, (16,), w, b, 1e-12)
    assert torch.allclose(y, ref, atol=1e-6)


def test_masked_softmax():
    scores = torch.randn(2, 2, 4, 4)
    mask = torch.zeros(2, 1, 1, 4)
    mask[0, 0, 0, 2:] = -10000.0
    p = eager.masked_softmax(scores, mask)
    assert torch.allclose(p.sum(-1), torch.ones(2, 2, 4), atol=1e-6)
    assert p[0, :, :, 2:].max() < 1e-3


def test_attention_context_matches_sdpa():
    q, k, v = (torch.randn(2, 4, 8, 16) for _ in range(3))
    out = eager.attention_context(q, k, v, None)
    ref = torch.nn.functional.scaled_dot_product_attention(q, k, v)
    assert torch.allclose(out, ref, atol=1e-5)


def test_linear_act_gelu():
    x = torch.randn(4, 8)
    w = torch.randn(16, 8)
    b = torch.randn(16)
    y = eager.linear_act(x, w, b, "gelu")
    ref = eager.gelu(torch.nn.functional.linear(x, w, b))
    assert torch.allclose(y, ref, atol=1e-6)


def test_embedding_fused():
    V, H, S = 50, 16, 8
    we, pe, te = torch.randn(V, H), torch.randn(32, H), torch.randn(2, H)
    ids = torch.randint(0, V, (2, S))
    tids = torch.zeros(2, S, dtype=torch.long)
    pids = torch.arange(S).unsqueeze(0).expand(2, S)
    w, b = torch.ones(H), torch.zeros(H)
    y = eager.embedding_fused(ids, tids, pids, we, pe, te, w, b)
    ref = torch.nn.functional.layer_norm(
        we[ids] + pe[pids] + te[tids], (H,), w, b, 1e-12
    )
    assert torch.allclose(y, ref, atol=1e-5)


def test_sgd_step_with_master_weights():
    p = torch.randn(10, dtype=torch.bfloat16)
    master = p.detach().float()
    g = torch.randn(10, dtype=torch.bfloat16)
    p0 = master.clone()
    eager.sgd_step([p], [g], lr=0.1, master_params=[master])
    assert torch.allclose(master, p0 - 0.1 * g.float())
    assert torch.allclose(p.float(), master, atol=1e-2)


def test_dispatch_uses_eager_on_cpu():
    x = torch.randn(4, 8, requires_grad=True)
    w, b = torch.ones(8, requires_grad=True), torch.zeros(8, requires_grad=True)
    y = ops.layer_norm(x, w, b)
    y.sum().backward()
    assert x.grad is not None and w.grad is not None
