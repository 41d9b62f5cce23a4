#!/usr/bin/env python3
"""Exerciser for PMC runs: attention bwd, fused SGD, LN fwd at bench shapes."""
import sys
import torch
sys.path.insert(0, "/root/repo")
from skycomputing_amd.ops.functions import FusedAttentionFn, LayerNormFn
from skycomputing_amd.optim import FusedSGD

torch.manual_seed(0)
# attention fwd+bwd (bench shape)
qkv = torch.randn(32, 128, 3, 16, 64, dtype=torch.bfloat16, device="cuda", requires_grad=True)
mask = torch.zeros(32, 1, 1, 128, dtype=torch.bfloat16, device="cuda")
for _ in range(8):
    out = FusedAttentionFn.apply(qkv, mask, 0.125, 0.1, True)
    out.backward(torch.ones_like(out))
    qkv.grad = None
# LN fwd+bwd (bench shape, residual+dropout like the layer)
x = torch.randn(4096, 1024, dtype=torch.bfloat16, device="cuda", requires_grad=True)
res = torch.randn_like(x).requires_grad_(True)
w = torch.randn(1024, dtype=torch.bfloat16, device="cuda", requires_grad=True)
b = torch.randn(1024, dtype=torch.bfloat16, device="cuda", requires_grad=True)
for _ in range(8):
    y = LayerNormFn.apply(x, w, b, 1e-12, res, 0.1)
    y.backward(torch.ones_like(y))
    x.grad = res.grad = None
# fused SGD, ~0.5 GB of bf16 params with masters
params = [torch.randn(64 << 20, dtype=torch.bfloat16, device="cuda", requires_grad=True)
          for _ in range(4)]
for p in params:
    p.grad = torch.randn_like(p)
opt = FusedSGD(params, lr=1e-3)
for _ in range(5):
    opt.step()
torch.cuda.synchronize()
print("done")
