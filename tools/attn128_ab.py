#!/usr/bin/env python3
"""Fused S=128 attention vs torch sdpa at the exact bench shape."""
import sys, time
import torch
sys.path.insert(0, "/root/repo")
from skycomputing_amd.ops.functions import FusedAttentionFn

def bench(fn, iters=30):
    for _ in range(5): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / iters * 1e6

torch.manual_seed(0)
B, S, h, d = 32, 128, 16, 64
qkv = torch.randn(B, S, 3, h, d, dtype=torch.bfloat16, device="cuda", requires_grad=True)
mask = torch.zeros(B, 1, 1, S, dtype=torch.bfloat16, device="cuda")
scale = 0.125

def ours_fwd():
    return FusedAttentionFn.apply(qkv, mask, scale, 0.1, True)

def ours_step():
    out = FusedAttentionFn.apply(qkv, mask, scale, 0.1, True)
    out.backward(torch.ones_like(out))
    qkv.grad = None

def sdpa_fwd():
    q = qkv[:, :, 0].permute(0, 2, 1, 3)
    k = qkv[:, :, 1].permute(0, 2, 1, 3)
    v = qkv[:, :, 2].permute(0, 2, 1, 3)
    return torch.nn.functional.scaled_dot_product_attention(q, k, v, attn_mask=mask, dropout_p=0.1)

def sdpa_step():
    out = sdpa_fwd().permute(0, 2, 1, 3).reshape(B, S, h * d)
    out.backward(torch.ones_like(out))
    qkv.grad = None

print(f"ours fwd      {bench(ours_fwd):7.1f} us")
print(f"sdpa fwd      {bench(sdpa_fwd):7.1f} us")
print(f"ours fwd+bwd  {bench(ours_step, 15):7.1f} us")
print(f"sdpa fwd+bwd  {bench(sdpa_step, 15):7.1f} us")
