#!/usr/bin/env python3
"""Standalone timing of the fused attention kernels at the bench shape."""

import sys
import time

import torch

sys.path.insert(0, "/root/repo")
from skycomputing_amd.ops import hiplib
from skycomputing_amd.ops.hiplib import check, ptr

lib = hiplib.require()
torch.manual_seed(0)
B, S, h, d = 32, 128, 16, 64
qkv = torch.randn(B, S, 3, h, d, dtype=torch.bfloat16, device="cuda")
mask = torch.zeros(B, 1, 1, S, dtype=torch.bfloat16, device="cuda")
out = torch.empty(B, S, h, d, dtype=torch.bfloat16, device="cuda")
m = torch.empty(B, h, S, dtype=torch.float32, device="cuda")
l = torch.empty_like(m)
P = torch.empty(B, h, S, S, dtype=torch.bfloat16, device="cuda")
stream = torch.cuda.current_stream().cuda_stream
scale = 0.125


def run_fwd(keep=0.9):
    check(lib.sky_attn_fwd(stream, ptr(qkv), ptr(mask), ptr(out), ptr(m),
                           ptr(l), B, S, h, d, scale, keep, 123, 0), "fwd")


def run_probs(keep=0.9):
    check(lib.sky_attn_probs(stream, ptr(qkv), ptr(mask), ptr(m), ptr(l),
                             ptr(P), ptr(P), B, S, h, d, scale, keep, 123, 0),
          "probs")


def bench(fn, iters=50):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


run_fwd()  # populate m/l for probs mode
torch.cuda.synchronize()
print(f"attn_fwd  keep=0.9: {bench(lambda: run_fwd(0.9)):8.1f} us")
print(f"attn_fwd  keep=1.0: {bench(lambda: run_fwd(1.0)):8.1f} us")
print(f"attn_probs keep=.9: {bench(lambda: run_probs(0.9)):8.1f} us")

# reference: torch sdpa on the same problem
q = qkv[:, :, 0].permute(0, 2, 1, 3).contiguous()
k = qkv[:, :, 1].permute(0, 2, 1, 3).contiguous()
v = qkv[:, :, 2].permute(0, 2, 1, 3).contiguous()
print(f"torch sdpa (no drop): "
      f"{bench(lambda: torch.nn.functional.scaled_dot_product_attention(q, k, v)):8.1f} us")

# fused backward timing
dout = torch.randn(B, S, h, d, dtype=torch.bfloat16, device="cuda")
pdT = torch.empty(B, h, S, S, dtype=torch.bfloat16, device="cuda")
dsT = torch.empty_like(pdT)
dqkv = torch.empty_like(qkv)


def run_bwd(keep=0.9):
    check(lib.sky_attn_bwd(stream, ptr(qkv), ptr(dout), ptr(mask), ptr(m),
                           ptr(l), ptr(pdT), ptr(dsT), ptr(dqkv),
                           B, S, h, d, scale, keep, 123, 0), "bwd")


print(f"attn_bwd  keep=0.9: {bench(lambda: run_bwd(0.9)):8.1f} us")
print(f"attn_bwd  keep=1.0: {bench(lambda: run_bwd(1.0)):8.1f} us")

# ---------------- backward A/B: wide (one block per bh) vs split-q ----------------
import os

dout = torch.randn_like(out)
pdT = torch.empty(B, h, S, S, dtype=torch.bfloat16, device="cuda")
dsT = torch.empty_like(pdT)
dqkv = torch.empty_like(qkv)


def run_bwd(keep=0.9):
    check(lib.sky_attn_bwd(stream, ptr(qkv), ptr(dout), ptr(mask), ptr(m),
                           ptr(l), ptr(pdT), ptr(dsT), ptr(dqkv),
                           B, S, h, d, scale, keep, 123, 0), "bwd")


results = {}
for mode in ("wide", "split"):
    if mode == "wide":
        os.environ["SKY_ATTN_BWD1"] = "wide"
    else:
        os.environ.pop("SKY_ATTN_BWD1", None)
    run_bwd()
    torch.cuda.synchronize()
    results[mode] = (dqkv.clone(), dsT.clone(), pdT.clone())
    print(f"attn_bwd [{mode}] keep=0.9: {bench(lambda: run_bwd(0.9)):8.1f} us "
          f"(bwd1+bwd2)")
    print(f"attn_bwd [{mode}] keep=1.0: {bench(lambda: run_bwd(1.0)):8.1f} us")

for name, a, b_ in zip(("dqkv", "dsT", "pdT"),
                       results["wide"], results["split"]):
    md = (a.float() - b_.float()).abs().max().item()
    print(f"wide-vs-split max|d{''}| {name}: {md:.2e}")

# ---------------- forward A/B: wide vs split-q ----------------
fwd_res = {}
for mode in ("wide", "split"):
    if mode == "wide":
        os.environ["SKY_ATTN_FWD"] = "wide"
    else:
        os.environ.pop("SKY_ATTN_FWD", None)
    run_fwd()
    torch.cuda.synchronize()
    fwd_res[mode] = (out.clone(), m.clone(), l.clone())
    print(f"attn_fwd [{mode}] keep=0.9: {bench(lambda: run_fwd(0.9)):8.1f} us")
    print(f"attn_fwd [{mode}] keep=1.0: {bench(lambda: run_fwd(1.0)):8.1f} us")
    print(f"attn_probs [{mode}] k=.9 : {bench(lambda: run_probs(0.9)):8.1f} us")
for name, a, b_ in zip(("out", "m", "l"), fwd_res["wide"], fwd_res["split"]):
    print(f"fwd wide-vs-split max|d| {name}: {(a.float()-b_.float()).abs().max().item():.2e}")
