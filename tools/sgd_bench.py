#!/usr/bin/env python3
"""A/B the fused SGD kernel variants on a ~2B-param-scale problem."""
import os, sys, time
import torch
sys.path.insert(0, "/root/repo")

def run(tag):
    # re-import fresh so env takes effect in multi_tensor
    import importlib
    import skycomputing_amd.ops.multi_tensor as mt
    importlib.reload(mt)
    from skycomputing_amd.optim import FusedSGD
    torch.manual_seed(0)
    params = [torch.randn(128 << 20, dtype=torch.bfloat16, device="cuda", requires_grad=True)
              for _ in range(8)]  # 1B params, 2 GB bf16
    for p in params:
        p.grad = torch.randn_like(p)
    opt = FusedSGD(params, lr=1e-3)
    for _ in range(3):
        opt.step()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(10):
        opt.step()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / 10
    n = sum(p.numel() for p in params)
    gb = n * (2 + 4 + 4 + 4 + 2) / 1e9  # g + m(r+w) + p(w)... per elem: g2 + m4r + m4w + p2w = 12 B
    print(f"{tag}: {dt*1e3:.2f} ms  ({n*12/dt/1e12:.2f} TB/s effective)")
    del params, opt
    torch.cuda.empty_cache()

for unroll in (2, 4, 8):
    for slab in (1 << 16, 1 << 18, 1 << 20):
        os.environ["SKY_SGD_UNROLL"] = str(unroll)
        os.environ["SKY_SGD_SLAB"] = str(slab)
        run(f"unroll={unroll} slab={slab>>10}k")
