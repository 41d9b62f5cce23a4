"""Standalone timing of the LayerNorm kernels at the bench shape."""
import sys, time

sys.path.insert(0, "/root/repo")
import torch
from skycomputing_amd.ops import hiplib
from skycomputing_amd.ops.functions import LayerNormFn

hiplib.require()
torch.manual_seed(0)
rows, cols = 4096, 1024
x = torch.randn(rows, cols, dtype=torch.bfloat16, device="cuda", requires_grad=True)
res = torch.randn_like(x).requires_grad_(True)
w = torch.rand(cols, dtype=torch.bfloat16, device="cuda").requires_grad_(True)
b = torch.randn(cols, dtype=torch.bfloat16, device="cuda").requires_grad_(True)


def bench(fn, iters=100):
    for _ in range(10):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


def fwd():
    return LayerNormFn.apply(x, w, b, 1e-12, res, 0.1)


y = fwd()
dy = torch.randn_like(y)


def full():
    out = fwd()
    (g,) = torch.autograd.grad(out, x, dy, retain_graph=False)
    return g


print(f"ln fwd (res+drop)      : {bench(fwd):8.1f} us")
print(f"ln fwd+bwd (res+drop)  : {bench(full):8.1f} us")
