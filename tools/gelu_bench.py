"""Standalone timing of bias_gelu fwd/bwd at the bench shape."""
import sys, time
sys.path.insert(0, "/root/repo")
import torch
from skycomputing_amd.ops import hiplib
from skycomputing_amd.ops.hiplib import check, ptr
from skycomputing_amd.ops.functions import _red_scratch

lib = hiplib.require()
torch.manual_seed(0)
rows, cols = 4096, 4096
x = torch.randn(rows, cols, dtype=torch.bfloat16, device="cuda")
b = torch.randn(cols, dtype=torch.bfloat16, device="cuda")
dy = torch.randn_like(x)
y = torch.empty_like(x)
dx = torch.empty_like(x)
db = torch.empty(cols, dtype=torch.bfloat16, device="cuda")
scratch = _red_scratch(cols, 1, x.device)
stream = torch.cuda.current_stream().cuda_stream


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
    check(lib.sky_bias_gelu_fwd(stream, ptr(x), ptr(b), ptr(y), rows, cols, 1), "f")


def bwd():
    check(lib.sky_bias_gelu_bwd(stream, ptr(dy), ptr(x), ptr(b), ptr(dx),
                                ptr(db), ptr(scratch), rows, cols, 1), "b")


print(f"bias_gelu_fwd : {bench(fwd):8.1f} us")
print(f"bias_gelu_bwd : {bench(bwd):8.1f} us  (fused dx+db partials + final)")
# correctness vs fp32
xf = x.float().requires_grad_(True)
bf = b.float().requires_grad_(True)
yr = torch.nn.functional.gelu(xf + bf)
yr.backward(dy.float())
bwd(); torch.cuda.synchronize()
print("max|dx err|:", (dx.float() - xf.grad).abs().max().item())
print("max|db err|:", (db.float() - bf.grad).abs().max().item() /
      max(1.0, bf.grad.abs().max().item()), "(rel)")
