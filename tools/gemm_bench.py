#!/usr/bin/env python3
"""Correctness + A/B timing of sky_gemm vs hipBLASLt (torch.matmul) on the
BERT bench shapes. Run on an MI355X box."""

import sys
import time

import torch

sys.path.insert(0, "/root/repo")
from skycomputing_amd.ops import hiplib
from skycomputing_amd.ops.hiplib import check, ptr


import os
USE_GLDS = int(os.environ.get('SKY_GEMM_GLDS', '1'))


def sky_gemm(a, b, bias=None, transA=0, transB=1, epi=0, z=None):
    lib = hiplib.require()
    if transA == 0:
        M, K = a.shape
    else:
        K, M = a.shape
    if transB == 1:
        N = b.shape[0]
    else:
        N = b.shape[1]
    c = torch.empty(M, N, dtype=a.dtype, device=a.device)
    check(
        lib.sky_gemm(
            torch.cuda.current_stream().cuda_stream, ptr(a), ptr(b), ptr(c),
            ptr(bias), ptr(z), M, N, K, a.stride(0), b.stride(0), c.stride(0),
            transA, transB, epi, 1, USE_GLDS,
        ),
        "sky_gemm",
    )
    return c


_WK = {}


def sky_gemm2(a, b, bias=None, epi=0, z=None, gsu=1, transA=0, transB=0):
    """v2 256^2 8-phase GEMM. transA/transB: 0 = operand stored [out][red],
    1 = stored [red][out] (kmajor; read via ds_read_b64_tr_b16).
    NT fwd: C=a@b.T (tA=0,tB=0); dgrad NN: C=a@b (tA=0,tB=1);
    wgrad TN: C=a.T@b (tA=1,tB=1)."""
    lib = hiplib.require()
    M = a.shape[1] if transA else a.shape[0]
    K = a.shape[0] if transA else a.shape[1]
    N = b.shape[1] if transB else b.shape[0]
    c = torch.empty(M, N, dtype=a.dtype, device=a.device)
    wk = None
    if gsu > 1:
        key = (M, N, gsu)
        wk = _WK.get(key)
        if wk is None:
            wk = torch.empty(gsu * M * N, dtype=torch.float32, device=a.device)
            _WK[key] = wk
    check(
        lib.sky_gemm2(
            torch.cuda.current_stream().cuda_stream, ptr(a), ptr(b), ptr(c),
            ptr(bias), ptr(z), ptr(wk), M, N, K, a.stride(0), b.stride(0),
            c.stride(0), transA, transB, epi, gsu,
        ),
        "sky_gemm2",
    )
    return c


def bench_v2():
    print("\n-- v2 256^2 8-phase (NT fwd) --")
    print(f"{'shape':<18} {'gsu':>3} {'torch us':>9} {'v2 us':>9} {'ratio':>6} {'v2 TF':>8}  max_err")
    shapes = [
        ("qkv 4096x3072", 4096, 3072, 1024, (1, 2)),
        ("proj 4096x1024", 4096, 1024, 1024, (1, 2, 4)),
        ("ffnup 4096x4096", 4096, 4096, 1024, (1, 2)),
        ("ffndn 4096x1024", 4096, 1024, 4096, (1, 2, 4, 8)),
        ("sq 4096x4096x4096", 4096, 4096, 4096, (1,)),
        ("sq 8192x8192x8192", 8192, 8192, 8192, (1,)),
    ]
    for name, M, N, K, gsus in shapes:
        x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
        w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda") * 0.03
        ref = x.float() @ w.float().t()
        tt = bench(lambda: torch.nn.functional.linear(x, w))
        for gsu in gsus:
            c = sky_gemm2(x, w, gsu=gsu)
            err = (c.float() - ref).abs().max().item()
            rel = err / ref.abs().max().item()
            ts = bench(lambda: sky_gemm2(x, w, gsu=gsu))
            tf = 2 * M * N * K / ts / 1e12
            print(f"{name:<18} {gsu:>3} {tt*1e6:9.1f} {ts*1e6:9.1f} {tt/ts:6.2f} {tf:8.0f}  {err:.3f} ({rel:.2e})")
    # transposed orientations (dgrad NN, wgrad TN) via tr16 kmajor reads
    print("  -- transposed (kmajor) --")
    for name, M, N, K, tA, tB, gsus in [
        ("dgradNN 4096x1024", 4096, 1024, 4096, 0, 1, (1, 2, 4)),
        ("dgradNN 4096x4096", 4096, 4096, 1024, 0, 1, (1,)),
        ("wgradTN 4096x1024", 4096, 1024, 4096, 1, 1, (1, 2, 4)),
        ("wgradTN 1024x4096", 1024, 4096, 4096, 1, 1, (1, 2, 4)),
    ]:
        a = (torch.randn(K, M, dtype=torch.bfloat16, device="cuda") * 0.05
             if tA else torch.randn(M, K, dtype=torch.bfloat16, device="cuda") * 0.05)
        b = torch.randn(K, N, dtype=torch.bfloat16, device="cuda") * 0.03
        ref = (a.float().t() if tA else a.float()) @ b.float()
        tt = bench(lambda: (a.t() if tA else a) @ b)
        for gsu in gsus:
            c = sky_gemm2(a, b, gsu=gsu, transA=tA, transB=tB)
            err = (c.float() - ref).abs().max().item()
            rel = err / ref.abs().max().item()
            ts = bench(lambda: sky_gemm2(a, b, gsu=gsu, transA=tA, transB=tB))
            tf = 2 * M * N * K / ts / 1e12
            print(f"{name:<18} {gsu:>3} {tt*1e6:9.1f} {ts*1e6:9.1f} {tt/ts:6.2f} {tf:8.0f}  {err:.3f} ({rel:.2e})")

    # fused bias+gelu epilogue + Z store
    M, N, K = 4096, 4096, 1024
    x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda") * 0.03
    b = torch.randn(N, dtype=torch.bfloat16, device="cuda")
    z = torch.empty(M, N, dtype=torch.bfloat16, device="cuda")
    import math
    zr = x.float() @ w.float().t() + b.float()
    ref = zr * 0.5 * (1 + torch.erf(zr / math.sqrt(2)))
    c = sky_gemm2(x, w, bias=b, epi=2, z=z)
    err = (c.float() - ref).abs().max().item()
    errz = (z.float() - zr).abs().max().item()
    ts = bench(lambda: sky_gemm2(x, w, bias=b, epi=2, z=z))
    print(f"{'ffnup+gelu(+z)':<18} {1:>3} {'':>9} {ts*1e6:9.1f} {'':>6} "
          f"{2*M*N*K/ts/1e12:8.0f}  {err:.3f}/{errz:.3f}")


def bench(fn, iters=30):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    torch.manual_seed(0)
    shapes = [
        ("qkv fwd NT", 4096, 3072, 1024),
        ("proj fwd NT", 4096, 1024, 1024),
        ("ffn-up fwd NT", 4096, 4096, 1024),
        ("ffn-dn fwd NT", 4096, 1024, 4096),
    ]
    print(f"{'shape':<16} {'torch us':>9} {'sky us':>9} {'ratio':>6} {'sky TF':>8}  max_err")
    for name, M, N, K in shapes:
        x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
        w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda") * 0.03
        ref = x.float() @ w.float().t()
        c = sky_gemm(x, w)
        err = (c.float() - ref).abs().max().item()
        tt = bench(lambda: torch.nn.functional.linear(x, w))
        ts = bench(lambda: sky_gemm(x, w))
        tf = 2 * M * N * K / ts / 1e12
        print(f"{name:<16} {tt*1e6:9.1f} {ts*1e6:9.1f} {tt/ts:6.2f} {tf:8.0f}  {err:.3f}")

    # dgrad NN: dx = dy @ w  (w [N,K] stored row-major, transB=0)
    M, N, K = 4096, 4096, 1024  # dy [M,N], w [N,K] -> dx [M,K]
    dy = torch.randn(M, N, dtype=torch.bfloat16, device="cuda") * 0.05
    w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda") * 0.03
    ref = dy.float() @ w.float()
    c = sky_gemm(dy, w, transA=0, transB=0)
    err = (c.float() - ref).abs().max().item()
    tt = bench(lambda: dy @ w)
    ts = bench(lambda: sky_gemm(dy, w, transA=0, transB=0))
    print(f"{'dgrad NN':<16} {tt*1e6:9.1f} {ts*1e6:9.1f} {tt/ts:6.2f} "
          f"{2*M*N*K/ts/1e12:8.0f}  {err:.3f}")

    # wgrad TN: dw = dy^T @ x  (dy [R,M... dy [Mtok,N] transA=1 -> C [N,K])
    R, N, K = 4096, 4096, 1024
    dy = torch.randn(R, N, dtype=torch.bfloat16, device="cuda") * 0.05
    x = torch.randn(R, K, dtype=torch.bfloat16, device="cuda")
    ref = dy.float().t() @ x.float()
    c = sky_gemm(dy, x, transA=1, transB=0)
    err = (c.float() - ref).abs().max().item()
    tt = bench(lambda: dy.t() @ x)
    ts = bench(lambda: sky_gemm(dy, x, transA=1, transB=0))
    print(f"{'wgrad TN':<16} {tt*1e6:9.1f} {ts*1e6:9.1f} {tt/ts:6.2f} "
          f"{2*R*N*K/ts/1e12:8.0f}  {err:.3f}")

    # fused bias+gelu epilogue
    M, N, K = 4096, 4096, 1024
    x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda") * 0.03
    b = torch.randn(N, dtype=torch.bfloat16, device="cuda")
    z = torch.empty(M, N, dtype=torch.bfloat16, device="cuda")
    zr = x.float() @ w.float().t() + b.float()
    import math

    ref = zr * 0.5 * (1 + torch.erf(zr / math.sqrt(2)))
    c = sky_gemm(x, w, bias=b, epi=2, z=z)
    err = (c.float() - ref).abs().max().item()
    errz = (z.float() - zr).abs().max().item()
    ts = bench(lambda: sky_gemm(x, w, bias=b, epi=2, z=z))
    print(f"{'ffn-up+gelu':<16} {'':>9} {ts*1e6:9.1f} {'':>6} "
          f"{2*M*N*K/ts/1e12:8.0f}  {err:.3f}/{errz:.3f}")


if __name__ == "__main__":
    if "--v2-only" in sys.argv:
        torch.manual_seed(0)
        bench_v2()
    else:
        main()
        bench_v2()
