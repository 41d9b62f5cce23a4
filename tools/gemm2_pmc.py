#!/usr/bin/env python3
"""Minimal v2-GEMM kernel exerciser for rocprofv3 PMC runs."""

import sys

import torch

sys.path.insert(0, "/root/repo")
sys.path.insert(0, "/root/repo/tools")
from gemm_bench import sky_gemm2  # noqa: E402

torch.manual_seed(0)
for M, N, K, gsu in [(4096, 4096, 4096, 1), (4096, 4096, 1024, 1),
                     (4096, 1024, 4096, 4)]:
    x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda") * 0.03
    for _ in range(5):
        sky_gemm2(x, w, gsu=gsu)
torch.cuda.synchronize()
print("done")
