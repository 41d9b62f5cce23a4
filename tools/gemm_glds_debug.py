import os, sys, torch
sys.path.insert(0, "/root/repo")
os.environ["SKY_GEMM_GLDS"] = "1"
from tools.gemm_bench import sky_gemm
for (M, N, K) in [(128, 128, 64), (128, 128, 128), (256, 256, 128), (1024, 1024, 1024), (4096, 1024, 1024)]:
    x = torch.randn(M, K, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(N, K, dtype=torch.bfloat16, device="cuda") * 0.03
    c = sky_gemm(x, w)
    torch.cuda.synchronize()
    err = (c.float() - x.float() @ w.float().t()).abs().max().item()
    print(f"glds NT {M}x{N}x{K}: max_err={err:.4f}", flush=True)
