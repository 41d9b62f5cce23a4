import sys, time, torch, os
sys.path.insert(0, "/root/repo")
from skycomputing_amd import ops

def bench(fn, iters=30):
    for _ in range(5): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / iters * 1e6

def bench_bwd(fn, iters=10):
    for _ in range(3): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / iters * 1e6

torch.manual_seed(0)
print("forward-only (us):")
for S in (256, 512, 1024, 2048, 4096):
    B = max(1, 4096 // S)  # constant token count
    qkv = torch.randn(B, S, 3, 16, 64, dtype=torch.bfloat16, device="cuda")
    t_sdpa_d = bench(lambda: ops.attention(qkv, None, 0.0, False))  # default: sdpa
    os.environ["SKY_FLASH_ATTN"] = "1"
    t_flash = bench(lambda: ops.attention(qkv, None, 0.0, False))
    del os.environ["SKY_FLASH_ATTN"]
    os.environ["SKY_NO_SDPA"] = "1"
    t_dec = bench(lambda: ops.attention(qkv, None, 0.0, False))
    del os.environ["SKY_NO_SDPA"]
    print(f"S={S:5d} B={B:2d}: default(sdpa) {t_sdpa_d:7.1f}  skyflash {t_flash:7.1f}  decomposed {t_dec:7.1f}")

print("fwd+bwd (us):")
for S in (512, 2048):
    B = max(1, 4096 // S)
    qkv = torch.randn(B, S, 3, 16, 64, dtype=torch.bfloat16, device="cuda", requires_grad=True)
    def step():
        out = ops.attention(qkv, None, 0.0, True)
        out.backward(torch.ones_like(out))
        qkv.grad = None
    t_def = bench_bwd(step)
    os.environ["SKY_NO_SDPA"] = "1"
    t_dec = bench_bwd(step)
    del os.environ["SKY_NO_SDPA"]
    print(f"S={S:5d} B={B:2d}: default(sdpa) {t_def:8.1f}  decomposed {t_dec:8.1f}")
