import sys, time, torch, os
sys.path.insert(0, "/root/repo")
from skycomputing_amd import ops

def bench(fn, iters=30):
    for _ in range(5): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / iters * 1e6

torch.manual_seed(0)
for S in (256, 512, 1024, 2048):
    B = max(1, 4096 // S)  # constant token count
    qkv = torch.randn(B, S, 3, 16, 64, dtype=torch.bfloat16, device="cuda")
    t_flash = bench(lambda: ops.attention(qkv, None, 0.0, False))
    os.environ["SKY_NO_FUSED_ATTN"] = "1"
    t_dec = bench(lambda: ops.attention(qkv, None, 0.0, False))
    del os.environ["SKY_NO_FUSED_ATTN"]
    q = qkv[:, :, 0].permute(0, 2, 1, 3).contiguous()
    k = qkv[:, :, 1].permute(0, 2, 1, 3).contiguous()
    v = qkv[:, :, 2].permute(0, 2, 1, 3).contiguous()
    t_sdpa = bench(lambda: torch.nn.functional.scaled_dot_product_attention(q, k, v))
    print(f"S={S:5d} B={B:2d}: flash {t_flash:7.1f} us  decomposed {t_dec:7.1f} us  sdpa(core only) {t_sdpa:7.1f} us")
