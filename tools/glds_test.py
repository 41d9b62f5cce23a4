import sys, torch
sys.path.insert(0, "/root/repo")
from skycomputing_amd.ops import hiplib
from skycomputing_amd.ops.hiplib import ptr
lib = hiplib.require()
src = torch.arange(2048, dtype=torch.float32).to(torch.bfloat16).cuda()
stream = torch.cuda.current_stream().cuda_stream
for mode in (0, 1):
    dst = torch.zeros_like(src)
    rc = lib.sky_glds_probe(stream, ptr(src), ptr(dst), mode)
    torch.cuda.synchronize()
    ok = torch.equal(dst, src)
    print(f"glds mode {mode}: rc={rc} roundtrip_ok={ok}", flush=True)
    if not ok:
        bad = (dst != src).nonzero().flatten()
        print("   mismatches:", bad.numel(), "first:", bad[:6].tolist())
