"""Probe MX (BlockWise1x32) fp8 scaled_mm support + rate on gfx950."""
import sys, os, time
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import torch
import torch.nn.functional as F

def bench(fn, iters=20):
    fn(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters

M, K, N = 16384, 4096, 4096
a16 = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
b16 = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
t = bench(lambda: a16 @ b16.t())
print(f"bf16 matmul: {2*M*K*N/t/1e12:.0f} TF")

a8 = a16.to(torch.float8_e4m3fn)
b8 = b16.to(torch.float8_e4m3fn)
# tensorwise for reference
sa = torch.ones((), device="cuda"); sb = torch.ones((), device="cuda")
try:
    t = bench(lambda: torch._scaled_mm(a8, b8.t(), scale_a=sa, scale_b=sb, out_dtype=torch.bfloat16))
    print(f"fp8 tensorwise _scaled_mm: {2*M*K*N/t/1e12:.0f} TF")
except Exception as e:
    print("tensorwise failed:", e)
# MX: e8m0 scales, 1 per 32 along K
e8m0 = torch.uint8
sa_mx = torch.full((M, K // 32), 127, device="cuda", dtype=e8m0).view(torch.float8_e8m0fnu)
sb_mx = torch.full((N, K // 32), 127, device="cuda", dtype=e8m0).view(torch.float8_e8m0fnu)
ST = F.ScalingType
for swz in [None]:
    try:
        t = bench(lambda: F.scaled_mm(a8, b8.t(), sa_mx, ST.BlockWise1x32, sb_mx, ST.BlockWise1x32,
                                      output_dtype=torch.bfloat16))
        out = F.scaled_mm(a8, b8.t(), sa_mx, ST.BlockWise1x32, sb_mx, ST.BlockWise1x32,
                          output_dtype=torch.bfloat16)
        ref = (a8.float() @ b8.float().t()).bfloat16()
        err = (out.float() - ref.float()).abs().max().item()
        print(f"MX1x32 scaled_mm: {2*M*K*N/t/1e12:.0f} TF  maxerr={err:.4f}")
    except Exception as e:
        print("MX failed:", repr(e)[:300])
