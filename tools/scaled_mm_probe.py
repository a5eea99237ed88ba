import torch, time
torch.manual_seed(0)
b, n, d = 32768, 8192, 768
g16 = torch.randn(b, n, device="cuda", dtype=torch.bfloat16)
zt16 = torch.randn(n, d, device="cuda", dtype=torch.bfloat16)
sg = torch.tensor(1.0, device="cuda"); st = torch.tensor(1.0, device="cuda")
g8 = (g16.float()).to(torch.float8_e4m3fn)
ztT8 = zt16.t().contiguous().t().float().to(torch.float8_e4m3fn)  # col-major (n,d)
def t(f, it=10):
    for _ in range(3): f()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(it): f()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/it*1000
ms16 = t(lambda: g16 @ zt16)
try:
    out = torch._scaled_mm(g8, ztT8, scale_a=sg, scale_b=st, out_dtype=torch.bfloat16)
    ms8 = t(lambda: torch._scaled_mm(g8, ztT8, scale_a=sg, scale_b=st, out_dtype=torch.bfloat16))
    ref = (g8.float() @ ztT8.float())
    err = (out.float()-ref).abs().max().item()
    print(f"bf16 matmul {ms16:.3f} ms   scaled_mm fp8 {ms8:.3f} ms  speedup {ms16/ms8:.2f}x  maxerr {err:.4f}")
except Exception as e:
    print("scaled_mm failed:", type(e).__name__, str(e)[:300])
    print(f"bf16 matmul {ms16:.3f} ms")
