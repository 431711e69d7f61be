"""Probe fp8 (e4m3) GEMM support + speed on gfx950 via torch._scaled_mm."""
import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import time
import torch

print("float8 dtypes:", hasattr(torch, "float8_e4m3fn"), hasattr(torch, "float8_e4m3fnuz"))
M, N, K = 8192, 3072, 768
a = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
b = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
try:
    a8 = a.to(torch.float8_e4m3fn)
    b8 = b.to(torch.float8_e4m3fn).t()
    sa = torch.tensor(1.0, device="cuda")
    out = torch._scaled_mm(a8, b8, scale_a=sa, scale_b=sa, out_dtype=torch.bfloat16)
    ref = a @ b.t()
    err = (out.float() - ref.float()).abs().mean() / ref.float().abs().mean()
    print("scaled_mm works, mean rel err:", float(err))
    for _ in range(5): torch._scaled_mm(a8, b8, scale_a=sa, scale_b=sa, out_dtype=torch.bfloat16)
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(30): torch._scaled_mm(a8, b8, scale_a=sa, scale_b=sa, out_dtype=torch.bfloat16)
    torch.cuda.synchronize(); dt = (time.perf_counter()-t0)/30
    for _ in range(5): a @ b.t()
    torch.cuda.synchronize(); t1 = time.perf_counter()
    for _ in range(30): a @ b.t()
    torch.cuda.synchronize(); db = (time.perf_counter()-t1)/30
    fl = 2.0*M*N*K
    print(f"fp8 {dt*1e6:.1f}us {fl/dt/1e12:.1f}TF | bf16 {db*1e6:.1f}us {fl/db/1e12:.1f}TF")
    # larger shape
    M2,N2,K2 = 16384,8192,8192
    a2 = torch.randn(M2,K2,device="cuda",dtype=torch.bfloat16).to(torch.float8_e4m3fn)
    b2 = torch.randn(N2,K2,device="cuda",dtype=torch.bfloat16).to(torch.float8_e4m3fn).t()
    ab2 = torch.randn(M2,K2,device="cuda",dtype=torch.bfloat16); bb2 = torch.randn(N2,K2,device="cuda",dtype=torch.bfloat16)
    for _ in range(3): torch._scaled_mm(a2, b2, scale_a=sa, scale_b=sa, out_dtype=torch.bfloat16)
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(10): torch._scaled_mm(a2, b2, scale_a=sa, scale_b=sa, out_dtype=torch.bfloat16)
    torch.cuda.synchronize(); dt = (time.perf_counter()-t0)/10
    for _ in range(3): ab2 @ bb2.t()
    torch.cuda.synchronize(); t1 = time.perf_counter()
    for _ in range(10): ab2 @ bb2.t()
    torch.cuda.synchronize(); db = (time.perf_counter()-t1)/10
    fl = 2.0*M2*N2*K2
    print(f"big fp8 {dt*1e6:.1f}us {fl/dt/1e12:.1f}TF | bf16 {db*1e6:.1f}us {fl/db/1e12:.1f}TF")
except Exception as e:
    print("scaled_mm failed:", type(e).__name__, e)
