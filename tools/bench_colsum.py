"""Microbench: colsum (bias-grad) kernel GB/s on bench shapes."""
import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, time
from quintnet_amd.ops._backend import ext

def bench(rows, n, iters=50):
    x = torch.randn(rows, n, device="cuda", dtype=torch.bfloat16)
    lib = ext()
    for _ in range(5):
        lib.colsum(x)
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters):
        lib.colsum(x)
    torch.cuda.synchronize(); dt = (time.perf_counter() - t0) / iters
    gb = rows * n * 2 / 1e9
    print(f"rows={rows} n={n}: {dt*1e6:8.1f} us  {gb/dt:7.1f} GB/s")

def check(rows, n):
    x = torch.randn(rows, n, device="cuda", dtype=torch.bfloat16)
    out = ext().colsum(x)
    ref = x.float().sum(0)
    err = (out - ref).abs().max().item() / max(ref.abs().max().item(), 1e-6)
    assert err < 2e-2, (rows, n, err)
    print(f"check rows={rows} n={n} ok (rel {err:.2e})")

if __name__ == "__main__":
    for n in (768, 2304, 3072, 50264, 50257, 10):
        check(16384, n)
    check(100, 768); check(3, 8)
    for n in (768, 2304, 3072):
        bench(16384, n)
    bench(8192, 768)
