"""wgrad_tn: refcheck + race screen + perf vs hipBLASLt."""
import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import time
import torch

def check(M, N, K, runs=5):
    from quintnet_amd import _C
    torch.manual_seed(0)
    dy = (torch.randn(M, N, device="cuda") / M**0.25).to(torch.bfloat16)
    x = (torch.randn(M, K, device="cuda") / M**0.25).to(torch.bfloat16)
    ref = dy.float().t() @ x.float()
    outs = [_C.wgrad_tn(dy, x) for _ in range(runs)]
    for i, o in enumerate(outs):
        err = (o.float() - ref).abs().max().item()
        sc = ref.abs().max().item()
        assert err / sc < 2e-2, (M, N, K, i, err, sc)
        if i: assert torch.equal(o, outs[0]), f"NONDETERMINISM {(M,N,K)} run {i}"
    print(f"check {M}x{N}x{K}: ok (rel {err/sc:.1e}, {runs} runs bit-identical)")

def bench(M, N, K, iters=30):
    from quintnet_amd import _C
    dy = torch.randn(M, N, device="cuda", dtype=torch.bfloat16)
    x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    def t(fn):
        for _ in range(5): fn()
        torch.cuda.synchronize(); t0 = time.perf_counter()
        for _ in range(iters): fn()
        torch.cuda.synchronize(); return (time.perf_counter() - t0) / iters
    tc = t(lambda: _C.wgrad_tn(dy, x))
    tl = t(lambda: torch.matmul(dy.t(), x))
    fl = 2.0 * M * N * K
    print(f"{M}x{N:<5}x{K:<5}: custom {tc*1e6:7.1f}us {fl/tc/1e12:6.1f}TF | library {tl*1e6:7.1f}us {fl/tl/1e12:6.1f}TF")

if __name__ == "__main__":
    for s in [(128,128,128),(256,384,128),(4096,768,768),(16384,2304,768),(16384,768,3072)]:
        check(*s)
    for s in [(16384,2304,768),(16384,768,768),(16384,3072,768),(16384,768,3072)]:
        bench(*s)
