"""8-phase 256^2 GEMM: refcheck + race screen + perf vs torch/old kernel.

Guide discipline for a NEW sync structure: multi-run race-screen at
256/512/4096 + within-probe A/B vs the unmodified template."""
import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import time
import torch

def check(M, N, K, act=0, runs=5):
    from quintnet_amd import _C
    torch.manual_seed(0)
    a = (torch.randn(M, K, device="cuda", dtype=torch.float32) / K**0.25).to(torch.bfloat16)
    b = (torch.randn(N, K, device="cuda", dtype=torch.float32) / K**0.25).to(torch.bfloat16)
    bias = torch.randn(N, device="cuda", dtype=torch.bfloat16)
    ref = torch.nn.functional.linear(a.float(), b.float(), bias.float())
    if act == 1:
        ref = torch.nn.functional.gelu(ref, approximate="tanh")
    outs = []
    for _ in range(runs):
        o = _C.gemm_nt(a, b, bias, act)[0]
        outs.append(o)
    for i, o in enumerate(outs):
        err = (o.float() - ref).abs().max().item()
        scale = ref.abs().max().item()
        assert err / scale < 2e-2, (M, N, K, act, i, err, scale)
        if i > 0:
            assert torch.equal(o, outs[0]), f"NONDETERMINISM run {i} {(M,N,K)}"
    print(f"check {M}x{N}x{K} act={act}: ok (rel {err/scale:.1e}, {runs} runs bit-identical)")

def bench(M, N, K, act=0, iters=30):
    from quintnet_amd import _C
    a = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
    bias = torch.randn(N, device="cuda", dtype=torch.bfloat16)
    def run():
        _C.gemm_nt(a, b, bias, act)
    for _ in range(5): run()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): run()
    torch.cuda.synchronize(); dt = (time.perf_counter() - t0) / iters
    tref = torch.nn.functional.linear
    for _ in range(5): tref(a, b, bias)
    torch.cuda.synchronize(); t1 = time.perf_counter()
    for _ in range(iters): tref(a, b, bias)
    torch.cuda.synchronize(); dr = (time.perf_counter() - t1) / iters
    fl = 2.0 * M * N * K
    print(f"{M:>6}x{N:<6}x{K:<5} act{act}: {dt*1e6:7.1f}us {fl/dt/1e12:7.1f} TF   (torch {dr*1e6:7.1f}us {fl/dr/1e12:6.1f} TF)")

if __name__ == "__main__":
    for shape in [(256,256,128),(512,512,256),(512,256,384),(4096,4096,4096)]:
        check(*shape)
    check(512, 512, 256, act=1)
    check(8192, 2304, 768); check(8192, 3072, 768, act=1); check(8192, 768, 3072)
    print("--- perf (QN_GEMM_8P=%s) ---" % os.environ.get("QN_GEMM_8P", "1"))
    for shape in [(8192,2304,768),(8192,768,768),(8192,3072,768),(8192,768,3072),(4096,4096,4096),(8192,8192,8192)]:
        bench(*shape)
    bench(8192, 3072, 768, act=1)
