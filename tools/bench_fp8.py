"""fp8 (OCP e4m3) forward-path evidence: where it wins and where it loses.

GPU: python tools/bench_fp8.py
Times the fused fp8 linear (cached weight cast + delayed activation
scaling, ops/linear.py) vs bf16 on GPT-2-family hidden sizes.  Expected
(NOTES r1): net-negative at n_embd=768, wins at n_embd>=2048 where the
activation cast pass is small next to the GEMM.
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from quintnet_amd.ops import linear as fused_linear


def bench(fn, iters=20):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    dev = torch.device("cuda")
    print(f"{'model-class':>12} {'M':>6} {'N':>6} {'K':>6}  bf16(us)  fp8(us)  speedup")
    for name, n_embd in [("gpt2-124M", 768), ("gpt2-medium", 1024),
                         ("gpt2-xl", 1600), ("2048-class", 2048),
                         ("4096-class", 4096)]:
        M = 8192
        for (n, k) in [(4 * n_embd, n_embd), (n_embd, 4 * n_embd)]:
            x = torch.randn(M, k, device=dev, dtype=torch.bfloat16)
            w = torch.randn(n, k, device=dev, dtype=torch.bfloat16) * 0.02
            b = torch.randn(n, device=dev, dtype=torch.bfloat16)
            t_bf = bench(lambda: fused_linear(x, w, b))
            t_f8 = bench(lambda: fused_linear(x, w, b, fp8=True))
            print(f"{name:>12} {M:>6} {n:>6} {k:>6}  {t_bf*1e6:8.1f} "
                  f"{t_f8*1e6:8.1f}  {t_bf/t_f8:6.2f}x", flush=True)


if __name__ == "__main__":
    main()
