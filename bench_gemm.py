"""GEMM kernel microbench: custom NT (glds/reg modes) vs torch.matmul
on the real GPT-2 bench shapes.  Run on GPU: python bench_gemm.py"""
import os
import time

import torch

SHAPES = [  # (M, N, K) of the forward linears + lm_head
    (8192, 2304, 768),   # c_attn
    (8192, 768, 768),    # attn c_proj
    (8192, 3072, 768),   # c_fc
    (8192, 768, 3072),   # mlp c_proj
    (8192, 50257, 768),  # lm_head
]


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
    from quintnet_amd import _C

    dev = torch.device("cuda")
    print(f"{'shape':>22} {'torch':>9} {'custom':>9}  TF(custom)")
    for M, N, K in SHAPES:
        a = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
        b = torch.randn(N, K, device=dev, dtype=torch.bfloat16)
        bias = torch.randn(N, device=dev, dtype=torch.bfloat16)
        t_ref = bench(lambda: torch.nn.functional.linear(a, b, bias))
        t_cus = bench(lambda: _C.gemm_nt(a, b, bias, 0))
        fl = 2.0 * M * N * K
        print(f"{(M,N,K)!s:>22} {t_ref*1e6:8.1f}u {t_cus*1e6:8.1f}u  {fl/t_cus/1e12:7.1f} (torch {fl/t_ref/1e12:.1f})")


if __name__ == "__main__":
    main()
