"""Measure hipBLASLt (torch.matmul) on the bench's backward GEMM shapes:
dgrad dX[M,K] = dY[M,N] @ W[N,K];  wgrad dW[N,K] = dY[M,N]^T @ X[M,K]."""
import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import time
import torch

M = 16384
SHAPES = [(2304, 768, "c_attn"), (768, 768, "c_proj"), (3072, 768, "c_fc"), (768, 3072, "mlp_proj")]

def t(fn, iters=30):
    for _ in range(5): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / iters

for N, K, name in SHAPES:
    dy = torch.randn(M, N, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
    x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    fl = 2.0 * M * N * K
    td = t(lambda: torch.matmul(dy, w))
    tw = t(lambda: torch.matmul(dy.t(), x))
    # also the layout torch autograd actually uses for nn.Linear wgrad:
    tw2 = t(lambda: torch.matmul(dy.transpose(-2, -1), x))
    print(f"{name:>9} N={N:<5} K={K:<5} dgrad {td*1e6:7.1f}us {fl/td/1e12:6.1f}TF | "
          f"wgrad {tw*1e6:7.1f}us {fl/tw/1e12:6.1f}TF (alt {fl/tw2/1e12:6.1f}TF)")
