import sys, os, subprocess
for s in ["auto", "4", "8", "16", "32"]:
    env = dict(os.environ)
    if s != "auto": env["QN_WGRAD_SPLITS"] = s
    print(f"== splits {s}", flush=True)
    subprocess.run([sys.executable, "-c", '''
import sys, os; sys.path.insert(0, "/root/repo")
import torch, time
from quintnet_amd import _C
for (M,N,K) in [(16384,2304,768),(16384,768,768),(16384,3072,768),(16384,768,3072)]:
    dy = torch.randn(M, N, device="cuda", dtype=torch.bfloat16)
    x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    for _ in range(5): _C.wgrad_tn(dy, x)
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(30): _C.wgrad_tn(dy, x)
    torch.cuda.synchronize(); dt = (time.perf_counter()-t0)/30
    print(f"  {N}x{K}: {dt*1e6:7.1f}us {2.0*M*N*K/dt/1e12:6.1f}TF", flush=True)
'''], env=env)
