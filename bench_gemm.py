"""GEMM microbench: custom NT kernel variants vs hipBLASLt (TunableOp-tuned)
on the real GPT-2 bench shapes (micro 16 x seq 1024 -> M=16384).

Run on GPU:  python bench_gemm.py [--iters 20] [--quick]

Modes (csrc/gemm.hip gemm_nt_launch):
  0 auto   1 tile 128x128   2 tile 256x128   3 tile 128x256   4 8-phase 256^2
"""
import argparse
import os
import time

os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
_CSV = os.path.join(os.path.dirname(os.path.abspath(__file__)), "profiles",
                    "tunableop_gfx950_0.csv")
os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING",
                      "0" if os.path.exists(_CSV) else "1")
os.environ.setdefault("PYTORCH_TUNABLEOP_FILENAME", _CSV)

import torch

SHAPES = [  # (name, M, N, K, act)
    ("c_attn", 16384, 2304, 768, 0),
    ("c_proj", 16384, 768, 768, 0),
    ("c_fc+gelu", 16384, 3072, 768, 1),
    ("mlp_proj", 16384, 768, 3072, 0),
]


def bench(fn, iters):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=20)
    ap.add_argument("--quick", action="store_true")
    ap.add_argument("--modes", type=str, default="1,4,5")
    ap.add_argument("--shapes", type=str, default=None,
                    help="semicolon list M,N,K[,act] overriding the default set")
    args = ap.parse_args()
    modes = [int(m) for m in args.modes.split(",")]
    global SHAPES
    if args.shapes:
        SHAPES = []
        for s in args.shapes.split(";"):
            parts = [int(x) for x in s.split(",")]
            m, n, k = parts[:3]
            act = parts[3] if len(parts) > 3 else 0
            SHAPES.append((f"M{m}N{n}K{k}", m, n, k, act))

    from quintnet_amd import _C

    dev = torch.device("cuda")
    results = {}
    for name, M, N, K, act in SHAPES:
        a = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
        b = torch.randn(N, K, device=dev, dtype=torch.bfloat16) * 0.02
        bias = torch.randn(N, device=dev, dtype=torch.bfloat16)
        fl = 2.0 * M * N * K

        def lib():
            out = torch.nn.functional.linear(a, b, bias)
            if act == 1:
                out = torch.nn.functional.gelu(out, approximate="tanh")
            return out

        ref = lib()
        t_lib = bench(lib, args.iters)
        row = {"library": fl / t_lib / 1e12}

        for mode in modes:
            if mode in (4, 5) and not (
                M % 256 == 0 and N % 256 == 0 and K % 128 == 0
            ):
                continue
            if mode == 2 and M % 256:
                continue
            if mode == 3 and N % 256:
                continue
            try:
                res = _C.gemm_nt(a, b, bias, act, mode)
                out = res[0]
                err = (out.float() - ref.float()).abs()
                rel = (err / (ref.float().abs() + 1e-3)).max().item()
                ok = rel < 0.05
                t = bench(lambda: _C.gemm_nt(a, b, bias, act, mode), args.iters)
                row[f"mode{mode}"] = fl / t / 1e12 if ok else float("nan")
                if not ok:
                    print(f"  !! mode{mode} WRONG on {name}: relerr {rel:.3f}")
            except Exception as e:  # noqa: BLE001
                print(f"  !! mode{mode} failed on {name}: {e}")
        results[name] = row
        cols = "  ".join(f"{k}={v:7.1f}" for k, v in row.items())
        print(f"{name:>10} M{M} N{N} K{K}: {cols} TF", flush=True)

    best = {n: max(((v, k) for k, v in r.items() if k != "library"), default=(0, ""))
            for n, r in results.items()}
    for n, (v, k) in best.items():
        lib_v = results[n]["library"]
        print(f"{n:>10}: best custom {k} {v:.1f} TF vs library {lib_v:.1f} "
              f"({'WIN' if v >= lib_v else 'lose'} {v / lib_v:.2f}x)")


if __name__ == "__main__":
    main()
