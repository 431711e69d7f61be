"""Pre-tune hipBLASLt (TunableOp) for every GEMM shape the bench meshes
use — including the TP-halved shapes of the [2,2,2] run — and merge the
results into profiles/tunableop_gfx950_0.csv so multi-rank warmups load
instead of re-tuning.  Run on GPU."""
import os
import sys

HERE = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, HERE)
CSV = os.path.join(HERE, "profiles", "tunableop_gfx950_0.csv")
os.environ["PYTORCH_TUNABLEOP_ENABLED"] = "1"
os.environ["PYTORCH_TUNABLEOP_TUNING"] = "1"
os.environ["PYTORCH_TUNABLEOP_FILENAME"] = CSV

import torch


def main():
    # bound per-solution tuning cost; the full sweep must fit a gpurun call
    torch.cuda.tunable.set_max_tuning_duration(10)
    torch.cuda.tunable.set_max_tuning_iterations(30)
    if os.path.exists(CSV):
        torch.cuda.tunable.read_file(CSV)
    dev = torch.device("cuda")
    shapes = []
    # NEW (likely-untuned) shapes first so a short timeout still banks
    # them: the padded-vocab lm-head GEMMs at every mesh M
    for M in (8192, 16384, 32768):
        shapes += [(M, 50304, 768)]
    for M in (32768, 16384, 8192):          # micro 16 (pp1) and micro 8 (pp2)
        for tp in (1, 2):            # full and TP-halved
            E, I, V = 768, 3072, 50257
            shapes += [
                (M, 3 * E // tp, E),         # c_attn fwd
                (M, E, E // tp),             # attn c_proj fwd
                (M, I // tp, E),             # c_fc fwd
                (M, E, I // tp),             # mlp c_proj fwd
            ]
        # lm_head shapes: logical 50257 (vocab_pad_to=0) and padded
        # 50304 (bench default) — dp meshes run M=32768, pp meshes 8192
        shapes += [(M, 50257, 768), (M, 50304, 768)]
    def flush():
        # write INCREMENTALLY so a timeout keeps everything tuned so far
        res = torch.cuda.tunable.get_results()
        with open(CSV, "w") as f:
            for k, v in torch.cuda.tunable.get_validators():
                f.write(f"Validator,{k},{v}\n")
            for r in res:
                f.write(",".join(str(x) for x in r) + "\n")
        return len(res)

    done = set()
    for (M, N, K) in shapes:
        if (M, N, K) in done:
            continue
        done.add((M, N, K))
        a = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
        b = torch.randn(N, K, device=dev, dtype=torch.bfloat16)
        bias = torch.randn(N, device=dev, dtype=torch.bfloat16)
        torch.nn.functional.linear(a, b, bias)   # fwd (bias epilogue)
        torch.nn.functional.linear(a, b)         # fwd (plain)
        g = torch.randn(M, N, device=dev, dtype=torch.bfloat16)
        g @ b                                    # dgrad
        g.t() @ a                                # wgrad
        torch.cuda.synchronize()
        n = flush()
        print(f"tuned {M}x{N}x{K} ({n} entries)", flush=True)
    print(f"wrote {flush()} entries -> {CSV}")


if __name__ == "__main__":
    main()
