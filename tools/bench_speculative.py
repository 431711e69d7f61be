"""Speculative vs plain greedy decode throughput (GPU).

    gpurun -- 'python tools/bench_speculative.py --target base --draft tiny'

Target = GPT-2 preset (random init); draft = a 2-layer/256-wide model
of the same vocab.  Reports ms/token for plain KV-cached greedy,
graph-replayed decode (StaticKVDecoder) and speculative (draft_k 2/4/8)
with the measured acceptance rate.  Random-init models accept less than
trained pairs would, so the speedup here is a LOWER bound."""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--target", default="base")
    ap.add_argument("--tokens", type=int, default=128)
    ap.add_argument("--ctx", type=int, default=64)
    args = ap.parse_args()

    from quintnet_amd.models import GPT2Config, GPT2Stage
    from quintnet_amd.models.gpt2.speculative import speculative_generate

    dev = "cuda" if torch.cuda.is_available() else "cpu"
    dt = torch.bfloat16 if dev == "cuda" else torch.float32
    torch.manual_seed(0)
    tcfg = GPT2Config.from_name(args.target, dropout=0.0)
    target = GPT2Stage(tcfg, device=dev, dtype=dt).eval()
    dcfg = GPT2Config(n_embd=256, n_layer=2, n_head=4, dropout=0.0,
                      vocab_size=tcfg.vocab_size,
                      n_positions=tcfg.n_positions)
    draft = GPT2Stage(dcfg, device=dev, dtype=dt).eval()
    ids = torch.randint(0, tcfg.vocab_size, (1, args.ctx), device=dev)

    def timed(fn, reps=3):
        fn()  # warm
        if dev == "cuda":
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(reps):
            out = fn()
        if dev == "cuda":
            torch.cuda.synchronize()
        dt_ = (time.perf_counter() - t0) / reps
        return dt_, out

    t_plain, out_p = timed(
        lambda: target.generate(ids, max_new_tokens=args.tokens,
                                temperature=0.0))
    n_new = out_p.shape[1] - args.ctx
    print(f"plain greedy      : {1000*t_plain/n_new:7.3f} ms/token", flush=True)

    for k in (2, 4, 8):
        t_spec, out_s = timed(
            lambda: speculative_generate(target, draft, ids,
                                         max_new_tokens=args.tokens,
                                         draft_k=k))
        n_s = out_s.shape[1] - args.ctx
        match = torch.equal(out_s, out_p)
        print(f"speculative k={k}  : {1000*t_spec/n_s:7.3f} ms/token "
              f"({t_plain/t_spec:4.2f}x, exact={match})", flush=True)


if __name__ == "__main__":
    main()
