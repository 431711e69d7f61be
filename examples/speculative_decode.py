"""Speculative decoding demo: a small draft accelerates a larger target
with token-identical greedy output.

    python -m examples.speculative_decode            # CPU demo sizes
    python -m examples.speculative_decode --cuda     # bf16 on GPU

Random-init models (no network for weights): the demo verifies greedy
EXACTNESS and prints the accept-rate; real draft/target pairs
(fine-tuned small model) accept far more and hence run faster — see
tools/bench_speculative.py for the timing harness.
"""

import argparse

import torch

from quintnet_amd.models import GPT2Config, GPT2Stage, speculative_generate


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--cuda", action="store_true")
    ap.add_argument("--tokens", type=int, default=32)
    ap.add_argument("--draft-k", type=int, default=4)
    ap.add_argument("--temperature", type=float, default=0.0)
    args = ap.parse_args()
    dev = "cuda" if args.cuda and torch.cuda.is_available() else "cpu"
    dt = torch.bfloat16 if dev == "cuda" else torch.float32

    torch.manual_seed(0)
    target = GPT2Stage(
        GPT2Config(n_embd=256, n_layer=8, n_head=8, vocab_size=1024,
                   n_positions=256, dropout=0.0), device=dev, dtype=dt,
    ).eval()
    draft = GPT2Stage(
        GPT2Config(n_embd=64, n_layer=2, n_head=2, vocab_size=1024,
                   n_positions=256, dropout=0.0), device=dev, dtype=dt,
    ).eval()

    ids = torch.randint(0, 1024, (1, 16), device=dev)
    out = speculative_generate(
        target, draft, ids, max_new_tokens=args.tokens,
        draft_k=args.draft_k, temperature=args.temperature, seed=1,
    )
    print("speculative:", out[0, 16:].tolist())
    if args.temperature == 0.0:
        want = target.generate(ids, max_new_tokens=args.tokens, temperature=0.0)
        print("greedy     :", want[0, 16:].tolist())
        print("exact match:", bool(torch.equal(out, want)))


if __name__ == "__main__":
    main()
