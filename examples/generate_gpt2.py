"""KV-cached GPT-2 generation demo (serving path).

    python -m examples.generate_gpt2 [--checkpoint merged.pt] [--prompt-len 8]

Without a checkpoint this runs a random-init model on synthetic token
prompts — enough to exercise the cache path end to end; with one (a
merge_checkpoints.py output) it loads real weights first.
"""

import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import argparse

import torch

from quintnet_amd.models import GPT2Config, GPT2Stage


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--checkpoint", default=None, help="merged checkpoint (.pt)")
    ap.add_argument("--prompt-len", type=int, default=8)
    ap.add_argument("--max-new", type=int, default=32)
    ap.add_argument("--temperature", type=float, default=0.8)
    ap.add_argument("--top-k", type=int, default=40)
    ap.add_argument("--top-p", type=float, default=0.0)
    ap.add_argument("--num-beams", type=int, default=0,
                    help=">1: beam search (overrides sampling)")
    ap.add_argument("--tiny", action="store_true", help="tiny random model (CPU demo)")
    args = ap.parse_args()

    dev = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    dtype = torch.bfloat16 if dev.type == "cuda" else torch.float32
    if args.tiny or dev.type == "cpu":
        cfg = GPT2Config(vocab_size=512, n_positions=128, n_embd=64, n_layer=2,
                         n_head=2, dropout=0.0)
    else:
        cfg = GPT2Config(dropout=0.0)
    stage = GPT2Stage(cfg, pp_rank=0, pp_size=1, tp_group=None, device=dev, dtype=dtype)
    if args.checkpoint:
        sd = torch.load(args.checkpoint, map_location="cpu", weights_only=False)
        sd = sd.get("model_state_dict", sd)
        stage.load_state_dict(sd, strict=False)
    stage.eval()

    ids = torch.randint(0, cfg.vocab_size, (1, args.prompt_len), device=dev)
    if args.num_beams > 1:
        from quintnet_amd.models import beam_search

        out = beam_search(stage, ids, max_new_tokens=args.max_new,
                          num_beams=args.num_beams)
    else:
        out = stage.generate(ids, max_new_tokens=args.max_new,
                             temperature=args.temperature, top_k=args.top_k,
                             top_p=args.top_p)
    print("prompt :", ids[0].tolist())
    print("output :", out[0, args.prompt_len:].tolist())


if __name__ == "__main__":
    main()
