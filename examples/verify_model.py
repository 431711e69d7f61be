"""Reload a checkpoint and run a quick evaluation loop.

Auto-detects the family: HF-format GPT-2 keys (a merge_checkpoints.py
output) -> perplexity on synthetic token sequences; otherwise the ViT
classifier -> accuracy on synthetic MNIST (reference
examples/verify_model.py covered only the latter).
"""

import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import argparse
import math

import torch
from torch.utils.data import DataLoader


def _verify_vit(state, n):
    from quintnet_amd.models import Model
    from quintnet_amd.utils.data import SyntheticMNIST

    model = Model()
    model.load_state_dict(state)
    model.eval()
    dev = "cuda" if torch.cuda.is_available() else "cpu"
    model.to(dev)
    dl = DataLoader(SyntheticMNIST(n=n, seed=3), batch_size=64)
    correct = total = 0
    with torch.no_grad():
        for b in dl:
            pred = model(b["images"].to(dev)).argmax(-1).cpu()
            correct += int((pred == b["labels"]).sum())
            total += len(b["labels"])
    print(f"accuracy: {100.0 * correct / total:.2f}% over {total}")


def _verify_gpt2(state, n):
    from quintnet_amd.checkpoint.distributed_loading import _strip
    from quintnet_amd.models import GPT2Config, GPT2Stage
    from quintnet_amd.ops import causal_lm_loss
    from quintnet_amd.utils.data import SyntheticCLM

    sd = {_strip(k): v for k, v in state.items()}
    vocab, n_embd = sd["wte.weight"].shape
    n_pos = sd["wpe.weight"].shape[0]
    n_layer = max(int(k.split(".")[1]) for k in sd if k.startswith("h.")) + 1
    n_head = max(2, n_embd // 64)
    cfg = GPT2Config(vocab_size=vocab, n_positions=n_pos, n_embd=n_embd,
                     n_layer=n_layer, n_head=n_head, dropout=0.0)
    stage = GPT2Stage(cfg, pp_rank=0, pp_size=1, tp_group=None)
    tgt = {"embedding.wte.weight": sd["wte.weight"],
           "embedding.wpe.weight": sd["wpe.weight"],
           "ln_f.weight": sd["ln_f.weight"], "ln_f.bias": sd["ln_f.bias"]}
    for i in range(n_layer):
        for k in ("ln_1.weight", "ln_1.bias", "ln_2.weight", "ln_2.bias"):
            tgt[f"blocks.{i}.{k}"] = sd[f"h.{i}.{k}"]
        # HF Conv1D [in, out] -> Linear [out, in]
        for a, b in (("attn.c_attn", "attn.c_attn"), ("attn.c_proj", "attn.c_proj"),
                     ("mlp.c_fc", "mlp.c_fc"), ("mlp.c_proj", "mlp.c_proj")):
            tgt[f"blocks.{i}.{a}.weight"] = sd[f"h.{i}.{b}.weight"].t().contiguous()
            tgt[f"blocks.{i}.{a}.bias"] = sd[f"h.{i}.{b}.bias"]
    stage.load_state_dict(tgt, strict=False)
    stage.eval()
    dev = "cuda" if torch.cuda.is_available() else "cpu"
    stage.to(dev)
    seq = min(n_pos, 64)
    dl = DataLoader(SyntheticCLM(n=min(n, 64), seq_len=seq, vocab_size=vocab,
                                 seed=3), batch_size=4)
    tot, steps = 0.0, 0
    with torch.no_grad():
        for b in dl:
            loss = causal_lm_loss(stage(b["input_ids"].to(dev)),
                                  b["labels"].to(dev), ignore_index=-100)
            tot += float(loss)
            steps += 1
    avg = tot / max(steps, 1)
    print(f"val loss: {avg:.4f}  ppl: {math.exp(min(avg, 20.0)):.2f} over {steps * 4} seqs")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--checkpoint", required=True)
    ap.add_argument("--n", type=int, default=512)
    args = ap.parse_args()
    ckpt = torch.load(args.checkpoint, map_location="cpu", weights_only=False)
    state = ckpt.get("model_state_dict", ckpt)
    if any(k.startswith("transformer.") or k.startswith("wte.") for k in state):
        _verify_gpt2(state, args.n)
    else:
        _verify_vit(state, args.n)


if __name__ == "__main__":
    main()
