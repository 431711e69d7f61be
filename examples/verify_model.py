"""Reload a ViT checkpoint and run a plain accuracy loop
(reference examples/verify_model.py)."""

import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import argparse

import torch
from torch.utils.data import DataLoader

from quintnet_amd.models import Model
from quintnet_amd.utils.data import SyntheticMNIST


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--checkpoint", required=True)
    ap.add_argument("--n", type=int, default=512)
    args = ap.parse_args()
    ckpt = torch.load(args.checkpoint, map_location="cpu", weights_only=False)
    state = ckpt.get("model_state_dict", ckpt)
    model = Model()
    model.load_state_dict(state)
    model.eval()
    dev = "cuda" if torch.cuda.is_available() else "cpu"
    model.to(dev)
    dl = DataLoader(SyntheticMNIST(n=args.n, seed=3), batch_size=64)
    correct = total = 0
    with torch.no_grad():
        for b in dl:
            pred = model(b["images"].to(dev)).argmax(-1).cpu()
            correct += int((pred == b["labels"]).sum())
            total += len(b["labels"])
    print(f"accuracy: {100.0 * correct / total:.2f}% over {total}")


if __name__ == "__main__":
    main()
