"""Single-GPU ViT baseline (reference examples/train_on_single_gpu.py)."""

import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from examples.common import build_loaders, build_model, parse_args
from quintnet_amd import Trainer, load_config


def main():
    args = parse_args()
    cfg = load_config(args.config)
    model = build_model(cfg)
    if torch.cuda.is_available():
        model = model.to("cuda")
    train, val = build_loaders(cfg, args)
    Trainer(model, train, val, cfg, None).fit()


if __name__ == "__main__":
    main()
