"""Tensor-parallel ViT training (reference examples/simple_tp.py).

    torchrun --nproc_per_node=2 --master-addr 127.0.0.1 -m examples.simple_tp
"""

import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch.distributed as dist

from examples.common import build_loaders, build_model, device_type, parse_args
from quintnet_amd import Trainer, get_strategy, init_process_groups, load_config


def main():
    args = parse_args()
    cfg = load_config(args.config)
    world = int(os.environ.get("WORLD_SIZE", 1))
    pg = init_process_groups(device_type(), [world], ["tp"])
    model = build_model(cfg)
    for p in model.parameters():
        dist.broadcast(p.data, src=0)
    pmodel = get_strategy("tp", pg, cfg).apply(model)
    train, val = build_loaders(cfg, args, pg)
    Trainer(pmodel, train, val, cfg, pg).fit()


if __name__ == "__main__":
    main()
