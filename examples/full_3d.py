"""Full 3D-parallel ViT training on mesh [2,2,2] (reference examples/full_3d.py).

    torchrun --nproc_per_node=8 --master-addr 127.0.0.1 -m examples.full_3d \
        --config examples/config.yaml
"""

import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import time

import torch.distributed as dist

from examples.common import build_loaders, build_model, device_type, parse_args
from quintnet_amd import Trainer, get_strategy, init_process_groups, load_config


def main():
    args = parse_args()
    cfg = load_config(args.config)
    pg = init_process_groups(
        device_type(), cfg.get("mesh_dim", [2, 2, 2]), cfg.get("mesh_name", ["dp", "tp", "pp"])
    )
    pg.print_mesh_info()
    model = build_model(cfg)
    for p in model.parameters():
        dist.broadcast(p.data, src=0)
    strategy = get_strategy(cfg.get("strategy_name", "3d"), pg, cfg)
    pmodel = strategy.apply(model)
    train, val = build_loaders(cfg, args, pg)
    t0 = time.time()
    Trainer(pmodel, train, val, cfg, pg).fit()
    if pg.rank == 0:
        print(f"total training time: {time.time() - t0:.2f}s")


if __name__ == "__main__":
    main()
