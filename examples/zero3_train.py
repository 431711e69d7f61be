"""ZeRO-3 sharded training demo: every transformer block's params, grads
and activations live at 1/world between uses.

    torchrun --nproc_per_node=2 --master-addr 127.0.0.1 -m examples.zero3_train
"""

import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist

from quintnet_amd import init_process_groups
from quintnet_amd.models import Model
from quintnet_amd.parallel import apply_zero3
from quintnet_amd.utils.data import SyntheticMNIST


def main():
    dev_type = "cuda" if torch.cuda.is_available() else "cpu"
    world = int(os.environ.get("WORLD_SIZE", 1))
    pg = init_process_groups(dev_type, [world], ["dp"])
    torch.manual_seed(0)
    model = Model(hidden_dim=64, n_heads=4, depth=6).to(pg.device)
    for p in model.parameters():
        dist.broadcast(p.data, src=0)
    apply_zero3(model, dp_group=pg.get_group("dp"))
    shard = sum(p.numel() for p in model.parameters() if p.requires_grad)
    print(f"[rank {pg.rank}] trainable numel after sharding: {shard}")

    opt = torch.optim.AdamW(
        [p for p in model.parameters() if p.requires_grad], lr=1e-3
    )
    ds = SyntheticMNIST(n=64, seed=pg.rank)
    dl = torch.utils.data.DataLoader(ds, batch_size=8)
    for step, batch in enumerate(dl):
        x = batch["image" if "image" in batch else "images"]
        y = batch["label" if "label" in batch else "labels"]
        x, y = x.to(pg.device), y.to(pg.device)
        loss = torch.nn.functional.cross_entropy(model(x), y)
        (loss / world).backward()  # ZeRO-3 grads reduce-SUM across dp
        opt.step(); opt.zero_grad()
        if pg.rank == 0 and step % 4 == 0:
            print(f"step {step}: loss {float(loss):.4f}")
        if step >= 8:
            break


if __name__ == "__main__":
    main()
