"""MoE GPT-2 with expert parallelism (EP) demo.

    torchrun --nproc_per_node=2 --master-addr 127.0.0.1 -m examples.moe_train
"""

import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist
from torch.utils.data import DataLoader

from quintnet_amd import GPT2Trainer, init_process_groups
from quintnet_amd.models import GPT2Config, GPT2Stage
from quintnet_amd.utils.data import SyntheticCLM


def main():
    dev_type = "cuda" if torch.cuda.is_available() else "cpu"
    world = int(os.environ.get("WORLD_SIZE", 1))
    pg = init_process_groups(dev_type, [world], ["tp"])  # axis reused as EP
    ep_group = pg.get_group("tp") if world > 1 else None

    torch.manual_seed(0)
    cfg = GPT2Config(vocab_size=512, n_positions=64, n_embd=64, n_layer=2,
                     n_head=2, dropout=0.0, n_experts=4, moe_top_k=2)
    dtype = torch.bfloat16 if dev_type == "cuda" else torch.float32
    stage = GPT2Stage(cfg, pp_rank=0, pp_size=1, tp_group=None,
                      ep_group=ep_group, device=pg.device, dtype=dtype)
    for p in stage.parameters():
        dist.broadcast(p.data, src=0)

    ds = SyntheticCLM(n=32, seq_len=64, vocab_size=512, seed=pg.rank)
    cfg_t = {"num_epochs": 2, "grad_acc_steps": 2, "zero1": False,
             "learning_rate": 1e-3, "task_type": "clm",
             "moe_aux_weight": 0.01,
             "model_config": {"n_embd": 64, "n_positions": 64}}
    GPT2Trainer(stage, DataLoader(ds, batch_size=4), None, cfg_t, pg).fit()


if __name__ == "__main__":
    main()
