"""Context-parallel long-context training demo: the sequence is sharded
across ranks end to end (attention sees the full context through the
gathered-KV kernels), gradients are exact.

    torchrun --nproc_per_node=2 --master-addr 127.0.0.1 -m examples.long_context_cp
    # flavors: --ring (O(T/cp) KV memory) or --zigzag (balanced causal work)
"""

import argparse
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist
from torch.utils.data import DataLoader

from quintnet_amd import GPT2Trainer, init_process_groups
from quintnet_amd.models import GPT2Config, GPT2Stage
from quintnet_amd.parallel import DataParallel, DistributedConfig
from quintnet_amd.utils.data import SyntheticCLM


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--ring", action="store_true",
                    help="ring KV exchange (peak KV memory O(T/cp))")
    ap.add_argument("--zigzag", action="store_true",
                    help="zigzag load-balanced ring (equal causal work/rank)")
    args = ap.parse_args()
    dev_type = "cuda" if torch.cuda.is_available() else "cpu"
    world = int(os.environ.get("WORLD_SIZE", 1))
    pg = init_process_groups(dev_type, [world], ["cp"])

    seq = 4096 if dev_type == "cuda" else 256
    torch.manual_seed(0)
    cfg = GPT2Config(vocab_size=512, n_positions=seq, n_embd=64, n_layer=2,
                     n_head=2, dropout=0.0,
                     cp_ring=args.ring or args.zigzag,
                     cp_zigzag=args.zigzag)
    dtype = torch.bfloat16 if dev_type == "cuda" else torch.float32
    stage = GPT2Stage(cfg, pp_rank=0, pp_size=1, tp_group=None,
                      cp_group=pg.get_group("cp") if world > 1 else None,
                      device=pg.device, dtype=dtype)
    for p in stage.parameters():
        dist.broadcast(p.data, src=0)
    model = (
        DataParallel(stage, DistributedConfig(pg.rank, world, pg.get_group("cp")))
        if world > 1 else stage
    )

    ds = SyntheticCLM(n=8, seq_len=seq, vocab_size=512, seed=1)
    cfg_t = {"num_epochs": 1, "grad_acc_steps": 2, "zero1": False,
             "learning_rate": 1e-3, "task_type": "clm",
             "context_parallel": world > 1, "max_seq_length": seq,
             "model_config": {"n_embd": 64, "n_positions": seq}}
    GPT2Trainer(model, DataLoader(ds, batch_size=2), None, cfg_t, pg).fit()


if __name__ == "__main__":
    main()
