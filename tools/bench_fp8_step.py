"""Measure the GPT-2 training step with fp8 forward GEMMs (experimental,
not the headline bench — that stays pure bf16)."""
import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import time
import torch

from quintnet_amd.models import GPT2Config, GPT2Stage
from quintnet_amd.ops import causal_lm_loss
from quintnet_amd.optim import ZeroRedundancyAdamW

def run(fp8):
    torch.manual_seed(0)
    cfg = GPT2Config(n_embd=768, n_layer=12, n_head=12, vocab_size=50257,
                     dropout=0.0, fp8=fp8)
    stage = GPT2Stage(cfg, pp_rank=0, pp_size=1, tp_group=None,
                      device=torch.device("cuda"), dtype=torch.bfloat16)
    opt = ZeroRedundancyAdamW(stage.parameters(), lr=1e-4, dp_group=None)
    ids = torch.randint(0, 50257, (16, 1024), device="cuda")
    def step():
        logits = stage(ids)
        loss = causal_lm_loss(logits, ids, ignore_index=-100)
        loss.backward()
        opt.step()
        opt.zero_grad()
    for _ in range(3): step()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(10): step()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / 10
    print(f"fp8={fp8}: {dt*1e3:.2f} ms per micro-batch (16x1024)")
    del stage, opt
    torch.cuda.empty_cache()

run(False)
run(True)
