"""Bisect the GPT-2 hipGraph capture fault (NOTES_ROUND2 item 4).

Runs on GPU.  Builds the bench GPT-2 (124M, micro 16) and tries to
capture, in order: forward only; forward+backward; +optimizer step —
at tiny and full size — printing which stage faults.  Each attempt runs
its replay twice and checks the loss is finite.
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def try_capture(tag, step_fn, warmup=2):
    try:
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(warmup):
                step_fn()
        torch.cuda.current_stream().wait_stream(side)
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            step_fn()
        g.replay()
        g.replay()
        torch.cuda.synchronize()
        print(f"[OK]   {tag}", flush=True)
        return True
    except Exception as e:  # noqa: BLE001
        torch.cuda.synchronize()
        print(f"[FAIL] {tag}: {type(e).__name__}: {e}", flush=True)
        return False


def main():
    from quintnet_amd.models import GPT2Config, GPT2Stage
    from quintnet_amd.optim import ZeroRedundancyAdamW
    from quintnet_amd.ops import causal_lm_loss

    dev = torch.device("cuda")
    size = os.environ.get("QN_BISECT_SIZE", "full")
    if size == "tiny":
        cfg = GPT2Config(n_embd=256, n_layer=2, n_head=4, vocab_size=1024,
                         n_positions=1024, dropout=0.0)
    else:
        cfg = GPT2Config(dropout=0.0)
    micro = int(os.environ.get("QN_BISECT_MICRO", "16"))
    model = GPT2Stage(cfg, device=dev, dtype=torch.bfloat16)
    opt = ZeroRedundancyAdamW(model.parameters(), lr=1e-4)
    ids = torch.randint(0, cfg.vocab_size, (micro, 1024), device=dev)
    labels = ids.clone()

    state = {}

    def fwd():
        state["loss"] = causal_lm_loss(model(ids), labels, -100)

    def fwd_bwd():
        opt.zero_grad()
        loss = causal_lm_loss(model(ids), labels, -100)
        loss.backward()

    def full_step():
        opt.zero_grad()
        loss = causal_lm_loss(model(ids), labels, -100)
        loss.backward()
        opt.step()

    print(f"== size={size} micro={micro} n_layer={cfg.n_layer} "
          f"n_embd={cfg.n_embd}", flush=True)
    ok_f = try_capture("forward", fwd)
    ok_fb = try_capture("forward+backward", fwd_bwd)
    ok_all = try_capture("full step (+ZeRO AdamW)", full_step)
    if ok_all:
        print("CAPTURE_ALL_OK")


if __name__ == "__main__":
    sys.exit(main())
