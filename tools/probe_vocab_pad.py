"""A/B the GPT-2 LM-head GEMMs + CE at vocab 50257 (odd rows) vs 50304
(128-aligned).  Motivates the padded-vocab option (models/gpt2/config.py
``vocab_pad_to``): every logits-sized tensor has odd-element rows at
50257, so vectorized row access is unaligned in the GEMM epilogues and
the CE kernels."""
import sys, time, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

def bench(fn, iters=20):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3

def main():
    dev = "cuda"
    M, K = 32768, 768
    torch.manual_seed(0)
    x = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
    tgt = torch.randint(0, 50257, (M,), device=dev)
    from quintnet_amd.ops import causal_lm_loss  # noqa: F401  (ext load)
    from quintnet_amd.ops.cross_entropy import cross_entropy
    for N in (50257, 50304):
        w = torch.randn(N, K, device=dev, dtype=torch.bfloat16)
        g = torch.randn(M, N, device=dev, dtype=torch.bfloat16)
        t_fwd = bench(lambda: x @ w.t())
        t_dgrad = bench(lambda: g @ w)
        t_wgrad = bench(lambda: g.t() @ x)
        logits = (x @ w.t()).requires_grad_(True)
        def ce_fb():
            loss = cross_entropy(logits, tgt)
            loss.backward()
            logits.grad = None
        t_ce = bench(ce_fb, iters=10)
        print(f"N={N}: fwd {t_fwd:.3f} ms  dgrad {t_dgrad:.3f} ms  "
              f"wgrad {t_wgrad:.3f} ms  ce f+b {t_ce:.3f} ms  "
              f"sum {t_fwd+t_dgrad+t_wgrad+t_ce:.3f} ms", flush=True)

if __name__ == "__main__":
    main()
