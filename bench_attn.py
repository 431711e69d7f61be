"""Isolated fused-attention kernel timings at the bench shape."""
import time

import torch


def bench(fn, iters=30):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    from quintnet_amd import _C

    dev = torch.device("cuda")
    B, H, T, D = 8, 12, 1024, 64
    q = torch.randn(B, H, T, D, device=dev, dtype=torch.bfloat16)
    k = torch.randn_like(q)
    v = torch.randn_like(q)
    out = torch.empty_like(q)
    dout = torch.randn_like(q)
    dq = torch.empty_like(q)
    dk = torch.empty_like(q)
    dv = torch.empty_like(q)
    scale = 0.125
    lse = _C.attn_fwd(q, k, v, out, scale, True)

    t_f = bench(lambda: _C.attn_fwd(q, k, v, out, scale, True))
    t_b = bench(lambda: _C.attn_bwd(q, k, v, out, dout, lse, dq, dk, dv, scale, True))
    fl_f = 4.0 * B * H * T * T * D * 0.5  # causal
    fl_b = fl_f * 2.5
    print(f"fwd {t_f*1e6:8.1f}us  {fl_f/t_f/1e12:6.1f} TF")
    print(f"bwd {t_b*1e6:8.1f}us  {fl_b/t_b/1e12:6.1f} TF (dq+dkv+delta)")

    def sdpa():
        torch.nn.functional.scaled_dot_product_attention(q, k, v, is_causal=True)

    print(f"sdpa fwd {bench(sdpa)*1e6:8.1f}us (torch reference)")

    qr = q.clone().requires_grad_(True)
    kr = k.clone().requires_grad_(True)
    vr = v.clone().requires_grad_(True)
    o = torch.nn.functional.scaled_dot_product_attention(qr, kr, vr, is_causal=True)

    def sdpa_bwd():
        o.backward(dout, retain_graph=True)
        qr.grad = kr.grad = vr.grad = None

    print(f"sdpa fwd+bwd-bwd {bench(sdpa_bwd)*1e6:8.1f}us (torch bwd reference)")


if __name__ == "__main__":
    main()
