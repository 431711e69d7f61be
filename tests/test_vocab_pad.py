"""Padded-vocab layout (GPT2Config.vocab_pad_to) is EXACTLY the
unpadded model: pad logits columns are masked to -inf, pad embedding
rows stay zero and get zero gradient (models/gpt2/config.py,
models/gpt2/stage.py mask_pad_logits, ops/linear.py logical_out)."""

import torch

from conftest import run_distributed

from quintnet_amd.models import GPT2Config, GPT2Stage
from quintnet_amd.ops import causal_lm_loss


def _mk(pad):
    torch.manual_seed(7)
    cfg = GPT2Config(
        n_embd=64, n_layer=2, n_head=2, vocab_size=300, n_positions=64,
        dropout=0.0, vocab_pad_to=pad,
    )
    torch.manual_seed(11)
    stage = GPT2Stage(cfg)
    return cfg, stage


def test_padded_matches_unpadded_exactly():
    cfg0, m0 = _mk(0)
    cfg1, m1 = _mk(128)
    assert cfg1.padded_vocab_size == 384
    # copy the unpadded weights into the padded model (pad rows zero)
    sd0, sd1 = m0.state_dict(), m1.state_dict()
    for k, v in sd0.items():
        if v.shape != sd1[k].shape:
            assert k == "embedding.wte.weight"
            sd1[k].zero_()
            sd1[k][: v.shape[0]] = v
        else:
            sd1[k] = v
    m1.load_state_dict(sd1)

    ids = torch.randint(0, 300, (2, 64))
    labels = torch.randint(0, 300, (2, 64))
    out0 = m0(ids)
    out1 = m1(ids)
    assert out1.shape[-1] == 384
    assert torch.equal(out0, out1[..., :300])
    assert bool(torch.isneginf(out1[..., 300:]).all())

    l0 = causal_lm_loss(out0, labels)
    l1 = causal_lm_loss(out1, labels)
    assert torch.equal(l0, l1)
    l0.backward()
    l1.backward()
    g0 = m0.embedding.wte.weight.grad
    g1 = m1.embedding.wte.weight.grad
    # grads equal up to BLAS blocking (the wgrad GEMM tiles differently
    # at N=384 vs N=300, reordering the fp32 reduction)
    assert torch.allclose(g0, g1[:300], rtol=1e-5, atol=1e-7)
    assert bool((g1[300:] == 0).all())
    for (k0, p0), (k1, p1) in zip(
        m0.named_parameters(), m1.named_parameters()
    ):
        if k0 == "embedding.wte.weight":
            continue
        assert torch.allclose(p0.grad, p1.grad, rtol=1e-5, atol=1e-7), k0


def test_generate_never_emits_pad_tokens():
    _, m = _mk(128)
    m.eval()
    ids = torch.randint(0, 300, (2, 8))
    out = m.generate(ids, max_new_tokens=6)
    assert int(out.max()) < 300


def _pp2_padded(rank, world):
    """pp=2 pipeline with a PADDED-vocab config vs an UNPADDED
    single-process oracle: tied wte/lm_head grad sync runs on padded
    shapes at both ends; train loss must match the unpadded math."""
    import torch.distributed as dist

    from quintnet_amd import GPT2Trainer, init_process_groups
    from quintnet_amd.models import GPT2Config, GPT2Stage
    from quintnet_amd.ops import causal_lm_loss
    from quintnet_amd.parallel import PipelineParallelWrapper
    from quintnet_amd.utils.data import SyntheticCLM

    pg = init_process_groups("cpu", [1, 1, 2], ["dp", "tp", "pp"])
    base = dict(n_embd=32, n_layer=4, n_head=2, vocab_size=96,
                n_positions=32, dropout=0.0)
    cfg0 = GPT2Config(**base)                     # oracle
    cfg1 = GPT2Config(**base, vocab_pad_to=128)   # pipeline under test
    assert cfg1.padded_vocab_size == 128
    seq = 16
    torch.manual_seed(33)
    full = GPT2Stage(cfg0, pp_rank=0, pp_size=1)
    for p in full.parameters():
        dist.broadcast(p.data, src=0)

    stage = GPT2Stage(
        cfg1, pp_rank=pg.pp_rank, pp_size=pg.pp_size,
        tied_group=pg.get_tied_embedding_group(),
    )

    def pad_rows(w):
        out = torch.zeros(cfg1.padded_vocab_size, w.shape[1])
        out[: w.shape[0]] = w
        return out

    with torch.no_grad():
        sd = full.state_dict()
        offset = stage.layer_distribution[pg.pp_rank][0]
        tgt = {}
        for i, _ in enumerate(stage.my_layers):
            for k in (
                "ln_1.weight", "ln_1.bias", "ln_2.weight", "ln_2.bias",
                "attn.c_attn.weight", "attn.c_attn.bias",
                "attn.c_proj.weight", "attn.c_proj.bias",
                "mlp.c_fc.weight", "mlp.c_fc.bias",
                "mlp.c_proj.weight", "mlp.c_proj.bias",
            ):
                tgt[f"blocks.{i}.{k}"] = sd[f"blocks.{i + offset}.{k}"]
        if stage.is_first_stage:
            tgt["embedding.wte.weight"] = pad_rows(sd["embedding.wte.weight"])
            tgt["embedding.wpe.weight"] = sd["embedding.wpe.weight"]
        if stage.is_last_stage and not stage.is_first_stage:
            tgt["ln_f.weight"] = sd["ln_f.weight"]
            tgt["ln_f.bias"] = sd["ln_f.bias"]
            tgt["lm_head"] = pad_rows(sd["embedding.wte.weight"])
        stage.load_state_dict(tgt, strict=False)

    wrapper = PipelineParallelWrapper(
        stage_module=stage, pp_rank=pg.pp_rank, pp_group=pg.get_group("pp"),
        pp_size=pg.pp_size,
    )
    wrapper.seq_len, wrapper.hidden_dim = seq, cfg1.n_embd

    ds = SyntheticCLM(n=8, seq_len=seq, vocab_size=96, seed=5)
    dl = torch.utils.data.DataLoader(ds, batch_size=2, shuffle=False)
    tcfg = {
        "batch_size": 2, "num_epochs": 1, "learning_rate": 1e-3,
        "grad_acc_steps": 2, "max_grad_norm": None, "schedule": "1f1b",
        "zero1": True, "max_seq_length": seq,
        "model_config": {"n_embd": cfg1.n_embd},
    }
    trainer = GPT2Trainer(wrapper, dl, None, tcfg, pg)
    metrics = trainer.fit()

    if rank == 1:
        from quintnet_amd.optim import ZeroRedundancyAdamW

        opt = ZeroRedundancyAdamW(full.parameters(), lr=1e-3, weight_decay=0.01)
        it = iter(dl)
        step_losses = []
        for _ in range(len(dl) // 2):
            tot = 0.0
            for _ in range(2):
                b = next(it)
                loss = causal_lm_loss(full(b["input_ids"]), b["labels"])
                (loss / 2).backward()
                tot += float(loss.detach())
            opt.step()
            opt.zero_grad()
            step_losses.append(tot / 2)
        ref = sum(step_losses) / len(step_losses)
        assert abs(metrics["train_loss"] - ref) < 2e-3, (metrics["train_loss"], ref)


def test_pp2_padded_matches_unpadded_oracle():
    run_distributed(_pp2_padded, 2)
