"""Real-HF-weights round trip (reference test.py:28-120 parity).

The staged loader (checkpoint/distributed_loading.py) is exercised here
against an ACTUAL ``transformers`` GPT-2 safetensors file (random-init —
no network — but the genuine HF key names, Conv1D [in,out] layout and
tied lm_head), so a layout mismatch against HF's on-disk format fails
these tests.  Logits and CE loss are compared against the
``transformers`` model itself, and the merge CLI must reproduce the HF
key set from our shards.
"""

import os
import tempfile

import pytest
import torch

from conftest import run_distributed

transformers = pytest.importorskip("transformers")


def _make_hf_checkpoint(tmpdir, n_embd=64, n_layer=3, n_head=2, vocab=128, n_pos=64):
    from transformers import GPT2Config as HFConfig
    from transformers import GPT2LMHeadModel

    torch.manual_seed(42)
    hf_cfg = HFConfig(
        n_embd=n_embd,
        n_layer=n_layer,
        n_head=n_head,
        vocab_size=vocab,
        n_positions=n_pos,
        bos_token_id=0,
        eos_token_id=0,
        attn_pdrop=0.0,
        embd_pdrop=0.0,
        resid_pdrop=0.0,
    )
    hf = GPT2LMHeadModel(hf_cfg).eval()
    hf.save_pretrained(tmpdir, safe_serialization=True)
    assert os.path.exists(os.path.join(tmpdir, "model.safetensors"))
    return hf


def _our_config(n_embd=64, n_layer=3, n_head=2, vocab=128, n_pos=64):
    from quintnet_amd.models import GPT2Config

    return GPT2Config(
        vocab_size=vocab,
        n_positions=n_pos,
        n_embd=n_embd,
        n_layer=n_layer,
        n_head=n_head,
        dropout=0.0,
    )


def test_hf_safetensors_pp1_logits_match():
    from quintnet_amd.checkpoint.distributed_loading import load_gpt2_distributed
    from quintnet_amd.models import GPT2Stage

    with tempfile.TemporaryDirectory() as d:
        hf = _make_hf_checkpoint(d)
        cfg = _our_config()
        sd = load_gpt2_distributed(d, cfg, pp_rank=0, pp_size=1, tp_rank=0, tp_size=1)
        stage = GPT2Stage.from_sharded_state_dict(cfg, sd, pp_rank=0, pp_size=1)
        stage.eval()

        torch.manual_seed(0)
        ids = torch.randint(0, cfg.vocab_size, (2, 32))
        with torch.no_grad():
            ours = stage(ids)
            theirs = hf(ids).logits
        assert ours.shape == theirs.shape
        assert torch.allclose(ours, theirs, atol=2e-4), (
            (ours - theirs).abs().max().item()
        )

        # CE loss parity (the reference's test.py perplexity check)
        from quintnet_amd.ops import causal_lm_loss

        labels = ids.clone()
        with torch.no_grad():
            our_loss = causal_lm_loss(ours, labels, ignore_index=-100)
            hf_loss = hf(ids, labels=ids).loss
        assert abs(float(our_loss) - float(hf_loss)) < 1e-3, (
            float(our_loss),
            float(hf_loss),
        )


def test_hf_safetensors_pp2_chain_match():
    """Two pipeline stages loaded from the HF file chain to the same logits
    (tied lm_head copy on the last stage included)."""
    from quintnet_amd.checkpoint.distributed_loading import load_gpt2_distributed
    from quintnet_amd.models import GPT2Stage

    with tempfile.TemporaryDirectory() as d:
        hf = _make_hf_checkpoint(d)
        cfg = _our_config()
        stages = []
        for pp_rank in range(2):
            sd = load_gpt2_distributed(
                d, cfg, pp_rank=pp_rank, pp_size=2, tp_rank=0, tp_size=1
            )
            st = GPT2Stage.from_sharded_state_dict(cfg, sd, pp_rank=pp_rank, pp_size=2)
            st.eval()
            stages.append(st)

        torch.manual_seed(1)
        ids = torch.randint(0, cfg.vocab_size, (2, 16))
        with torch.no_grad():
            mid = stages[0](ids)
            ours = stages[1](mid)
            theirs = hf(ids).logits
        assert torch.allclose(ours, theirs, atol=2e-4), (
            (ours - theirs).abs().max().item()
        )


def _hf_tp2_worker(rank, world, ckpt_dir, want):
    import torch.distributed as dist

    from quintnet_amd.checkpoint.distributed_loading import load_gpt2_distributed
    from quintnet_amd.models import GPT2Stage

    cfg = _our_config()
    sd = load_gpt2_distributed(
        ckpt_dir, cfg, pp_rank=0, pp_size=1, tp_rank=rank, tp_size=world
    )
    stage = GPT2Stage.from_sharded_state_dict(
        cfg, sd, pp_rank=0, pp_size=1, tp_group=dist.group.WORLD
    )
    stage.eval()
    torch.manual_seed(2)
    ids = torch.randint(0, cfg.vocab_size, (2, 16))
    with torch.no_grad():
        ours = stage(ids)
    want = torch.as_tensor(want)
    assert torch.allclose(ours, want, atol=5e-4), (ours - want).abs().max().item()


def test_hf_safetensors_tp2_logits_match():
    with tempfile.TemporaryDirectory() as d:
        hf = _make_hf_checkpoint(d)
        torch.manual_seed(2)
        ids = torch.randint(0, 128, (2, 16))
        with torch.no_grad():
            want = hf(ids).logits
        run_distributed(_hf_tp2_worker, 2, d, want.numpy())


def test_merge_cli_emits_hf_keyset():
    """Our shards, merged, must reproduce the HF checkpoint's key set and
    tensors (merge CLI layout parity against the REAL HF naming)."""
    from safetensors import safe_open

    from quintnet_amd.checkpoint.distributed_loading import load_gpt2_distributed
    from quintnet_amd.checkpoint.merge import merge_checkpoints
    from quintnet_amd.models import GPT2Stage

    with tempfile.TemporaryDirectory() as d:
        _make_hf_checkpoint(d)
        cfg = _our_config()
        shard_dir = os.path.join(d, "shards")
        os.makedirs(shard_dir)
        for pp_rank in range(2):
            sd = load_gpt2_distributed(
                d, cfg, pp_rank=pp_rank, pp_size=2, tp_rank=0, tp_size=1
            )
            st = GPT2Stage.from_sharded_state_dict(cfg, sd, pp_rank=pp_rank, pp_size=2)
            torch.save(
                {
                    "model_state_dict": st.state_dict(),
                    "parallelism_info": {
                        "pp_rank": pp_rank,
                        "pp_size": 2,
                        "tp_rank": 0,
                        "tp_size": 1,
                        "dp_rank": 0,
                    },
                },
                os.path.join(shard_dir, f"final_model_pp{pp_rank}_tp0.pt"),
            )
        out = os.path.join(d, "merged.pt")
        merge_checkpoints(shard_dir, out, prefix="final_model")
        merged = torch.load(out, map_location="cpu", weights_only=False)
        # merge_checkpoints already emits HF-format keys/layouts
        hf_state = merged.get("model_state_dict", merged)

        with safe_open(
            os.path.join(d, "model.safetensors"), framework="pt", device="cpu"
        ) as f:
            ref_keys = set(f.keys())
            # HF omits the tied lm_head tensor from the file
            for k in ref_keys:
                want = f.get_tensor(k)
                have = hf_state.get(k)
                if have is None and k.startswith("transformer."):
                    have = hf_state.get(k[len("transformer."):])
                assert have is not None, f"merged checkpoint missing HF key {k}"
                # HF stores Conv1D weights [in, out]; accept either layout
                if have.shape != want.shape and have.t().shape == want.shape:
                    have = have.t()
                assert have.shape == want.shape, k
                assert torch.allclose(have.float(), want.float(), atol=1e-5), k


def test_hf_load_into_padded_config_and_export_slices_back():
    """HF file (logical vocab) -> padded-vocab stage: logits over the
    real vocab match HF exactly, pad columns are -inf; and the merge
    CLI's HF export slices wte/lm_head back to the logical vocab."""
    from quintnet_amd.checkpoint.distributed_loading import load_gpt2_distributed
    from quintnet_amd.checkpoint.merge import convert_to_hf_format
    from quintnet_amd.models import GPT2Config, GPT2Stage

    with tempfile.TemporaryDirectory() as d:
        hf = _make_hf_checkpoint(d, vocab=100)
        cfg = GPT2Config(
            vocab_size=100, n_positions=64, n_embd=64, n_layer=3, n_head=2,
            dropout=0.0, vocab_pad_to=64,
        )
        assert cfg.padded_vocab_size == 128
        sd = load_gpt2_distributed(d, cfg, pp_rank=0, pp_size=1, tp_rank=0, tp_size=1)
        assert sd["embedding.wte.weight"].shape[0] == 128
        assert bool((sd["embedding.wte.weight"][100:] == 0).all())
        stage = GPT2Stage.from_sharded_state_dict(cfg, sd, pp_rank=0, pp_size=1)
        stage.eval()
        ids = torch.randint(0, 100, (2, 16))
        with torch.no_grad():
            ours = stage(ids)
            theirs = hf(ids).logits
        assert ours.shape[-1] == 128
        assert torch.allclose(ours[..., :100], theirs, atol=2e-4)
        assert bool(torch.isneginf(ours[..., 100:]).all())

        # HF export: padded rows must be sliced off
        merged = {
            "wte.weight": stage.embedding.wte.weight.detach(),
            "wpe.weight": stage.embedding.wpe.weight.detach(),
        }
        out = convert_to_hf_format(merged, vocab_size=100)
        assert out["transformer.wte.weight"].shape[0] == 100
        assert out["lm_head.weight"].shape[0] == 100


def test_merge_to_safetensors_loads_in_transformers():
    """merge_checkpoints --output x.safetensors produces a file
    transformers loads directly (from_pretrained on the directory)."""
    from transformers import GPT2Config as HFConfig
    from transformers import GPT2LMHeadModel

    from quintnet_amd.checkpoint.distributed_loading import load_gpt2_distributed
    from quintnet_amd.checkpoint.merge import merge_checkpoints
    from quintnet_amd.models import GPT2Stage

    with tempfile.TemporaryDirectory() as d:
        hf = _make_hf_checkpoint(d)
        cfg = _our_config()
        shard_dir = os.path.join(d, "shards")
        os.makedirs(shard_dir)
        sd = load_gpt2_distributed(d, cfg, pp_rank=0, pp_size=1, tp_rank=0, tp_size=1)
        st = GPT2Stage.from_sharded_state_dict(cfg, sd, pp_rank=0, pp_size=1)
        torch.save(
            {"model_state_dict": st.state_dict(),
             "parallelism_info": {"pp_rank": 0, "pp_size": 1, "tp_rank": 0,
                                  "tp_size": 1, "dp_rank": 0}},
            os.path.join(shard_dir, "final_model_pp0_tp0.pt"),
        )
        outdir = os.path.join(d, "export")
        os.makedirs(outdir)
        merge_checkpoints(shard_dir, os.path.join(outdir, "model.safetensors"),
                          prefix="final_model")
        # config.json so from_pretrained can instantiate
        HFConfig(n_embd=64, n_layer=3, n_head=2, vocab_size=128,
                 n_positions=64).save_pretrained(outdir)
        reloaded = GPT2LMHeadModel.from_pretrained(outdir).eval()
        ids = torch.randint(0, 128, (1, 16))
        with torch.no_grad():
            a = hf(ids).logits
            b = reloaded(ids).logits
        assert torch.allclose(a, b, atol=1e-5)
