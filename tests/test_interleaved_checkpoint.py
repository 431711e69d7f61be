"""Interleaved (virtual-pipeline) checkpoint save + merge (round-2 gap:
chunk-local names chunks.c.i.* must merge to the HF layout)."""

import os
import tempfile

import torch


class _PG:
    """Minimal pg_manager stand-in for per-rank shard writing."""

    def __init__(self, pp_rank, pp_size):
        self.pp_rank, self.pp_size = pp_rank, pp_size
        self.tp_rank = self.dp_rank = 0
        self.tp_size = 1
        self.mesh_name = ("dp", "tp", "pp")

    def axis_rank(self, ax):
        return 0


def test_interleaved_save_merge_roundtrip():
    from quintnet_amd.checkpoint import merge_checkpoints, save_sharded_checkpoint
    from quintnet_amd.models import GPT2Config, GPT2ForInterleaving
    from quintnet_amd.parallel import InterleavedPipelineWrapper

    torch.manual_seed(3)
    cfg = GPT2Config(n_embd=64, n_layer=8, n_head=2, vocab_size=256,
                     n_positions=64, dropout=0.0)
    full = GPT2ForInterleaving(cfg)
    pp_size, chunks = 2, 2
    with tempfile.TemporaryDirectory() as d:
        for r in range(pp_size):
            w = InterleavedPipelineWrapper(full, pp_rank=r, pp_size=pp_size,
                                           num_chunks=chunks)
            save_sharded_checkpoint(w, d, name="final_model", pg_manager=_PG(r, pp_size))
        out = os.path.join(d, "merged.pt")
        merge_checkpoints(d, out, prefix="final_model")
        hf = torch.load(out, map_location="cpu", weights_only=False)["model_state_dict"]

    # every layer present under HF names with the ORIGINAL weights
    for L in range(cfg.n_layer):
        want = full.blocks[L].ln_1.weight
        have = hf[f"transformer.h.{L}.ln_1.weight"]
        assert torch.equal(have, want), L
        # Conv1D transpose on weights
        assert torch.equal(hf[f"transformer.h.{L}.attn.c_attn.weight"],
                           full.blocks[L].attn.c_attn.weight.t())
    assert torch.equal(hf["transformer.wte.weight"], full.embedding.wte.weight)
    assert torch.equal(hf["transformer.wpe.weight"], full.embedding.wpe.weight)
    assert torch.equal(hf["transformer.ln_f.weight"], full.head[0].weight)
    assert torch.equal(hf["lm_head.weight"], full.embedding.wte.weight)


def test_ep_shard_merge_renumbers_experts():
    """EP shards (rank-local expert indices) fold into standard pp/tp
    shards with global expert numbering; replicated keys from ep0."""
    import tempfile

    from quintnet_amd.checkpoint.merge import merge_ep_shards

    with tempfile.TemporaryDirectory() as d:
        for ep in range(2):
            sd = {
                "blocks.0.mlp.router.weight": torch.full((4, 8), float(ep)),
                "blocks.0.mlp.experts.0.c_fc.weight": torch.full((8, 8), ep * 10.0),
                "blocks.0.mlp.experts.1.c_fc.weight": torch.full((8, 8), ep * 10.0 + 1),
                "blocks.0.ln_1.weight": torch.full((8,), float(ep)),
            }
            torch.save({"model_state_dict": sd, "parallelism_info": {}},
                       os.path.join(d, f"final_model_pp0_tp0_ep{ep}.pt"))
        out_dir = merge_ep_shards(d)
        merged = torch.load(os.path.join(out_dir, "final_model_pp0_tp0.pt"),
                            map_location="cpu", weights_only=False)
        st = merged["model_state_dict"]
        # global experts 0..3 with the right sources
        assert float(st["blocks.0.mlp.experts.0.c_fc.weight"][0, 0]) == 0.0
        assert float(st["blocks.0.mlp.experts.1.c_fc.weight"][0, 0]) == 1.0
        assert float(st["blocks.0.mlp.experts.2.c_fc.weight"][0, 0]) == 10.0
        assert float(st["blocks.0.mlp.experts.3.c_fc.weight"][0, 0]) == 11.0
        # replicated keys come from ep0
        assert float(st["blocks.0.ln_1.weight"][0]) == 0.0
        assert merged["parallelism_info"]["ep_merged"]["ep_size"] == 2
