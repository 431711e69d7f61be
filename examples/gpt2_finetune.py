"""GPT-2 3D-parallel training/fine-tuning (reference examples/gpt2_finetune.py).

Staged mode loads a pretrained HF GPT-2 checkpoint sharded per rank
(each rank reads only its PP layers / TP slices); otherwise weights are
random-init.  Data: a summarization CSV (article/highlights) when
``dataset_path`` is set, else synthetic token sequences.

    torchrun --nproc_per_node=8 --master-addr 127.0.0.1 -m examples.gpt2_finetune \
        --config examples/gpt2_config.yaml [--checkpoint path/to/gpt2 --staged]
"""

import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import argparse

import torch
from torch.utils.data import DataLoader

from quintnet_amd import GPT2Trainer, get_strategy, init_process_groups, load_config
from quintnet_amd.models import GPT2Config, GPT2Stage
from quintnet_amd.parallel import DataParallel, DistributedConfig, PipelineParallelWrapper
from quintnet_amd.utils.data import SyntheticCLM


def build_loaders(cfg, pg):
    bs = cfg.get("batch_size", 8)
    seq = int(cfg.get("max_seq_length", 512))
    path = cfg.get("dataset_path")
    if path:
        from transformers import GPT2Tokenizer

        from quintnet_amd.utils.data import SummarizationDataLoader, SummarizationDataset

        tok = GPT2Tokenizer.from_pretrained(cfg.get("tokenizer_path", "gpt2"))
        train = SummarizationDataset(path)
        return (
            SummarizationDataLoader(train, tok, batch_size=bs, max_length=seq,
                                    shuffle=False, dp_rank=pg.dp_rank,
                                    dp_size=pg.dp_size),
            None,
        )
    vocab = cfg.get("model_config", {}).get("vocab_size", 50257)
    n_train = int(cfg.get("n_train", 512))
    train = SyntheticCLM(n=n_train, seq_len=seq, vocab_size=vocab, seed=pg.dp_rank)
    val = SyntheticCLM(n=64, seq_len=seq, vocab_size=vocab, seed=1000)
    return (
        DataLoader(train, batch_size=bs, shuffle=False),
        DataLoader(val, batch_size=bs, shuffle=False),
    )


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--config", default="examples/gpt2_config.yaml")
    ap.add_argument("--checkpoint", default=None)
    ap.add_argument("--staged", action="store_true")
    args = ap.parse_args()
    cfg = load_config(args.config)
    from quintnet_amd.core.config import validate_config

    problems = validate_config(cfg, world_size=int(os.environ.get("WORLD_SIZE", "1")))
    if problems:
        raise SystemExit("config problems:\n  " + "\n  ".join(problems))

    dev_type = "cuda" if torch.cuda.is_available() else "cpu"
    pg = init_process_groups(
        dev_type, cfg.get("mesh_dim", [2, 2, 2]), cfg.get("mesh_name", ["dp", "tp", "pp"])
    )

    if args.staged and args.checkpoint:
        pmodel = get_strategy("3d", pg, cfg, checkpoint_path=args.checkpoint, is_staged=True).apply(None)
    else:
        mc = cfg.get("model_config", {})
        gcfg = GPT2Config(
            vocab_size=mc.get("vocab_size", 50257),
            n_positions=mc.get("n_positions", 1024),
            n_embd=mc.get("n_embd", 768),
            n_layer=mc.get("n_layer", 12),
            n_head=mc.get("n_head", 12),
            dropout=mc.get("dropout", 0.1),
            vocab_pad_to=mc.get("vocab_pad_to", 0),
            activation_checkpointing=mc.get("activation_checkpointing", False),
        )
        dtype = torch.bfloat16 if dev_type == "cuda" else torch.float32
        stage = GPT2Stage(
            gcfg,
            pp_rank=pg.pp_rank,
            pp_size=pg.pp_size,
            tp_group=pg.get_group("tp") if "tp" in pg.mesh_name else None,
            tied_group=pg.get_tied_embedding_group(),
            device=pg.device,
            dtype=dtype,
        )
        stage.seq_len = int(cfg.get("max_seq_length", 512))
        stage.hidden_dim = gcfg.n_embd
        pmodel = stage
        if pg.pp_size > 1:
            pmodel = PipelineParallelWrapper(
                stage_module=stage, pp_rank=pg.pp_rank, pp_group=pg.get_group("pp"),
                pp_size=pg.pp_size, device=pg.device,
            )
            pmodel.seq_len, pmodel.hidden_dim = stage.seq_len, gcfg.n_embd
        if pg.dp_size > 1:
            pmodel = DataParallel(
                pmodel, DistributedConfig(pg.dp_rank, pg.dp_size, pg.get_group("dp"))
            )

    train, val = build_loaders(cfg, pg)
    trainer = GPT2Trainer(pmodel, train, val, cfg, pg)
    trainer.fit()


if __name__ == "__main__":
    main()
