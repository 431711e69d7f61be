#!/usr/bin/env python3
"""Merge per-rank ``{name}_pp{p}_tp{t}.pt`` shards into HF GPT-2 format.

Parity with reference merge_checkpoints.py (same shard filename layout,
TP cat + PP key remap + Conv1D re-transpose), implemented in
quintnet_amd.checkpoint.merge.

    python merge_checkpoints.py --input-dir checkpoints/ \
        --output merged.pt --prefix final_model
"""

import argparse

from quintnet_amd.checkpoint import merge_checkpoints


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--input-dir", required=True, help="directory with *_pp{p}_tp{t}.pt shards")
    ap.add_argument("--output", required=True, help="output path: .pt (state dict + config) or .safetensors (transformers-loadable)")
    ap.add_argument("--prefix", default="final_model", help="shard filename prefix")
    args = ap.parse_args()
    out = merge_checkpoints(args.input_dir, args.output, prefix=args.prefix)
    print(f"merged -> {out}")


if __name__ == "__main__":
    main()
