"""Offline merge of ``{name}_pp{p}_tp{t}.pt`` shards into HF GPT-2 format.

Parity with reference merge_checkpoints.py:33-188.  Differences owned
end-to-end by this framework:

* c_attn/c_fc (column-parallel) shards are merged interleave-aware —
  each shard's rows are ``[q_loc; k_loc; v_loc]`` (see
  models/gpt2/attention.py), so the merge splits each shard into its
  Q/K/V thirds and concatenates per component (a naive cat would
  scramble head order);
* c_proj (row-parallel) weights cat along dim 1; its bias (and every
  replicated tensor) is taken from tp rank 0;
* PP merge remaps ``blocks.X`` → ``h.{X+offset}``; stage 0 contributes
  wte/wpe, the last stage ln_f + lm_head;
* HF conversion adds the ``transformer.`` prefix and re-transposes the
  Conv1D weights (HF stores [in, out]).
"""

from __future__ import annotations

import glob
import os
import re
from typing import Dict, Optional

import torch

__all__ = ["load_shards", "merge_tp_shards", "merge_pp_stages", "remap_interleaved_stage", "merge_ep_shards", "convert_to_hf_format", "merge_checkpoints"]

_SHARD_RE = re.compile(r"_pp(\d+)_tp(\d+)\.pt$")


def load_shards(input_dir: str, prefix: Optional[str] = None) -> Dict[int, Dict[int, dict]]:
    out: Dict[int, Dict[int, dict]] = {}
    for path in sorted(glob.glob(os.path.join(input_dir, "*.pt"))):
        base = os.path.basename(path)
        if prefix and not base.startswith(prefix):
            continue
        m = _SHARD_RE.search(base)
        if not m:
            continue
        pp, tp = int(m.group(1)), int(m.group(2))
        out.setdefault(pp, {})[tp] = torch.load(path, map_location="cpu", weights_only=False)
    if not out:
        raise FileNotFoundError(
            f"no *_pp*_tp*.pt shards under {input_dir} (shards with an _ep* "
            "suffix must first be folded with merge_ep_shards(); _cp* "
            "shards are full replicas along cp — rename one set)"
        )
    return out


def _merge_qkv_rows(tensors):
    """cat column-parallel fused-QKV shards: split each into thirds, cat per part."""
    parts = [t.chunk(3, dim=0) for t in tensors]
    return torch.cat(
        [torch.cat([p[i] for p in parts], dim=0) for i in range(3)], dim=0
    )


def merge_tp_shards(tp_shards: Dict[int, Dict[str, torch.Tensor]]) -> Dict[str, torch.Tensor]:
    tp_size = len(tp_shards)
    keys = tp_shards[0].keys()
    merged: Dict[str, torch.Tensor] = {}
    for key in keys:
        ts = [tp_shards[r][key] for r in range(tp_size)]
        if tp_size == 1:
            merged[key] = ts[0]
        elif "attn.c_attn.weight" in key:
            merged[key] = _merge_qkv_rows(ts)
        elif "attn.c_attn.bias" in key:
            merged[key] = _merge_qkv_rows([t.unsqueeze(1) for t in ts]).squeeze(1)
        elif "c_fc.weight" in key or "c_fc.bias" in key:
            merged[key] = torch.cat(ts, dim=0)
        elif "c_proj.weight" in key:
            merged[key] = torch.cat(ts, dim=1)
        else:
            # replicated (LN, embeddings, c_proj bias, lm_head): rank 0 copy
            merged[key] = ts[0]
    return merged


def merge_pp_stages(
    pp_stages: Dict[int, Dict[str, torch.Tensor]], n_layer: Optional[int] = None
) -> Dict[str, torch.Tensor]:
    pp_size = len(pp_stages)
    merged: Dict[str, torch.Tensor] = {}
    offset = 0
    for pp_rank in sorted(pp_stages):
        state = pp_stages[pp_rank]
        max_block = -1
        for key, value in state.items():
            new_key = key
            m = re.match(r"blocks\.(\d+)\.(.*)", key)
            if m:
                idx = int(m.group(1))
                max_block = max(max_block, idx)
                new_key = f"h.{idx + offset}.{m.group(2)}"
            elif key.startswith("embedding.wte."):
                new_key = "wte." + key.split(".")[-1]
            elif key.startswith("embedding.wpe."):
                new_key = "wpe." + key.split(".")[-1]
            elif key.startswith("ln_f."):
                new_key = key
            elif key.startswith("lm_head"):
                new_key = "lm_head.weight"
            merged[new_key] = value
        offset += max_block + 1
    return merged


def convert_to_hf_format(
    merged: Dict[str, torch.Tensor], vocab_size: Optional[int] = None
) -> Dict[str, torch.Tensor]:
    hf: Dict[str, torch.Tensor] = {}
    for key, value in merged.items():
        new_key = key if key.startswith("lm_head") else "transformer." + key
        if (
            "c_attn.weight" in key
            or "c_proj.weight" in key
            or "c_fc.weight" in key
        ):
            value = value.t().contiguous()  # Linear [out,in] -> HF Conv1D [in,out]
        if (
            vocab_size is not None
            and (key.startswith("lm_head") or key.startswith("wte."))
            and value.shape[0] > vocab_size
        ):
            # drop the padded-vocab layout rows (config vocab_pad_to):
            # they are zero / never trained — HF files carry logical vocab
            value = value[:vocab_size].contiguous()
        hf[new_key] = value
    if "lm_head.weight" not in hf and "transformer.wte.weight" in hf:
        hf["lm_head.weight"] = hf["transformer.wte.weight"]
    return hf


def remap_interleaved_stage(
    state: Dict[str, torch.Tensor], info: Dict
) -> Dict[str, torch.Tensor]:
    """Rename an InterleavedPipelineWrapper shard's chunk-local names
    (``chunks.c.i.*``) to the global merged naming (wte/wpe, h.L.*, ln_f,
    lm_head) using the recorded chunk→global-stage layer map.

    Chunk module order (parallel/pipeline/wrapper.py): [embedding (global
    stage 0 only)] + blocks + [head = Sequential(ln_f, TiedLMHead) (last
    global stage only)].
    """
    meta = info["interleaved"]
    pp_rank = info["pp_rank"]
    pp_size = meta["pp_size"]
    num_chunks = meta["num_chunks"]
    dist_layers = meta["layer_distribution"]
    n_stages = pp_size * num_chunks
    out: Dict[str, torch.Tensor] = {}
    for key, value in state.items():
        m = re.match(r"chunks\.(\d+)\.(\d+)\.(.*)", key)
        if not m:
            out[key] = value
            continue
        c, i, rest = int(m.group(1)), int(m.group(2)), m.group(3)
        g = c * pp_size + pp_rank
        has_emb = g == 0
        blocks = dist_layers[g]
        if has_emb and i == 0:
            # GPT2Embedding: wte.weight / wpe.weight
            out[rest] = value
        elif i - (1 if has_emb else 0) < len(blocks):
            layer = blocks[i - (1 if has_emb else 0)]
            out[f"h.{layer}.{rest}"] = value
        elif g == n_stages - 1:
            # head Sequential: 0 = ln_f (FusedLayerNorm), 1 = TiedLMHead
            hm = re.match(r"0\.(.*)", rest)
            if hm:
                out[f"ln_f.{hm.group(1)}"] = value
            else:
                hm = re.match(r"1\.wte\.(.*)", rest)
                if hm:  # tied copy of the embedding matrix
                    out["lm_head.weight"] = value
        else:
            raise ValueError(f"unmappable interleaved key {key}")
    return out


_SHARD_EP_RE = re.compile(r"_pp(\d+)_tp(\d+)_ep(\d+)\.pt$")
_EXPERT_KEY_RE = re.compile(r"(.*\bexperts)\.(\d+)\.(.*)")


def merge_ep_shards(input_dir: str, prefix: str = "final_model",
                    output_dir: Optional[str] = None) -> str:
    """Fold expert-parallel shards ``{name}_pp{p}_tp{t}_ep{r}.pt`` into
    standard ``{name}_pp{p}_tp{t}.pt`` shards with GLOBALLY numbered
    experts (expert e lives on ep rank e // n_local —
    parallel/expert_parallel.py), so the regular TP/PP merge CLI and the
    EP=1 ``ExpertParallelMLP`` can consume them.  Replicated (non-expert)
    tensors are taken from ep rank 0."""
    groups: Dict[tuple, Dict[int, dict]] = {}
    for path in sorted(glob.glob(os.path.join(input_dir, "*.pt"))):
        base = os.path.basename(path)
        if prefix and not base.startswith(prefix):
            continue
        m = _SHARD_EP_RE.search(base)
        if not m:
            continue
        pp, tp, ep = int(m.group(1)), int(m.group(2)), int(m.group(3))
        groups.setdefault((pp, tp), {})[ep] = torch.load(
            path, map_location="cpu", weights_only=False)
    if not groups:
        raise FileNotFoundError(f"no *_pp*_tp*_ep*.pt shards under {input_dir}")
    out_dir = output_dir or os.path.join(input_dir, "ep_merged")
    os.makedirs(out_dir, exist_ok=True)
    for (pp, tp), eps in groups.items():
        ep_size = len(eps)
        # n_local from the rank-0 shard's expert key set
        n_local = 0
        for k in eps[0]["model_state_dict"]:
            m = _EXPERT_KEY_RE.match(k)
            if m:
                n_local = max(n_local, int(m.group(2)) + 1)
        merged: Dict[str, torch.Tensor] = {}
        for ep in range(ep_size):
            for k, v in eps[ep]["model_state_dict"].items():
                m = _EXPERT_KEY_RE.match(k)
                if m:
                    g = ep * n_local + int(m.group(2))
                    merged[f"{m.group(1)}.{g}.{m.group(3)}"] = v
                elif ep == 0:
                    merged[k] = v
        payload = dict(eps[0])
        payload["model_state_dict"] = merged
        info = dict(payload.get("parallelism_info", {}))
        info["ep_merged"] = {"ep_size": ep_size, "n_local": n_local}
        payload["parallelism_info"] = info
        torch.save(payload, os.path.join(out_dir, f"{prefix}_pp{pp}_tp{tp}.pt"))
    return out_dir


def merge_checkpoints(input_dir: str, output_path: str, prefix: str = "final_model") -> str:
    shards = load_shards(input_dir, prefix)
    first = next(iter(next(iter(shards.values())).values()))
    interleaved = "interleaved" in first.get("parallelism_info", {})
    pp_stages = {pp: merge_tp_shards({t: s["model_state_dict"] for t, s in tps.items()})
                 for pp, tps in shards.items()}
    if interleaved:
        merged: Dict[str, torch.Tensor] = {}
        for pp, tps in shards.items():
            info = next(iter(tps.values()))["parallelism_info"]
            merged.update(remap_interleaved_stage(pp_stages[pp], info))
    else:
        merged = merge_pp_stages(pp_stages)
    config = first.get("config", {})
    mc = config.get("model_config", {}) if isinstance(config, dict) else {}
    hf_state = convert_to_hf_format(merged, vocab_size=mc.get("vocab_size"))
    os.makedirs(os.path.dirname(os.path.abspath(output_path)), exist_ok=True)
    if output_path.endswith(".safetensors"):
        # the on-disk format transformers loads directly: tensors only,
        # HF key names minus the tied lm_head (HF re-ties from wte), and
        # Linear weights already transposed back to Conv1D layout above
        from safetensors.torch import save_file

        st = {k: v.contiguous() for k, v in hf_state.items()
              if k != "lm_head.weight"}
        save_file(st, output_path, metadata={"format": "pt"})
    else:
        torch.save({"model_state_dict": hf_state, "config": config}, output_path)
    return output_path
