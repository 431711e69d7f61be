"""Staged (distributed) GPT-2 checkpoint loading from safetensors.

Parity with reference core/distributed_loading.py:203-376: each rank
memory-maps the safetensors file and reads ONLY its slice — PP stage
picks its layers, TP rank slices its columns/rows (with the HF
Conv1D → Linear transpose), stage 0 gets wte/wpe, the last stage ln_f +
a tied wte copy for lm_head.

TP slicing honors the per-head fused-QKV layout of
models/gpt2/attention.py: each rank's c_attn shard is
``[q_loc; k_loc; v_loc]`` (its n_embd/tp rows of each component), NOT a
naive contiguous third of the 3·n_embd output.
"""

from __future__ import annotations

import os
from typing import Dict

import torch

__all__ = ["load_gpt2_distributed"]


def _open_state(path: str):
    """Return a dict-like reader over a safetensors file or a .pt/.bin."""
    if os.path.isdir(path):
        for cand in ("model.safetensors", "pytorch_model.bin"):
            p = os.path.join(path, cand)
            if os.path.exists(p):
                path = p
                break
    if path.endswith(".safetensors"):
        from safetensors import safe_open

        f = safe_open(path, framework="pt", device="cpu")

        class Reader:
            def keys(self):
                return f.keys()

            def get(self, k):
                return f.get_tensor(k)

        return Reader()
    sd = torch.load(path, map_location="cpu", weights_only=False)
    if "model_state_dict" in sd:
        sd = sd["model_state_dict"]

    class DictReader:
        def keys(self):
            return sd.keys()

        def get(self, k):
            return sd[k]

    return DictReader()


def _strip(k: str) -> str:
    return k[len("transformer."):] if k.startswith("transformer.") else k


def load_gpt2_distributed(
    checkpoint_path: str,
    config,
    pp_rank: int,
    pp_size: int,
    tp_rank: int,
    tp_size: int,
) -> Dict[str, torch.Tensor]:
    """Build this rank's GPT2Stage state dict from an HF GPT-2 checkpoint."""
    from ..parallel.pipeline.wrapper import distribute_layers

    reader = _open_state(checkpoint_path)
    keymap = {_strip(k): k for k in reader.keys()}

    def get(name: str) -> torch.Tensor:
        return reader.get(keymap[name]).float()

    H = config.n_embd
    h_loc = H // tp_size
    inner_loc = config.n_inner // tp_size
    layers = distribute_layers(config.n_layer, pp_size)[pp_rank]
    out: Dict[str, torch.Tensor] = {}

    is_first = pp_rank == 0
    is_last = pp_rank == pp_size - 1

    def _pad_vocab(w: torch.Tensor) -> torch.Tensor:
        # padded-vocab layout (config.vocab_pad_to): HF files carry the
        # logical vocab rows; zero-fill the pad rows (they are masked out
        # of the logits and never receive gradient — models/gpt2/stage.py)
        pv = getattr(config, "padded_vocab_size", w.shape[0])
        if pv == w.shape[0]:
            return w
        return torch.cat([w, w.new_zeros(pv - w.shape[0], w.shape[1])], dim=0)

    if is_first:
        out["embedding.wte.weight"] = _pad_vocab(get("wte.weight"))
        out["embedding.wpe.weight"] = get("wpe.weight")
    if is_last:
        out["ln_f.weight"] = get("ln_f.weight")
        out["ln_f.bias"] = get("ln_f.bias")
        if pp_size > 1:
            out["lm_head"] = _pad_vocab(get("wte.weight").clone())

    for local_idx, gl in enumerate(layers):
        src = f"h.{gl}"
        dst = f"blocks.{local_idx}"
        for ln in ("ln_1", "ln_2"):
            out[f"{dst}.{ln}.weight"] = get(f"{src}.{ln}.weight")
            out[f"{dst}.{ln}.bias"] = get(f"{src}.{ln}.bias")

        # c_attn: HF Conv1D [in, 3H] -> Linear [3H, in]; per-head TP slice
        w = get(f"{src}.attn.c_attn.weight").t().contiguous()  # [3H, H]
        b = get(f"{src}.attn.c_attn.bias")  # [3H]
        qw, kw, vw = w.chunk(3, dim=0)
        qb, kb, vb = b.chunk(3, dim=0)
        sl = slice(tp_rank * h_loc, (tp_rank + 1) * h_loc)
        out[f"{dst}.attn.c_attn.weight"] = torch.cat([qw[sl], kw[sl], vw[sl]], dim=0)
        out[f"{dst}.attn.c_attn.bias"] = torch.cat([qb[sl], kb[sl], vb[sl]], dim=0)

        # attn c_proj: row-parallel — slice input dim; bias replicated
        w = get(f"{src}.attn.c_proj.weight").t().contiguous()  # [H, H]
        out[f"{dst}.attn.c_proj.weight"] = w[:, sl].contiguous()
        out[f"{dst}.attn.c_proj.bias"] = get(f"{src}.attn.c_proj.bias")

        # mlp c_fc: column-parallel
        w = get(f"{src}.mlp.c_fc.weight").t().contiguous()  # [4H, H]
        b = get(f"{src}.mlp.c_fc.bias")
        isl = slice(tp_rank * inner_loc, (tp_rank + 1) * inner_loc)
        out[f"{dst}.mlp.c_fc.weight"] = w[isl].contiguous()
        out[f"{dst}.mlp.c_fc.bias"] = b[isl].contiguous()

        # mlp c_proj: row-parallel
        w = get(f"{src}.mlp.c_proj.weight").t().contiguous()  # [H, 4H]
        out[f"{dst}.mlp.c_proj.weight"] = w[:, isl].contiguous()
        out[f"{dst}.mlp.c_proj.bias"] = get(f"{src}.mlp.c_proj.bias")

    return out
