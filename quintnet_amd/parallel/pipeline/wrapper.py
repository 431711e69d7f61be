"""Pipeline stage wrapper: splits a model into per-rank stages.

Parity with reference parallelism/pipeline_parallel/wrapper.py:40-250.
Contract: the model exposes ``.embedding``, ``.blocks`` (ModuleList) and
a head (``.classification_head`` / ``.head`` / ``.lm_head``); blocks are
distributed evenly with the remainder going to EARLY stages, the first
stage gets the embedding, the last the head.  A pre-built
``stage_module`` (e.g. GPT2Stage from a staged checkpoint load) can be
passed instead.
"""

from __future__ import annotations

from typing import List, Optional

import torch
import torch.nn as nn

__all__ = ["PipelineParallelWrapper", "InterleavedPipelineWrapper", "distribute_layers"]


def distribute_layers(depth: int, pp_size: int) -> List[List[int]]:
    """Even block split; remainder blocks go to early stages."""
    base = depth // pp_size
    rem = depth % pp_size
    out: List[List[int]] = []
    start = 0
    for r in range(pp_size):
        n = base + (1 if r < rem else 0)
        out.append(list(range(start, start + n)))
        start += n
    return out


def _find_head(model: nn.Module) -> Optional[nn.Module]:
    for name in ("classification_head", "head", "lm_head"):
        if hasattr(model, name):
            return getattr(model, name)
    return None


class PipelineParallelWrapper(nn.Module):
    def __init__(
        self,
        model: Optional[nn.Module] = None,
        pp_rank: int = 0,
        pp_group=None,
        pp_size: int = 1,
        device: Optional[torch.device] = None,
        stage_module: Optional[nn.Module] = None,
    ):
        super().__init__()
        self.pp_rank = pp_rank
        self.pp_size = pp_size
        self.pp_group = pp_group
        self.device = device
        self.is_first_stage = pp_rank == 0
        self.is_last_stage = pp_rank == pp_size - 1

        # shape hints for the schedules (inter-stage activation is [B, seq, hidden])
        src = model if model is not None else stage_module
        self.seq_len = getattr(src, "seq_len", None)
        self.hidden_dim = getattr(src, "hidden_dim", None)

        if stage_module is not None:
            self.local_module = stage_module
            self.layer_distribution = None
        else:
            assert model is not None, "need model or stage_module"
            if not hasattr(model, "blocks"):
                raise ValueError(
                    "PipelineParallelWrapper needs a model with a .blocks ModuleList"
                )
            depth = len(model.blocks)
            self.layer_distribution = distribute_layers(depth, pp_size)
            my_blocks = self.layer_distribution[pp_rank]
            mods: List[nn.Module] = []
            if self.is_first_stage and hasattr(model, "embedding"):
                mods.append(model.embedding)
            mods.extend(model.blocks[i] for i in my_blocks)
            if self.is_last_stage:
                head = _find_head(model)
                if head is not None:
                    mods.append(head)
            self.local_module = nn.Sequential(*mods)
        if device is not None:
            self.local_module.to(device)

    # ------------------------------------------------------------------
    def forward(self, x):
        return self.local_module(x)

    def backward(self, input_tensor, output_tensor, output_tensor_grad):
        """Manual micro-batch backward (reference wrapper.py:214-250).

        On the last stage ``output_tensor`` is the (scaled) loss and
        ``output_tensor_grad`` is None; elsewhere it is the stage output
        with the grad received from the next stage.
        """
        if input_tensor is not None and not input_tensor.requires_grad:
            raise RuntimeError("pipeline input tensor must require grad")
        if input_tensor is not None:
            input_tensor.retain_grad()
        torch.autograd.backward(output_tensor, grad_tensors=output_tensor_grad)
        return input_tensor.grad if input_tensor is not None else None


class InterleavedPipelineWrapper(nn.Module):
    """Virtual-pipeline stage holder for the interleaved-1F1B schedule.

    The model's blocks are split into ``pp_size * num_chunks`` global
    stages; this rank holds stages ``{c * pp_size + pp_rank}`` for chunk
    c in [0, num_chunks) (Megatron-style assignment: inter-stage edges
    are always rank r -> r+1 with a ring wrap between chunks).  Global
    stage 0 gets the embedding, the last global stage the head.  Beyond
    reference parity (the reference has no interleaved schedule).
    """

    def __init__(
        self,
        model: nn.Module,
        pp_rank: int = 0,
        pp_group=None,
        pp_size: int = 1,
        num_chunks: int = 2,
        device: Optional[torch.device] = None,
        tied_group=None,
    ):
        super().__init__()
        if not hasattr(model, "blocks"):
            raise ValueError("InterleavedPipelineWrapper needs a model with .blocks")
        self.tied_group = tied_group
        self.pp_rank = pp_rank
        self.pp_size = pp_size
        self.pp_group = pp_group
        self.num_chunks = num_chunks
        self.device = device
        self.seq_len = getattr(model, "seq_len", None)
        self.hidden_dim = getattr(model, "hidden_dim", None)

        n_stages = pp_size * num_chunks
        depth = len(model.blocks)
        if depth < n_stages:
            raise ValueError(
                f"{depth} blocks cannot fill {n_stages} virtual stages"
            )
        self.layer_distribution = distribute_layers(depth, n_stages)
        chunks: List[nn.Module] = []
        for c in range(num_chunks):
            g = c * pp_size + pp_rank
            mods: List[nn.Module] = []
            if g == 0 and hasattr(model, "embedding"):
                mods.append(model.embedding)
            mods.extend(model.blocks[i] for i in self.layer_distribution[g])
            if g == n_stages - 1:
                head = _find_head(model)
                if head is not None:
                    mods.append(head)
            chunks.append(nn.Sequential(*mods))
        self.chunks = nn.ModuleList(chunks)
        if device is not None:
            self.chunks.to(device)

        # tied embedding/LM-head: global stage 0 holds the embedding
        # (module with .wte), the last global stage a TiedLMHead-style
        # module (.tied_weight); when they live on DIFFERENT ranks their
        # grads must be summed over the first+last-stage subgroup.
        self._tied_params: List[torch.nn.Parameter] = []
        for chunk in self.chunks:
            for m in chunk.modules():
                if hasattr(m, "tied_weight"):
                    self._tied_params.append(m.tied_weight)
                    if pp_size > 1 and pp_rank != 0:
                        # head-side copy of the tied weight: after the
                        # tied-group grad sync it duplicates stage 0's
                        # embedding grad — count once in the global norm
                        m.tied_weight._tied_copy = True
                elif hasattr(m, "wte") and isinstance(getattr(m, "wte"), nn.Embedding):
                    self._tied_params.append(m.wte.weight)

    def sync_tied_weights_grad(self) -> None:
        import torch.distributed as dist

        if self.tied_group is None or len(self._tied_params) != 1:
            return  # both copies local (p==1) or no tied weight
        p = self._tied_params[0]
        if p.grad is not None:
            dist.all_reduce(p.grad, op=dist.ReduceOp.SUM, group=self.tied_group)

    def forward(self, x, chunk_id: int = 0):
        return self.chunks[chunk_id](x)

    # same manual-backward contract as PipelineParallelWrapper
    backward = PipelineParallelWrapper.backward
