"""GPT-2 pipeline stage container (reference utils/GPT2/gpt2_stage.py).

First stage owns the embedding (wte+wpe); the last stage owns ln_f and
lm_head — a tied COPY of wte when pp>1 (its gradient is averaged with
the embedding's over the first+last-stage subgroup each step —
``sync_tied_weights_grad``; reference gpt2_stage.py:112-141, fixed to a
dedicated 2-rank group per SURVEY.md §8.3 so pp>2 cannot deadlock).
With pp==1 lm_head IS wte (true tying, no sync needed).

State-dict naming contract (kept for the merge CLI / checkpoint layout
parity): ``embedding.wte.weight``, ``embedding.wpe.weight``,
``blocks.N.{ln_1,attn.c_attn,attn.c_proj,ln_2,mlp.c_fc,mlp.c_proj}.*``,
``ln_f.*``, ``lm_head.weight``.
"""

from __future__ import annotations

from typing import Dict, List

import torch
import torch.distributed as dist
import torch.nn as nn

from ...core.comm import All_Gather, scatter_to_sequence
from ...ops import FusedLayerNorm, linear as fused_linear
from ...parallel.pipeline.wrapper import distribute_layers
from .block import GPT2Block
from .config import GPT2Config
from .embeddings import GPT2Embedding

__all__ = ["GPT2Stage", "mask_pad_logits"]


def mask_pad_logits(logits: torch.Tensor, config: GPT2Config) -> torch.Tensor:
    """-inf the pad columns of a padded-vocab logits tensor (no-op when
    vocab_pad_to is off).  exp(-inf)=0 makes the CE/logsumexp/argmax math
    EXACTLY the unpadded model's, and the pad columns get zero gradient
    (so the zero-init pad rows of wte/lm_head never train).  Writes 47
    columns of 50304 — ~3 MB at bench shape, noise next to the 1.16 ms
    the aligned GEMM/CE rows save (tools/probe_vocab_pad.py)."""
    real = config.vocab_size
    if logits.shape[-1] == real:
        return logits
    assert not logits.requires_grad, (
        "grad-mode logits must be masked via linear(..., logical_out=) "
        "(in-place on a reshape view costs a full-logits CopySlices clone)"
    )
    logits[..., real:] = float("-inf")
    return logits


class GPT2Stage(nn.Module):
    def __init__(
        self,
        config: GPT2Config,
        pp_rank: int = 0,
        pp_size: int = 1,
        tp_group=None,
        tied_group=None,
        device=None,
        dtype=None,
        cp_group=None,
        ep_group=None,
    ):
        super().__init__()
        self.config = config
        self.pp_rank = pp_rank
        self.pp_size = pp_size
        self.tp_group = tp_group
        self.tied_group = tied_group
        self.cp_group = cp_group
        if cp_group is not None:
            assert not config.sequence_parallel, "CP and Megatron-SP are exclusive"
        if getattr(config, "activation_checkpointing", False):
            assert not config.n_experts, (
                "activation_checkpointing is incompatible with MoE blocks "
                "(aux_loss is read outside the recomputed region)"
            )
        self.sequence_parallel = config.sequence_parallel
        self.is_first_stage = pp_rank == 0
        self.is_last_stage = pp_rank == pp_size - 1
        kw = {"device": device, "dtype": dtype}

        self.layer_distribution = distribute_layers(config.n_layer, pp_size)
        self.my_layers: List[int] = self.layer_distribution[pp_rank]

        if self.is_first_stage:
            self.embedding = GPT2Embedding(config, **kw)
        self.blocks = nn.ModuleList(
            GPT2Block(config, tp_group=tp_group, cp_group=cp_group,
                      ep_group=ep_group, **kw)
            for _ in self.my_layers
        )
        if self.is_last_stage:
            self.ln_f = FusedLayerNorm(config.n_embd, eps=config.layer_norm_epsilon, **kw)
            if self.is_first_stage:
                # pp==1: true weight tying
                self.lm_head = None
            else:
                self.lm_head = nn.Parameter(
                    torch.empty(config.padded_vocab_size, config.n_embd, **kw)
                )
                nn.init.normal_(self.lm_head, std=config.initializer_range)
                if config.padded_vocab_size != config.vocab_size:
                    with torch.no_grad():
                        self.lm_head[config.vocab_size:].zero_()
                # grad-norm bookkeeping: after sync_tied_weights_grad this
                # grad equals the first stage's wte grad — count it once
                # (on the first stage) in the PP-reduced global norm
                self.lm_head._tied_copy = True

    # ------------------------------------------------------------------
    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self.is_first_stage:
            if self.cp_group is not None:
                # context parallelism: x is this rank's ids SHARD (caller
                # scatters, see parallel/context_parallel.py); positions
                # start at the shard's global offset (contiguous shards)
                # or follow the zigzag chunk map
                import torch.distributed as dist

                if getattr(self.config, "cp_zigzag", False):
                    from ...parallel.context_parallel import zigzag_positions

                    cp = dist.get_world_size(group=self.cp_group)
                    pos = zigzag_positions(x.shape[1] * cp, self.cp_group,
                                           x.device)
                    x = self.embedding(x, positions=pos)
                else:
                    cp_rank = dist.get_rank(group=self.cp_group)
                    x = self.embedding(x, pos_offset=cp_rank * x.shape[1])
            else:
                x = self.embedding(x)
            if self.sequence_parallel:
                # enter SP: keep only this rank's sequence shard
                x = scatter_to_sequence(x, self.tp_group, 1)
        # residual-fused block chain: each block's trailing residual add
        # rides the NEXT LayerNorm kernel (ops/layernorm.py)
        ckpt = (
            getattr(self.config, "activation_checkpointing", False)
            and self.training
            and torch.is_grad_enabled()
            # ZeRO-3 blocks already recompute inside their own gather
            # region — a second wrapper would recompute the recompute
            and not (len(self.blocks)
                     and type(self.blocks[0]).__name__ == "ZeRO3Block")
        )
        pending = None
        for blk in self.blocks:
            if ckpt:
                from torch.utils.checkpoint import checkpoint

                pending, x = checkpoint(
                    blk.forward_fused, x, pending, use_reentrant=False
                )
            else:
                pending, x = blk.forward_fused(x, pending)
        if self.is_last_stage:
            if pending is None:
                x = self.ln_f(x)
            else:
                x, _ = self.ln_f(x, residual=pending)
            if self.sequence_parallel:
                # leave SP: full sequence for the LM head.  Backward mode
                # is "slice": everything downstream (lm_head + loss) is
                # REPLICATED across TP ranks, so each rank already holds
                # the complete grad — reduce-scatter would double-count.
                x = All_Gather.apply(x, self.tp_group, 1, "slice")
            w = self.embedding.wte.weight if self.lm_head is None else self.lm_head
            # plain GEMM: hipBLASLt; logical_out masks padded-vocab columns
            x = fused_linear(x, w, None, None, prefer_library=True,
                             logical_out=self.config.vocab_size)
        elif pending is not None:
            x = x + pending  # fold before the PP send
        return x

    # ------------------------------------------------------------------
    @torch.no_grad()
    def generate(
        self,
        input_ids: torch.Tensor,
        max_new_tokens: int = 32,
        temperature: float = 0.0,
        top_k: int = 0,
        top_p: float = 0.0,
        repetition_penalty: float = 1.0,
        eos_token_id: int = None,
        cache_dtype: str = None,
    ) -> torch.Tensor:
        """KV-cached autoregressive generation (serving path; pp==1).

        temperature 0 = greedy; otherwise softmax sampling with optional
        top-k and/or top-p (nucleus) truncation.  Works under TP (all TP ranks call this and run
        the same collectives; sampling is made rank-consistent by seeding
        from the tokens).  The cache attends via the q_offset attention
        path (models/gpt2/attention.py), positions via wpe slicing.
        """
        assert self.is_first_stage and self.is_last_stage, "generate needs pp==1"
        assert not self.sequence_parallel, "generate: disable sequence_parallel"
        assert input_ids.shape[1] <= self.config.n_positions, (
            "prompt longer than the model's context window"
        )
        was_training = self.training
        self.eval()
        caches = [
            {"k": None, "v": None, "k8": None,
             "quant": "int8" if cache_dtype == "int8" else None}
            for _ in self.blocks
        ]
        ids = input_ids
        out = input_ids
        past = 0
        try:
            for _ in range(max_new_tokens):
                x = self.embedding(ids, pos_offset=past)
                for blk, cache in zip(self.blocks, caches):
                    x = blk.forward_cached(x, cache)
                x = self.ln_f(x[:, -1:])
                w = self.embedding.wte.weight if self.lm_head is None else self.lm_head
                logits = fused_linear(x, w, None, None, prefer_library=True)[:, -1]
                logits = mask_pad_logits(logits, self.config)
                if repetition_penalty and repetition_penalty != 1.0:
                    # CTRL-style: push down tokens already in the output
                    logits = logits.float()
                    seen = logits.gather(1, out)
                    seen = torch.where(seen > 0, seen / repetition_penalty,
                                       seen * repetition_penalty)
                    logits = logits.scatter(1, out, seen)
                # temperatures below 1e-5 behave as greedy (dividing by
                # them overflows the logits to inf -> NaN softmax)
                if temperature and temperature > 1e-5:
                    logits = logits.float() / temperature
                    if top_k and top_k > 0:
                        k_ = min(int(top_k), logits.shape[-1])
                        kth = torch.topk(logits, k_, dim=-1).values[..., -1:]
                        logits = logits.masked_fill(logits < kth, float("-inf"))
                    if top_p and 0.0 < top_p < 1.0:
                        # nucleus: keep the smallest prefix of the sorted
                        # distribution whose mass reaches top_p (always
                        # keeps the argmax)
                        sl, si = torch.sort(logits, dim=-1, descending=True)
                        cum = torch.softmax(sl, dim=-1).cumsum(dim=-1)
                        drop_sorted = cum - torch.softmax(sl, -1) >= top_p
                        # tiny top_p underflows to 0 in fp32 (0 >= 0 would
                        # drop EVERYTHING): the argmax is always kept
                        drop_sorted[..., 0] = False
                        drop = torch.zeros_like(drop_sorted).scatter(
                            -1, si, drop_sorted
                        )
                        logits = logits.masked_fill(drop, float("-inf"))
                    # rank-consistent sampling under TP: derive the RNG from
                    # the current sequence so every TP rank draws the same
                    gen = torch.Generator(device="cpu")
                    gen.manual_seed(int(out.sum().item()) & 0x7FFFFFFF)
                    probs = torch.softmax(logits, dim=-1).cpu()
                    nxt = torch.multinomial(probs, 1, generator=gen).to(out.device)
                else:
                    nxt = logits.argmax(dim=-1, keepdim=True)
                past += ids.shape[1]
                ids = nxt
                out = torch.cat([out, nxt], dim=1)
                if eos_token_id is not None and bool((nxt == eos_token_id).all()):
                    break
                if past + 1 >= self.config.n_positions:
                    break
        finally:
            if was_training:
                self.train()
        return out

    # ------------------------------------------------------------------
    def moe_aux_loss(self) -> torch.Tensor:
        """Sum of the MoE load-balancing losses of this stage's blocks
        (zero tensor when the model is dense)."""
        total = None
        for blk in self.blocks:
            aux = getattr(blk.mlp, "aux_loss", None)
            if aux is not None:
                total = aux if total is None else total + aux
        if total is None:
            dev = next(self.parameters()).device
            return torch.zeros((), device=dev)
        return total

    # ------------------------------------------------------------------
    def sync_sequence_parallel_grads(self) -> None:
        """Under SP the LayerNorms and row-parallel biases compute their
        grads from sequence SHARDS — all-reduce them over the TP group
        (the Megatron "sequence-parallel params" sync)."""
        if not self.sequence_parallel or self.tp_group is None:
            return
        if not dist.is_initialized() or dist.get_world_size(group=self.tp_group) == 1:
            return
        params = []
        for blk in self.blocks:
            params += [blk.ln_1.weight, blk.ln_1.bias, blk.ln_2.weight, blk.ln_2.bias]
            params += [blk.attn.c_proj.bias, blk.mlp.c_proj.bias]
        if self.is_last_stage:
            params += [self.ln_f.weight, self.ln_f.bias]
        for p in params:
            if p is not None and p.grad is not None:
                dist.all_reduce(p.grad, op=dist.ReduceOp.SUM, group=self.tp_group)

    def sync_tied_weights_grad(self) -> None:
        """Average the tied wte/lm_head gradient between first & last
        stage (and run the SP grad sync on every stage)."""
        self.sync_sequence_parallel_grads()
        if self.pp_size == 1 or not (self.is_first_stage or self.is_last_stage):
            return
        if not dist.is_initialized():
            return
        grad = None
        if self.is_first_stage and self.embedding.wte.weight.grad is not None:
            grad = self.embedding.wte.weight.grad
        elif self.is_last_stage and self.lm_head is not None and self.lm_head.grad is not None:
            grad = self.lm_head.grad
        if grad is None:
            return
        group = self.tied_group
        dist.all_reduce(grad, op=dist.ReduceOp.SUM, group=group)
        grad.div_(2.0)

    # ------------------------------------------------------------------
    @classmethod
    def from_sharded_state_dict(
        cls,
        config: GPT2Config,
        state_dict: Dict[str, torch.Tensor],
        pp_rank: int,
        pp_size: int,
        tp_group=None,
        tied_group=None,
        device=None,
        dtype=None,
    ) -> "GPT2Stage":
        """Build a stage and load pre-sharded (rank-local) tensors.

        ``state_dict`` uses the stage-local naming contract above with
        already-TP-sliced c_attn/c_fc/c_proj tensors (see
        checkpoint/distributed_loading.py).
        """
        stage = cls(
            config,
            pp_rank=pp_rank,
            pp_size=pp_size,
            tp_group=tp_group,
            tied_group=tied_group,
            device=device,
            dtype=dtype,
        )
        tgt = stage.state_dict()
        missing = [k for k in state_dict if k not in tgt]
        if missing:
            raise KeyError(f"staged load: unexpected keys {missing[:5]}...")
        stage.load_state_dict(state_dict, strict=False)
        return stage
