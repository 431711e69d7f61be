"""GPT-2 MLP with Megatron TP (reference utils/GPT2/gpt2_mlp.py:51-162).

c_fc is column-parallel with the GELU fused into the GEMM epilogue
(gather_output=False); c_proj is row-parallel with the closing
all-reduce.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from ...ops import FusedDropout
from ...parallel.tensor_parallel import ColumnParallelLinear, RowParallelLinear
from .config import GPT2Config

__all__ = ["GPT2MLP"]


class GPT2MLP(nn.Module):
    def __init__(self, config: GPT2Config, tp_group=None, device=None, dtype=None):
        super().__init__()
        sp = config.sequence_parallel
        self.c_fc = ColumnParallelLinear(
            config.n_embd,
            config.n_inner,
            tp_group=tp_group,
            gather_output=False,
            activation="gelu",  # fused into the MFMA GEMM epilogue
            sequence_parallel=sp,
            device=device,
            dtype=dtype,
        )
        self.c_proj = RowParallelLinear(
            config.n_inner,
            config.n_embd,
            tp_group=tp_group,
            input_is_parallel=True,
            sequence_parallel=sp,
            device=device,
            dtype=dtype,
        )
        self.c_fc.fp8 = self.c_proj.fp8 = config.fp8
        self.dropout = FusedDropout(config.dropout)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.dropout(self.c_proj(self.c_fc(x)))
