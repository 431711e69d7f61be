"""hipGraph capture of a whole training step (MI355X launch-bound loops).

``CapturedStep`` records one invocation of a step function into a HIP
graph and replays it afterwards — collapsing thousands of kernel
launches (the ViT-scale regime: hidden 64 kernels take ~2-5 µs but cost
~20 µs of launch each) into a single graph launch.  Fresh inputs are
copied into the captured static buffers before each replay.

Requirements for the step function (same as CUDA graph capture):
static shapes, no host syncs (.item()), no prints; the ZeRO-1 AdamW and
all quintnet_amd ops are capture-safe (device-side step counter /
n_valid — see optim/zero.py, csrc/adamw.hip).
"""

from __future__ import annotations

from typing import Callable, Dict, List, Optional

import torch

__all__ = ["CapturedStep"]


class CapturedStep:
    """Capture ``step_fn`` (which reads from ``static_inputs``) into a
    hipGraph; ``__call__(feed)`` copies ``feed`` tensors into the static
    buffers and replays.

    static_inputs: list of dicts of tensors the step reads (e.g. the
    micro-batches of one optimizer step).
    """

    def __init__(
        self,
        step_fn: Callable[[], None],
        static_inputs: List[Dict[str, torch.Tensor]],
        warmup: int = 2,
    ):
        self.static_inputs = static_inputs
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(warmup):
                step_fn()
        torch.cuda.current_stream().wait_stream(side)
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            step_fn()

    def __call__(self, feed: Optional[List[Dict[str, torch.Tensor]]] = None) -> None:
        if feed is not None:
            for dst, src in zip(self.static_inputs, feed):
                for k, v in dst.items():
                    v.copy_(src[k], non_blocking=True)
        self.graph.replay()
