"""Loader for the in-tree HIP extension (quintnet_amd._C, gfx950).

Policy: on a GPU (ROCm) box the hand-written CDNA4 kernels are THE
compute path — if the extension is missing we raise loudly rather than
silently fall back to eager PyTorch.  On CPU (unit tests, gloo
multi-process logic tests) ops use their PyTorch reference
implementations.

Set ``QUINTNET_FORCE_EAGER=1`` to force the PyTorch path everywhere
(debugging only).
"""

from __future__ import annotations

import os
from typing import Optional

import torch

_EXT = None
_EXT_ERR: Optional[BaseException] = None
_TRIED = False


def _try_load():
    global _EXT, _EXT_ERR, _TRIED
    if _TRIED:
        return
    _TRIED = True
    try:
        from quintnet_amd import _C  # built in-tree by setup.py build_ext --inplace

        _EXT = _C
    except Exception as e:  # noqa: BLE001
        _EXT_ERR = e


def has_ext() -> bool:
    _try_load()
    return _EXT is not None


def ext():
    _try_load()
    if _EXT is None:
        raise RuntimeError(
            "quintnet_amd._C HIP extension is not built. Build it in-tree with "
            "`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950). "
            f"Import error was: {_EXT_ERR!r}"
        )
    return _EXT


def force_eager() -> bool:
    return os.environ.get("QUINTNET_FORCE_EAGER", "0") == "1"


def use_native(*tensors: torch.Tensor) -> bool:
    """True when the hand-written HIP path must run (any input on GPU)."""
    if force_eager():
        return False
    return any(isinstance(t, torch.Tensor) and t.is_cuda for t in tensors)
