"""First-class step/phase timing (the reference left this as TODO stubs).

``StepTimer`` measures wall step time; ``PhaseTimer`` brackets named
phases (fwd/bwd/comm/opt) with HIP events on GPU so timings are
device-accurate without global synchronizes; kernels launched by this
framework carry informative names for rocprofv3 correlation.
"""

from __future__ import annotations

import time
from collections import defaultdict
from typing import Dict, List

import torch

__all__ = ["StepTimer", "PhaseTimer"]


class StepTimer:
    def __init__(self):
        self.times: List[float] = []
        self._t0 = None

    def start(self):
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        self._t0 = time.perf_counter()

    def stop(self) -> float:
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        dt = time.perf_counter() - self._t0
        self.times.append(dt)
        return dt

    @property
    def mean_ms(self) -> float:
        return 1000.0 * sum(self.times) / max(len(self.times), 1)


class PhaseTimer:
    """HIP-event-based per-phase timing; CPU clock fallback."""

    def __init__(self, use_cuda: bool = None):
        self.use_cuda = torch.cuda.is_available() if use_cuda is None else use_cuda
        self._events: Dict[str, List] = defaultdict(list)
        self._cpu: Dict[str, float] = {}
        self.totals_ms: Dict[str, float] = defaultdict(float)

    def start(self, phase: str):
        if self.use_cuda:
            e = torch.cuda.Event(enable_timing=True)
            e.record()
            self._events[phase].append([e, None])
        else:
            self._cpu[phase] = time.perf_counter()

    def stop(self, phase: str):
        if self.use_cuda:
            e = torch.cuda.Event(enable_timing=True)
            e.record()
            self._events[phase][-1][1] = e
        else:
            self.totals_ms[phase] += 1000.0 * (time.perf_counter() - self._cpu[phase])

    def summary(self) -> Dict[str, float]:
        if self.use_cuda:
            torch.cuda.synchronize()
            for phase, pairs in self._events.items():
                for s, e in pairs:
                    if e is not None:
                        self.totals_ms[phase] += s.elapsed_time(e)
            self._events.clear()
        return dict(self.totals_ms)
