"""Hang/failure detection for distributed training (the
reference's only recovery story was torchrun restart + timeouts).

``Watchdog`` is a daemon-thread heartbeat monitor: the training loop
calls ``beat()`` every step; if no beat arrives within ``timeout_s`` the
watchdog dumps all Python thread stacks (faulthandler) to stderr —
turning a silent RCCL/collective deadlock into an actionable trace —
and optionally kills the process so torchrun's elastic agent can
restart the job instead of burning the full collective timeout.
"""

from __future__ import annotations

import faulthandler
import os
import sys
import threading
import time

__all__ = ["Watchdog"]


class Watchdog:
    def __init__(self, timeout_s: float = 600.0, kill_on_hang: bool = False):
        self.timeout_s = float(timeout_s)
        self.kill_on_hang = kill_on_hang
        self._last = time.monotonic()
        self._stop = threading.Event()
        self._fired = threading.Event()
        self._thread: threading.Thread | None = None

    # ------------------------------------------------------------------
    def start(self) -> "Watchdog":
        self._last = time.monotonic()
        self._thread = threading.Thread(target=self._run, daemon=True,
                                        name="quintnet-watchdog")
        self._thread.start()
        return self

    def beat(self) -> None:
        self._last = time.monotonic()

    def stop(self) -> None:
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=2.0)

    @property
    def fired(self) -> bool:
        return self._fired.is_set()

    def __enter__(self) -> "Watchdog":
        return self.start()

    def __exit__(self, *exc) -> None:
        self.stop()

    # ------------------------------------------------------------------
    def _run(self) -> None:
        poll = min(self.timeout_s / 4.0, 5.0)
        while not self._stop.wait(poll):
            if time.monotonic() - self._last > self.timeout_s:
                self._fired.set()
                rank = os.environ.get("RANK", "?")
                print(
                    f"[quintnet watchdog] rank {rank}: no heartbeat for "
                    f"{self.timeout_s:.0f}s — dumping thread stacks",
                    file=sys.stderr, flush=True,
                )
                try:
                    faulthandler.dump_traceback(file=sys.stderr)
                except Exception:
                    # captured/wrapped stderr has no fileno (pytest, some
                    # launcher tees) — fall back to the traceback module
                    import traceback

                    for tid, frame in sys._current_frames().items():
                        print(f"--- thread {tid} ---", file=sys.stderr)
                        traceback.print_stack(frame, file=sys.stderr)
                if self.kill_on_hang:
                    os._exit(42)  # let the elastic agent restart the job
                self._last = time.monotonic()  # rearm


def assert_finite_grads(model) -> None:
    """Raise naming the first parameter whose gradient is NaN/Inf —
    failure-detection companion to the hang watchdog (enable with
    config ``detect_nan_grads: true``; costs one reduction per param
    per optimizer step, so leave off in production benches)."""
    import torch

    for name, p in model.named_parameters():
        if p.grad is not None and not bool(torch.isfinite(p.grad).all()):
            raise FloatingPointError(
                f"non-finite gradient in {name!r} (shape {tuple(p.grad.shape)}):"
                " divergence or bad input batch — lower the lr, re-check the"
                " data pipeline, or resume from the last checkpoint"
            )
