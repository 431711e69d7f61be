"""Test fixtures: gpu marker + self-launching multi-process gloo harness.

Unlike the reference (whose world_size>1 tests silently skip unless the
whole pytest session runs under torchrun — SURVEY.md §4), distributed
logic tests here spawn their own process group over gloo/CPU via
``run_distributed``, so plain ``pytest`` exercises world_size 2-8.
GPU-only tests are marked ``@pytest.mark.gpu``.
"""

from __future__ import annotations

import os
import socket
import sys

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs an MI355X (ROCm) GPU")
    config.addinivalue_line("markers", "slow: long-running test")


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _worker(rank: int, world_size: int, port: int, fn_name: str, mod_name: str, args: tuple, q):
    """Child entry: init gloo, run fn, report exception if any."""
    import importlib
    import traceback

    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    os.environ["LOCAL_RANK"] = str(rank)
    try:
        dist.init_process_group(
            backend="gloo",
            init_method=f"tcp://127.0.0.1:{port}",
            rank=rank,
            world_size=world_size,
        )
        mod = importlib.import_module(mod_name)
        fn = getattr(mod, fn_name)
        fn(rank, world_size, *args)
        q.put((rank, None))
    except Exception:  # noqa: BLE001
        q.put((rank, traceback.format_exc()))
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def run_distributed(fn, world_size: int, *args, timeout: float = 180.0):
    """Spawn ``world_size`` gloo/CPU processes running module-level ``fn``.

    ``fn`` must be a MODULE-LEVEL function (picklable by name) with
    signature ``fn(rank, world_size, *args)``.
    """
    import torch.multiprocessing as mp

    # children are fresh interpreters: make tests/ and the repo root importable
    here = os.path.dirname(os.path.abspath(__file__))
    root = os.path.dirname(here)
    extra = f"{here}{os.pathsep}{root}"
    prev = os.environ.get("PYTHONPATH")
    os.environ["PYTHONPATH"] = f"{extra}{os.pathsep}{prev}" if prev else extra

    port = _free_port()
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [
        ctx.Process(
            target=_worker,
            args=(r, world_size, port, fn.__name__, fn.__module__, args, q),
            daemon=False,
        )
        for r in range(world_size)
    ]
    for p in procs:
        p.start()
    errs = []
    for _ in range(world_size):
        rank, err = q.get()
        if err is not None:
            errs.append(f"--- rank {rank} ---\n{err}")
    for p in procs:
        p.join(timeout)
        if p.is_alive():
            p.terminate()
            errs.append(f"rank process {p.pid} timed out")
    if errs:
        raise AssertionError("\n".join(errs))


@pytest.fixture
def device():
    if torch.cuda.is_available():
        return torch.device("cuda", int(os.environ.get("LOCAL_RANK", 0)))
    return torch.device("cpu")


@pytest.fixture
def dummy_model():
    return torch.nn.Sequential(
        torch.nn.Linear(10, 16), torch.nn.ReLU(), torch.nn.Linear(16, 5)
    )
