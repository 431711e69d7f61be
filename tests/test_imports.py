"""Smoke: every module in the package imports (reference tests/test_imports.py)."""

import importlib
import pkgutil

import pytest

import quintnet_amd


def _modules():
    out = ["quintnet_amd"]
    for m in pkgutil.walk_packages(quintnet_amd.__path__, prefix="quintnet_amd."):
        out.append(m.name)
    return out


@pytest.mark.parametrize("name", _modules())
def test_import(name):
    importlib.import_module(name)


def test_public_api():
    for sym in [
        "init_process_groups",
        "get_strategy",
        "Trainer",
        "GPT2Trainer",
        "DataParallel",
        "TensorParallel",
        "PipelineParallelWrapper",
        "PipelineTrainer",
        "PipelineDataLoader",
        "ColumnParallelLinear",
        "RowParallelLinear",
        "VocabParallelEmbedding",
        "ZeroRedundancyAdamW",
    ]:
        assert hasattr(quintnet_amd, sym), sym
