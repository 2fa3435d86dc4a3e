"""Pytest fixtures for the sparkagd_amd suite.

Tiering mirrors the reference's: local correctness tests run
everywhere on CPU (the analog of Spark `local[2]`); multi-process CPU tests
use the gloo backend with world_size 2 (the analog of `local-cluster[2,1,512]`
— real separate processes, real collectives, no GPU needed); everything that
needs an MI355X is marked `gpu`.
"""

import os
import sys

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test requires an AMD GPU (MI355X)")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip_gpu = pytest.mark.skip(reason="no GPU available")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)


@pytest.fixture
def device():
    return torch.device("cuda:0" if torch.cuda.is_available() else "cpu")


def assert_rel(a: float, b: float, rel: float, msg: str = ""):
    """MLlib TestingUtils `~= relTol` analog (Suite.scala:28,88)."""
    denom = max(abs(a), abs(b), 1e-300)
    assert abs(a - b) / denom <= rel, f"{msg}: |{a} - {b}| relerr {abs(a-b)/denom:.4g} > {rel}"
