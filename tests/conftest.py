import os
import sys

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: needs a real MI355X GPU (run via gpurun)")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this container")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture(autouse=True)
def _repo_root_cwd():
    """Keep cwd at the repo root (several tests use relative model paths
    and some e2e tests chdir into tmp dirs)."""
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    prev = os.getcwd()
    os.chdir(root)
    yield
    os.chdir(prev if os.path.isdir(prev) else root)
