import os
import sys

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: needs a ROCm GPU (run on the MI355X box)"
    )


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this container")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture(autouse=True)
def _eager_on_cpu(monkeypatch):
    """CPU tests always run the eager oracle; GPU tests opt in to hip."""
    from wap_amd import ops
    if not torch.cuda.is_available():
        ops.set_backend("eager")
    yield


@pytest.fixture
def vocab():
    from wap_amd.data import Vocab
    return Vocab.default()
