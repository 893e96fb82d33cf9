"""Pytest configuration: registers the `gpu` marker and shared fixtures.

CPU tests (the default) must run in a GPU-less container; tests marked
`@pytest.mark.gpu` run only on a real MI355X box (`pytest -m gpu`).
"""
import logging
import os
import sys

import pytest

# repo root on sys.path so `import pipeedge_amd` and root apps work from tests/
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

logging.disable(logging.WARNING)


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires a ROCm GPU (run with `pytest -m gpu`)")


def pytest_collection_modifyitems(config, items):
    """Skip gpu-marked tests automatically when no GPU is present."""
    import torch
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU available")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture(scope="session")
def tmp_weights_dir(tmp_path_factory):
    return tmp_path_factory.mktemp("weights")
