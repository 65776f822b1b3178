import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run with -m gpu on a GPU box)"
    )


def pytest_collection_modifyitems(config, items):
    # -m "not gpu" / -m gpu filtering is done by pytest; additionally skip
    # gpu-marked tests automatically when no GPU is present and no -m was given.
    if config.getoption("-m"):
        return
    try:
        import torch
        has_gpu = torch.cuda.is_available()
    except Exception:
        has_gpu = False
    if not has_gpu:
        skip = pytest.mark.skip(reason="no GPU in this container")
        for item in items:
            if "gpu" in item.keywords:
                item.add_marker(skip)
