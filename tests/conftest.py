import os
import sys

import pytest

# make the repo root importable regardless of invocation dir
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an AMD GPU (run on an MI355X box)"
    )
    config.addinivalue_line(
        "markers",
        "gpu_experimental: GPU test for code pending hardware validation "
        "(not part of the standard gpu tier)",
    )


def pytest_collection_modifyitems(config, items):
    import torch

    if torch.cuda.is_available():
        return
    skip_gpu = pytest.mark.skip(reason="no GPU available")
    for item in items:
        if "gpu" in item.keywords or "gpu_experimental" in item.keywords:
            item.add_marker(skip_gpu)
