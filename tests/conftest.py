import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test requires an AMD MI355X GPU (run via gpurun)")
    config.addinivalue_line(
        "markers",
        "e2e: full-loop out-of-process e2e (controller + apiserver stand-in + "
        "emulator + prometheus stand-in); CPU-only but slower",
    )


def pytest_collection_modifyitems(config, items):
    try:
        import torch

        has_gpu = torch.cuda.is_available()
    except Exception:
        has_gpu = False
    if has_gpu:
        return
    skip_gpu = pytest.mark.skip(reason="no GPU available in this container")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)
