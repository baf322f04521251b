import os
import sys

# Load torch's ROCm runtime stack before any test module imports lux_amd /
# numpy native libs (see lux_amd/__init__.py note: other load orders break
# HIP device detection process-wide on the GPU boxes).
import torch  # noqa: F401
import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if ROOT not in sys.path:
    sys.path.insert(0, ROOT)


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: needs an MI355X GPU (run via gpurun / driver)")


def pytest_collection_modifyitems(config, items):
    try:
        import torch
        has_gpu = torch.cuda.is_available()
    except Exception:
        has_gpu = False
    if has_gpu:
        return
    skip = pytest.mark.skip(reason="no GPU in this container")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)
