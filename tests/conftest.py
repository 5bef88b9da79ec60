import os
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if ROOT not in sys.path:
    sys.path.insert(0, ROOT)


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: needs a real MI355X (run via gpurun / driver)")


def pytest_collection_modifyitems(config, items):
    # skip gpu tests automatically when no device is visible and the gpu
    # marker was not explicitly requested
    if config.option.markexpr:
        return
    try:
        import ceph_amd
        has_gpu = ceph_amd.device_count() > 0
    except Exception:
        has_gpu = False
    if not has_gpu:
        skip = pytest.mark.skip(reason="no GPU visible")
        for item in items:
            if "gpu" in item.keywords:
                item.add_marker(skip)
