import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run via gpurun)")


def have_gpu():
    try:
        import minio_amd
        return minio_amd.device_count() > 0
    except ImportError:
        return False


def pytest_collection_modifyitems(config, items):
    if have_gpu():
        return
    skip = pytest.mark.skip(reason="no GPU visible")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)
