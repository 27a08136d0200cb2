import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X / HIP device")


def pytest_collection_modifyitems(config, items):
    try:
        import cimba_amd

        has_gpu = cimba_amd.has_gpu()
    except Exception:
        has_gpu = False
    skip_gpu = pytest.mark.skip(reason="no HIP device visible")
    for item in items:
        if "gpu" in item.keywords and not has_gpu:
            item.add_marker(skip_gpu)
