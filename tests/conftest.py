import os
import sys

import pytest
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

REFERENCE_DIR = "/root/reference"


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs an MI355X GPU (run via gpurun)")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this container")
    for item in items:
        if "gpu" in item.keywords and not any(
            m in (config.getoption("-m") or "") for m in ("gpu",)
        ):
            item.add_marker(skip)


@pytest.fixture
def reference_dir():
    if not os.path.isdir(REFERENCE_DIR):
        pytest.skip("reference repo not mounted")
    return REFERENCE_DIR
