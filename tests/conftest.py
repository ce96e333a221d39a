import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_addoption(parser):
    parser.addoption("--run-soak", action="store_true", default=False,
                     help="run extended soak/fuzz tests")


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an AMD GPU (run on MI355X box)"
    )
    config.addinivalue_line(
        "markers", "soak: extended fuzz/soak (opt-in via --run-soak)"
    )


def pytest_collection_modifyitems(config, items):
    import torch

    if not config.getoption("--run-soak"):
        skip_soak = pytest.mark.skip(reason="soak tests need --run-soak")
        for item in items:
            if "soak" in item.keywords:
                item.add_marker(skip_soak)
    if torch.cuda.is_available():
        return
    skip_gpu = pytest.mark.skip(reason="no GPU available")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)
