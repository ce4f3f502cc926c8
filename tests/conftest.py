import os
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
sys.path.insert(0, os.path.join(REPO, "oracle"))

GOLDEN = os.path.join(REPO, "tests", "golden")


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: needs a real MI355X (run with `pytest -m gpu` on the GPU box)"
    )


def pytest_collection_modifyitems(config, items):
    if config.getoption("-m"):
        return
    # default runs (no -m filter) skip gpu tests when no device is present
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


@pytest.fixture(scope="session")
def params15():
    with open(os.path.join(GOLDEN, "params_15"), "rb") as fh:
        data = fh.read()
    assert len(data) == 2097220
    return data
