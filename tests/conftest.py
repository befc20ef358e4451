import os
import sys

import pytest

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: needs an MI355X (run with `pytest -m gpu` on a GPU box)")


def pytest_collection_modifyitems(config, items):
    if config.getoption("-m") and "gpu" in config.getoption("-m"):
        return
    # nothing else: gpu tests are excluded by `-m "not gpu"` in the driver


@pytest.fixture(scope="session")
def golden():
    import numpy as np
    path = os.path.join(os.path.dirname(__file__), "golden", "golden.npz")
    if not os.path.exists(path):
        pytest.skip("golden.npz missing (run tests/golden/make_golden.py)")
    return np.load(path)


def golden_tags(g):
    return sorted({k.split("/")[0] for k in g.files})
