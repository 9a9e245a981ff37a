import os
import sys

import numpy as np
import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an AMD GPU (run on an MI355X box)")


@pytest.fixture(scope="session")
def binary_data():
    """Synthetic binary classification set with a nonlinear boundary."""
    rng = np.random.RandomState(0)
    n = 8000
    x1 = rng.randn(n).astype(np.float32)
    x2 = rng.randn(n).astype(np.float32)
    x3 = rng.randn(n).astype(np.float32)
    y = (2 * x1 - x2 + 0.5 * x1 * x2 + 0.3 * rng.randn(n) > 0)
    return {
        "x1": x1, "x2": x2, "x3": x3,
        "label": np.where(y, "yes", "no"),
    }


@pytest.fixture(scope="session")
def regression_data():
    rng = np.random.RandomState(1)
    n = 8000
    x1 = rng.randn(n).astype(np.float32)
    x2 = rng.randn(n).astype(np.float32)
    y = (3 * x1 + x2 * x2 + 0.1 * rng.randn(n)).astype(np.float32)
    return {"x1": x1, "x2": x2, "label": y}


@pytest.fixture(scope="session")
def adult_paths():
    base = "/root/reference/yggdrasil_decision_forests/test_data/dataset"
    tr = os.path.join(base, "adult_train.csv")
    te = os.path.join(base, "adult_test.csv")
    if not (os.path.exists(tr) and os.path.exists(te)):
        pytest.skip("reference adult.csv not available")
    return tr, te
