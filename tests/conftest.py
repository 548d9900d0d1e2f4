import sys
from pathlib import Path

import numpy as np
import pytest

REPO_ROOT = Path(__file__).resolve().parent.parent
if str(REPO_ROOT) not in sys.path:
    sys.path.insert(0, str(REPO_ROOT))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: needs a real MI355X GPU (run via gpurun)")


@pytest.fixture(scope="session")
def digits_std():
    """Standardized digits dataset (shared fixture, loaded once)."""
    from sklearn import datasets

    d = datasets.load_digits()
    X = d.data.astype(np.float64)
    X = (X - X.mean(axis=0)) / (X.std(axis=0) + 1e-9)
    return X, d.target


@pytest.fixture(scope="session")
def binary_blobs():
    """Easy separable binary problem for solver-parity tests."""
    rng = np.random.default_rng(0)
    n, f = 2000, 32
    X = rng.standard_normal((n, f))
    w = rng.standard_normal(f)
    y = (X @ w + 0.3 * rng.standard_normal(n) > 0).astype(np.int64)
    return X.astype(np.float64), y
