import numpy as np
import pandas as pd
import pytest


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs a ROCm GPU (run on MI355X box)")


@pytest.fixture(scope="session")
def binary_df():
    rng = np.random.default_rng(0)
    n, nf = 3000, 10
    X = rng.normal(size=(n, nf)).astype(np.float32)
    w = rng.normal(size=nf)
    y = ((X @ w + rng.normal(size=n) * 0.5) > 0).astype(np.float32)
    return pd.DataFrame({"features": list(X), "label": y})


@pytest.fixture(scope="session")
def regression_df():
    rng = np.random.default_rng(1)
    n, nf = 2000, 8
    X = rng.normal(size=(n, nf)).astype(np.float32)
    y = (X[:, 0] * 2 + X[:, 1] ** 2 + rng.normal(size=n) * 0.1).astype(np.float32)
    return pd.DataFrame({"features": list(X), "label": y})
