import numpy as np
import pytest


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an AMD GPU (MI355X); run with -m gpu")


def pytest_collection_modifyitems(config, items):
    import torch
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU available")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture
def rng():
    return np.random.RandomState(42)


def make_classification(n=2000, f=10, seed=0, n_class=2):
    rng = np.random.RandomState(seed)
    X = rng.randn(n, f).astype(np.float32)
    w = rng.randn(f, n_class if n_class > 2 else 1)
    logits = X @ w + 0.3 * rng.randn(n, w.shape[1])
    if n_class == 2:
        y = (logits[:, 0] > 0).astype(np.float32)
    else:
        y = logits.argmax(axis=1).astype(np.float32)
    return X, y


def make_regression(n=2000, f=10, seed=0):
    rng = np.random.RandomState(seed)
    X = rng.randn(n, f).astype(np.float32)
    w = rng.randn(f)
    y = (X @ w + 0.1 * rng.randn(n)).astype(np.float32)
    return X, y
