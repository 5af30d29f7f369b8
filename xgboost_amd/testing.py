"""Synthetic data generators for tests, demos and benchmarks.

Offline analog of the reference's public ``xgboost.testing`` helpers
(python-package/xgboost/testing/data.py): same roles — batched
regression data, learning-to-rank groups, sparse regression matrices,
categorical frames — generated locally instead of downloaded.
"""
from typing import List, Tuple

import numpy as np

__all__ = ["make_batches", "make_ltr", "make_sparse_regression",
           "make_categorical", "make_regression", "make_classification"]


def make_regression(n_samples: int = 1000, n_features: int = 10,
                    noise: float = 0.1, seed: int = 0
                    ) -> Tuple[np.ndarray, np.ndarray]:
    """Dense regression data with a random linear + quadratic signal."""
    rng = np.random.RandomState(seed)
    X = rng.randn(n_samples, n_features).astype(np.float32)
    w = rng.randn(n_features)
    y = X @ w + 0.3 * X[:, 0] * X[:, min(1, n_features - 1)]
    y = (y + noise * rng.randn(n_samples)).astype(np.float32)
    return X, y


def make_classification(n_samples: int = 1000, n_features: int = 10,
                        n_classes: int = 2, seed: int = 0
                        ) -> Tuple[np.ndarray, np.ndarray]:
    """Dense classification data (class = argmax of random linear
    scores plus noise)."""
    rng = np.random.RandomState(seed)
    X = rng.randn(n_samples, n_features).astype(np.float32)
    W = rng.randn(n_features, n_classes)
    scores = X @ W + 0.5 * rng.randn(n_samples, n_classes)
    y = np.argmax(scores, axis=1).astype(np.float32)
    return X, y


def make_batches(n_samples_per_batch: int, n_features: int,
                 n_batches: int, use_cupy: bool = False, *,
                 vector_leaf: bool = False, n_targets: int = 1,
                 seed: int = 0
                 ) -> Tuple[List[np.ndarray], List[np.ndarray],
                            List[np.ndarray]]:
    """Batched (X, y, w) lists for DataIter / external-memory tests
    (reference testing/data.py make_batches)."""
    X, y, w = [], [], []
    if use_cupy:
        raise NotImplementedError(
            "cupy is not available on this stack; pass torch cuda "
            "tensors to DataIter instead")
    rng = np.random.RandomState(seed)
    t = n_targets if (vector_leaf or n_targets > 1) else 1
    for _ in range(n_batches):
        _X = rng.randn(n_samples_per_batch, n_features).astype(np.float32)
        _y = rng.randn(n_samples_per_batch, t).astype(np.float32)
        if t == 1:
            _y = _y.ravel()
        _w = rng.uniform(low=0.0, high=1.0,
                         size=n_samples_per_batch).astype(np.float32)
        X.append(_X)
        y.append(_y)
        w.append(_w)
    return X, y, w


def make_ltr(n_samples: int, n_features: int, n_query_groups: int,
             max_rel: int, sort_qid: bool = True, seed: int = 0
             ) -> Tuple[np.ndarray, np.ndarray, np.ndarray]:
    """Learning-to-rank data: (X, y, qid) with integer relevance in
    [0, max_rel] correlated with the features (reference
    testing/data.py make_ltr)."""
    rng = np.random.RandomState(seed)
    X = rng.randn(n_samples, n_features).astype(np.float32)
    qid = rng.randint(0, n_query_groups, size=n_samples)
    if sort_qid:
        qid = np.sort(qid)
    score = X[:, 0] + 0.5 * rng.randn(n_samples)
    edges = np.quantile(score, np.linspace(0, 1, max_rel + 2)[1:-1])
    y = np.digitize(score, edges).astype(np.float32)
    return X, y, qid.astype(np.int64)


def make_sparse_regression(n_samples: int, n_features: int,
                           sparsity: float, as_dense: bool = False,
                           seed: int = 0):
    """CSR regression data with the given fraction of MISSING entries
    (reference testing/data.py make_sparse_regression)."""
    from scipy import sparse

    rng = np.random.RandomState(seed)
    density = 1.0 - sparsity
    csr = sparse.random(n_samples, n_features, density=density,
                        format="csr", dtype=np.float32, random_state=rng)
    w = rng.randn(n_features).astype(np.float32)
    y = np.asarray(csr @ w).ravel() + 0.05 * rng.randn(n_samples)
    y = y.astype(np.float32)
    if as_dense:
        X = np.full((n_samples, n_features), np.nan, dtype=np.float32)
        coo = csr.tocoo()
        X[coo.row, coo.col] = coo.data
        return X, y
    return csr, y


def make_categorical(n_samples: int, n_features: int, n_categories: int,
                     onehot: bool = False, sparsity: float = 0.0,
                     cat_ratio: float = 1.0, shuffle: bool = False,
                     seed: int = 0):
    """Mixed categorical/numeric frame for categorical-split tests
    (reference testing/dask.py make_categorical, without dask): returns
    (X, y) where X is a pandas DataFrame with ``category`` dtype
    columns, or a one-hot-encoded numpy array when ``onehot``."""
    import pandas as pd

    rng = np.random.RandomState(seed)
    n_cat = max(1, int(n_features * cat_ratio))
    cols = {}
    signal = np.zeros(n_samples)
    for f in range(n_features):
        if f < n_cat:
            codes = rng.randint(0, n_categories, size=n_samples)
            if sparsity > 0:
                mask = rng.rand(n_samples) < sparsity
                vals = pd.array(codes, dtype="Int64")
                vals[mask] = pd.NA
                col = pd.Series(vals).astype("category")
            else:
                col = pd.Series(codes).astype("category")
            cols[f"c{f}"] = col
            signal += (codes % 3) * 0.5
        else:
            v = rng.randn(n_samples).astype(np.float32)
            cols[f"n{f}"] = v
            signal += v * 0.3
    X = pd.DataFrame(cols)
    if shuffle:
        X = X.sample(frac=1.0, random_state=rng).reset_index(drop=True)
    y = (signal + 0.1 * rng.randn(n_samples)).astype(np.float32)
    if onehot:
        return pd.get_dummies(X).to_numpy(dtype=np.float32), y
    return X, y
