"""DeviceSketch (gpu_sketch.py) vs the exact sort-based oracle
(quantile.make_cuts).  The sketch is pure torch ops, so the CPU suite
runs the very code the GPU executes (reference analog:
tests/cpp/common/test_quantile.cu)."""
import numpy as np
import pytest
import torch

from xgboost_amd.gpu_sketch import DeviceSketch, device_cuts
from xgboost_amd.quantile import make_cuts


def _rank(vals, weights, c):
    """Weighted rank of cut c = total weight of values <= ... < c."""
    return float(weights[vals < c].sum())


def test_exact_match_small_cardinality():
    rng = np.random.RandomState(0)
    X = np.round(rng.randn(5000, 6), 1).astype(np.float32)  # heavy ties
    X[rng.rand(5000, 6) < 0.1] = np.nan
    for max_bin in (16, 64, 256):
        ref = make_cuts(X, max_bin)
        got = device_cuts(torch.from_numpy(X), max_bin)
        assert np.array_equal(ref.ptrs, got.ptrs), max_bin
        assert np.allclose(ref.values, got.values), max_bin
        assert np.allclose(ref.min_vals, got.min_vals)


def test_exact_match_weighted():
    rng = np.random.RandomState(1)
    X = np.round(rng.randn(3000, 4), 1).astype(np.float32)
    w = (rng.rand(3000) + 0.05).astype(np.float32)
    ref = make_cuts(X, 32, weights=w)
    got = device_cuts(torch.from_numpy(X), 32,
                      weights=torch.from_numpy(w))
    assert np.array_equal(ref.ptrs, got.ptrs)
    assert np.allclose(ref.values, got.values)


def test_eps_bound_large_distinct():
    """All-distinct data larger than the sketch budget: every interior
    cut's rank must sit within total/B + 1 of its query rank (the
    documented prune error bound)."""
    rng = np.random.RandomState(2)
    n, max_bin = 200_000, 32
    X = rng.randn(n, 2).astype(np.float32)
    got = device_cuts(torch.from_numpy(X), max_bin)
    B = max(64, 8 * max_bin)
    w = np.ones(n)
    for f in range(2):
        cuts = got.feature_cuts(f)[:-1]  # drop sentinel
        vals = X[:, f]
        step = n / max_bin
        for i, c in enumerate(cuts):
            # cut i answers query (i+1)*step: first distinct with
            # rmax >= q -> rank in [q - eps, q + eps + w_max]
            q = (i + 1) * step
            r = _rank(vals, w, c)
            assert abs(r - q) <= n / B + 2, (f, i, r, q)


def test_eps_bound_weighted_skewed():
    """Skewed weighted data (the reference's DeviceSketchWithHessian
    case): rank error of every cut stays inside the documented bound."""
    rng = np.random.RandomState(3)
    n, max_bin = 100_000, 64
    X = np.exp(rng.randn(n, 1) * 2).astype(np.float32)  # log-normal skew
    w = (rng.rand(n) ** 2 + 1e-3).astype(np.float32)
    got = device_cuts(torch.from_numpy(X), max_bin,
                      weights=torch.from_numpy(w))
    B = max(64, 8 * max_bin)
    total = float(w.sum())
    step = total / max_bin
    cuts = got.feature_cuts(0)[:-1]
    wmax = float(w.max())
    for i, c in enumerate(cuts):
        q = (i + 1) * step
        r = _rank(X[:, 0], w, c)
        assert abs(r - q) <= 2 * total / B + wmax + 1e-6, (i, r, q)


def test_multi_batch_equals_single_small():
    rng = np.random.RandomState(4)
    X = np.round(rng.randn(4000, 5), 1).astype(np.float32)
    one = device_cuts(torch.from_numpy(X), 32)
    sk = DeviceSketch(5, 32)
    sk.push(torch.from_numpy(X[:1500]))
    sk.push(torch.from_numpy(X[1500:2500]))
    sk.push(torch.from_numpy(X[2500:]))
    multi = sk.make_cuts()
    assert np.array_equal(one.ptrs, multi.ptrs)
    assert np.allclose(one.values, multi.values)


def test_multi_batch_eps_large():
    rng = np.random.RandomState(5)
    n, max_bin = 120_000, 32
    X = rng.randn(n, 1).astype(np.float32)
    sk = DeviceSketch(1, max_bin)
    for lo in range(0, n, 40_000):
        sk.push(torch.from_numpy(X[lo:lo + 40_000]))
    got = sk.make_cuts()
    B = max(64, 8 * max_bin)
    step = n / max_bin
    for i, c in enumerate(got.feature_cuts(0)[:-1]):
        r = _rank(X[:, 0], np.ones(n), c)
        assert abs(r - (i + 1) * step) <= 2 * n / B + 2, (i, r)


def test_categorical_and_empty_columns():
    rng = np.random.RandomState(6)
    X = np.stack([rng.randint(0, 7, 1000).astype(np.float32),
                  np.full(1000, np.nan, np.float32),
                  rng.randn(1000).astype(np.float32)], axis=1)
    ft = ["c", "q", "q"]
    ref = make_cuts(X, 16, feature_types=ft)
    got = device_cuts(torch.from_numpy(X), 16, feature_types=ft)
    assert np.array_equal(ref.ptrs, got.ptrs)
    assert np.allclose(ref.values, got.values)
    assert got.feature_types == ft


def test_custom_missing_value():
    rng = np.random.RandomState(7)
    X = np.round(rng.randn(2000, 3), 1).astype(np.float32)
    X[rng.rand(2000, 3) < 0.2] = -999.0
    ref = make_cuts(np.where(X == -999.0, np.nan, X), 24)
    got = device_cuts(torch.from_numpy(X), 24, missing=-999.0)
    assert np.array_equal(ref.ptrs, got.ptrs)
    assert np.allclose(ref.values, got.values)


def test_deterministic():
    rng = np.random.RandomState(8)
    X = rng.randn(50_000, 3).astype(np.float32)
    a = device_cuts(torch.from_numpy(X), 64)
    b = device_cuts(torch.from_numpy(X), 64)
    assert np.array_equal(a.values, b.values)
    assert np.array_equal(a.ptrs, b.ptrs)
