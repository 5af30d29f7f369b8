"""SHAP correctness: efficiency axiom (sum of contributions == margin
prediction), interactions consistency (reference analog:
tests/cpp/predictor/test_shap.cc, python test_shap.py)."""
import numpy as np
import pytest

import xgboost_amd as xgb
from conftest import make_classification, make_regression


def _small_model(rounds=3, depth=3, n=150, f=4):
    X, y = make_regression(n, f)
    d = xgb.DMatrix(X, label=y)
    bst = xgb.train({"objective": "reg:squarederror", "max_depth": depth,
                     "eta": 0.5}, d, rounds, verbose_eval=False)
    return bst, d, X, y


def test_shap_sums_to_margin():
    bst, d, X, y = _small_model()
    contribs = bst.predict(d, pred_contribs=True)
    assert contribs.shape == (150, 5)
    margin = bst.predict(d, output_margin=True)
    assert np.allclose(contribs.sum(axis=1), margin, atol=1e-4)


def test_shap_missing_values():
    X, y = make_regression(100, 4)
    X[np.random.RandomState(0).rand(*X.shape) < 0.2] = np.nan
    d = xgb.DMatrix(X, label=y)
    bst = xgb.train({"objective": "reg:squarederror", "max_depth": 3},
                    d, 3, verbose_eval=False)
    contribs = bst.predict(d, pred_contribs=True)
    margin = bst.predict(d, output_margin=True)
    assert np.allclose(contribs.sum(axis=1), margin, atol=1e-4)


def test_shap_unused_feature_zero():
    rng = np.random.RandomState(0)
    X = rng.randn(200, 3).astype(np.float32)
    y = X[:, 0].astype(np.float32)  # only feature 0 matters
    d = xgb.DMatrix(X, label=y)
    bst = xgb.train({"objective": "reg:squarederror", "max_depth": 2,
                     "colsample_bytree": 1.0}, d, 3, verbose_eval=False)
    contribs = bst.predict(d, pred_contribs=True)
    used = set()
    for t in bst.trees:
        for nid in range(t.n_nodes):
            if not t.is_leaf(nid):
                used.add(int(t.split_index[nid]))
    for f in range(3):
        if f not in used:
            assert np.allclose(contribs[:, f], 0.0)


def test_approx_contribs_sum():
    bst, d, X, y = _small_model()
    contribs = bst.predict(d, pred_contribs=True, approx_contribs=True)
    margin = bst.predict(d, output_margin=True)
    assert np.allclose(contribs.sum(axis=1), margin, atol=1e-4)


def test_interactions_sum_to_shap():
    bst, d, X, y = _small_model(rounds=2, depth=3, n=60)
    inter = bst.predict(d, pred_interactions=True)
    assert inter.shape == (60, 5, 5)
    contribs = bst.predict(d, pred_contribs=True)
    # rows of the interaction matrix sum to the SHAP values
    assert np.allclose(inter.sum(axis=2), contribs, atol=1e-3)
    # symmetry
    assert np.allclose(inter, np.transpose(inter, (0, 2, 1)), atol=1e-5)


def test_multiclass_contribs():
    X, y = make_classification(200, 4, n_class=3)
    d = xgb.DMatrix(X, label=y)
    bst = xgb.train({"objective": "multi:softprob", "num_class": 3,
                     "max_depth": 3}, d, 2, verbose_eval=False)
    contribs = bst.predict(d, pred_contribs=True)
    assert contribs.shape == (200, 3, 5)
    margin = bst.predict(d, output_margin=True)
    assert np.allclose(contribs.sum(axis=2), margin, atol=1e-4)
