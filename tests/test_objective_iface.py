"""Experimental class-based objective interface (reference
python-package/xgboost/objective.py + core.py:2320 dispatch): plain
Objective classes get (iteration, raw_margin, dtrain); TreeObjective
may return a REDUCED split gradient while the full gradient values
vector leaves (XGBoosterTrainOneIterWithSplitGrad)."""
import numpy as np
import pytest

import xgboost_amd as xgb
from xgboost_amd.objective import Objective, TreeObjective


def _mt_data(n=1500, C=3, seed=0):
    rng = np.random.RandomState(seed)
    X = rng.randn(n, 5).astype(np.float32)
    W = rng.randn(5, C).astype(np.float32)
    Y = (X @ W + 0.05 * rng.randn(n, C)).astype(np.float32)
    return X, Y


class SqErr(Objective):
    def __call__(self, iteration, y_pred, dtrain):
        y = dtrain.get_label().reshape(y_pred.shape)
        return y_pred - y, np.ones_like(y_pred)


def test_class_objective_equals_plain_callable():
    X, Y = _mt_data()
    d = xgb.DMatrix(X, label=Y)
    p = {"max_depth": 4, "eta": 0.3, "num_target": Y.shape[1],
         "base_score": 0.0, "seed": 1}

    def plain(preds, dmat):
        y = dmat.get_label().reshape(preds.shape)
        return preds - y, np.ones_like(preds)

    b1 = xgb.train(dict(p), d, 5, obj=SqErr())
    d2 = xgb.DMatrix(X, label=Y)
    b2 = xgb.train(dict(p), d2, 5, obj=plain)
    assert b1.get_dump(with_stats=True) == b2.get_dump(with_stats=True)


class Reduced(TreeObjective):
    """Full squared-error gradient per target; structure found from the
    target-summed gradient."""

    def __call__(self, iteration, y_pred, dtrain):
        y = dtrain.get_label().reshape(y_pred.shape)
        return y_pred - y, np.ones_like(y_pred)

    def split_grad(self, iteration, grad, hess):
        return grad.sum(axis=1, keepdims=True), hess.sum(
            axis=1, keepdims=True)


def test_split_grad_builds_vector_leaves():
    import torch
    X, Y = _mt_data(seed=3)
    C = Y.shape[1]
    d = xgb.DMatrix(X, label=Y)
    p = {"max_depth": 4, "eta": 0.3, "num_target": C, "base_score": 0.0,
         "reg_lambda": 1.0, "seed": 2}
    bst = xgb.train(dict(p), d, 4, obj=Reduced())
    assert len(bst.trees) == 4  # ONE vector-leaf tree per iteration
    for t in bst.trees:
        assert t.leaf_values is not None
        assert t.leaf_values.shape[1] == C
    # incremental margin cache == fresh weighted predict
    cached, _ = bst._cache[id(d)]
    fresh = bst._predict_margin(d)
    assert torch.allclose(cached, fresh, atol=1e-4)


def test_split_grad_leaf_value_oracle():
    """Round-1 leaf values must be -G_c/(H_c+lambda)*eta computed from
    the FULL gradient within each structure leaf."""
    X, Y = _mt_data(n=800, seed=5)
    C = Y.shape[1]
    lam, eta = 1.5, 0.4
    d = xgb.DMatrix(X, label=Y)
    bst = xgb.train({"max_depth": 3, "eta": eta, "num_target": C,
                     "base_score": 0.0, "reg_lambda": lam, "seed": 7},
                    d, 1, obj=Reduced())
    t = bst.trees[0]
    pos = t.predict_leaf_np(X, float("nan"))
    grad0 = 0.0 - Y  # base margin 0 -> grad = pred - y
    for nid in np.unique(pos):
        rows = pos == nid
        G = grad0[rows].sum(axis=0)
        H = float(rows.sum())
        expect = -G / (H + lam) * eta
        assert np.allclose(t.leaf_values[nid], expect, atol=1e-4), nid


def test_split_grad_requires_multi_target():
    X, Y = _mt_data(n=300)
    d = xgb.DMatrix(X, label=Y[:, 0])

    class Bad(TreeObjective):
        def __call__(self, iteration, y_pred, dtrain):
            y = dtrain.get_label().reshape(y_pred.shape)
            return y_pred - y, np.ones_like(y_pred)

        def split_grad(self, iteration, grad, hess):
            return grad, hess

    with pytest.raises(ValueError, match="vector-leaf"):
        xgb.train({"max_depth": 3, "base_score": 0.0}, d, 2, obj=Bad())


def test_split_grad_with_dart():
    """Reduced-gradient boosting composes with DART: the vector-leaf
    tree's margin add is scaled by the DART weight and the dropped
    trees are renormalized."""
    import torch
    X, Y = _mt_data(n=600, seed=9)
    C = Y.shape[1]
    d = xgb.DMatrix(X, label=Y)
    bst = xgb.train({"max_depth": 3, "eta": 0.3, "num_target": C,
                     "base_score": 0.0, "rate_drop": 1.0, "seed": 3},
                    d, 4, obj=Reduced())
    assert len(bst.weight_drop) == 4
    assert any(w != 1.0 for w in bst.weight_drop)
    cached, _ = bst._cache[id(d)]
    fresh = bst._predict_margin(d)
    assert torch.allclose(cached, fresh, atol=1e-4)
