"""Reference demo-idiom coverage (reference: demo/guide-python/*): the
usage patterns upstream documents must work verbatim here."""
import numpy as np
import pytest

import xgboost_amd as xgb


def _reg_data(n=400, f=6, seed=0):
    rng = np.random.RandomState(seed)
    X = rng.randn(n, f).astype(np.float32)
    y = (X[:, 0] * 2 + np.sin(X[:, 1]) + 0.1 * rng.randn(n)).astype(
        np.float32)
    return X, y


def test_boost_from_prediction():
    """demo/guide-python/boost_from_prediction.py: stage-2 training from
    stage-1 margins via base_margin."""
    X, y = _reg_data()
    d1 = xgb.DMatrix(X, label=y)
    bst1 = xgb.train({"objective": "reg:squarederror", "max_depth": 3}, d1, 5)
    ptrain = bst1.predict(d1, output_margin=True)
    d2 = xgb.DMatrix(X, label=y, base_margin=ptrain)
    bst2 = xgb.train({"objective": "reg:squarederror", "max_depth": 3}, d2, 5)
    # combined margins == continued training from stage 1
    final = bst2.predict(d2, output_margin=True)
    assert np.sqrt(np.mean((final - y) ** 2)) < \
        np.sqrt(np.mean((ptrain - y) ** 2))


def test_custom_objective_and_metric_together():
    """demo custom_rmsle.py: fobj + feval through xgb.train."""
    X, y = _reg_data()
    y = np.abs(y)
    d = xgb.DMatrix(X, label=y)

    def squared_log(preds, dtrain):
        yv = dtrain.get_label()
        p = np.maximum(preds, -1 + 1e-6)
        grad = (np.log1p(p) - np.log1p(yv)) / (p + 1)
        hess = ((-np.log1p(p) + np.log1p(yv) + 1) / (p + 1) ** 2)
        return grad, np.maximum(hess, 1e-6)

    def rmsle(preds, dtrain):
        yv = dtrain.get_label()
        p = np.maximum(preds, -1 + 1e-6)
        return "my-rmsle", float(np.sqrt(np.mean(
            (np.log1p(p) - np.log1p(yv)) ** 2)))

    res = {}
    xgb.train({"max_depth": 3, "base_score": 0.5, "disable_default_eval_metric": 1},
              d, 10, obj=squared_log, custom_metric=rmsle,
              evals=[(d, "t")], evals_result=res, verbose_eval=False)
    vals = res["t"]["my-rmsle"]
    assert vals[-1] < vals[0]


def test_predict_first_ntree():
    """demo predict_first_ntree.py: iteration_range slicing of
    prediction equals a model trained with fewer rounds."""
    X, y = _reg_data()
    d = xgb.DMatrix(X, label=y)
    params = {"objective": "reg:squarederror", "max_depth": 3, "seed": 1}
    bst = xgb.train(params, d, 10)
    p3 = bst.predict(d, iteration_range=(0, 3))
    bst3 = xgb.train(params, xgb.DMatrix(X, label=y), 3)
    np.testing.assert_allclose(p3, bst3.predict(d), rtol=1e-5, atol=1e-5)


def test_predict_leaf_indices_and_individual_trees():
    """demo predict_leaf_indices.py + individual_trees.py."""
    X, y = _reg_data()
    d = xgb.DMatrix(X, label=y)
    bst = xgb.train({"objective": "reg:squarederror", "max_depth": 3}, d, 6)
    leaves = bst.predict(d, pred_leaf=True)
    assert leaves.shape == (400, 6)
    # every leaf id must be a leaf node of its tree
    t0 = bst.trees[0]
    assert all(t0.left[int(i)] == -1 for i in np.unique(leaves[:, 0]))
    # summing individual sliced trees reproduces the full margin
    total = np.zeros(400, dtype=np.float64)
    for i in range(6):
        sub = bst[i: i + 1]
        total += sub.predict(d, output_margin=True) - bst.base_score
    full = bst.predict(d, output_margin=True) - bst.base_score
    np.testing.assert_allclose(total, full, rtol=1e-4, atol=1e-4)


def test_prediction_intervals_via_quantile():
    """demo prediction_intervals.py: multi-alpha quantile regression."""
    X, y = _reg_data(n=2000)
    d = xgb.DMatrix(X, label=y)
    bst = xgb.train({"objective": "reg:quantileerror",
                     "quantile_alpha": [0.05, 0.5, 0.95],
                     "max_depth": 4}, d, 30)
    p = bst.predict(d)
    assert p.shape == (2000, 3)
    # coverage: ~90% of labels inside [q05, q95]
    inside = ((y >= p[:, 0]) & (y <= p[:, 2])).mean()
    assert inside > 0.75
    # monotone quantiles on average
    assert (p[:, 0] <= p[:, 2]).mean() > 0.95


def test_feature_weights_demo():
    """demo feature_weights.py: higher weight -> more splits on that
    feature."""
    rng = np.random.RandomState(0)
    X = rng.randn(2000, 8).astype(np.float32)
    y = np.sum(X, axis=1).astype(np.float32)  # all features equal
    fw = np.ones(8, dtype=np.float32)
    fw[0] = 50.0
    d = xgb.DMatrix(X, label=y, feature_weights=fw)
    bst = xgb.train({"objective": "reg:squarederror", "max_depth": 4,
                     "colsample_bynode": 0.4, "seed": 0}, d, 20)
    score = bst.get_score(importance_type="weight")
    w0 = score.get("f0", 0)
    others = [score.get(f"f{i}", 0) for i in range(1, 8)]
    assert w0 > np.mean(others)


def test_custom_softmax_idiom():
    """demo/guide-python/custom_softmax.py: multiclass custom objective
    via raw-margin softmax gradients matches the builtin."""
    rng = np.random.RandomState(4)
    n, f, C = 900, 6, 3
    X = rng.randn(n, f).astype(np.float32)
    y = np.argmax(X[:, :C] + 0.3 * rng.randn(n, C), axis=1).astype(
        np.float32)

    def softprob_obj(preds, dmat):
        labels = dmat.get_label().astype(int)
        m = preds.reshape(n, C)
        e = np.exp(m - m.max(axis=1, keepdims=True))
        p = e / e.sum(axis=1, keepdims=True)
        grad = p.copy()
        grad[np.arange(n), labels] -= 1.0
        hess = np.maximum(2.0 * p * (1.0 - p), 1e-6)
        return grad, hess

    params = {"max_depth": 4, "eta": 0.3, "num_class": C,
              "base_score": 0.5, "seed": 0}
    d1 = xgb.DMatrix(X, label=y)
    custom = xgb.train(dict(params, objective="multi:softmax",
                            disable_default_eval_metric=1),
                       d1, 8, obj=softprob_obj)
    d2 = xgb.DMatrix(X, label=y)
    builtin = xgb.train(dict(params, objective="multi:softprob"), d2, 8)
    pc = custom.predict(d1, output_margin=True).reshape(n, C).argmax(1)
    pb = builtin.predict(d2).reshape(n, C).argmax(1)
    assert (pc == pb).mean() > 0.95
    assert (pc == y).mean() > 0.85


def test_reduced_gradient_demo_runs():
    import demo.reduced_gradient as rg
    rg.main()
