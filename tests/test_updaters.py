"""Tree methods & updaters: exact/approx, prune/refresh,
process_type=update (reference analog: tests/python/test_updaters.py)."""
import numpy as np
import pytest

import xgboost_amd as xgb
from conftest import make_classification, make_regression


def test_exact_learns():
    X, y = make_regression(2000, 6)
    d = xgb.DMatrix(X, label=y)
    res = {}
    xgb.train({"objective": "reg:squarederror", "tree_method": "exact",
               "max_depth": 5, "eta": 0.3}, d, 15, evals=[(d, "t")],
              evals_result=res, verbose_eval=False)
    assert res["t"]["rmse"][-1] < 0.4 * np.std(y)


def test_exact_matches_hist_quality():
    X, y = make_classification(2000, 6)
    d = xgb.DMatrix(X, label=y)
    out = {}
    for m in ("exact", "hist", "approx"):
        res = {}
        xgb.train({"objective": "binary:logistic", "tree_method": m,
                   "max_depth": 4}, d, 10, evals=[(d, "t")],
                  evals_result=res, verbose_eval=False)
        out[m] = res["t"]["logloss"][-1]
    assert all(v < 0.4 for v in out.values()), out
    assert abs(out["exact"] - out["hist"]) < 0.1


def test_exact_with_missing():
    X, y = make_classification(1000, 5)
    X[np.random.RandomState(0).rand(*X.shape) < 0.2] = np.nan
    d = xgb.DMatrix(X, label=y)
    res = {}
    xgb.train({"objective": "binary:logistic", "tree_method": "exact",
               "max_depth": 4}, d, 10, evals=[(d, "t")], evals_result=res,
              verbose_eval=False)
    assert res["t"]["logloss"][-1] < 0.55


def test_process_type_update_refresh():
    X, y = make_regression(1000, 5)
    d = xgb.DMatrix(X, label=y)
    bst = xgb.train({"objective": "reg:squarederror", "max_depth": 4},
                    d, 5, verbose_eval=False)
    p_before = bst.predict(d)
    # refresh on the SAME data should keep predictions roughly unchanged
    X2, y2 = make_regression(1000, 5, seed=99)
    d2 = xgb.DMatrix(X2, label=y2)
    bst2 = xgb.train({"objective": "reg:squarederror",
                      "process_type": "update", "updater": "refresh"},
                     d2, 5, xgb_model=bst, verbose_eval=False)
    assert len(bst2.trees) == len(bst.trees)
    # structures identical, leaf values refreshed on new data
    for t1, t2 in zip(bst.trees, bst2.trees):
        assert t1.n_nodes == t2.n_nodes
        assert np.array_equal(t1.split_index[:t1.n_nodes],
                              t2.split_index[:t2.n_nodes])
    p_after = bst2.predict(d2)
    rmse_new = np.sqrt(np.mean((p_after - y2) ** 2))
    assert rmse_new < np.std(y2)  # refreshed leaves fit the new data


def test_prune_updater():
    from xgboost_amd.updaters import prune_tree
    from xgboost_amd.params import make_train_param
    X, y = make_classification(1000, 6)
    d = xgb.DMatrix(X, label=y)
    bst = xgb.train({"objective": "binary:logistic", "max_depth": 6},
                    d, 2, verbose_eval=False)
    big = prune_tree(bst.trees[0], make_train_param({"gamma": 0.0}))
    small = prune_tree(bst.trees[0], make_train_param({"gamma": 1e9}))
    assert small.n_nodes == 1  # everything pruned to root leaf
    assert big.n_nodes == bst.trees[0].n_nodes


def test_refresh_leaf_false_keeps_leaves():
    X, y = make_regression(500, 4)
    d = xgb.DMatrix(X, label=y)
    bst = xgb.train({"objective": "reg:squarederror", "max_depth": 3},
                    d, 3, verbose_eval=False)
    p1 = bst.predict(d)
    bst2 = xgb.train({"objective": "reg:squarederror",
                      "process_type": "update", "updater": "refresh",
                      "refresh_leaf": False}, d, 3, xgb_model=bst,
                     verbose_eval=False)
    assert np.allclose(bst2.predict(d), p1, atol=1e-5)


def test_multi_output_tree():
    rng = np.random.RandomState(0)
    X = rng.randn(1500, 6).astype(np.float32)
    W = rng.randn(6, 3)
    Y = (X @ W + 0.1 * rng.randn(1500, 3)).astype(np.float32)
    d = xgb.DMatrix(X, label=Y)
    res = {}
    bst = xgb.train({"objective": "reg:squarederror", "max_depth": 5,
                     "multi_strategy": "multi_output_tree", "eta": 0.3},
                    d, 15, evals=[(d, "t")], evals_result=res,
                    verbose_eval=False)
    assert len(bst.trees) == 15  # one vector-leaf tree per round
    assert bst.trees[0].leaf_values is not None
    p = bst.predict(d)
    assert p.shape == (1500, 3)
    assert res["t"]["rmse"][-1] < 0.6
    # JSON round-trip with size_leaf_vector
    raw = bst.save_raw("json")
    b2 = xgb.Booster()
    b2.load_model(bytes(raw))
    assert np.allclose(b2.predict(d), p, atol=1e-6)


def test_multi_target_one_output_per_tree():
    rng = np.random.RandomState(0)
    X = rng.randn(1000, 5).astype(np.float32)
    W = rng.randn(5, 2)
    Y = (X @ W).astype(np.float32)
    d = xgb.DMatrix(X, label=Y)
    bst = xgb.train({"objective": "reg:squarederror", "max_depth": 4},
                    d, 10, verbose_eval=False)
    assert len(bst.trees) == 20  # one tree per target per round
    p = bst.predict(d)
    assert p.shape == (1000, 2)


def test_exact_rejects_unsupported():
    """reference updater_colmaker.cc:104-113: exact rejects categorical
    data, external memory and colsample_bynode."""
    import pandas as pd
    rng = np.random.RandomState(0)
    y = rng.randn(60).astype(np.float32)
    Xc = pd.DataFrame({
        "c": pd.Series(rng.randint(0, 3, 60)).astype("category"),
        "n": rng.randn(60).astype(np.float32)})
    dc = xgb.DMatrix(Xc, label=y, enable_categorical=True)
    with pytest.raises(ValueError, match="categorical"):
        xgb.train({"tree_method": "exact", "max_depth": 2}, dc, 1)
    X = rng.randn(60, 3).astype(np.float32)
    d = xgb.DMatrix(X, label=y)
    with pytest.raises(ValueError, match="colsample_bynode|column sample"):
        xgb.train({"tree_method": "exact", "max_depth": 2,
                   "colsample_bynode": 0.5}, d, 1)


def test_hist_equals_exact_on_discrete_data():
    """Differential oracle: with few distinct feature values (every
    value its own bin), the hist updater must find the SAME trees as
    the exact enumeration (reference relationship between
    grow_quantile_histmaker and grow_colmaker)."""
    rng = np.random.RandomState(0)
    X = rng.randint(0, 20, (500, 4)).astype(np.float32)
    y = (X[:, 0] * 0.5 - X[:, 1] * 0.2 + rng.randn(500)).astype(
        np.float32)
    p = {"max_depth": 4, "base_score": 0.0, "reg_lambda": 1.0}
    bh = xgb.train(dict(p, tree_method="hist", max_bin=256),
                   xgb.DMatrix(X, label=y), 3)
    be = xgb.train(dict(p, tree_method="exact"),
                   xgb.DMatrix(X, label=y), 3)
    d = xgb.DMatrix(X)
    assert np.allclose(bh.predict(d), be.predict(d), atol=1e-6)
    for a, b in zip(bh.trees, be.trees):
        assert np.array_equal(a.split_index[:a.n_nodes],
                              b.split_index[:b.n_nodes])


def test_approx_equals_hist_with_uniform_hessian():
    """With reg:squarederror the hessian is 1 everywhere, so the
    hessian-weighted re-sketch reduces to plain quantiles and approx
    must reproduce hist exactly (same max_bin)."""
    rng = np.random.RandomState(0)
    X = rng.randn(800, 4).astype(np.float32)
    y = (X[:, 0] - X[:, 1]).astype(np.float32)
    p = {"max_depth": 3, "base_score": 0.0, "max_bin": 64}
    bh = xgb.train(dict(p, tree_method="hist"),
                   xgb.DMatrix(X, label=y), 3)
    ba = xgb.train(dict(p, tree_method="approx"),
                   xgb.DMatrix(X, label=y), 3)
    d = xgb.DMatrix(X)
    assert np.allclose(bh.predict(d), ba.predict(d), atol=1e-7)


def test_explicit_updater_selects_growth_algorithm():
    """reference gbtree specified_updater_: updater="grow_colmaker,
    prune" runs the exact enumeration; grow_quantile_histmaker -> hist."""
    rng = np.random.RandomState(0)
    X = rng.randn(150, 3).astype(np.float32)
    d = xgb.DMatrix(X, label=X[:, 0])
    b = xgb.train({"updater": "grow_colmaker,prune", "max_depth": 3},
                  d, 2)
    assert b.tparam.tree_method == "exact"
    b2 = xgb.train({"updater": "grow_colmaker,prune", "max_depth": 3,
                    "tree_method": "exact"}, d, 2)
    assert b.get_dump() == b2.get_dump()
    b3 = xgb.train({"updater": "grow_quantile_histmaker",
                    "max_depth": 3}, d, 2)
    assert b3.tparam.tree_method == "hist"
