"""End-to-end CPU training tests (reference analog:
tests/python/test_basic.py, test_basic_models.py)."""
import numpy as np
import pytest

import xgboost_amd as xgb
from conftest import make_classification, make_regression


def test_binary_classification_learns():
    X, y = make_classification()
    dtrain = xgb.DMatrix(X[:1500], label=y[:1500])
    dvalid = xgb.DMatrix(X[1500:], label=y[1500:])
    res = {}
    bst = xgb.train({"objective": "binary:logistic", "max_depth": 4,
                     "eta": 0.3, "eval_metric": ["logloss", "auc"]},
                    dtrain, 30, evals=[(dvalid, "valid")],
                    evals_result=res, verbose_eval=False)
    assert res["valid"]["logloss"][-1] < 0.35
    assert res["valid"]["auc"][-1] > 0.93
    p = bst.predict(dvalid)
    acc = ((p > 0.5) == y[1500:]).mean()
    assert acc > 0.85
    # monotone improvement in train loss
    res2 = {}
    xgb.train({"objective": "binary:logistic", "max_depth": 4, "eta": 0.3},
              dtrain, 10, evals=[(dtrain, "train")], evals_result=res2,
              verbose_eval=False)
    ll = res2["train"]["logloss"]
    assert ll[-1] < ll[0]


def test_regression_squarederror():
    X, y = make_regression()
    dtrain = xgb.DMatrix(X[:1500], label=y[:1500])
    dvalid = xgb.DMatrix(X[1500:], label=y[1500:])
    res = {}
    bst = xgb.train({"objective": "reg:squarederror", "max_depth": 6,
                     "eta": 0.3}, dtrain, 40, evals=[(dvalid, "valid")],
                    evals_result=res, verbose_eval=False)
    base_rmse = float(np.std(y[1500:]))
    assert res["valid"]["rmse"][-1] < 0.5 * base_rmse
    p = bst.predict(dvalid)
    assert p.shape == (500,)


def test_multiclass():
    X, y = make_classification(n_class=4)
    dtrain = xgb.DMatrix(X[:1500], label=y[:1500])
    dvalid = xgb.DMatrix(X[1500:], label=y[1500:])
    res = {}
    bst = xgb.train({"objective": "multi:softprob", "num_class": 4,
                     "max_depth": 4, "eta": 0.4}, dtrain, 15,
                    evals=[(dvalid, "valid")], evals_result=res,
                    verbose_eval=False)
    p = bst.predict(dvalid)
    assert p.shape == (500, 4)
    assert np.allclose(p.sum(axis=1), 1.0, atol=1e-5)
    acc = (p.argmax(axis=1) == y[1500:]).mean()
    assert acc > 0.7
    # softmax returns class ids
    bst2 = xgb.train({"objective": "multi:softmax", "num_class": 4,
                      "max_depth": 4, "eta": 0.4}, dtrain, 15,
                     verbose_eval=False)
    p2 = bst2.predict(dvalid)
    assert p2.shape == (500,)
    assert set(np.unique(p2)) <= {0.0, 1.0, 2.0, 3.0}


def test_base_margin_and_base_score():
    X, y = make_regression(500, 5)
    d = xgb.DMatrix(X, label=y)
    bst = xgb.train({"objective": "reg:squarederror", "base_score": 3.0},
                    d, 1, verbose_eval=False)
    # first tree fits residuals vs 3.0
    m = bst.predict(d, output_margin=True)
    assert abs(np.mean(m) - (3.0 + np.mean(y - 3.0) * 0.3)) < 0.5


def test_eta_zero_keeps_base():
    X, y = make_regression(200, 3)
    d = xgb.DMatrix(X, label=y)
    bst = xgb.train({"objective": "reg:squarederror", "eta": 0.0}, d, 2,
                    verbose_eval=False)
    p = bst.predict(d)
    assert np.allclose(p, p[0])


def test_max_depth_respected():
    X, y = make_classification(1000, 8)
    d = xgb.DMatrix(X, label=y)
    for depth in (1, 3):
        bst = xgb.train({"objective": "binary:logistic", "max_depth": depth},
                        d, 3, verbose_eval=False)
        for t in bst.trees:
            assert t.max_depth() <= depth


def test_max_leaves_lossguide():
    X, y = make_classification(1000, 8)
    d = xgb.DMatrix(X, label=y)
    bst = xgb.train({"objective": "binary:logistic", "grow_policy": "lossguide",
                     "max_leaves": 8, "max_depth": 0}, d, 3, verbose_eval=False)
    for t in bst.trees:
        n_leaves = sum(1 for nid in range(t.n_nodes) if t.is_leaf(nid))
        assert n_leaves <= 8


def test_custom_objective_and_metric():
    X, y = make_regression(500, 5)
    d = xgb.DMatrix(X, label=y)

    def sq_obj(preds, dtrain):
        g = preds - dtrain.get_label()
        h = np.ones_like(g)
        return g, h

    def mae_metric(preds, dtrain):
        return "my-mae", float(np.abs(preds - dtrain.get_label()).mean())

    res = {}
    xgb.train({"max_depth": 3, "eta": 0.3, "disable_default_eval_metric": 1},
              d, 10, obj=sq_obj, custom_metric=mae_metric,
              evals=[(d, "train")], evals_result=res, verbose_eval=False)
    assert res["train"]["my-mae"][-1] < res["train"]["my-mae"][0]


def test_early_stopping():
    X, y = make_classification()
    dtrain = xgb.DMatrix(X[:1500], label=y[:1500])
    dvalid = xgb.DMatrix(X[1500:], label=y[1500:])
    bst = xgb.train({"objective": "binary:logistic", "eta": 0.5,
                     "max_depth": 6}, dtrain, 500,
                    evals=[(dvalid, "valid")],
                    early_stopping_rounds=5, verbose_eval=False)
    assert bst.num_boosted_rounds() < 500
    assert bst.best_iteration is not None


def test_missing_values():
    X, y = make_classification(1000, 5)
    Xm = X.copy()
    mask = np.random.RandomState(0).rand(*X.shape) < 0.2
    Xm[mask] = np.nan
    d = xgb.DMatrix(Xm, label=y)
    res = {}
    xgb.train({"objective": "binary:logistic", "max_depth": 4}, d, 10,
              evals=[(d, "train")], evals_result=res, verbose_eval=False)
    assert res["train"]["logloss"][-1] < 0.6


def test_weights():
    X, y = make_classification(1000, 5)
    w = np.where(y == 1, 10.0, 1.0).astype(np.float32)
    d = xgb.DMatrix(X, label=y, weight=w)
    bst = xgb.train({"objective": "binary:logistic", "max_depth": 3}, d, 5,
                    verbose_eval=False)
    p = bst.predict(d)
    # heavily weighting positives shifts predictions up
    d0 = xgb.DMatrix(X, label=y)
    bst0 = xgb.train({"objective": "binary:logistic", "max_depth": 3}, d0, 5,
                     verbose_eval=False)
    assert p.mean() > bst0.predict(d0).mean()


def test_subsample_colsample():
    X, y = make_classification(2000, 10)
    d = xgb.DMatrix(X, label=y)
    res = {}
    xgb.train({"objective": "binary:logistic", "max_depth": 4,
               "subsample": 0.5, "colsample_bytree": 0.5,
               "colsample_bylevel": 0.7, "colsample_bynode": 0.8,
               "seed": 7}, d, 10, evals=[(d, "train")], evals_result=res,
              verbose_eval=False)
    assert res["train"]["logloss"][-1] < 0.5


def test_determinism_same_seed():
    X, y = make_classification(1000, 6)
    d = xgb.DMatrix(X, label=y)
    params = {"objective": "binary:logistic", "max_depth": 4,
              "subsample": 0.8, "colsample_bytree": 0.8, "seed": 3}
    p1 = xgb.train(params, d, 5, verbose_eval=False).predict(d)
    d2 = xgb.DMatrix(X, label=y)
    p2 = xgb.train(params, d2, 5, verbose_eval=False).predict(d2)
    assert np.array_equal(p1, p2)


def test_num_parallel_tree():
    X, y = make_regression(800, 6)
    d = xgb.DMatrix(X, label=y)
    bst = xgb.train({"objective": "reg:squarederror", "num_parallel_tree": 4,
                     "max_depth": 3}, d, 3, verbose_eval=False)
    assert len(bst.trees) == 12
    assert bst.num_boosted_rounds() == 3


def test_iteration_range_predict():
    X, y = make_regression(500, 5)
    d = xgb.DMatrix(X, label=y)
    bst = xgb.train({"objective": "reg:squarederror"}, d, 10, verbose_eval=False)
    p5 = bst.predict(d, iteration_range=(0, 5))
    sliced = bst[:5]
    assert np.allclose(sliced.predict(d), p5, atol=1e-6)


def test_pred_leaf():
    X, y = make_classification(300, 4)
    d = xgb.DMatrix(X, label=y)
    bst = xgb.train({"objective": "binary:logistic", "max_depth": 3}, d, 4,
                    verbose_eval=False)
    leaves = bst.predict(d, pred_leaf=True)
    assert leaves.shape == (300, 4)
    for t in range(4):
        tree = bst.trees[t]
        for leaf_id in np.unique(leaves[:, t]).astype(int):
            assert tree.is_leaf(leaf_id)


def test_inplace_predict_zero_copy_proxy():
    """inplace_predict wraps numpy/torch input in a proxy with no
    DMatrix materialization (reference GBTree::InplacePredict via
    ProxyDMatrix) and matches the DMatrix prediction exactly."""
    import torch
    rng = np.random.RandomState(3)
    X = rng.randn(500, 6).astype(np.float32)
    y = (X[:, 0] > 0).astype(np.float32)
    bst = xgb.train({"objective": "binary:logistic", "max_depth": 4},
                    xgb.DMatrix(X, label=y), 5, verbose_eval=False)
    ref = bst.predict(xgb.DMatrix(X))
    got_np = bst.inplace_predict(X)
    assert np.allclose(got_np, ref, atol=1e-7)
    got_t = bst.inplace_predict(torch.from_numpy(X))
    assert np.allclose(got_t, ref, atol=1e-7)
    # margin type
    m_ref = bst.predict(xgb.DMatrix(X), output_margin=True)
    m = bst.inplace_predict(X, predict_type="margin")
    assert np.allclose(m, m_ref, atol=1e-6)
    # base_margin honored
    bm = np.full(500, 0.7, np.float32)
    r1 = bst.predict(xgb.DMatrix(X, base_margin=bm))
    r2 = bst.inplace_predict(X, base_margin=bm)
    assert np.allclose(r1, r2, atol=1e-6)
    # feature mismatch raises
    import pytest as _pytest
    with _pytest.raises(ValueError):
        bst.inplace_predict(X[:, :4])
    # custom missing value
    Xm = X.copy()
    Xm[rng.rand(500, 6) < 0.1] = -999.0
    Xn = np.where(Xm == -999.0, np.nan, Xm)
    r3 = bst.inplace_predict(Xm, missing=-999.0)
    r4 = bst.predict(xgb.DMatrix(Xn))
    assert np.allclose(r3, r4, atol=1e-6)


def test_arrow_table_ingestion():
    """pyarrow Table ingestion (reference _from_arrow_table) incl.
    dictionary-encoded categoricals and nulls-as-missing."""
    import pyarrow as pa
    rng = np.random.RandomState(4)
    a = rng.randn(300).astype(np.float32)
    b = rng.randn(300).astype(np.float64)
    b[5] = np.nan
    cat = pa.array(rng.choice(["x", "y", "z"], 300)).dictionary_encode()
    tbl = pa.table({"a": pa.array(a), "b": pa.array(b), "c": cat})
    y = (a > 0).astype(np.float32)
    d = xgb.DMatrix(tbl, label=y, enable_categorical=True)
    assert d.num_col() == 3
    assert d.feature_names == ["a", "b", "c"]
    assert d.feature_types == ["float", "float", "c"]
    bst = xgb.train({"objective": "binary:logistic", "max_depth": 3}, d, 5,
                    verbose_eval=False)
    acc = ((bst.predict(d) > 0.5) == y).mean()
    assert acc > 0.9
    # nulls became missing
    X = d.raw_data()
    assert np.isnan(X[5, 1])
    # plain (non-categorical) tables work without the flag
    tbl2 = pa.table({"a": pa.array(a), "b": pa.array(b)})
    d2 = xgb.DMatrix(tbl2, label=y)
    assert d2.num_col() == 2


def test_deprecated_setters():
    """set_label/set_weight/set_base_margin/set_group aliases
    (reference core.py deprecated per-field setters)."""
    rng = np.random.RandomState(0)
    X = rng.randn(60, 3).astype(np.float32)
    d = xgb.DMatrix(X)
    d.set_label(np.arange(60, dtype=np.float32))
    d.set_weight(np.ones(60, dtype=np.float32))
    d.set_base_margin(np.zeros(60, dtype=np.float32))
    assert d.get_label()[-1] == 59
    assert d.get_weight().sum() == 60
    d2 = xgb.DMatrix(X, label=np.zeros(60, np.float32))
    d2.set_group([30, 30])
    assert list(d2.get_group()) == [30, 30]  # group sizes


def test_label_validation():
    """reference data.cc:566: labels must be finite; AFT censoring
    bounds (which may be +inf) go through label_lower/upper_bound."""
    X = np.ones((3, 2), np.float32)
    with pytest.raises(ValueError, match="NaN, infinity"):
        xgb.DMatrix(X, label=np.array([1, np.nan, 2], np.float32))
    d = xgb.DMatrix(X)
    with pytest.raises(ValueError, match="NaN, infinity"):
        d.set_info(label=np.array([np.inf, 0, 1], np.float32))
    d.set_info(label=np.array([0, 1, 2], np.float32),
               label_lower_bound=np.zeros(3, np.float32),
               label_upper_bound=np.full(3, np.inf, np.float32))


def test_quantile_dmatrix_hist_only():
    """reference IterativeDMatrix: no SparsePage for exact
    (iterative_dmatrix.cc:159), no cut regeneration for approx
    (:129) — QuantileDMatrix trains with hist only."""
    rng = np.random.RandomState(0)
    X = rng.randn(100, 4).astype(np.float32)
    y = rng.randn(100).astype(np.float32)
    qd = xgb.QuantileDMatrix(X, label=y, max_bin=32)
    for tm in ("exact", "approx"):
        with pytest.raises(ValueError, match="QuantileDMatrix"):
            xgb.train({"tree_method": tm, "max_depth": 2}, qd, 1)
    xgb.train({"tree_method": "hist", "max_depth": 2}, qd, 1)


def test_feature_names_validation():
    """reference _validate_features: predict-frame names (incl. order)
    must match training names."""
    import pandas as pd
    rng = np.random.RandomState(0)
    X = pd.DataFrame({"a": rng.randn(50).astype(np.float32),
                      "b": rng.randn(50).astype(np.float32)})
    y = rng.randn(50).astype(np.float32)
    bst = xgb.train({"max_depth": 2}, xgb.DMatrix(X, label=y), 2)
    with pytest.raises(ValueError, match="feature_names mismatch"):
        bst.predict(xgb.DMatrix(X.rename(columns={"b": "c"})))
    with pytest.raises(ValueError, match="feature_names mismatch"):
        bst.predict(xgb.DMatrix(X[["b", "a"]]))
    p = bst.predict(xgb.DMatrix(X))  # matching names fine
    # validate_features=False skips the check
    p2 = bst.predict(xgb.DMatrix(X[["b", "a"]]), validate_features=False)
    assert p.shape == p2.shape


def test_trees_to_dataframe_columns():
    """reference core.py:3259 column set, incl. Category for
    categorical splits."""
    import pandas as pd
    rng = np.random.RandomState(0)
    Xc = pd.DataFrame({
        "c": pd.Series(rng.randint(0, 6, 300)).astype("category"),
        "n": rng.randn(300).astype(np.float32)})
    y = (Xc["c"].cat.codes.to_numpy() % 2 +
         rng.randn(300) * 0.1).astype(np.float32)
    bst = xgb.train({"max_depth": 3, "max_cat_to_onehot": 1}, 
                    xgb.DMatrix(Xc, label=y, enable_categorical=True), 3)
    df = bst.trees_to_dataframe()
    assert list(df.columns) == [
        "Tree", "Target", "Node", "ID", "Feature", "Split", "Yes", "No",
        "Missing", "Gain", "Cover", "Category"]
    cat_rows = df[df["Category"].notna()]
    assert len(cat_rows) > 0  # categorical splits present
    assert all(isinstance(c, list) for c in cat_rows["Category"])


def test_pandas_nullable_and_bad_dtypes():
    """pandas nullable extension dtypes (Int64/Float64/boolean) convert
    with NA -> NaN; non-numeric dtypes are rejected (reference pandas
    adapter semantics)."""
    import pandas as pd
    df = pd.DataFrame({
        "i": pd.array([1, None, 3, 4] * 10, dtype="Int64"),
        "f": pd.array([0.5, 1.5, None, 2.5] * 10, dtype="Float64"),
        "b": pd.array([True, False, None, True] * 10, dtype="boolean"),
    })
    y = np.arange(40, dtype=np.float32)
    d = xgb.DMatrix(df, label=y)
    X0 = d.raw_data()
    assert np.isnan(X0[1, 0]) and np.isnan(X0[2, 1]) and np.isnan(X0[2, 2])
    assert X0[0, 0] == 1.0 and X0[0, 2] == 1.0
    xgb.train({"max_depth": 2}, d, 2)
    with pytest.raises(ValueError, match="int, float, bool or category"):
        xgb.DMatrix(pd.DataFrame({"t": pd.date_range("2020", periods=5)}),
                    label=y[:5])


def test_pred_leaf_strict_shape():
    """reference strict_shape leaf predictions:
    (n, n_iterations, n_groups, n_parallel_tree)."""
    rng = np.random.RandomState(0)
    X = rng.randn(40, 3).astype(np.float32)
    y3 = rng.randint(0, 3, 40).astype(np.float32)
    d = xgb.DMatrix(X, label=y3)
    bst = xgb.train({"objective": "multi:softprob", "num_class": 3,
                     "num_parallel_tree": 2, "max_depth": 2}, d, 4)
    leaves = bst.predict(d, pred_leaf=True)
    assert leaves.shape == (40, 24)  # 4 iters * 3 classes * 2 trees
    strict = bst.predict(d, pred_leaf=True, strict_shape=True)
    assert strict.shape == (40, 4, 3, 2)
    assert np.array_equal(strict.reshape(40, -1), leaves)


def test_booster_iter_and_index_bounds():
    """Booster.__iter__ yields per-iteration slices; out-of-range int
    indexing raises IndexError (the legacy iteration protocol otherwise
    loops forever)."""
    rng = np.random.RandomState(0)
    X = rng.randn(40, 3).astype(np.float32)
    y = X[:, 0].astype(np.float32)
    bst = xgb.train({"max_depth": 2}, xgb.DMatrix(X, label=y), 3)
    parts = list(bst)
    assert len(parts) == 3
    assert all(p.num_boosted_rounds() == 1 for p in parts)
    with pytest.raises(IndexError):
        bst[3]
    assert bst[-1].num_boosted_rounds() == 1  # negative indexing
    d = xgb.DMatrix(X)
    total = sum(p.predict(d, output_margin=True) - bst.base_score
                for p in parts) + bst.base_score
    assert np.allclose(total, bst.predict(d, output_margin=True),
                       atol=1e-5)


def test_inplace_predict_options():
    """iteration_range slices and base_margin adds (reference
    inplace_predict surface)."""
    rng = np.random.RandomState(0)
    X = rng.randn(50, 3).astype(np.float32)
    y = X[:, 0].astype(np.float32)
    bst = xgb.train({"max_depth": 2, "base_score": 0.0, "seed": 1},
                    xgb.DMatrix(X, label=y), 4)
    p2 = bst.inplace_predict(X, iteration_range=(0, 2))
    bst2 = xgb.train({"max_depth": 2, "base_score": 0.0, "seed": 1},
                     xgb.DMatrix(X, label=y), 2)
    assert np.allclose(p2, bst2.predict(xgb.DMatrix(X)), atol=1e-6)
    bm = np.full(50, 1.5, np.float32)
    assert np.allclose(bst.inplace_predict(X, base_margin=bm),
                       bst.inplace_predict(X) + 1.5, atol=1e-5)


def test_dmatrix_slice_propagates_meta():
    """DMatrix.slice(rindex) carries labels/weights for the selected
    rows (reference DMatrix::SliceCol/Slice semantics)."""
    rng = np.random.RandomState(0)
    X = rng.randn(100, 3).astype(np.float32)
    y = X[:, 0].astype(np.float32)
    d = xgb.DMatrix(X, label=y, weight=np.arange(100, dtype=np.float32))
    sub = d.slice([5, 10, 20])
    assert sub.num_row() == 3
    assert np.allclose(sub.get_label(), y[[5, 10, 20]])
    assert np.allclose(sub.get_weight(), [5, 10, 20])
    bst = xgb.train({"max_depth": 2}, d, 2)
    assert np.allclose(bst.predict(sub),
                       bst.predict(d)[[5, 10, 20]], atol=1e-6)
