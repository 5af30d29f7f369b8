"""DART dropout (reference gbm/gbtree.cc: DropTrees :474, NormalizeTrees
:539, weighted PredictBatch :632).  The `dart` booster name is a
deprecated alias; dropout is driven by rate_drop/one_drop/skip_drop on
the tree booster."""
import json

import numpy as np
import pytest
import torch

import xgboost_amd as xgb


def _data(n=2000, seed=0):
    rng = np.random.RandomState(seed)
    X = rng.randn(n, 6).astype(np.float32)
    y = (X[:, 0] * 1.5 + X[:, 1] ** 2 + 0.1 * rng.randn(n)).astype(
        np.float32)
    return X, y


def test_skip_drop_one_is_plain_gbtree():
    X, y = _data()
    d = xgb.DMatrix(X, label=y)
    p = {"max_depth": 4, "eta": 0.3, "seed": 7}
    plain = xgb.train(dict(p), d, 10)
    dart = xgb.train(dict(p, rate_drop=0.5, skip_drop=1.0), d, 10)
    assert plain.get_dump(with_stats=True) == dart.get_dump(with_stats=True)
    # weights tracked but all 1.0
    assert dart.weight_drop == [1.0] * 10


def test_margin_cache_matches_fresh_weighted_predict():
    """The incremental margin cache (new-tree adds scaled by the DART
    weight + (factor-1)*dropped fix-up) must equal a from-scratch
    weighted prediction after every kind of round."""
    X, y = _data(1500, seed=3)
    d = xgb.DMatrix(X, label=y)
    bst = xgb.train({"max_depth": 4, "eta": 0.3, "seed": 5,
                     "rate_drop": 0.4, "one_drop": True}, d, 12)
    assert len(bst.weight_drop) == len(bst.trees)
    assert any(w != 1.0 for w in bst.weight_drop)
    cached, _ = bst._cache[id(d)]
    fresh = bst._predict_margin(d)
    assert torch.allclose(cached, fresh, atol=1e-4), \
        (cached - fresh).abs().max()


def test_rate_drop_one_weights_recurrence():
    """rate_drop=1.0 drops EVERY tree each round -> deterministic
    NormalizeTrees recurrence (normalize_type=tree):
    k=len(trees), dropped *= k/(k+lr), new tree weight 1/(k+lr)."""
    X, y = _data(800, seed=1)
    d = xgb.DMatrix(X, label=y)
    lr = 0.5
    rounds = 5
    bst = xgb.train({"max_depth": 3, "eta": lr, "seed": 2,
                     "rate_drop": 1.0}, d, rounds)
    w = []
    for _ in range(rounds):
        k = len(w)
        if k == 0:
            w.append(1.0)
        else:
            factor = k / (k + lr)
            w = [x * factor for x in w]
            w.append(1.0 / (k + lr))
    assert np.allclose(bst.weight_drop, w, rtol=1e-6), (bst.weight_drop, w)


def test_normalize_type_forest():
    X, y = _data(800, seed=4)
    d = xgb.DMatrix(X, label=y)
    lr = 0.3
    bst = xgb.train({"max_depth": 3, "eta": lr, "seed": 2,
                     "rate_drop": 1.0, "normalize_type": "forest"}, d, 3)
    f = 1.0 / (1.0 + lr)
    # round2: w=[f, f]; round3: w=[f*f, f*f, f]
    assert np.allclose(bst.weight_drop, [f * f, f * f, f], rtol=1e-6)


def test_dart_save_load_roundtrip(tmp_path):
    X, y = _data(1000, seed=6)
    d = xgb.DMatrix(X, label=y)
    bst = xgb.train({"max_depth": 4, "eta": 0.3, "seed": 9,
                     "rate_drop": 0.3}, d, 8)
    p1 = bst.predict(d)
    fn = str(tmp_path / "m.json")
    bst.save_model(fn)
    j = json.load(open(fn))
    assert "weight_drop" in j["learner"]["gradient_booster"]
    bst2 = xgb.Booster(model_file=fn)
    assert bst2.weight_drop == pytest.approx(bst.weight_drop)
    assert np.allclose(bst2.predict(d), p1, atol=1e-6)


def test_legacy_dart_format_loads(tmp_path):
    """Old models: gradient_booster = {name: dart, gbtree: {...},
    weight_drop: [...]} (gbtree.cc:452-463 compat)."""
    X, y = _data(500, seed=8)
    d = xgb.DMatrix(X, label=y)
    bst = xgb.train({"max_depth": 3, "eta": 0.3, "rate_drop": 0.3,
                     "seed": 1}, d, 4)
    fn = str(tmp_path / "m.json")
    bst.save_model(fn)
    j = json.load(open(fn))
    gb = j["learner"]["gradient_booster"]
    wd = gb.pop("weight_drop")
    j["learner"]["gradient_booster"] = {
        "name": "dart", "gbtree": gb, "weight_drop": wd}
    fn2 = str(tmp_path / "legacy.json")
    json.dump(j, open(fn2, "w"))
    bst2 = xgb.Booster(model_file=fn2)
    assert bst2.weight_drop == pytest.approx(bst.weight_drop)
    assert np.allclose(bst2.predict(d), bst.predict(d), atol=1e-6)


def test_dart_booster_alias_and_quality():
    X, y = _data(3000, seed=10)
    ycls = (y > np.median(y)).astype(np.float32)
    d = xgb.DMatrix(X, label=ycls)
    res = {}
    bst = xgb.train({"booster": "dart", "objective": "binary:logistic",
                     "max_depth": 4, "eta": 0.3, "rate_drop": 0.2,
                     "seed": 3, "eval_metric": "auc"}, d, 25,
                    evals=[(d, "t")], evals_result=res, verbose_eval=False)
    assert res["t"]["auc"][-1] > 0.9
    assert len(bst.weight_drop) == len(bst.trees)


def test_dart_sklearn_wrapper():
    from xgboost_amd.sklearn import XGBRegressor
    X, y = _data(1200, seed=11)
    m = XGBRegressor(n_estimators=10, max_depth=4, booster="dart",
                     rate_drop=0.3, learning_rate=0.3)
    m.fit(X, y)
    p = m.predict(X)
    assert np.isfinite(p).all()
    assert np.corrcoef(p, y)[0, 1] > 0.8


def test_weighted_sample_type_runs():
    X, y = _data(800, seed=12)
    d = xgb.DMatrix(X, label=y)
    bst = xgb.train({"max_depth": 3, "eta": 0.3, "rate_drop": 0.4,
                     "sample_type": "weighted", "seed": 4}, d, 10)
    assert len(bst.weight_drop) == 10
    cached, _ = bst._cache[id(d)]
    fresh = bst._predict_margin(d)
    assert torch.allclose(cached, fresh, atol=1e-4)


def test_dart_slice_carries_weights():
    """Booster slicing keeps each tree's DART weight (reference
    GBTree::Slice, gbtree.cc:625-631)."""
    X, y = _data(800, seed=13)
    d = xgb.DMatrix(X, label=y)
    bst = xgb.train({"max_depth": 3, "eta": 0.3, "rate_drop": 1.0,
                     "seed": 4}, d, 6)
    sl = bst[2:5]
    assert sl.weight_drop == pytest.approx(bst.weight_drop[2:5])
    # sliced predictions = weighted sum over just those trees
    base = bst.base_score
    exp = np.full(len(y), base, np.float32)
    for i, t in enumerate(range(2, 5)):
        pos = bst.trees[t].predict_leaf_np(X, float("nan"))
        exp += bst.weight_drop[t] * \
            bst.trees[t].split_cond[:bst.trees[t].n_nodes][pos]
    assert np.allclose(sl.predict(d), exp, atol=1e-5)


def test_dart_pickle_and_continuation():
    """weight_drop survives pickling (save_raw embeds it) and training
    continuation extends it."""
    import pickle
    X, y = _data(400, seed=15)
    d = xgb.DMatrix(X, label=y)
    b = xgb.train({"max_depth": 3, "rate_drop": 0.5, "seed": 1}, d, 6)
    b2 = pickle.loads(pickle.dumps(b))
    assert b2.weight_drop == pytest.approx(b.weight_drop)
    assert np.allclose(b2.predict(d), b.predict(d))
    b3 = xgb.train({"max_depth": 3, "rate_drop": 0.5, "seed": 2}, d, 3,
                   xgb_model=b)
    assert len(b3.weight_drop) == len(b3.trees) == 9
    cached = b3._predict_margin(d)
    # margin from scratch must be consistent with incremental history
    assert np.isfinite(cached.numpy()).all()


def test_shap_ignores_dart_weights():
    """pred_contribs follow the reference: the predictor's contribution
    path sees the raw tree model (no weight_drop), so phi sums to the
    UNWEIGHTED margin."""
    X, y = _data(500, seed=17)
    d = xgb.DMatrix(X, label=y)
    bst = xgb.train({"max_depth": 3, "eta": 0.3, "rate_drop": 1.0,
                     "seed": 2}, d, 4)
    phi = bst.predict(d, pred_contribs=True)
    unweighted = np.full(len(y), bst.base_score, np.float64)
    for t in bst.trees:
        pos = t.predict_leaf_np(X, float("nan"))
        unweighted += t.split_cond[:t.n_nodes][pos]
    assert np.allclose(phi.sum(axis=1), unweighted, atol=1e-4)


def test_dart_with_cv_and_early_stopping():
    """xgb.cv and EarlyStopping work with dropout configured."""
    X, y = _data(900, seed=19)
    ycls = (y > np.median(y)).astype(np.float32)
    d = xgb.DMatrix(X, label=ycls)
    res = xgb.cv({"objective": "binary:logistic", "max_depth": 3,
                  "rate_drop": 0.3, "seed": 1, "eval_metric": "logloss"},
                 d, num_boost_round=8, nfold=3)
    assert len(res["test-logloss-mean"]) == 8
    es = xgb.callback.EarlyStopping(rounds=3, save_best=True)
    bst = xgb.train({"objective": "binary:logistic", "max_depth": 3,
                     "rate_drop": 0.3, "seed": 1,
                     "eval_metric": "logloss"}, d, 30,
                    evals=[(d, "t")], callbacks=[es], verbose_eval=False)
    assert bst.num_boosted_rounds() <= 30
    assert len(bst.weight_drop) == len(bst.trees)


def test_dart_heavy_feature_combination():
    """DART composed with multiclass, parallel trees, monotone,
    colsample, subsample and weights: margin cache stays consistent
    with a fresh weighted predict and the model round-trips."""
    import torch
    rng = np.random.RandomState(0)
    X = rng.randn(1000, 6).astype(np.float32)
    y = rng.randint(0, 3, 1000).astype(np.float32)
    w = rng.rand(1000).astype(np.float32) + 0.5
    d = xgb.DMatrix(X, label=y, weight=w)
    b = xgb.train({
        "objective": "multi:softprob", "num_class": 3, "max_depth": 4,
        "rate_drop": 0.3, "one_drop": True,
        "monotone_constraints": "(1,0,0,0,0,-1)",
        "colsample_bytree": 0.8, "subsample": 0.9,
        "num_parallel_tree": 2, "seed": 5}, d, 6)
    assert len(b.trees) == len(b.weight_drop) == 36
    cached, _ = b._cache[id(d)]
    assert torch.allclose(cached, b._predict_margin(d), atol=1e-3)
    b2 = xgb.Booster()
    b2.load_model(bytearray(b.save_raw("ubj")))
    assert np.allclose(b2.predict(d), b.predict(d), atol=1e-5)
