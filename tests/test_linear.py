"""gblinear booster tests (reference analog:
tests/python/test_linear.py — elastic-net coordinate descent over
updaters/selectors, convergence and IO)."""
import numpy as np
import pytest

import xgboost_amd as xgb


def _linear_data(n=2000, f=10, seed=0):
    rng = np.random.RandomState(seed)
    X = rng.randn(n, f).astype(np.float32)
    true_w = np.zeros(f, dtype=np.float32)
    true_w[:3] = [1.5, -2.0, 0.5]
    y = X @ true_w + 0.3 + 0.01 * rng.randn(n).astype(np.float32)
    return X, y, true_w


@pytest.mark.parametrize("updater", ["coord_descent", "shotgun"])
def test_gblinear_recovers_coefficients(updater):
    X, y, true_w = _linear_data()
    d = xgb.DMatrix(X, label=y)
    res = {}
    bst = xgb.train({"booster": "gblinear", "updater": updater,
                     "eta": 0.5, "reg_lambda": 0.0, "reg_alpha": 0.0,
                     "objective": "reg:squarederror"},
                    d, 40, evals=[(d, "t")], evals_result=res,
                    verbose_eval=False)
    rmse = res["t"]["rmse"]
    assert rmse[-1] < 0.1 * rmse[0]
    pred = bst.predict(d)
    assert np.sqrt(np.mean((pred - y) ** 2)) < 0.2


@pytest.mark.parametrize("selector", ["cyclic", "shuffle", "random",
                                      "greedy", "thrifty"])
def test_gblinear_feature_selectors(selector):
    X, y, _ = _linear_data(n=500)
    d = xgb.DMatrix(X, label=y)
    res = {}
    xgb.train({"booster": "gblinear", "feature_selector": selector,
               "top_k": 5 if selector in ("greedy", "thrifty") else 0,
               "objective": "reg:squarederror"},
              d, 10, evals=[(d, "t")], evals_result=res, verbose_eval=False)
    vals = res["t"]["rmse"]
    assert vals[-1] < vals[0]


def test_gblinear_l1_sparsity():
    X, y, true_w = _linear_data()
    d = xgb.DMatrix(X, label=y)
    bst = xgb.train({"booster": "gblinear", "eta": 0.5,
                     "reg_alpha": 2.0, "reg_lambda": 0.0,
                     "objective": "reg:squarederror"}, d, 40)
    j = bst._model_to_json()
    w = np.array(j["learner"]["gradient_booster"]["model"]["weights"],
                 dtype=np.float32)
    # strong L1 zeroes (or nearly) the 7 null coefficients
    null_w = np.abs(w[3:-1])
    assert np.all(null_w < 0.05)


def test_gblinear_json_roundtrip(tmp_path):
    X, y, _ = _linear_data(n=300)
    d = xgb.DMatrix(X, label=y)
    bst = xgb.train({"booster": "gblinear",
                     "objective": "reg:squarederror"}, d, 5)
    p = bst.predict(d)
    path = str(tmp_path / "lin.json")
    bst.save_model(path)
    bst2 = xgb.Booster(model_file=path)
    assert np.allclose(bst2.predict(d), p, atol=1e-6)


def test_gblinear_binary_classification():
    rng = np.random.RandomState(1)
    X = rng.randn(800, 6).astype(np.float32)
    y = (X[:, 0] - X[:, 1] > 0).astype(np.float32)
    d = xgb.DMatrix(X, label=y)
    res = {}
    bst = xgb.train({"booster": "gblinear", "objective": "binary:logistic",
                     "eta": 0.5}, d, 30, evals=[(d, "t")],
                    evals_result=res, verbose_eval=False)
    assert res["t"]["logloss"][-1] < 0.3
    p = bst.predict(d)
    assert ((p > 0.5) == (y > 0.5)).mean() > 0.9


def test_gblinear_multiclass():
    rng = np.random.RandomState(2)
    X = rng.randn(600, 5).astype(np.float32)
    y = np.abs(X[:, :3]).argmax(axis=1).astype(np.float32)
    d = xgb.DMatrix(X, label=y)
    bst = xgb.train({"booster": "gblinear", "objective": "multi:softprob",
                     "num_class": 3, "eta": 0.5}, d, 20)
    p = bst.predict(d)
    assert p.shape == (600, 3)
    assert np.allclose(p.sum(axis=1), 1.0, atol=1e-5)


def test_gblinear_rejects_categorical_and_shotgun_selectors():
    """reference gblinear.cc:129 (NoCategorical) and
    updater_shotgun.cc:20 (cyclic/shuffle only)."""
    import pandas as pd
    rng = np.random.RandomState(0)
    y = rng.randn(60).astype(np.float32)
    Xc = pd.DataFrame({
        "c": pd.Series(rng.randint(0, 3, 60)).astype("category"),
        "n": rng.randn(60).astype(np.float32)})
    dc = xgb.DMatrix(Xc, label=y, enable_categorical=True)
    with pytest.raises(ValueError, match="categorical"):
        xgb.train({"booster": "gblinear"}, dc, 1)
    d = xgb.DMatrix(rng.randn(60, 3).astype(np.float32), label=y)
    with pytest.raises(ValueError, match="shotgun"):
        xgb.train({"booster": "gblinear", "updater": "shotgun",
                   "feature_selector": "greedy"}, d, 1)


def test_gblinear_feature_importance_is_coefficients():
    """reference gblinear.cc:210 FeatureScore: weight importance = the
    coefficients (bias excluded); other types are undefined."""
    rng = np.random.RandomState(0)
    X = rng.randn(300, 4).astype(np.float32)
    y = (X[:, 0] * 2 + X[:, 1]).astype(np.float32)
    bst = xgb.train({"booster": "gblinear", "eta": 0.5}, 
                    xgb.DMatrix(X, label=y), 20)
    s = bst.get_score(importance_type="weight")
    assert set(s) == {"f0", "f1", "f2", "f3"}
    assert abs(s["f0"]) > abs(s["f2"])  # real coefficient magnitudes
    with pytest.raises(ValueError, match="weight"):
        bst.get_score(importance_type="gain")


def test_gblinear_converges_to_ridge_solution():
    """Differential oracle: coordinate descent must converge to the
    closed-form ridge solution with the reference's DENORMALIZED
    penalty (lambda * sum_instance_weight, gblinear.cc
    DenormalizePenalties)."""
    rng = np.random.RandomState(0)
    n, f = 500, 4
    X = rng.randn(n, f).astype(np.float32)
    w_true = np.array([1.0, -2.0, 0.5, 0.0], np.float32)
    y = (X @ w_true).astype(np.float32)
    lam = 1.0
    bst = xgb.train({"booster": "gblinear", "eta": 0.8, "lambda": lam,
                     "alpha": 0.0, "base_score": 0.0},
                    xgb.DMatrix(X, label=y), 300)
    w_fit = bst._linear.weights[:-1, 0].numpy()
    A = X.T @ X + lam * n * np.eye(f)
    w_ridge = np.linalg.solve(A, X.T @ y)
    assert np.abs(w_fit - w_ridge).max() < 5e-3
