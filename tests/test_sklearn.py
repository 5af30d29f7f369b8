"""sklearn wrapper tests (reference analog: tests/python/
test_with_sklearn.py)."""
import numpy as np
import pytest

import xgboost_amd as xgb
from xgboost_amd.sklearn import (XGBClassifier, XGBRanker, XGBRegressor,
                                 XGBRFClassifier, XGBRFRegressor)
from conftest import make_classification, make_regression


def test_regressor_fit_predict():
    X, y = make_regression(1500, 8)
    m = XGBRegressor(n_estimators=30, max_depth=4, learning_rate=0.3)
    m.fit(X[:1000], y[:1000])
    pred = m.predict(X[1000:])
    assert pred.shape == (500,)
    r2 = m.score(X[1000:], y[1000:])
    assert r2 > 0.7
    assert m.n_features_in_ == 8


def test_classifier_binary():
    X, y = make_classification(1500, 8)
    m = XGBClassifier(n_estimators=20, max_depth=4)
    m.fit(X[:1000], y[:1000])
    pred = m.predict(X[1000:])
    assert set(np.unique(pred)) <= {0.0, 1.0}
    proba = m.predict_proba(X[1000:])
    assert proba.shape == (500, 2)
    assert np.allclose(proba.sum(axis=1), 1.0, atol=1e-5)
    acc = m.score(X[1000:], y[1000:])
    assert acc > 0.8


def test_classifier_multiclass_label_encoding():
    X, _ = make_classification(900, 6, n_class=3)
    rng = np.random.RandomState(0)
    labels = np.array(["a", "b", "c"])[rng.randint(0, 3, 900)]
    # make labels learnable
    labels = np.array(["a", "b", "c"])[
        (X[:, :3].argmax(axis=1))]
    m = XGBClassifier(n_estimators=10, max_depth=3)
    m.fit(X, labels)
    assert list(m.classes_) == ["a", "b", "c"]
    pred = m.predict(X)
    assert set(pred) <= {"a", "b", "c"}
    assert (pred == labels).mean() > 0.8
    proba = m.predict_proba(X)
    assert proba.shape == (900, 3)


def test_early_stopping_sklearn():
    X, y = make_classification(1500, 8)
    m = XGBClassifier(n_estimators=500, max_depth=5, learning_rate=0.5,
                      early_stopping_rounds=5)
    m.fit(X[:1000], y[:1000], eval_set=[(X[1000:], y[1000:])], verbose=False)
    assert m.get_booster().num_boosted_rounds() < 500
    assert hasattr(m, "best_iteration")


def test_feature_importances():
    X, y = make_regression(500, 5)
    m = XGBRegressor(n_estimators=5, max_depth=3).fit(X, y)
    imp = m.feature_importances_
    assert imp.shape == (5,)
    assert abs(imp.sum() - 1.0) < 1e-5


def test_get_set_params():
    m = XGBRegressor(n_estimators=10, max_depth=3)
    params = m.get_params()
    assert params["n_estimators"] == 10
    m.set_params(max_depth=7)
    assert m.max_depth == 7


def test_sklearn_clone_compat():
    from sklearn.base import clone
    m = XGBRegressor(n_estimators=5, max_depth=3)
    m2 = clone(m)
    assert m2.get_params()["n_estimators"] == 5


def test_sklearn_model_io(tmp_path):
    X, y = make_regression(300, 4)
    m = XGBRegressor(n_estimators=5).fit(X, y)
    path = str(tmp_path / "m.json")
    m.save_model(path)
    m2 = XGBRegressor()
    m2.load_model(path)
    assert np.allclose(m.predict(X), m2.predict(X), atol=1e-6)


def test_ranker():
    rng = np.random.RandomState(0)
    n, f = 1200, 6
    X = rng.randn(n, f).astype(np.float32)
    qid = np.repeat(np.arange(60), 20)
    rel = (X[:, 0] + 0.5 * rng.randn(n))
    y = np.zeros(n, np.float32)
    for q in range(60):
        m_ = qid == q
        y[m_] = np.argsort(np.argsort(rel[m_])) // 5  # 0..3 grades
    m = XGBRanker(n_estimators=20, max_depth=4, learning_rate=0.3)
    m.fit(X, y, qid=qid)
    scores = m.predict(X)
    # ranking should correlate with relevance
    from scipy.stats import spearmanr
    rho = spearmanr(scores, y).statistic
    assert rho > 0.5, rho


def test_rf_regressor():
    X, y = make_regression(1000, 6)
    m = XGBRFRegressor(n_estimators=20, max_depth=5)
    m.fit(X, y)
    b = m.get_booster()
    assert len(b.trees) == 20
    assert b.num_boosted_rounds() == 1
    pred = m.predict(X)
    rmse = np.sqrt(np.mean((pred - y) ** 2))
    assert rmse < np.std(y)


def test_rf_classifier():
    X, y = make_classification(1000, 6)
    m = XGBRFClassifier(n_estimators=10, max_depth=5)
    m.fit(X, y)
    assert len(m.get_booster().trees) == 10
    assert m.score(X, y) > 0.8


def test_unfitted_raises():
    m = XGBRegressor()
    with pytest.raises(ValueError):
        m.get_booster()


def test_sklearn_extended_surface():
    import numpy as np
    import xgboost_amd as xgb
    rng = np.random.RandomState(0)
    X = rng.randn(200, 4).astype(np.float32)
    y = X[:, 0].astype(np.float32)
    m = xgb.XGBRegressor(n_estimators=5, booster="gblinear")
    m.fit(X, y, verbose=False)
    assert m.coef_.shape == (4,)
    assert m.get_num_boosting_rounds() == 5
    # best_iteration raises without early stopping
    import pytest
    with pytest.raises(AttributeError):
        _ = m.best_iteration
    m2 = xgb.XGBRegressor(n_estimators=50, early_stopping_rounds=3)
    m2.fit(X, y, eval_set=[(X, y)], verbose=False)
    assert isinstance(m2.best_iteration, int)
    assert np.isfinite(m2.best_score)
    # ranker score
    q = np.repeat(np.arange(10), 20)
    yr = (rng.rand(200) * 3).astype(np.float32)
    r = xgb.XGBRanker(n_estimators=4)
    r.fit(X, yr, qid=q, verbose=False)
    s = r.score(X, yr, qid=q)
    assert 0.0 <= s <= 1.0


def test_sklearn_callable_objective():
    """reference _objective_decorator: a callable objective on the
    sklearn wrapper takes (y_true, y_pred) -> (grad, hess)."""
    rng = np.random.RandomState(5)
    X = rng.randn(800, 5).astype(np.float32)
    y = (X[:, 0] * 2 + rng.randn(800) * 0.1).astype(np.float32)

    def sq_err(y_true, y_pred):
        return (y_pred - y_true), np.ones_like(y_true)

    m1 = xgb.XGBRegressor(n_estimators=8, max_depth=4,
                          objective=sq_err).fit(X, y)
    m2 = xgb.XGBRegressor(n_estimators=8, max_depth=4,
                          objective="reg:squarederror").fit(X, y)
    p1, p2 = m1.predict(X), m2.predict(X)
    assert np.allclose(p1, p2, atol=1e-5), np.abs(p1 - p2).max()


def test_sklearn_meta_estimator_compat():
    """sklearn >= 1.6 tags protocol: GridSearchCV / cross_val_score /
    Pipeline work (estimators inherit BaseEstimator + mixins like the
    reference's XGBModelBase)."""
    from sklearn.model_selection import GridSearchCV, cross_val_score
    from sklearn.pipeline import Pipeline
    from sklearn.preprocessing import StandardScaler
    rng = np.random.RandomState(0)
    X = rng.randn(150, 4).astype(np.float32)
    y = (X[:, 0] > 0).astype(np.float32)
    gs = GridSearchCV(xgb.XGBClassifier(n_estimators=4),
                      {"max_depth": [2, 3]}, cv=2)
    gs.fit(X, y)
    assert gs.best_score_ > 0.8
    sc = cross_val_score(xgb.XGBRegressor(n_estimators=4), X, X[:, 0], cv=2)
    assert (sc > 0.5).all()
    pipe = Pipeline([("s", StandardScaler()),
                     ("m", xgb.XGBRegressor(n_estimators=4))])
    pipe.fit(X, X[:, 0])
    assert pipe.score(X, X[:, 0]) > 0.5


def test_sklearn_callable_eval_metric():
    """reference _metric_decorator: eval_metric may be an sklearn-style
    callable (y_true, y_pred) used as the custom metric."""
    from sklearn.metrics import mean_absolute_error
    rng = np.random.RandomState(3)
    X = rng.randn(120, 4).astype(np.float32)
    y = X[:, 0].astype(np.float32)
    m = xgb.XGBRegressor(n_estimators=5,
                         eval_metric=mean_absolute_error)
    m.fit(X, y, eval_set=[(X, y)], verbose=False)
    vals = m.evals_result_["validation_0"]["mean_absolute_error"]
    assert len(vals) == 5 and vals[-1] < vals[0]
    # early stopping on the callable metric
    m2 = xgb.XGBRegressor(n_estimators=50,
                          eval_metric=mean_absolute_error,
                          early_stopping_rounds=3)
    m2.fit(X, y, eval_set=[(X, y)], verbose=False)
    assert m2.get_booster().best_iteration is not None


def test_rf_wrapper_defaults():
    """reference XGBRF defaults: one boosting round of n_estimators
    parallel trees, eta=1, subsample=0.8, colsample_bynode=0.8,
    reg_lambda=1e-5 (sklearn.py:2053-2056)."""
    rng = np.random.RandomState(0)
    X = rng.randn(80, 3).astype(np.float32)
    m = xgb.XGBRFRegressor(n_estimators=6).fit(X, X[:, 0])
    b = m.get_booster()
    assert b.num_boosted_rounds() == 1
    assert len(b.trees) == 6
    tp = b.tparam
    assert (tp.num_parallel_tree, tp.eta, tp.subsample,
            tp.colsample_bynode) == (6, 1.0, 0.8, 0.8)
    assert tp.reg_lambda == pytest.approx(1e-5)
