"""Metric value spot checks (reference analog: tests/cpp/metric/*)."""
import math

import numpy as np
import pytest

from xgboost_amd.data import MetaInfo
from xgboost_amd.metrics import create_metric


def _info(y, weights=None, group=None, **kw):
    y = np.asarray(y, np.float32)
    info = MetaInfo(num_row=len(y), num_col=1, labels=y)
    if weights is not None:
        info.weights = np.asarray(weights, np.float32)
    if group is not None:
        info.group_ptr = np.concatenate([[0], np.cumsum(group)]).astype(np.int64)
    for k, v in kw.items():
        setattr(info, k, v)
    return info


def test_rmse():
    m = create_metric("rmse")
    assert m(np.array([0.0, 2.0]), _info([1.0, 1.0])) == pytest.approx(1.0)


def test_mae():
    m = create_metric("mae")
    assert m(np.array([0.0, 3.0]), _info([1.0, 1.0])) == pytest.approx(1.5)


def test_logloss():
    m = create_metric("logloss")
    v = m(np.array([0.9, 0.1]), _info([1.0, 0.0]))
    assert v == pytest.approx(-math.log(0.9), rel=1e-6)


def test_error_threshold():
    m = create_metric("error")
    preds = np.array([0.3, 0.7])
    assert m(preds, _info([0.0, 0.0])) == pytest.approx(0.5)
    m2 = create_metric("error@0.8")
    assert m2(preds, _info([0.0, 0.0])) == pytest.approx(0.0)


def test_auc_perfect_and_random():
    m = create_metric("auc")
    y = np.array([0, 0, 1, 1], np.float32)
    assert m(np.array([0.1, 0.2, 0.8, 0.9]), _info(y)) == pytest.approx(1.0)
    assert m(np.array([0.9, 0.8, 0.2, 0.1]), _info(y)) == pytest.approx(0.0)
    # ties give 0.5
    assert m(np.array([0.5, 0.5, 0.5, 0.5]), _info(y)) == pytest.approx(0.5)


def test_auc_weighted():
    m = create_metric("auc")
    y = np.array([0, 1], np.float32)
    v = m(np.array([0.4, 0.6]), _info(y, weights=[2.0, 3.0]))
    assert v == pytest.approx(1.0)


def test_aucpr_range():
    m = create_metric("aucpr")
    y = np.array([0, 0, 1, 1], np.float32)
    v = m(np.array([0.1, 0.2, 0.8, 0.9]), _info(y))
    assert 0.99 <= v <= 1.0


def test_merror_mlogloss():
    y = np.array([0, 1, 2], np.float32)
    p = np.array([[0.8, 0.1, 0.1], [0.1, 0.8, 0.1], [0.8, 0.1, 0.1]])
    assert create_metric("merror")(p, _info(y)) == pytest.approx(1 / 3)
    assert create_metric("mlogloss")(p, _info(y)) == pytest.approx(
        -(math.log(0.8) * 2 + math.log(0.1)) / 3, rel=1e-6)


def test_ndcg():
    m = create_metric("ndcg")
    y = np.array([3, 2, 1, 0], np.float32)
    perfect = m(np.array([4.0, 3.0, 2.0, 1.0]), _info(y, group=[4]))
    assert perfect == pytest.approx(1.0)
    worse = m(np.array([1.0, 2.0, 3.0, 4.0]), _info(y, group=[4]))
    assert worse < 1.0


def test_ndcg_topn():
    m = create_metric("ndcg@2")
    y = np.array([0, 0, 1, 1], np.float32)
    v = m(np.array([0.9, 0.8, 0.2, 0.1]), _info(y, group=[4]))
    assert v == pytest.approx(0.0)


def test_map():
    m = create_metric("map")
    y = np.array([1, 0, 1, 0], np.float32)
    v = m(np.array([0.9, 0.8, 0.7, 0.1]), _info(y, group=[4]))
    # AP = (1/1 + 2/3) / 2
    assert v == pytest.approx((1.0 + 2 / 3) / 2, rel=1e-6)


def test_rmsle():
    m = create_metric("rmsle")
    v = m(np.array([math.e - 1]), _info([0.0]))
    assert v == pytest.approx(1.0, rel=1e-6)


def test_mape():
    m = create_metric("mape")
    assert m(np.array([2.0]), _info([1.0])) == pytest.approx(1.0)


def test_quantile_metric():
    m = create_metric("quantile@0.9")
    # under-prediction penalized by alpha
    assert m(np.array([0.0]), _info([1.0])) == pytest.approx(0.9)
    assert m(np.array([1.0]), _info([0.0])) == pytest.approx(0.1)


def test_poisson_nloglik():
    m = create_metric("poisson-nloglik")
    v = m(np.array([2.0]), _info([2.0]))
    expected = 2.0 - 2.0 * math.log(2.0) + math.lgamma(3.0)
    assert v == pytest.approx(expected, rel=1e-6)


def test_interval_accuracy():
    m = create_metric("interval-regression-accuracy")
    info = _info([1.0, 1.0],
                 label_lower_bound=np.array([0.5, 2.0], np.float32),
                 label_upper_bound=np.array([1.5, 3.0], np.float32))
    assert m(np.array([1.0, 1.0]), info) == pytest.approx(0.5)


def test_unknown_metric_raises():
    with pytest.raises(ValueError):
        create_metric("bogus")


def test_torch_metrics_match_numpy():
    """Device-resident metric implementations (torch path) must equal
    the numpy path bit-for-near (same fp64 math, different backend)."""
    import torch
    from xgboost_amd.data import MetaInfo
    from xgboost_amd.metrics import create_metric
    rng = np.random.RandomState(3)
    n = 5000
    p = rng.rand(n)
    y = (rng.rand(n) > 0.4).astype(np.float64)
    w = rng.rand(n) + 0.1
    info = MetaInfo(num_row=n)
    info.labels = y
    info.weights = w.astype(np.float32)
    for mname in ("rmse", "mae", "logloss", "error", "auc"):
        m = create_metric(mname)
        v_np = m(p, info)
        v_t = m(torch.from_numpy(p), info)
        assert abs(v_np - v_t) < 1e-10, (mname, v_np, v_t)
    # multiclass pair
    k = 4
    pm = rng.rand(n, k)
    pm /= pm.sum(1, keepdims=True)
    ym = rng.randint(0, k, n).astype(np.float64)
    info2 = MetaInfo(num_row=n)
    info2.labels = ym
    for mname in ("mlogloss", "merror"):
        m = create_metric(mname)
        v_np = m(pm, info2)
        v_t = m(torch.from_numpy(pm), info2)
        assert abs(v_np - v_t) < 1e-10, (mname, v_np, v_t)
    # ties in predictions exercise the AUC tie-merge
    pt = np.round(rng.rand(n), 2)
    m = create_metric("auc")
    assert abs(m(pt, info) - m(torch.from_numpy(pt), info)) < 1e-10


def test_degenerate_auc_is_nan():
    """reference auc.cc:351: single-class AUC -> NaN + warning."""
    import warnings
    import xgboost_amd as xgb
    X = np.random.RandomState(0).randn(30, 2).astype(np.float32)
    d = xgb.DMatrix(X, label=np.ones(30, np.float32))
    res = {}
    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        xgb.train({"objective": "binary:logistic", "eval_metric": "auc",
                   "max_depth": 2}, d, 1, evals=[(d, "t")],
                  evals_result=res, verbose_eval=False)
    assert np.isnan(res["t"]["auc"][0])


def test_rank_metric_minus_suffix():
    """reference rank_metric.cc:385/:446: a group with no relevant docs
    scores 1 by default and 0 with the `-` suffix, in both cases
    remaining in the denominator."""
    from xgboost_amd.data import MetaInfo
    from xgboost_amd.metrics import create_metric
    info = MetaInfo()
    info.labels = np.array([1, 0, 0, 0, 0, 0], np.float32)
    info.num_row = 6
    info.group_ptr = np.array([0, 3, 6], np.int64)  # group 2: no positives
    preds = np.array([0.9, 0.1, 0.2, 0.5, 0.4, 0.3], np.float64)
    for name in ("ndcg", "map"):
        full = create_metric(f"{name}@3")(preds, info)
        minus = create_metric(f"{name}@3-")(preds, info)
        # group1 scores 1.0 (perfect); group2: 1 vs 0
        assert full == pytest.approx((1.0 + 1.0) / 2)
        assert minus == pytest.approx((1.0 + 0.0) / 2)
