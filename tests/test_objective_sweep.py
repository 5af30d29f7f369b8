"""Integration sweep: every registered objective trains, predicts and
round-trips (reference analog: objective coverage across
tests/python/test_objectives.py)."""
import numpy as np
import pytest

import xgboost_amd as xgb
from xgboost_amd.objectives import _REGISTRY


def _data_for(name, n=400, seed=0):
    rng = np.random.RandomState(seed)
    X = rng.randn(n, 5).astype(np.float32)
    kwargs = {}
    params = {"max_depth": 3, "eta": 0.3}
    if name.startswith("binary"):
        y = (X[:, 0] > 0).astype(np.float32)
    elif name.startswith("multi"):
        y = np.abs(X[:, :3]).argmax(axis=1).astype(np.float32)
        params["num_class"] = 3
    elif name.startswith("rank"):
        if name == "rank:map":
            # reference CheckPreLabels: MAP requires binary relevance
            y = (X[:, 0] > 0).astype(np.float32)
        else:
            y = np.clip((X[:, 0] * 2 + 2).astype(int), 0,
                        3).astype(np.float32)
        kwargs["qid"] = np.repeat(np.arange(20), n // 20)
    elif name == "survival:cox":
        t = np.exp(X[:, 0] * 0.5 + 2)
        event = rng.rand(n) > 0.3
        y = np.where(event, t, -t).astype(np.float32)
    elif name == "survival:aft":
        t = np.exp(X[:, 0] * 0.5 + 2).astype(np.float32)
        y = t
        kwargs["label_lower_bound"] = t
        kwargs["label_upper_bound"] = np.where(rng.rand(n) > 0.2, t,
                                               np.inf).astype(np.float32)
    elif name in ("count:poisson",):
        y = rng.poisson(np.exp(0.3 * X[:, 0]) + 0.5).astype(np.float32)
    elif name in ("reg:gamma", "reg:tweedie", "reg:squaredlogerror"):
        y = (np.exp(0.3 * X[:, 0]) + 0.1).astype(np.float32)
    else:
        y = (X[:, 0] + 0.1 * rng.randn(n)).astype(np.float32)
    return xgb.DMatrix(X, label=y, **kwargs), params


@pytest.mark.parametrize("name", sorted(_REGISTRY.keys()))
def test_objective_end_to_end(name, tmp_path):
    d, params = _data_for(name)
    params["objective"] = name
    res = {}
    bst = xgb.train(params, d, 5, evals=[(d, "t")], evals_result=res,
                    verbose_eval=False)
    # loss decreases (or at least training produced trees)
    assert len(bst.trees) >= 5
    metric = list(res["t"].keys())[0]
    vals = res["t"][metric]
    assert np.isfinite(vals[-1])
    p = bst.predict(d)
    assert np.isfinite(p).all()
    # round-trip
    path = str(tmp_path / "m.json")
    bst.save_model(path)
    bst2 = xgb.Booster(model_file=path)
    assert np.allclose(bst2.predict(d), p, atol=1e-6)
