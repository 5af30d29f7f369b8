"""Model serialization round-trips (reference analog:
tests/cpp/test_serialization.cc, tests/python/test_model_io.py)."""
import json
import os
import pickle

import numpy as np
import pytest

import xgboost_amd as xgb
from conftest import make_classification, make_regression


def _train(tmp_path, params=None, rounds=5):
    X, y = make_classification(800, 6)
    d = xgb.DMatrix(X, label=y)
    p = {"objective": "binary:logistic", "max_depth": 4}
    p.update(params or {})
    bst = xgb.train(p, d, rounds, verbose_eval=False)
    return bst, d, X, y


def test_json_roundtrip(tmp_path):
    bst, d, X, y = _train(tmp_path)
    path = str(tmp_path / "model.json")
    bst.save_model(path)
    bst2 = xgb.Booster(model_file=path)
    assert np.allclose(bst.predict(d), bst2.predict(d), atol=1e-7)
    # schema sanity: xgboost key layout
    with open(path) as fh:
        j = json.load(fh)
    assert "learner" in j and "version" in j
    lrn = j["learner"]
    assert "gradient_booster" in lrn and "learner_model_param" in lrn
    model = lrn["gradient_booster"]["model"]
    assert int(model["gbtree_model_param"]["num_trees"]) == 5
    t0 = model["trees"][0]
    for key in ("left_children", "right_children", "parents", "split_indices",
                "split_conditions", "default_left", "loss_changes",
                "sum_hessian", "base_weights", "tree_param"):
        assert key in t0, key


def test_ubjson_roundtrip(tmp_path):
    bst, d, X, y = _train(tmp_path)
    path = str(tmp_path / "model.ubj")
    bst.save_model(path)
    bst2 = xgb.Booster(model_file=path)
    assert np.allclose(bst.predict(d), bst2.predict(d), atol=1e-7)


def test_save_raw_load():
    bst, d, X, y = _train(None)
    raw = bst.save_raw("json")
    bst2 = xgb.Booster()
    bst2.load_model(bytes(raw))
    assert np.allclose(bst.predict(d), bst2.predict(d), atol=1e-7)
    raw_ubj = bst.save_raw("ubj")
    bst3 = xgb.Booster()
    bst3.load_model(bytes(raw_ubj))
    assert np.allclose(bst.predict(d), bst3.predict(d), atol=1e-7)


def test_pickle_roundtrip():
    bst, d, X, y = _train(None)
    blob = pickle.dumps(bst)
    bst2 = pickle.loads(blob)
    assert np.allclose(bst.predict(d), bst2.predict(d), atol=1e-7)


def test_multiclass_roundtrip(tmp_path):
    X, y = make_classification(600, 5, n_class=3)
    d = xgb.DMatrix(X, label=y)
    bst = xgb.train({"objective": "multi:softprob", "num_class": 3,
                     "max_depth": 3}, d, 4, verbose_eval=False)
    path = str(tmp_path / "m.json")
    bst.save_model(path)
    bst2 = xgb.Booster(model_file=path)
    assert np.allclose(bst.predict(d), bst2.predict(d), atol=1e-7)
    with open(path) as fh:
        j = json.load(fh)
    assert j["learner"]["learner_model_param"]["num_class"] == "3"


def test_continue_training(tmp_path):
    X, y = make_regression(600, 5)
    d = xgb.DMatrix(X, label=y)
    bst = xgb.train({"objective": "reg:squarederror"}, d, 5, verbose_eval=False)
    path = str(tmp_path / "m.json")
    bst.save_model(path)
    bst2 = xgb.train({"objective": "reg:squarederror"}, d, 5,
                     xgb_model=path, verbose_eval=False)
    assert bst2.num_boosted_rounds() == 10
    assert len(bst2.trees) == 10


def test_dump_formats():
    bst, d, X, y = _train(None, rounds=2)
    dumps = bst.get_dump()
    assert len(dumps) == 2
    assert "leaf=" in dumps[0]
    jd = bst.get_dump(dump_format="json")
    parsed = json.loads(jd[0])
    assert "nodeid" in parsed
    dot = bst.get_dump(dump_format="dot")
    assert dot[0].startswith("digraph")
    with_stats = bst.get_dump(with_stats=True)
    assert "cover=" in with_stats[0]


def test_attributes_roundtrip(tmp_path):
    bst, d, X, y = _train(None, rounds=2)
    bst.set_attr(foo="bar", n="1")
    path = str(tmp_path / "m.json")
    bst.save_model(path)
    bst2 = xgb.Booster(model_file=path)
    assert bst2.attr("foo") == "bar"
    assert bst2.attributes() == {"foo": "bar", "n": "1"}


def test_get_score():
    bst, d, X, y = _train(None)
    for imp in ("weight", "gain", "cover", "total_gain", "total_cover"):
        s = bst.get_score(importance_type=imp)
        assert len(s) > 0
        assert all(v > 0 for v in s.values())


def test_feature_names_mismatch_raises():
    X, y = make_classification(100, 5)
    d = xgb.DMatrix(X, label=y)
    bst = xgb.train({"objective": "binary:logistic"}, d, 2, verbose_eval=False)
    bad = xgb.DMatrix(X[:, :4], label=y)
    with pytest.raises(ValueError):
        bst.predict(bad)


def test_attributes_and_names_roundtrip(tmp_path):
    import numpy as np
    import xgboost_amd as xgb
    rng = np.random.RandomState(0)
    X = rng.randn(120, 3).astype(np.float32)
    y = X[:, 0].astype(np.float32)
    d = xgb.DMatrix(X, label=y, feature_names=["alpha", "beta", "gamma"],
                    feature_types=["q", "q", "q"])
    bst = xgb.train({"objective": "reg:squarederror", "max_depth": 3}, d, 3)
    bst.set_attr(best_iteration="2", my_tag="hello")
    path = str(tmp_path / "m.json")
    bst.save_model(path)
    bst2 = xgb.Booster(model_file=path)
    assert bst2.attr("my_tag") == "hello"
    assert bst2.attr("best_iteration") == "2"
    assert bst2.feature_names == ["alpha", "beta", "gamma"]
    assert bst2.feature_types == ["q", "q", "q"]
    score = bst2.get_score(importance_type="gain")
    assert all(k in ("alpha", "beta", "gamma") for k in score)
