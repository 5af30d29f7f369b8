"""BASELINE.json config 1: agaricus, tree_method=hist, CPU, 10 rounds

Data: the classic UCI mushroom dataset in libsvm form (the exact
files BASELINE.json's config 1 names, demo/data in the reference;
public-domain UCI data, included here as test fixtures).
binary:logistic — the reference's canonical plumbing test
(demo/guide-python; tests/python/test_basic.py uses the same files)."""
import os

import numpy as np
import pytest

import xgboost_amd as xgb

HERE = os.path.dirname(os.path.abspath(__file__))
TRAIN = os.path.join(HERE, "data", "agaricus.txt.train")
TEST = os.path.join(HERE, "data", "agaricus.txt.test")


@pytest.fixture(scope="module")
def agaricus():
    dtrain = xgb.DMatrix(TRAIN)
    dtest = xgb.DMatrix(TEST)
    return dtrain, dtest


def test_agaricus_train(agaricus):
    dtrain, dtest = agaricus
    assert dtrain.num_row() == 6513
    res = {}
    bst = xgb.train({"objective": "binary:logistic", "tree_method": "hist",
                     "max_depth": 2, "eta": 1.0,
                     "eval_metric": ["error", "logloss"]},
                    dtrain, 10, evals=[(dtest, "eval")],
                    evals_result=res, verbose_eval=False)
    # mushroom is almost perfectly separable: error must go ~0
    assert res["eval"]["error"][-1] < 0.01
    pred = bst.predict(dtest)
    y = dtest.get_label()
    acc = ((pred > 0.5) == y).mean()
    assert acc > 0.99


def test_agaricus_model_roundtrip(agaricus, tmp_path):
    dtrain, dtest = agaricus
    bst = xgb.train({"objective": "binary:logistic", "max_depth": 3},
                    dtrain, 5, verbose_eval=False)
    path = str(tmp_path / "agaricus.json")
    bst.save_model(path)
    bst2 = xgb.Booster(model_file=path)
    assert np.allclose(bst.predict(dtest), bst2.predict(dtest), atol=1e-7)
