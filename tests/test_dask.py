"""Dask-integration orchestration tests.

`dask` is not installed in this image, so these tests drive
xgboost_amd.dask with a minimal fake client implementing the small
Client surface the module uses (submit/gather/scheduler_info) over a
real ProcessPoolExecutor — the workers are REAL separate processes
that rendezvous through the builder-owned RabitTracker and train with
gloo collectives, exactly like a dask cluster would."""
import multiprocessing
import os
from concurrent.futures import ProcessPoolExecutor

import numpy as np
import pytest

import xgboost_amd as xgb
from xgboost_amd.dask import DaskDMatrix, DaskXGBRegressor, predict, train


class FakeClient:
    def __init__(self, n_workers=2):
        # spawn: forking after torch has spun up OpenMP threads can
        # deadlock the children (locks held across fork)
        self._pool = ProcessPoolExecutor(
            max_workers=n_workers,
            mp_context=multiprocessing.get_context("spawn"))
        self._workers = [f"tcp://worker-{i}" for i in range(n_workers)]

    def scheduler_info(self):
        return {"workers": {w: {} for w in self._workers}}

    def submit(self, fn, *args, workers=None, pure=False):
        return self._pool.submit(fn, *args)

    def gather(self, futures):
        return [f.result(timeout=180) for f in futures]

    def close(self):
        self._pool.shutdown()


@pytest.fixture
def fake_client():
    c = FakeClient(2)
    yield c
    c.close()


def _data():
    rng = np.random.RandomState(11)
    n, f = 4000, 6
    X = rng.randn(n, f).astype(np.float32)
    w = rng.randn(f)
    y = (X @ w > 0).astype(np.float32)
    return X, y


def test_dask_train_two_real_workers(fake_client):
    X, y = _data()
    Xp = np.array_split(X, 4)
    yp = np.array_split(y, 4)
    d = DaskDMatrix(fake_client, Xp, yp)
    assert len(d.workers()) == 2  # partitions spread over both workers
    out = train(fake_client, {"objective": "binary:logistic",
                              "max_depth": 4, "eta": 0.3},
                d, num_boost_round=5, eval_train=True)
    bst = out["booster"]
    assert bst.num_boosted_rounds() == 5
    pred = bst.predict(xgb.DMatrix(X))
    acc = ((pred > 0.5) == y).mean()
    assert acc > 0.85, acc
    assert "train" in out["history"]

    # distributed prediction over partitions matches local predict
    pd = predict(fake_client, out, Xp)
    assert np.allclose(pd, pred, atol=1e-6)


def test_dask_sklearn_wrapper(fake_client):
    X, y = _data()
    reg = DaskXGBRegressor(client=fake_client, n_estimators=5, max_depth=4)
    reg.fit(np.array_split(X, 4), np.array_split(y, 4))
    p = reg.predict(np.array_split(X, 2))
    assert p.shape == (4000,)
    rmse = float(np.sqrt(((p - y) ** 2).mean()))
    assert rmse < 0.45, rmse
