"""External-memory / DataIter tests (reference analog:
tests/python/test_data_iterator.py)."""
import numpy as np
import pytest

import xgboost_amd as xgb
from xgboost_amd.extmem import DataIter, ExtMemQuantileDMatrix
from conftest import make_classification


class NumpyBatchIter(DataIter):
    def __init__(self, Xs, ys):
        super().__init__()
        self.Xs = Xs
        self.ys = ys
        self.i = 0

    def reset(self):
        self.i = 0

    def next(self, input_data) -> bool:
        if self.i >= len(self.Xs):
            return False
        input_data(data=self.Xs[self.i], label=self.ys[self.i])
        self.i += 1
        return True


@pytest.fixture
def batched_data():
    X, y = make_classification(4000, 8, seed=5)
    Xs = np.array_split(X, 4)
    ys = np.array_split(y, 4)
    return X, y, Xs, ys


def test_extmem_construction(batched_data):
    X, y, Xs, ys = batched_data
    d = ExtMemQuantileDMatrix(NumpyBatchIter(Xs, ys), max_bin=64)
    assert d.num_row() == 4000
    assert d.num_col() == 8
    assert len(d.pages) == 4
    assert d.get_label().shape == (4000,)


def test_extmem_cuts_close_to_incore(batched_data):
    X, y, Xs, ys = batched_data
    d_ext = ExtMemQuantileDMatrix(NumpyBatchIter(Xs, ys), max_bin=64)
    d_in = xgb.DMatrix(X, label=y)
    cuts_in = d_in.quantized(64).cuts
    # bin counts should agree approximately (sketch vs exact)
    for f in range(8):
        assert abs(d_ext.cuts.n_bins(f) - cuts_in.n_bins(f)) <= 4


def test_extmem_training_matches_incore(batched_data):
    X, y, Xs, ys = batched_data
    d_ext = ExtMemQuantileDMatrix(NumpyBatchIter(Xs, ys), max_bin=64)
    params = {"objective": "binary:logistic", "max_depth": 4, "eta": 0.3,
              "max_bin": 64}
    res_ext = {}
    bst_ext = xgb.train(params, d_ext, 10, evals=[(d_ext, "t")],
                        evals_result=res_ext, verbose_eval=False)
    d_in = xgb.DMatrix(X, label=y)
    res_in = {}
    xgb.train(params, d_in, 10, evals=[(d_in, "t")],
              evals_result=res_in, verbose_eval=False)
    # same algorithm, cuts differ slightly (sketch); quality must match
    assert abs(res_ext["t"]["logloss"][-1] - res_in["t"]["logloss"][-1]) < 0.05
    # fresh prediction on the ext-mem matrix (bin-based traversal) must
    # agree with the training cache
    p = bst_ext.predict(d_ext)
    cached = bst_ext._cache[id(d_ext)][0]
    import torch
    pm = bst_ext.objective.pred_transform(cached).cpu().numpy().reshape(-1)
    assert np.allclose(p, pm, atol=1e-5)


def test_extmem_with_ref(batched_data):
    X, y, Xs, ys = batched_data
    d_train = xgb.DMatrix(X, label=y)
    d_train.quantized(64)
    d_val = ExtMemQuantileDMatrix(NumpyBatchIter(Xs, ys), max_bin=64,
                                  ref=d_train)
    # shares the training cut points exactly
    assert np.array_equal(d_val.cuts.values,
                          d_train.quantized(64).cuts.values)


def test_quantile_dmatrix_from_iterator(batched_data):
    X, y, Xs, ys = batched_data
    d = xgb.QuantileDMatrix(NumpyBatchIter(Xs, ys), max_bin=64)
    assert d.num_row() == 4000
    res = {}
    xgb.train({"objective": "binary:logistic", "max_bin": 64,
               "max_depth": 4}, d, 5, evals=[(d, "t")],
              evals_result=res, verbose_eval=False)
    assert res["t"]["logloss"][-1] < 0.5


@pytest.mark.gpu
def test_extmem_gpu_training(batched_data):
    X, y, Xs, ys = batched_data
    d_ext = ExtMemQuantileDMatrix(NumpyBatchIter(Xs, ys), max_bin=64)
    res = {}
    xgb.train({"objective": "binary:logistic", "max_depth": 4,
               "device": "cuda", "max_bin": 64}, d_ext, 10,
              evals=[(d_ext, "t")], evals_result=res, verbose_eval=False)
    assert res["t"]["logloss"][-1] < 0.45


class CachedBatchIter(NumpyBatchIter):
    def __init__(self, Xs, ys, cache_prefix):
        super().__init__(Xs, ys)
        self.cache_prefix = cache_prefix


def test_extmem_disk_spill_matches_incore(batched_data, tmp_path):
    """Pages beyond the host budget spill to <cache_prefix>.pageN.bin
    and stream back through the 2-deep read-ahead ring (reference
    sparse_page_source.h disk cache): training must match the all-RAM
    external-memory path exactly."""
    X, y, Xs, ys = batched_data
    prefix = str(tmp_path / "cache")
    # budget fits only the first page -> pages 1..3 go to disk
    page_bytes = Xs[0].shape[0] * Xs[0].shape[1]  # u8 bins
    d_disk = ExtMemQuantileDMatrix(
        CachedBatchIter(Xs, ys, prefix), max_bin=64,
        max_host_cache_bytes=page_bytes + 1)
    assert sum(d_disk.store.is_disk(i) for i in range(4)) == 3
    import os
    assert os.path.exists(prefix + ".page1.bin")
    d_ram = ExtMemQuantileDMatrix(NumpyBatchIter(Xs, ys), max_bin=64)
    params = {"objective": "binary:logistic", "max_depth": 5, "eta": 0.3}
    b1 = xgb.train(params, d_disk, 5, verbose_eval=False)
    b2 = xgb.train(params, d_ram, 5, verbose_eval=False)
    assert bytes(b1.save_raw("json")) == bytes(b2.save_raw("json"))
    # predict traverses disk pages through page_qm
    p1 = b1.predict(d_disk)
    p2 = b2.predict(d_ram)
    assert np.allclose(p1, p2, atol=1e-7)


def test_extmem_disk_spill_requires_prefix(batched_data):
    """Without a cache_prefix the budget is ignored (pages stay in
    host RAM) — spilling must be an explicit opt-in."""
    X, y, Xs, ys = batched_data
    d = ExtMemQuantileDMatrix(NumpyBatchIter(Xs, ys), max_bin=64,
                              max_host_cache_bytes=1)
    assert not any(d.store.is_disk(i) for i in range(4))


def test_dart_with_external_memory():
    """DART's dropped-tree contributions on quantized-only pages use
    the bin-based traversal (no raw data exists to re-read)."""
    import torch
    rng = np.random.RandomState(31)
    batches = [(rng.randn(300, 5).astype(np.float32),
                rng.randn(300).astype(np.float32)) for _ in range(3)]

    class It(DataIter):
        def __init__(self):
            super().__init__()
            self.i = 0

        def reset(self):
            self.i = 0

        def next(self, input_data):
            if self.i >= len(batches):
                return False
            X, y = batches[self.i]
            input_data(data=X, label=y)
            self.i += 1
            return self.i < len(batches)

    d = ExtMemQuantileDMatrix(It(), max_bin=64)
    bst = xgb.train({"max_depth": 3, "eta": 0.3, "rate_drop": 0.5,
                     "seed": 3}, d, 6)
    assert len(bst.weight_drop) == 6
    assert any(w != 1.0 for w in bst.weight_drop)
    cached, _ = bst._cache[id(d)]
    fresh = bst._predict_margin(d)
    assert torch.allclose(cached, fresh, atol=1e-4), \
        (cached - fresh).abs().max()
