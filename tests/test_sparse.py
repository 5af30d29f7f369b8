"""Sparse (CSR) path: quantization, training, predict consistency
(reference analog: sparse SparsePage tests; absent entry = missing)."""
import numpy as np
import pytest
import scipy.sparse as sp
import torch

import xgboost_amd as xgb
from xgboost_amd.sparse import CsrCpuOps, quantize_csr, sketch_csr
from xgboost_amd.backend.cpu import GradQuantizer


def _sparse_data(n=5000, f=200, nnz_row=10, seed=0):
    rng = np.random.RandomState(seed)
    rows = np.repeat(np.arange(n), nnz_row)
    cols = rng.randint(0, f, n * nnz_row)
    vals = rng.rand(n * nnz_row).astype(np.float32) + 0.1
    X = sp.csr_matrix((vals, (rows, cols)), shape=(n, f))
    X.sum_duplicates()
    # strong signal: value of feature 0 when present, else negative class
    has0 = np.asarray((X[:, 0] > 0.5).todense()).ravel()
    y = has0.astype(np.float32)
    return X, y


def test_quantize_csr_bins():
    X, y = _sparse_data(1000, 50, 5)
    cuts = sketch_csr(X, 16)
    sqm = quantize_csr(X, cuts)
    assert sqm.n_rows == 1000
    assert sqm.bin_idx.shape[0] == X.nnz
    bins = sqm.bin_idx.numpy()
    rp = sqm.row_ptr.numpy()
    # per-row bins sorted (column order) and within global range
    for r in range(0, 1000, 97):
        seg = bins[rp[r]:rp[r + 1]]
        assert np.all(np.diff(seg) > 0)
    assert bins.min() >= 0 and bins.max() < cuts.total_bins


def test_sparse_hist_matches_dense():
    """CSR histogram == dense histogram of the same data with NaN for
    absent entries."""
    X, y = _sparse_data(2000, 30, 6)
    cuts = sketch_csr(X, 32)
    sqm = quantize_csr(X, cuts)
    ops = CsrCpuOps(sqm)
    gpair = torch.tensor(
        np.stack([np.random.RandomState(1).randn(2000),
                  np.random.RandomState(2).rand(2000) + 0.1],
                 axis=1).astype(np.float32))
    quant = GradQuantizer(gpair)
    qg = quant.quantize(gpair)
    ops.reset(2000)
    h_sparse = ops.build_hist_nodes(qg, [0])
    # dense oracle
    from xgboost_amd.data import DMatrix, quantize_dense
    from xgboost_amd.backend.cpu import CpuOps
    Xd = np.full((2000, 30), np.nan, dtype=np.float32)
    coo = X.tocoo()
    Xd[coo.row, coo.col] = coo.data
    qm = quantize_dense(Xd, cuts)
    dops = CpuOps(qm)
    dops.reset(2000)
    h_dense = dops.build_hist_nodes(qg, [0])
    assert torch.equal(h_sparse, h_dense)


def test_sparse_training_learns():
    X, y = _sparse_data(5000, 200, 10)
    d = xgb.DMatrix(X, label=y)
    res = {}
    bst = xgb.train({"objective": "binary:logistic", "max_depth": 5,
                     "eta": 0.5, "eval_metric": "auc"}, d, 15,
                    evals=[(d, "t")], evals_result=res, verbose_eval=False)
    assert res["t"]["auc"][-1] > 0.95
    p = bst.predict(d)
    acc = ((p > 0.5) == y).mean()
    assert acc > 0.9


def test_sparse_predict_matches_cache():
    X, y = _sparse_data(3000, 100, 8)
    d = xgb.DMatrix(X, label=y)
    bst = xgb.train({"objective": "binary:logistic", "max_depth": 4},
                    d, 5, verbose_eval=False)
    p = bst.predict(d)
    cached = bst._cache[id(d)][0]
    pm = bst.objective.pred_transform(cached).cpu().numpy().reshape(-1)
    assert np.allclose(p, pm, atol=1e-5)


def test_sparse_matches_dense_training():
    """Same data sparse-CSR vs dense-with-NaN must give the same trees."""
    X, y = _sparse_data(3000, 40, 8)
    Xd = np.full((3000, 40), np.nan, dtype=np.float32)
    coo = X.tocoo()
    Xd[coo.row, coo.col] = coo.data
    params = {"objective": "binary:logistic", "max_depth": 4, "eta": 0.3,
              "max_bin": 64}
    b1 = xgb.train(params, xgb.DMatrix(X, label=y), 5, verbose_eval=False)
    b2 = xgb.train(params, xgb.DMatrix(Xd, label=y), 5, verbose_eval=False)
    for t1, t2 in zip(b1.trees, b2.trees):
        assert t1.n_nodes == t2.n_nodes
        assert np.array_equal(t1.split_index[:t1.n_nodes],
                              t2.split_index[:t2.n_nodes])


@pytest.mark.gpu
def test_sparse_gpu_matches_cpu():
    X, y = _sparse_data(20000, 300, 12)
    params = {"objective": "binary:logistic", "max_depth": 5, "eta": 0.3}
    b_cpu = xgb.train(dict(params, device="cpu"), xgb.DMatrix(X, label=y),
                      5, verbose_eval=False)
    b_gpu = xgb.train(dict(params, device="cuda"), xgb.DMatrix(X, label=y),
                      5, verbose_eval=False)
    for t1, t2 in zip(b_cpu.trees, b_gpu.trees):
        assert t1.n_nodes == t2.n_nodes
        assert np.array_equal(t1.split_index[:t1.n_nodes],
                              t2.split_index[:t2.n_nodes])
        assert np.allclose(t1.split_cond[:t1.n_nodes],
                           t2.split_cond[:t2.n_nodes], rtol=1e-4, atol=1e-6)


def test_dart_with_sparse_training():
    """DART dropped-tree margins on scipy CSR training data route
    through the used-features sparse traversal (no densification of
    the full matrix)."""
    import torch
    from scipy import sparse
    rng = np.random.RandomState(9)
    csr = sparse.random(800, 50, density=0.1, format="csr",
                        dtype=np.float32, random_state=rng)
    y = np.asarray(csr @ rng.randn(50)).ravel().astype(np.float32)
    d = xgb.DMatrix(csr, label=y)
    bst = xgb.train({"max_depth": 3, "eta": 0.3, "rate_drop": 0.5,
                     "seed": 5}, d, 6)
    assert len(bst.weight_drop) == 6
    cached, _ = bst._cache[id(d)]
    fresh = bst._predict_margin(d)
    assert torch.allclose(cached, fresh, atol=1e-4), \
        (cached - fresh).abs().max()


def test_sparse_equals_dense_nan_training():
    """CSR missing entries behave exactly like NaN in a dense matrix:
    first-round trees match exactly; later rounds may break exact gain
    TIES differently between the two code paths, but predictions
    coincide on the training data."""
    from scipy import sparse
    rng = np.random.RandomState(0)
    csr = sparse.random(500, 20, density=0.2, format="csr",
                        dtype=np.float32, random_state=rng)
    y = np.asarray(csr @ rng.randn(20)).ravel().astype(np.float32)
    p = {"max_depth": 4, "max_bin": 64, "seed": 1}
    Xd = np.full((500, 20), np.nan, np.float32)
    coo = csr.tocoo()
    Xd[coo.row, coo.col] = coo.data
    bs = xgb.train(p, xgb.DMatrix(csr, label=y), 1)
    bd = xgb.train(p, xgb.DMatrix(Xd, label=y), 1)
    assert bs.get_dump(with_stats=True) == bd.get_dump(with_stats=True)
    bs5 = xgb.train(p, xgb.DMatrix(csr, label=y), 5)
    bd5 = xgb.train(p, xgb.DMatrix(Xd, label=y), 5)
    assert np.allclose(bs5.predict(xgb.DMatrix(Xd)),
                       bd5.predict(xgb.DMatrix(Xd)), atol=1e-6)
