"""Tree grower internals: deterministic quantized histograms, sibling
subtraction, constraints (reference analog: tests/cpp/tree/*)."""
import numpy as np
import pytest
import torch

import xgboost_amd as xgb
from xgboost_amd.backend.cpu import CpuOps, GradQuantizer
from xgboost_amd.data import DMatrix
from xgboost_amd.params import make_train_param
from conftest import make_classification, make_regression


def _setup(n=500, f=4, seed=0):
    rng = np.random.RandomState(seed)
    X = rng.randn(n, f).astype(np.float32)
    y = (X[:, 0] > 0).astype(np.float32)
    d = DMatrix(X, label=y)
    qm = d.quantized(64)
    ops = CpuOps(qm)
    g = torch.tensor(0.5 - y).view(-1, 1)
    h = torch.full((n, 1), 0.25)
    gpair = torch.cat([g, h], dim=1)
    quant = GradQuantizer(gpair)
    qg = quant.quantize(gpair)
    return ops, quant, qg, X, y


def test_hist_deterministic_under_permutation():
    ops, quant, qg, X, y = _setup()
    n = X.shape[0]
    ridx1 = torch.arange(n)
    ridx2 = torch.randperm(n)
    h1 = ops.build_hist(qg, ridx1, [(0, n)])
    h2 = ops.build_hist(qg, ridx2, [(0, n)])
    assert torch.equal(h1, h2)  # bit-exact: int64 fixed point


def test_hist_sibling_subtraction_exact():
    ops, quant, qg, X, y = _setup()
    n = X.shape[0]
    ridx = ops.make_ridx(n)
    parent = ops.build_hist(qg, ridx, [(0, n)])
    # split on feature 1 at median bin
    mid = n // 2
    left_rows = torch.arange(0, mid)
    right_rows = torch.arange(mid, n)
    hl = ops.build_hist(qg, left_rows, [(0, mid)])
    hr = ops.build_hist(qg, right_rows, [(0, n - mid)])
    assert torch.equal(parent, hl + hr)


def test_quantizer_roundtrip_precision():
    gpair = torch.tensor([[0.5, 0.25], [-0.3, 0.25], [1e-4, 0.25]])
    q = GradQuantizer(gpair)
    qg = q.quantize(gpair)
    back_g = qg[:, 0].double() / q.g_scale
    assert torch.allclose(back_g.float(), gpair[:, 0], atol=1e-6)


def test_monotone_constraints():
    rng = np.random.RandomState(0)
    n = 3000
    X = rng.rand(n, 2).astype(np.float32)
    y = (X[:, 0] + 0.2 * rng.randn(n)).astype(np.float32)
    d = xgb.DMatrix(X, label=y)
    bst = xgb.train({"objective": "reg:squarederror", "max_depth": 4,
                     "monotone_constraints": [1, 0], "eta": 0.5},
                    d, 10, verbose_eval=False)
    # predictions must be non-decreasing in feature 0
    grid = np.linspace(0.01, 0.99, 50, dtype=np.float32)
    for x2 in (0.2, 0.8):
        Xq = np.stack([grid, np.full_like(grid, x2)], axis=1)
        p = bst.predict(xgb.DMatrix(Xq))
        assert np.all(np.diff(p) >= -1e-5)
    # decreasing constraint
    bst2 = xgb.train({"objective": "reg:squarederror", "max_depth": 4,
                      "monotone_constraints": [-1, 0], "eta": 0.5},
                     d, 10, verbose_eval=False)
    Xq = np.stack([grid, np.full_like(grid, 0.5)], axis=1)
    p = bst2.predict(xgb.DMatrix(Xq))
    assert np.all(np.diff(p) <= 1e-5)


def test_interaction_constraints():
    X, y = make_regression(2000, 4)
    d = xgb.DMatrix(X, label=y)
    bst = xgb.train({"objective": "reg:squarederror", "max_depth": 5,
                     "interaction_constraints": [[0, 1], [2, 3]]},
                    d, 5, verbose_eval=False)
    # no path may mix features from different constraint sets
    for tree in bst.trees:
        def walk(nid, path_feats):
            if tree.is_leaf(nid):
                return
            f = int(tree.split_index[nid])
            group = 0 if f in (0, 1) else 1
            for pf in path_feats:
                pg = 0 if pf in (0, 1) else 1
                assert pg == group, f"mixed features {path_feats} + {f}"
            walk(int(tree.left[nid]), path_feats + [f])
            walk(int(tree.right[nid]), path_feats + [f])
        walk(0, [])


def test_gamma_prunes_splits():
    X, y = make_classification(1000, 6)
    d = xgb.DMatrix(X, label=y)
    b0 = xgb.train({"objective": "binary:logistic", "max_depth": 6,
                    "gamma": 0.0}, d, 3, verbose_eval=False)
    b1 = xgb.train({"objective": "binary:logistic", "max_depth": 6,
                    "gamma": 10.0}, d, 3, verbose_eval=False)
    n0 = sum(t.n_nodes for t in b0.trees)
    n1 = sum(t.n_nodes for t in b1.trees)
    assert n1 < n0


def test_min_child_weight():
    X, y = make_classification(500, 4)
    d = xgb.DMatrix(X, label=y)
    bst = xgb.train({"objective": "binary:logistic", "max_depth": 8,
                     "min_child_weight": 20.0}, d, 2, verbose_eval=False)
    # every leaf must have cover >= min_child_weight
    for t in bst.trees:
        for nid in range(t.n_nodes):
            if t.is_leaf(nid) and t.sum_hess[nid] > 0:
                assert t.sum_hess[nid] >= 20.0 - 1e-3


def test_lossguide_vs_depthwise_gain_order():
    X, y = make_classification(2000, 8)
    d = xgb.DMatrix(X, label=y)
    res_d, res_l = {}, {}
    xgb.train({"objective": "binary:logistic", "grow_policy": "depthwise",
               "max_depth": 4}, d, 5, evals=[(d, "t")], evals_result=res_d,
              verbose_eval=False)
    xgb.train({"objective": "binary:logistic", "grow_policy": "lossguide",
               "max_leaves": 16, "max_depth": 0}, d, 5, evals=[(d, "t")],
              evals_result=res_l, verbose_eval=False)
    assert res_l["t"]["logloss"][-1] < 0.6
    assert res_d["t"]["logloss"][-1] < 0.6


def test_categorical_onehot():
    rng = np.random.RandomState(0)
    n = 2000
    cat = rng.randint(0, 5, n).astype(np.float32)
    noise = rng.randn(n).astype(np.float32)
    y = (np.isin(cat, [1, 3]).astype(np.float32) * 2.0
         + 0.1 * rng.randn(n)).astype(np.float32)
    X = np.stack([cat, noise], axis=1)
    d = xgb.DMatrix(X, label=y, feature_types=["c", "q"])
    bst = xgb.train({"objective": "reg:squarederror", "max_depth": 4,
                     "eta": 0.5}, d, 10, verbose_eval=False)
    p = bst.predict(d)
    rmse = np.sqrt(np.mean((p - y) ** 2))
    assert rmse < 0.5
    # model uses categorical split nodes
    has_cat = any(t.cat_segments for t in bst.trees)
    assert has_cat
    # round-trip keeps categories
    raw = bst.save_raw("json")
    bst2 = xgb.Booster()
    bst2.load_model(bytes(raw))
    assert np.allclose(bst2.predict(d), p, atol=1e-6)


def test_categorical_partition_split():
    """Wide categorical features use sorted-partition subsets
    (reference: max_cat_to_onehot threshold + SortHistogram path)."""
    rng = np.random.RandomState(0)
    n = 4000
    cat = rng.randint(0, 30, n).astype(np.float32)
    good = [1, 5, 7, 12, 19, 22, 28]
    y = (np.isin(cat, good).astype(np.float32) * 2.0
         + 0.1 * rng.randn(n)).astype(np.float32)
    X = np.stack([cat, rng.randn(n).astype(np.float32)], axis=1)
    d = xgb.DMatrix(X, label=y, feature_types=["c", "q"])
    bst = xgb.train({"objective": "reg:squarederror", "max_depth": 3,
                     "eta": 0.5}, d, 8, verbose_eval=False)
    p = bst.predict(d)
    assert np.sqrt(np.mean((p - y) ** 2)) < 0.3
    t = bst.trees[0]
    assert t.split_type[0] == 1
    cats_right = set(int(c) for c in t.cat_segments[0])
    # the good categories must be cleanly separated at the root
    assert cats_right == set(range(30)) - set(good) or \
        cats_right == set(good)
    raw = bst.save_raw("json")
    b2 = xgb.Booster()
    b2.load_model(bytes(raw))
    assert np.allclose(b2.predict(d), p, atol=1e-6)


def test_native_cpu_hist_matches_torch_oracle():
    """The C (OpenMP) CPU kernels must match the pure-torch oracle
    bit-for-bit (same int64 fixed-point scheme)."""
    rng = np.random.RandomState(3)
    X = rng.randn(5000, 7).astype(np.float32)
    X[rng.rand(5000, 7) < 0.1] = np.nan
    d = DMatrix(X)
    qm = d.quantized(64)
    native = CpuOps(qm, use_native=True)
    oracle = CpuOps(qm, use_native=False)
    if native.lib is None:
        pytest.skip("native CPU kernels unavailable")
    gpair = torch.tensor(np.stack([rng.randn(5000),
                                   rng.rand(5000) + 0.1],
                                  axis=1).astype(np.float32))
    quant = GradQuantizer(gpair)
    qg = quant.quantize(gpair)
    segs = [(0, 2000), (2000, 5000)]
    h1 = native.build_hist(qg, native.make_ridx(5000), segs)
    h2 = oracle.build_hist(qg, oracle.make_ridx(5000), segs)
    assert torch.equal(h1, h2)
    # partition equality (stable on both)
    param = make_train_param({"max_depth": 4})
    s = qg.to(torch.int64).sum(0)
    splits = oracle.evaluate_splits(h2[:1] * 0 + oracle.build_hist(
        qg, oracle.make_ridx(5000), [(0, 5000)]), quant,
        [(int(s[0]), int(s[1]))], [0], param)
    r1 = native.make_ridx(5000)
    r2 = oracle.make_ridx(5000)
    seg1 = native.partition(r1, [(0, 5000)], splits)
    seg2 = oracle.partition(r2, [(0, 5000)], splits)
    assert seg1 == seg2
    assert torch.equal(r1, r2)  # both stable -> identical order
