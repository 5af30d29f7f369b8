"""Cut-finding and binning semantics (reference analog:
tests/cpp/common/test_quantile.cc, test_hist_util.cc)."""
import numpy as np

from xgboost_amd.quantile import make_cuts, search_bins
from xgboost_amd.data import DMatrix, quantize_dense


def test_few_distinct_values_exact_cuts():
    X = np.array([[1.0], [2.0], [2.0], [3.0], [1.0]], dtype=np.float32)
    cuts = make_cuts(X, max_bin=256)
    fc = cuts.feature_cuts(0)
    # distinct minus the minimum, plus sentinel > max
    assert fc[0] == 2.0 and fc[1] == 3.0
    assert fc[-1] > 3.0
    assert len(fc) == 3
    gidx = search_bins(X, cuts)
    # value 1 -> bin 0, 2 -> bin 1, 3 -> bin 2
    assert gidx[:, 0].tolist() == [0, 1, 1, 2, 0]


def test_binning_boundary_right():
    X = np.array([[1.0], [2.0], [3.0]], dtype=np.float32)
    cuts = make_cuts(X, max_bin=256)
    # bin = #cuts <= value; value == cut goes to next bin
    g = search_bins(np.array([[1.999], [2.0], [2.001]], np.float32), cuts)
    assert g[0, 0] == 0 and g[1, 0] == 1 and g[2, 0] == 1


def test_quantile_reduction():
    rng = np.random.RandomState(0)
    X = rng.randn(10000, 1).astype(np.float32)
    cuts = make_cuts(X, max_bin=16)
    fc = cuts.feature_cuts(0)
    assert len(fc) <= 16
    assert np.all(np.diff(fc) > 0)  # strictly increasing
    gidx = search_bins(X, cuts)
    counts = np.bincount(gidx[:, 0], minlength=len(fc))
    # roughly balanced bins from rank queries
    assert counts.max() < 3 * 10000 / 16


def test_missing_handling():
    X = np.array([[1.0], [np.nan], [2.0]], dtype=np.float32)
    cuts = make_cuts(X, max_bin=8)
    g = search_bins(X, cuts)
    assert g[1, 0] == -1
    qm = quantize_dense(X, cuts)
    assert qm.has_missing
    gg = qm.global_gidx().numpy()
    assert gg[1, 0] == -1


def test_custom_missing_value():
    X = np.array([[1.0], [-999.0], [2.0]], dtype=np.float32)
    d = DMatrix(X, label=[0, 0, 0], missing=-999.0)
    qm = d.quantized(8)
    gg = qm.global_gidx().numpy()
    assert gg[1, 0] == -1
    cuts = qm.cuts
    assert cuts.feature_cuts(0).min() >= 2.0  # -999 excluded from sketch


def test_weighted_cuts():
    v = np.concatenate([np.zeros(100), np.ones(100)]).astype(np.float32)
    w = np.concatenate([np.full(100, 1e-6), np.ones(100)]).astype(np.float32)
    # many distinct values, few bins
    rng = np.random.RandomState(1)
    v2 = np.concatenate([rng.rand(1000) * 0.1,
                         0.9 + rng.rand(10) * 0.1]).astype(np.float32)
    w2 = np.concatenate([np.full(1000, 0.001), np.full(10, 100.0)]).astype(np.float32)
    cuts = make_cuts(v2.reshape(-1, 1), max_bin=8, weights=w2)
    fc = cuts.feature_cuts(0)
    # heavy-weight region (>=0.9) should receive most cuts
    assert (fc[:-1] >= 0.9).sum() >= 4


def test_sentinel_covers_max():
    rng = np.random.RandomState(2)
    X = rng.randn(500, 3).astype(np.float32)
    cuts = make_cuts(X, max_bin=32)
    g = search_bins(X, cuts)
    for f in range(3):
        local = g[:, f] - cuts.ptrs[f]
        assert local.max() < cuts.n_bins(f)
        assert local.min() >= 0


def test_quantile_dmatrix_trains_identically_to_dmatrix():
    """In-core QuantileDMatrix carries the same cuts/bins as quantizing
    a plain DMatrix, so hist training must be bit-identical; a ref'd
    validation QDM bins with the training cuts."""
    import xgboost_amd as xgb
    rng = np.random.RandomState(0)
    X = rng.randn(600, 5).astype(np.float32)
    y = X[:, 0].astype(np.float32)
    p = {"max_depth": 4, "max_bin": 64, "seed": 3}
    b1 = xgb.train(p, xgb.DMatrix(X, label=y), 5)
    b2 = xgb.train(p, xgb.QuantileDMatrix(X, label=y, max_bin=64), 5)
    assert b1.get_dump(with_stats=True) == b2.get_dump(with_stats=True)
    dtr = xgb.QuantileDMatrix(X, label=y, max_bin=64)
    dva = xgb.QuantileDMatrix(X[:100], label=y[:100], max_bin=64, ref=dtr)
    res = {}
    xgb.train(p, dtr, 3, evals=[(dva, "v")], evals_result=res,
              verbose_eval=False)
    assert len(res["v"]["rmse"]) == 3
