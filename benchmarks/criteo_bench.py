#!/usr/bin/env python
"""BASELINE config 4: Criteo-1TB-shaped sparse one-hot CSR via the
external-memory DataIter (sparse path: absent = missing).

Full config is 4e9 rows x 1e6 one-hot columns on 8 GPUs; --rows /
--cols scale it down for single-GPU runs (the data is synthetic one-hot
with ~--nnz nonzeros per row, streamed in batches)."""
import argparse
import json
import os
import sys
import time

import numpy as np
import scipy.sparse as sp
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import xgboost_amd as xgb  # noqa: E402


def synth_csr(n, f, nnz_row, seed):
    rng = np.random.RandomState(seed)
    rows = np.repeat(np.arange(n), nnz_row)
    cols = rng.randint(200, f, n * nnz_row).astype(np.int64)
    y = (rng.rand(n) > 0.5)
    # one indicator nonzero per row: cols [0,100) for positives,
    # [100,200) for negatives (with 20% label noise)
    first = np.zeros(len(rows), dtype=bool)
    first[::nnz_row] = True  # one indicator slot per row
    noisy = rng.rand(n) < 0.2
    pos_like = np.where(noisy, ~y, y)
    cols[first] = np.where(pos_like[rows[first]],
                           rng.randint(0, 100, n),
                           100 + rng.randint(0, 100, n))
    vals = np.ones(n * nnz_row, np.float32)
    X = sp.csr_matrix((vals, (rows, cols)), shape=(n, f))
    X.sum_duplicates()
    return X, y.astype(np.float32)


class _PreQuantizedCSR:
    """Bench-only DMatrix stand-in: the synthetic one-hot data is
    generated directly in quantized-CSR form ON DEVICE (bin id ==
    column index for one-hot), so a 4e9-row-class shard (5e8 rows x 30
    nnz = 60 GB of bin ids) never round-trips through host scipy.
    Quantization/ingest is outside the timed region either way; every
    boosting round runs the full sparse training path."""

    def __init__(self, sqm, labels):
        from xgboost_amd.data import MetaInfo
        self._sqm = sqm
        self._sparse_data = True  # core._ops_for routes to CsrGpuOps
        self.missing = float("nan")
        self.info = MetaInfo(num_row=sqm.n_rows, num_col=sqm.n_features)
        self.info.labels = labels
        self.feature_names = None
        self.feature_types = None

    def num_row(self):
        return self.info.num_row

    def num_col(self):
        return self.info.num_col

    def get_label(self):
        return self.info.labels

    def sparse_quantized(self, max_bin):
        return self._sqm


def synth_quantized_onehot(n, f, nnz_row, device, seed):
    """Device-resident quantized one-hot CSR: 1 bin per feature, so the
    global bin id IS the column index; per-row bins sorted."""
    from xgboost_amd.quantile import HistogramCuts
    from xgboost_amd.sparse import SparseQuantizedMatrix
    gen = torch.Generator(device=device)
    gen.manual_seed(seed)
    y = torch.rand(n, generator=gen, device=device) > 0.5
    # generate + per-row sort in chunks: a single [n, nnz] sort would
    # allocate int64 index tensors 2x the data (OOM at 5e8 rows)
    flat = torch.empty(n * nnz_row, dtype=torch.int32, device=device)
    chunk = 20_000_000
    for s0 in range(0, n, chunk):
        e0 = min(s0 + chunk, n)
        m = e0 - s0
        cols = torch.randint(200, f, (m, nnz_row), generator=gen,
                             device=device, dtype=torch.int32)
        noisy = torch.rand(m, generator=gen, device=device) < 0.2
        yy = y[s0:e0]
        pos_like = torch.where(noisy, ~yy, yy)
        ind = torch.randint(0, 100, (m,), generator=gen, device=device,
                            dtype=torch.int32)
        cols[:, 0] = torch.where(pos_like, ind, 100 + ind)
        cols, _ = torch.sort(cols, dim=1)
        flat[s0 * nnz_row:e0 * nnz_row] = cols.reshape(-1)
        del cols
    row_ptr = torch.arange(n + 1, device=device, dtype=torch.int64) * nnz_row
    cuts = HistogramCuts(
        values=np.full(f, 1.5, np.float32),  # sentinel above the 1.0s
        ptrs=np.arange(f + 1, dtype=np.int64),
        min_vals=np.zeros(f, np.float32))
    sqm = SparseQuantizedMatrix(row_ptr=row_ptr, bin_idx=flat,
                                cuts=cuts, n_features=f)
    return sqm, y.float().cpu().numpy()


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rows", type=int, default=500_000_000)  # 4e9 / 8 GPUs
    ap.add_argument("--cols", type=int, default=1_000_000)
    ap.add_argument("--nnz", type=int, default=30)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--max-depth", type=int, default=8)
    ap.add_argument("--host-csr", action="store_true",
                    help="ingest through scipy CSR + host sketch instead "
                         "of device-synthesized quantized pages")
    args = ap.parse_args()
    has_gpu = torch.cuda.is_available()
    n = args.rows if has_gpu else 100_000
    f = args.cols if has_gpu else 10_000
    if args.host_csr or not has_gpu:
        n = min(n, 20_000_000)  # host scipy path: bounded by RAM
        X, y = synth_csr(n, f, args.nnz, seed=3)
        d = xgb.DMatrix(X, label=y)
    else:
        sqm, y = synth_quantized_onehot(n, f, args.nnz, "cuda", seed=3)
        d = _PreQuantizedCSR(sqm, y)
    bst = xgb.Booster({"objective": "binary:logistic",
                       "max_depth": args.max_depth, "max_bin": 256,
                       "eta": 0.1, "device": "cuda" if has_gpu else "cpu"},
                      cache=[d])
    it = 0
    for _ in range(args.warmup):
        bst.update(d, it)
        it += 1
    if has_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        bst.update(d, it)
        it += 1
    if has_gpu:
        torch.cuda.synchronize()
    el = time.perf_counter() - t0
    from xgboost_amd.data import MetaInfo
    from xgboost_amd.metrics import create_metric
    if hasattr(d, "_sqm"):
        # train AUC from the training margin cache (updated from leaf
        # positions every round — no separate 5e8-row traversal);
        # sampled to bound the host-side sort
        margin, _ = bst._cache[bst._pin(d)]
        pred = bst.objective.pred_transform(
            margin).cpu().numpy().reshape(-1)
        n_eval = min(n, 2_000_000)
        sub_info = MetaInfo(num_row=n_eval)
        sub_info.labels = d.info.labels[:n_eval]
        auc = create_metric("auc")(pred[:n_eval], sub_info)
    else:
        auc = create_metric("auc")(bst.predict(d), d.info)
    print(json.dumps({
        "metric": "boosting_rounds_per_sec", "value": args.steps / el,
        "unit": "rounds/s", "n_gpus": 1 if has_gpu else 0,
        "steps": args.steps, "ms_per_step": el / args.steps * 1000,
        "higher_is_better": True,
        "data": ("synthetic-sparse-onehot-prequantized"
                 if hasattr(d, "_sqm") else "synthetic-sparse-onehot"),
        "config": {"model": "criteo-shape-sparse", "rows": n, "cols": f,
                   "nnz_per_row": args.nnz, "max_depth": args.max_depth,
                   "train_auc": round(float(auc), 5)},
    }))


if __name__ == "__main__":
    main()
