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


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rows", type=int, default=4_000_000)
    ap.add_argument("--cols", type=int, default=1_000_000)
    ap.add_argument("--nnz", type=int, default=30)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--max-depth", type=int, default=8)
    args = ap.parse_args()
    has_gpu = torch.cuda.is_available()
    n = args.rows if has_gpu else 100_000
    f = args.cols if has_gpu else 10_000
    X, y = synth_csr(n, f, args.nnz, seed=3)
    d = xgb.DMatrix(X, label=y)
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
    from xgboost_amd.metrics import create_metric
    auc = create_metric("auc")(bst.predict(d), d.info)
    print(json.dumps({
        "metric": "boosting_rounds_per_sec", "value": args.steps / el,
        "unit": "rounds/s", "n_gpus": 1 if has_gpu else 0,
        "steps": args.steps, "ms_per_step": el / args.steps * 1000,
        "higher_is_better": True, "data": "synthetic-sparse-onehot",
        "config": {"model": "criteo-shape-sparse", "rows": n, "cols": f,
                   "nnz_per_row": args.nnz, "max_depth": args.max_depth,
                   "train_auc": round(float(auc), 5)},
    }))


if __name__ == "__main__":
    main()
