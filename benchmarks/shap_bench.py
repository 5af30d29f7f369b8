#!/usr/bin/env python
"""BASELINE config 5: SHAP pred_contribs on 1e6x28, 500-tree model,
1 MI355X (reference: Quadrature-TreeSHAP shap.cu)."""
import argparse
import json
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import xgboost_amd as xgb  # noqa: E402
from bench import make_higgs_like  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rows", type=int, default=1_000_000)
    ap.add_argument("--trees", type=int, default=500)
    ap.add_argument("--depth", type=int, default=6)
    args = ap.parse_args()
    has_gpu = torch.cuda.is_available()
    n = args.rows if has_gpu else 20_000
    trees = args.trees if has_gpu else 20
    X, y = make_higgs_like(n, 28, seed=7)
    d = xgb.DMatrix(X, label=y)
    t0 = time.perf_counter()
    bst = xgb.train({"objective": "binary:logistic", "max_depth": args.depth,
                     "device": "cuda" if has_gpu else "cpu", "eta": 0.1},
                    d, trees, verbose_eval=False)
    t_train = time.perf_counter() - t0
    # warmup
    sub = d.slice(np.arange(min(n, 10000)))
    bst.predict(sub, pred_contribs=True)
    if has_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    contribs = bst.predict(d, pred_contribs=True)
    if has_gpu:
        torch.cuda.synchronize()
    t_shap = time.perf_counter() - t0
    margin = bst.predict(d, output_margin=True)
    err = float(np.abs(contribs.sum(axis=1) - margin).max())
    print(json.dumps({
        "metric": "shap_rows_per_sec", "value": n / t_shap, "unit": "rows/s",
        "n_gpus": 1 if has_gpu else 0, "rows": n, "trees": trees,
        "depth": args.depth, "shap_seconds": t_shap,
        "train_seconds": t_train, "efficiency_max_err": err,
        "higher_is_better": True, "data": "synthetic",
    }))


if __name__ == "__main__":
    main()
