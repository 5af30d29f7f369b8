#!/usr/bin/env python
"""Per-feature performance numbers on the Higgs-1M shape (1 MI355X):
gpu approx (device cut regen), lossguide / monotone (native driver
policy replay), rank:ndcg (device lambdarank), SHAP interactions
(shap_ix.hip), and inplace predict vs the DMatrix path.  Emits one
JSON line per measurement."""
import json
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import xgboost_amd as xgb  # noqa: E402


def timed_rounds(params, d, steps=60, warmup=10):
    bst = xgb.Booster(params, cache=[d])
    it = 0
    for _ in range(warmup):
        bst.update(d, it)
        it += 1
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        bst.update(d, it)
        it += 1
    torch.cuda.synchronize()
    el = time.perf_counter() - t0
    return bst, steps / el, el / steps * 1000


def main():
    rng = np.random.RandomState(7)
    n, f = 1_000_000, 28
    X = rng.randn(n, f).astype(np.float32)
    w = rng.randn(f) / np.sqrt(f)
    y = ((X @ w + 0.7 * (X * X) @ (rng.randn(f) / np.sqrt(f))) > 0
         ).astype(np.float32)
    base = {"objective": "binary:logistic", "max_depth": 8, "max_bin": 256,
            "eta": 0.1, "device": "cuda", "seed": 7}

    out = []

    d = xgb.DMatrix(X, label=y)
    bst, rps, ms = timed_rounds(dict(base), d)
    out.append({"bench": "hist-depthwise (headline)", "rounds_per_sec":
                round(rps, 1), "ms_per_round": round(ms, 3)})

    _, rps, ms = timed_rounds(dict(base, grow_policy="lossguide",
                                   max_leaves=255), d)
    out.append({"bench": "lossguide max_leaves=255 (native replay)",
                "rounds_per_sec": round(rps, 1),
                "ms_per_round": round(ms, 3)})

    mono = [1, -1] + [0] * (f - 2)
    _, rps, ms = timed_rounds(dict(base, monotone_constraints=mono), d)
    out.append({"bench": "monotone (native whole-tree)",
                "rounds_per_sec": round(rps, 1),
                "ms_per_round": round(ms, 3)})

    _, rps, ms = timed_rounds(dict(base, tree_method="approx"), d,
                              steps=20, warmup=3)
    out.append({"bench": "gpu approx (device cut regen per tree)",
                "rounds_per_sec": round(rps, 1),
                "ms_per_round": round(ms, 3)})

    qid = np.repeat(np.arange(10_000), n // 10_000)
    dr = xgb.DMatrix(X, label=np.clip((X[:, 0] * 2 + 2).astype(int), 0,
                                      4).astype(np.float32), qid=qid)
    _, rps, ms = timed_rounds(dict(base, objective="rank:ndcg"), dr,
                              steps=30, warmup=5)
    out.append({"bench": "rank:ndcg 1M docs / 10k groups (device pairs)",
                "rounds_per_sec": round(rps, 1),
                "ms_per_round": round(ms, 3)})

    # SHAP interactions: 500-tree depth-6 model (BASELINE config 5 shape)
    d5 = xgb.DMatrix(X[:200_000], label=y[:200_000])
    bst5 = xgb.train(dict(base, max_depth=6), d5, 500, verbose_eval=False)
    n_ix = 50_000
    dix = xgb.DMatrix(X[:n_ix])
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    ix = bst5.predict(dix, pred_interactions=True)
    torch.cuda.synchronize()
    el = time.perf_counter() - t0
    assert ix.shape == (n_ix, f + 1, f + 1)
    out.append({"bench": "SHAP interactions 500 trees d6 (shap_ix.hip)",
                "rows_per_sec": round(n_ix / el, 1),
                "s_total": round(el, 2)})
    t0 = time.perf_counter()
    _ = bst5.predict(dix, pred_contribs=True)
    torch.cuda.synchronize()
    out.append({"bench": "SHAP contribs 500 trees d6 (shap_paths.hip)",
                "rows_per_sec": round(n_ix / (time.perf_counter() - t0), 1)})

    # inplace predict (zero-copy device) vs DMatrix predict
    Xd = torch.from_numpy(X).cuda()
    bst.inplace_predict(Xd)  # warm caches
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(10):
        bst.inplace_predict(Xd)
    torch.cuda.synchronize()
    ip = 10 * n / (time.perf_counter() - t0)
    t0 = time.perf_counter()
    for _ in range(10):
        bst.predict(xgb.DMatrix(X))
    torch.cuda.synchronize()
    dp = 10 * n / (time.perf_counter() - t0)
    out.append({"bench": "inplace_predict cuda tensor (70 trees)",
                "rows_per_sec": round(ip, 0),
                "vs_dmatrix_path": round(ip / dp, 2)})

    for o in out:
        print(json.dumps(o), flush=True)


if __name__ == "__main__":
    main()
