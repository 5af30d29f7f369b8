"""Distributed-aware sketch -> HistogramCuts.

Reference behavior: src/common/quantile.cu:594 (SketchContainer::AllReduce
merges per-worker GK summaries before MakeCuts).  Our distributed merge:
each rank summarizes every feature into K weighted quantile points
(K = 8*max_bin candidates, weight = local finite-count/K), the summaries
are allgathered (small: n_features * K floats per rank) and the final
cuts answer weighted rank queries on the pooled summary.  Single-process
falls through to the exact sort-based cuts in quantile.py.
"""
from __future__ import annotations

from typing import Optional

import numpy as np

from . import collective
from .quantile import HistogramCuts, make_cuts, _categorical_cuts


def sketch_cuts(dmat, max_bin: int) -> HistogramCuts:
    if collective.get_world_size() <= 1:
        return make_cuts(dmat.raw_data(), max_bin,
                         feature_types=dmat.info.feature_types,
                         missing=dmat.missing)
    return _distributed_cuts(dmat, max_bin)


def _distributed_cuts(dmat, max_bin: int) -> HistogramCuts:
    X = dmat.raw_data()
    missing = dmat.missing
    ftypes = dmat.info.feature_types
    n_features = X.shape[1]
    K = max(64, 8 * max_bin)
    qs = (np.arange(K) + 0.5) / K
    summaries = []
    for f in range(n_features):
        col = X[:, f]
        if np.isnan(missing):
            vals = col[~np.isnan(col)]
        else:
            vals = col[(col != missing) & ~np.isnan(col)]
        if ftypes is not None and ftypes[f] == "c":
            cats = np.unique(vals).astype(np.float32)
            summaries.append(("c", cats, float(vals.size)))
        elif vals.size == 0:
            summaries.append(("q", np.zeros(0, np.float32), 0.0))
        else:
            pts = np.quantile(vals, qs).astype(np.float32)
            mn, mx = np.float32(vals.min()), np.float32(vals.max())
            summaries.append(("q", np.concatenate([[mn], pts, [mx]]),
                              float(vals.size)))
    gathered = collective.allgather_obj(summaries)

    all_values, ptrs, min_vals = [], [0], np.zeros(n_features, np.float32)
    for f in range(n_features):
        kind = summaries[f][0]
        if kind == "c":
            cats = np.unique(np.concatenate(
                [g[f][1] for g in gathered if g[f][1].size]))
            cuts = (_categorical_cuts(cats) if cats.size
                    else np.array([0.0], np.float32))
            min_vals[f] = float(cats.min()) if cats.size else 0.0
        else:
            vals_list, w_list = [], []
            for g in gathered:
                pts, cnt = g[f][1], g[f][2]
                if cnt > 0 and pts.size:
                    vals_list.append(pts)
                    w_list.append(np.full(pts.size, cnt / pts.size))
            if not vals_list:
                cuts = np.array([1e-5], np.float32)
            else:
                v = np.concatenate(vals_list).astype(np.float64)
                w = np.concatenate(w_list)
                order = np.argsort(v, kind="stable")
                v, w = v[order], w[order]
                min_vals[f] = float(v[0])
                cw = np.cumsum(w)
                total = cw[-1]
                distinct_mask = np.empty(v.shape, bool)
                distinct_mask[0] = True
                np.not_equal(v[1:], v[:-1], out=distinct_mask[1:])
                distinct = v[distinct_mask]
                if distinct.size <= max_bin:
                    cuts = distinct[1:]
                else:
                    queries = np.arange(1, max_bin) * (total / max_bin)
                    first_idx = np.nonzero(distinct_mask)[0]
                    last_idx = np.concatenate([first_idx[1:] - 1, [v.size - 1]])
                    rmax = cw[last_idx]
                    pos = np.clip(np.searchsorted(rmax, queries, "left"),
                                  0, distinct.size - 1)
                    chosen = distinct[pos]
                    keep = np.empty(chosen.shape, bool)
                    keep[0] = chosen[0] > distinct[0]
                    np.greater(chosen[1:], chosen[:-1], out=keep[1:])
                    cuts = chosen[keep]
                mx = float(distinct[-1])
                cuts = np.append(cuts, mx + (abs(mx) + 1e-5)).astype(np.float32)
        all_values.append(np.asarray(cuts, np.float32))
        ptrs.append(ptrs[-1] + len(cuts))
    return HistogramCuts(
        values=(np.concatenate(all_values) if all_values
                else np.zeros(0, np.float32)),
        ptrs=np.asarray(ptrs, np.int64), min_vals=min_vals,
        feature_types=list(ftypes) if ftypes else None)
