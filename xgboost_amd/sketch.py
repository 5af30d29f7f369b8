"""Sketch -> HistogramCuts: batch-incremental and distributed-aware.

Reference behavior: src/common/quantile.cu:594 (SketchContainer::AllReduce
merges per-worker GK summaries before MakeCuts); the same merge handles
external-memory batches (reference: sketching per SparsePage chunk,
hist_util.cu DeviceSketchWithHessian).

Each data chunk (a batch from a DataIter, and/or a rank's shard) is
summarized per feature into K weighted quantile points; summaries are
pooled (allgathered across ranks when distributed) and the final cuts
answer weighted rank queries on the pooled summary.  Single-process
single-chunk input falls through to the exact sort-based cuts in
quantile.py.
"""
from __future__ import annotations

from typing import List, Optional, Sequence, Tuple

import numpy as np

from . import collective
from .quantile import HistogramCuts, make_cuts, _categorical_cuts

# a per-feature summary: ("q", points, total_weight) | ("c", cats, count)
Summary = Tuple[str, np.ndarray, float]


def summarize_batch(X: np.ndarray, missing: float,
                    feature_types: Optional[List[str]], max_bin: int,
                    weights: Optional[np.ndarray] = None
                    ) -> List[Summary]:
    K = max(64, 8 * max_bin)
    qs = (np.arange(K) + 0.5) / K
    out: List[Summary] = []
    for f in range(X.shape[1]):
        col = X[:, f]
        if np.isnan(missing):
            mask = ~np.isnan(col)
        else:
            mask = (col != missing) & ~np.isnan(col)
        vals = col[mask]
        if feature_types is not None and feature_types[f] == "c":
            out.append(("c", np.unique(vals).astype(np.float32),
                        float(vals.size)))
        elif vals.size == 0:
            out.append(("q", np.zeros(0, np.float32), 0.0))
        else:
            if weights is not None:
                w = weights[mask].astype(np.float64)
                order = np.argsort(vals, kind="stable")
                sv, sw = vals[order], w[order]
                cw = np.cumsum(sw)
                total = cw[-1]
                targets = qs * total
                pos = np.clip(np.searchsorted(cw, targets), 0, sv.size - 1)
                pts = sv[pos].astype(np.float32)
                wsum = float(total)
            else:
                pts = np.quantile(vals, qs).astype(np.float32)
                wsum = float(vals.size)
            mn, mx = np.float32(vals.min()), np.float32(vals.max())
            out.append(("q", np.concatenate([[mn], pts, [mx]]), wsum))
    return out


def cuts_from_summaries(chunks: Sequence[List[Summary]], max_bin: int,
                        n_features: int,
                        feature_types: Optional[List[str]]) -> HistogramCuts:
    all_values, ptrs = [], [0]
    min_vals = np.zeros(n_features, np.float32)
    for f in range(n_features):
        kind = None
        for ch in chunks:
            if ch[f][2] > 0:
                kind = ch[f][0]
                break
        if kind == "c":
            cats_list = [ch[f][1] for ch in chunks if ch[f][1].size]
            cats = (np.unique(np.concatenate(cats_list))
                    if cats_list else np.zeros(0, np.float32))
            cuts = (_categorical_cuts(cats) if cats.size
                    else np.array([0.0], np.float32))
            min_vals[f] = float(cats.min()) if cats.size else 0.0
        else:
            vals_list, w_list = [], []
            for ch in chunks:
                pts, cnt = ch[f][1], ch[f][2]
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
                    last_idx = np.concatenate([first_idx[1:] - 1,
                                               [v.size - 1]])
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
        feature_types=list(feature_types) if feature_types else None)


def sketch_cuts(dmat, max_bin: int) -> HistogramCuts:
    """Cuts for an in-core DMatrix; merges across ranks if distributed."""
    if collective.get_world_size() <= 1:
        return make_cuts(dmat.raw_data(), max_bin,
                         feature_types=dmat.info.feature_types,
                         missing=dmat.missing)
    local = summarize_batch(dmat.raw_data(), dmat.missing,
                            dmat.info.feature_types, max_bin)
    gathered = collective.allgather_obj(local)
    return cuts_from_summaries(gathered, max_bin, dmat.num_col(),
                               dmat.info.feature_types)


def sketch_cuts_batches(summaries: Sequence[List[Summary]], max_bin: int,
                        n_features: int,
                        feature_types: Optional[List[str]]) -> HistogramCuts:
    """Cuts from per-batch summaries (external memory / DataIter);
    distributed: each rank pools its batches, then ranks allgather."""
    chunks = list(summaries)
    if collective.get_world_size() > 1:
        gathered = collective.allgather_obj(chunks)
        chunks = [c for rank_chunks in gathered for c in rank_chunks]
    return cuts_from_summaries(chunks, max_bin, n_features, feature_types)
