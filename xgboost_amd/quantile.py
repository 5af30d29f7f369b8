"""Quantile cut finding (the sketch -> HistogramCuts stage).

Reference behavior: src/common/quantile.h:287 (WQSummary::QueryCutValues),
src/common/quantile.cc:525 (AddCutPoints), src/common/hist_util.h:110
(HistogramCuts::SearchBin = upper_bound over per-feature cut values).

Semantics reproduced exactly:
- per feature, cut values are quantile points of the observed values:
  * if #distinct <= max_bin: all distinct values EXCEPT the minimum
  * else: values answering the interior rank queries i*W/max_bin
    (weighted rank), forced strictly increasing
  * plus a final sentinel cut = max + (|max| + 1e-5) so every finite
    value satisfies value < last_cut
- bin(value) = number of cuts <= value  (searchsorted side='right')
- split condition for bin b is cuts[b]; rows go left iff value < cuts[b]
  (equivalently bin <= b), matching the reference predictor.

This implementation computes quantiles exactly via sort (optionally on a
row sample for very large shards) instead of a streaming GK sketch; the
distributed path merges per-rank summaries (see collective.py usage in
data.py).
"""
from __future__ import annotations

import dataclasses
from typing import List, Optional

import numpy as np
import torch


@dataclasses.dataclass
class HistogramCuts:
    """Per-feature bin boundaries, concatenated.

    values: float32 [total_bins]   - concatenated cut values
    ptrs:   int64   [n_features+1] - feature f owns values[ptrs[f]:ptrs[f+1]]
    min_vals: float32 [n_features] - per-feature minimum (model-dump only)
    """

    values: np.ndarray
    ptrs: np.ndarray
    min_vals: np.ndarray
    feature_types: Optional[List[str]] = None  # None => all numeric; "c" = categorical

    @property
    def n_features(self) -> int:
        return len(self.ptrs) - 1

    @property
    def total_bins(self) -> int:
        return int(self.ptrs[-1])

    def n_bins(self, f: int) -> int:
        return int(self.ptrs[f + 1] - self.ptrs[f])

    def max_n_bins(self) -> int:
        return int(np.max(np.diff(self.ptrs))) if self.n_features else 0

    def feature_cuts(self, f: int) -> np.ndarray:
        return self.values[self.ptrs[f]:self.ptrs[f + 1]]

    def is_categorical(self, f: int) -> bool:
        return self.feature_types is not None and self.feature_types[f] == "c"

    def has_categorical(self) -> bool:
        return self.feature_types is not None and "c" in self.feature_types

    def to_torch(self, device) -> "TorchCuts":
        return TorchCuts(
            values=torch.from_numpy(self.values).to(device),
            ptrs=torch.from_numpy(self.ptrs).to(device),
        )


@dataclasses.dataclass
class TorchCuts:
    values: torch.Tensor
    ptrs: torch.Tensor


def _cuts_for_column(vals: np.ndarray, weights: Optional[np.ndarray],
                     max_bin: int) -> np.ndarray:
    """Cut values for one numeric feature column (finite values only)."""
    if vals.size == 0:
        return np.array([1e-5], dtype=np.float32)
    order = np.argsort(vals, kind="stable")
    svals = vals[order]
    distinct_mask = np.empty(svals.shape, dtype=bool)
    distinct_mask[0] = True
    np.not_equal(svals[1:], svals[:-1], out=distinct_mask[1:])
    distinct = svals[distinct_mask]

    if distinct.size <= max_bin:
        cuts = distinct[1:].astype(np.float64)
    else:
        if weights is None:
            # ranks of each distinct value's first occurrence
            total = float(svals.size)
            csum = np.arange(1, svals.size + 1, dtype=np.float64)
        else:
            w = weights[order].astype(np.float64)
            csum = np.cumsum(w)
            total = float(csum[-1])
        # rmax of each distinct value = cumulative weight through its last
        # occurrence; query rank i*total/max_bin, pick the distinct value
        # whose rank interval contains it.
        last_idx = np.nonzero(distinct_mask)[0]
        last_idx = np.concatenate([last_idx[1:] - 1, [svals.size - 1]])
        rmax = csum[last_idx]  # per distinct value
        queries = np.arange(1, max_bin, dtype=np.float64) * (total / max_bin)
        pos = np.searchsorted(rmax, queries, side="left")
        pos = np.clip(pos, 0, distinct.size - 1)
        chosen = distinct[pos].astype(np.float64)
        # force strictly increasing
        keep = np.empty(chosen.shape, dtype=bool)
        keep[0] = chosen[0] > distinct[0]
        np.greater(chosen[1:], chosen[:-1], out=keep[1:])
        cuts = chosen[keep]
    mx = float(distinct[-1])
    sentinel = mx + (abs(mx) + 1e-5)
    cuts = np.append(cuts, sentinel)
    return cuts.astype(np.float32)


def make_cuts(X: np.ndarray, max_bin: int,
              weights: Optional[np.ndarray] = None,
              feature_types: Optional[List[str]] = None,
              missing: float = np.nan) -> HistogramCuts:
    """Build HistogramCuts from a dense [n_rows, n_features] float array.

    NaN (or `missing`) entries are skipped.
    """
    n_features = X.shape[1]
    all_cuts: List[np.ndarray] = []
    min_vals = np.zeros(n_features, dtype=np.float32)
    for f in range(n_features):
        col = X[:, f]
        if np.isnan(missing):
            mask = ~np.isnan(col)
        else:
            mask = (col != missing) & ~np.isnan(col)
        vals = col[mask]
        w = weights[mask] if weights is not None else None
        if feature_types is not None and feature_types[f] == "c":
            cuts = _categorical_cuts(vals)
        else:
            cuts = _cuts_for_column(vals, w, max_bin)
        min_vals[f] = float(vals.min()) if vals.size else 0.0
        all_cuts.append(cuts)
    ptrs = np.zeros(n_features + 1, dtype=np.int64)
    np.cumsum([c.size for c in all_cuts], out=ptrs[1:])
    values = np.concatenate(all_cuts).astype(np.float32) if all_cuts else np.zeros(0, np.float32)
    return HistogramCuts(values=values, ptrs=ptrs, min_vals=min_vals,
                         feature_types=list(feature_types) if feature_types else None)


def _categorical_cuts(vals: np.ndarray) -> np.ndarray:
    """Categorical feature: one bin per category 0..max_cat (reference
    AddCategories, src/common/quantile.cc:531)."""
    if vals.size == 0:
        return np.array([0.0], dtype=np.float32)
    if np.any(vals < 0) or np.any(vals != np.floor(vals)):
        raise ValueError("categorical features must be non-negative integers")
    max_cat = int(vals.max())
    return np.arange(0, max_cat + 1, dtype=np.float32)


def search_bins(X: np.ndarray, cuts: HistogramCuts,
                missing: float = np.nan) -> np.ndarray:
    """Dense global-bin index matrix: int32 [n_rows, n_features].

    Numeric: gidx = ptrs[f] + (#cuts_f <= value), missing -> -1.
    Categorical: gidx = ptrs[f] + category value (clipped), missing -> -1.
    """
    n_rows, n_features = X.shape
    out = np.empty((n_rows, n_features), dtype=np.int32)
    for f in range(n_features):
        col = X[:, f]
        fcuts = cuts.feature_cuts(f)
        if cuts.is_categorical(f):
            local = np.clip(col, 0, fcuts.size - 1).astype(np.int32, copy=False)
        else:
            local = np.searchsorted(fcuts, col, side="right").astype(np.int32)
            np.clip(local, 0, fcuts.size - 1, out=local)
        local += int(cuts.ptrs[f])
        if np.isnan(missing):
            mask = np.isnan(col)
        else:
            mask = (col == missing) | np.isnan(col)
        local[mask] = -1
        out[:, f] = local
    return out
