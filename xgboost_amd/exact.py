"""Exact greedy tree method (tree_method=exact).

Reference behavior: src/tree/updater_colmaker.cc (ColMaker) —
enumerates every distinct feature value over pre-sorted columns instead
of histogram bins; split condition is the midpoint between adjacent
values; single-node (non-distributed) like the reference.

Vectorized numpy implementation: per (node, feature), cumulative grad
sums over the node's rows in sorted-value order; candidate gains at
value boundaries for both missing directions.
"""
from __future__ import annotations

from typing import Dict, Optional, Tuple

import numpy as np

from .params import TrainParam
from .splits import calc_gain, calc_gain_given_weight, calc_weight
from .tree_model import RegTree


class ExactGrower:
    def __init__(self, X: np.ndarray, param: TrainParam,
                 missing: float = np.nan):
        self.X = X
        self.param = param
        self.missing = missing
        n, f = X.shape
        self.sorted_idx = np.empty((f, n), dtype=np.int64)
        self.sorted_val = np.empty((f, n), dtype=np.float32)
        self.valid_len = np.empty(f, dtype=np.int64)
        for j in range(f):
            col = X[:, j]
            if np.isnan(missing):
                mask = ~np.isnan(col)
            else:
                mask = (col != missing) & ~np.isnan(col)
            idx = np.nonzero(mask)[0]
            order = np.argsort(col[idx], kind="stable")
            k = len(idx)
            self.sorted_idx[j, :k] = idx[order]
            self.sorted_val[j, :k] = col[idx[order]]
            self.valid_len[j] = k

    def grow(self, gpair: np.ndarray, tree: RegTree, eta: float) -> np.ndarray:
        param = self.param
        n, f = self.X.shape
        position = np.zeros(n, dtype=np.int32)
        g = gpair[:, 0].astype(np.float64)
        h = gpair[:, 1].astype(np.float64)
        node_sums: Dict[int, Tuple[float, float]] = {
            0: (float(g.sum()), float(h.sum()))}
        depth_nodes = [0]
        depth = 0
        n_leaves = 1
        max_depth = param.max_depth if param.max_depth > 0 else 31
        while depth_nodes and depth < max_depth:
            next_nodes = []
            for nid in depth_nodes:
                if param.max_leaves and n_leaves >= param.max_leaves:
                    break
                best = self._find_split(nid, position, g, h, node_sums[nid])
                if best is None or best[0] <= param.gamma:
                    continue
                (gain, feat, cond, default_left,
                 lg, lh, rg, rh) = best
                wl = float(calc_weight(lg, lh, param))
                wr = float(calc_weight(rg, rh, param))
                l, r = tree.add_split(nid, feat, cond, default_left, gain,
                                      float(tree.base_weight[nid]), wl, wr,
                                      lh + rh, lh, rh)
                node_sums[l] = (lg, lh)
                node_sums[r] = (rg, rh)
                n_leaves += 1
                # update positions
                rows = np.nonzero(position == nid)[0]
                v = self.X[rows, feat]
                miss = np.isnan(v) if np.isnan(self.missing) else (
                    (v == self.missing) | np.isnan(v))
                go_left = np.where(miss, default_left, v < cond)
                position[rows] = np.where(go_left, l, r)
                next_nodes.extend([l, r])
            depth_nodes = next_nodes
            depth += 1
        for nid in range(tree.n_nodes):
            if tree.is_leaf(nid):
                tree.set_leaf(nid, float(tree.base_weight[nid]) * eta)
        return position

    def _find_split(self, nid: int, position: np.ndarray, g: np.ndarray,
                    h: np.ndarray, parent: Tuple[float, float]):
        param = self.param
        pg, ph = parent
        parent_gain = float(calc_gain(pg, ph, param))
        best = None
        in_node = position == nid
        for j in range(self.X.shape[1]):
            k = int(self.valid_len[j])
            idx = self.sorted_idx[j, :k]
            sel = in_node[idx]
            rows = idx[sel]
            if len(rows) < 2:
                continue
            vals = self.sorted_val[j, :k][sel]
            cg = np.cumsum(g[rows])
            ch = np.cumsum(h[rows])
            # candidates between distinct adjacent values
            boundary = vals[1:] != vals[:-1]
            if not boundary.any():
                continue
            bidx = np.nonzero(boundary)[0]  # split after position i
            GL = cg[bidx]
            HL = ch[bidx]
            feat_g, feat_h = cg[-1], ch[-1]
            miss_g, miss_h = pg - feat_g, ph - feat_h
            conds = (vals[bidx] + vals[bidx + 1]) * 0.5
            for missing_left in (False, True):
                gl = GL + (miss_g if missing_left else 0.0)
                hl = HL + (miss_h if missing_left else 0.0)
                gr = pg - gl
                hr = ph - hl
                ok = (hl >= param.min_child_weight) & (hr >= param.min_child_weight)
                if not ok.any():
                    continue
                wl = calc_weight(gl, hl, param)
                wr = calc_weight(gr, hr, param)
                gains = (calc_gain_given_weight(gl, hl, wl, param)
                         + calc_gain_given_weight(gr, hr, wr, param)
                         - parent_gain)
                gains = np.where(ok, gains, -np.inf)
                bi = int(np.argmax(gains))
                gv = float(gains[bi])
                if np.isfinite(gv) and (best is None or gv > best[0]):
                    best = (gv, j, float(conds[bi]), missing_left,
                            float(gl[bi]), float(hl[bi]),
                            float(gr[bi]), float(hr[bi]))
        return best


def grow_exact(X: np.ndarray, gpair: np.ndarray, param: TrainParam,
               tree: RegTree, missing: float = np.nan,
               cache: Optional[dict] = None) -> np.ndarray:
    """Entry point used by Booster; presort cache keyed by matrix id."""
    grower = None
    if cache is not None:
        grower = cache.get("exact_grower")
    if grower is None:
        grower = ExactGrower(X, param, missing)
        if cache is not None:
            cache["exact_grower"] = grower
    else:
        grower.param = param
    return grower.grow(gpair, tree, param.eta)
