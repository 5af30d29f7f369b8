"""SHAP values for tree ensembles.

Reference behavior: src/predictor/interpretability/shap.{cc,cu}
(Quadrature-TreeSHAP) — we provide the same outputs (exact SHAP
contributions, interactions, and the Saabas approximation for
approx_contribs) via the classic path-dependent TreeSHAP recursion
(Lundberg & Lee) on CPU; a HIP kernel accelerates pred_contribs on GPU
(ops/cpp/shap.hip).

Output convention matches xgboost: contribs shape [n, f+1] with the
last column the bias (expected value incl. base_score); interactions
[n, f+1, f+1]; multiclass adds a group axis [n, k, f+1].
"""
from __future__ import annotations

from typing import List, Tuple

import numpy as np

from .tree_model import RegTree


def _extend(path: List[List[float]], zero: float, one: float, fi: int) -> None:
    length = len(path)
    path.append([fi, zero, one, 1.0 if length == 0 else 0.0])
    for i in range(length - 1, -1, -1):
        path[i + 1][3] += one * path[i][3] * (i + 1) / (length + 1)
        path[i][3] = zero * path[i][3] * (length - i) / (length + 1)


def _unwind(path: List[List[float]], idx: int) -> None:
    length = len(path) - 1
    one = path[idx][2]
    zero = path[idx][1]
    n = path[length][3]
    for j in range(length - 1, -1, -1):
        if one != 0:
            t = path[j][3]
            path[j][3] = n * (length + 1) / ((j + 1) * one)
            n = t - path[j][3] * zero * (length - j) / (length + 1)
        else:
            path[j][3] = path[j][3] * (length + 1) / (zero * (length - j))
    # shift feature/zero/one down; pweights stay in place (the canonical
    # UNWIND keeps the recomputed pweights at indices 0..length-1)
    for j in range(idx, length):
        path[j][0] = path[j + 1][0]
        path[j][1] = path[j + 1][1]
        path[j][2] = path[j + 1][2]
    del path[length]


def _unwound_sum(path: List[List[float]], idx: int) -> float:
    length = len(path) - 1
    one = path[idx][2]
    zero = path[idx][1]
    total = 0.0
    n = path[length][3]
    for j in range(length - 1, -1, -1):
        if one != 0:
            t = n * (length + 1) / ((j + 1) * one)
            total += t
            n = path[j][3] - t * zero * (length - j) / (length + 1)
        else:
            total += path[j][3] / (zero * (length - j) / (length + 1))
    return total


def _decision(tree: RegTree, nid: int, x: np.ndarray, missing: float
              ) -> Tuple[int, int]:
    """Return (hot child, cold child) for row x at node nid."""
    f = int(tree.split_index[nid])
    v = x[f]
    is_missing = bool(np.isnan(v)) if np.isnan(missing) else \
        bool(v == missing or np.isnan(v))
    left, right = int(tree.left[nid]), int(tree.right[nid])
    if is_missing:
        hot = left if tree.default_left[nid] else right
    elif tree.split_type[nid] == 1:
        cats = tree.cat_segments.get(nid)
        in_set = cats is not None and int(v) in cats
        hot = right if in_set else left
    else:
        hot = left if v < tree.split_cond[nid] else right
    cold = right if hot == left else left
    return hot, cold


def _tree_shap(tree: RegTree, x: np.ndarray, phi: np.ndarray,
               missing: float, condition: int = 0,
               condition_feature: int = -1) -> None:
    """Path-dependent TreeSHAP (Lundberg & Lee alg. 2, with the
    conditional variant used for interaction values)."""

    def recurse(nid: int, path: List[List[float]], zero: float, one: float,
                pfeat: int, cond_frac: float) -> None:
        if cond_frac == 0.0:
            return
        path = [list(e) for e in path]
        if condition == 0 or condition_feature != pfeat:
            _extend(path, zero, one, pfeat)
        if tree.is_leaf(nid):
            leaf = float(tree.split_cond[nid])
            for i in range(1, len(path)):
                w = _unwound_sum(path, i)
                phi[int(path[i][0])] += (
                    w * (path[i][2] - path[i][1]) * leaf * cond_frac)
            return
        hot, cold = _decision(tree, nid, x, missing)
        f = int(tree.split_index[nid])
        cover = max(float(tree.sum_hess[nid]), 1e-16)
        hot_zero = float(tree.sum_hess[hot]) / cover
        cold_zero = float(tree.sum_hess[cold]) / cover
        incoming_zero, incoming_one = 1.0, 1.0
        idx = -1
        for i in range(1, len(path)):
            if int(path[i][0]) == f:
                idx = i
                break
        if idx >= 0:
            incoming_zero, incoming_one = path[idx][1], path[idx][2]
            _unwind(path, idx)
        hot_cf, cold_cf = cond_frac, cond_frac
        if condition > 0 and f == condition_feature:
            cold_cf = 0.0
        elif condition < 0 and f == condition_feature:
            hot_cf *= hot_zero
            cold_cf *= cold_zero
        recurse(hot, path, incoming_zero * hot_zero, incoming_one, f, hot_cf)
        recurse(cold, path, incoming_zero * cold_zero, 0.0, f, cold_cf)

    recurse(0, [], 1.0, 1.0, -1, 1.0)


def _expected_value(tree: RegTree) -> float:
    """Cover-weighted mean of leaf values (phi_0 of the tree)."""
    total = 0.0
    root_cover = max(float(tree.sum_hess[0]), 1e-16)
    for nid in range(tree.n_nodes):
        if tree.is_leaf(nid) and (tree.parent[nid] != -1 or nid == 0):
            total += float(tree.split_cond[nid]) * float(tree.sum_hess[nid])
    return total / root_cover


def shap_values(booster, dmat, iteration_range=(0, 0),
                approx: bool = False) -> np.ndarray:
    lo, hi = booster._tree_range(iteration_range)
    X = dmat.raw_data()
    n, f = X.shape
    n_groups = booster.n_outputs
    phi = np.zeros((n, n_groups, f + 1), dtype=np.float64)
    phi[:, :, f] = booster._base_margin_value()
    if booster.device.type == "cuda" and not approx:
        try:
            from .backend.gpu import shap_gpu
            return shap_gpu(booster, dmat, lo, hi, phi)
        except (ImportError, AttributeError):
            pass
    for t in range(lo, hi):
        tree = booster.trees[t]
        k = booster.tree_info[t]
        phi[:, k, f] += _expected_value(tree)
        if approx:
            _saabas(tree, X, phi[:, k, :], dmat.missing)
        else:
            for i in range(n):
                _tree_shap(tree, X[i], phi[i, k], dmat.missing)
    if n_groups == 1:
        return phi[:, 0, :].astype(np.float32)
    return phi.astype(np.float32)


def _saabas(tree: RegTree, X: np.ndarray, phi: np.ndarray,
            missing: float) -> None:
    """Gain-path approximation (xgboost approx_contribs)."""
    n = X.shape[0]
    mean_val = np.zeros(tree.n_nodes)
    for nid in range(tree.n_nodes - 1, -1, -1):
        if tree.is_leaf(nid):
            mean_val[nid] = float(tree.split_cond[nid])
        else:
            l, r = int(tree.left[nid]), int(tree.right[nid])
            hl, hr = float(tree.sum_hess[l]), float(tree.sum_hess[r])
            tot = max(hl + hr, 1e-16)
            mean_val[nid] = (mean_val[l] * hl + mean_val[r] * hr) / tot
    for i in range(n):
        nid = 0
        while not tree.is_leaf(nid):
            hot, _ = _decision(tree, nid, X[i], missing)
            f = int(tree.split_index[nid])
            phi[i, f] += mean_val[hot] - mean_val[nid]
            nid = hot


def shap_interactions(booster, dmat, iteration_range=(0, 0)) -> np.ndarray:
    lo, hi = booster._tree_range(iteration_range)
    if booster.device.type == "cuda":
        try:
            from .backend.gpu import shap_interactions_gpu
            return shap_interactions_gpu(booster, dmat, lo, hi,
                                         iteration_range)
        except (ImportError, AttributeError):
            pass  # categorical / deep-path forests: exact CPU fallback
    X = dmat.raw_data()
    n, f = X.shape
    n_groups = booster.n_outputs
    out = np.zeros((n, n_groups, f + 1, f + 1), dtype=np.float64)
    base = shap_values(booster, dmat, iteration_range).astype(np.float64)
    if n_groups == 1:
        base = base[:, None, :]
    for t in range(lo, hi):
        tree = booster.trees[t]
        k = booster.tree_info[t]
        used = sorted(set(int(tree.split_index[nid])
                          for nid in range(tree.n_nodes)
                          if not tree.is_leaf(nid)))
        for j in used:
            phi_on = np.zeros((n, f + 1))
            phi_off = np.zeros((n, f + 1))
            for i in range(n):
                _tree_shap(tree, X[i], phi_on[i], dmat.missing,
                           condition=1, condition_feature=j)
                _tree_shap(tree, X[i], phi_off[i], dmat.missing,
                           condition=-1, condition_feature=j)
            diff = (phi_on - phi_off) / 2.0
            diff[:, j] = 0.0
            out[:, k, j, :] += diff
            out[:, k, :, j] += diff
    for i_ in range(f + 1):
        out[:, :, i_, i_] = base[:, :, i_] - (
            out[:, :, i_, :].sum(axis=-1) - out[:, :, i_, i_])
    if n_groups == 1:
        return out[:, 0].astype(np.float32)
    return out.astype(np.float32)
