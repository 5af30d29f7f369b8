"""Tree post-processing updaters: prune and refresh.

Reference behavior: src/tree/updater_prune.cc (TreePruner: recursively
drop leaf pairs whose split gain < min_split_loss), updater_refresh.cc
(TreeRefresher: recompute node stats / leaf values on new data).
Used by process_type=update (reference gbtree.cc updater config).
"""
from __future__ import annotations

import numpy as np

from .params import TrainParam
from .splits import calc_weight
from .tree_model import RegTree


def prune_tree(tree: RegTree, param: TrainParam) -> RegTree:
    """Recursively collapse splits with loss_chg < gamma whose children
    are both leaves.  Returns a compacted copy."""
    changed = True
    keep_split = np.array([not tree.is_leaf(n) for n in range(tree.n_nodes)])
    while changed:
        changed = False
        for nid in range(tree.n_nodes):
            if not keep_split[nid]:
                continue
            l, r = int(tree.left[nid]), int(tree.right[nid])
            l_leaf = not keep_split[l] if not tree.is_leaf(l) else True
            r_leaf = not keep_split[r] if not tree.is_leaf(r) else True
            if l_leaf and r_leaf and tree.loss_chg[nid] < param.gamma:
                keep_split[nid] = False
                changed = True
    # rebuild compacted tree (preorder; children ids assigned on visit)
    out = RegTree(tree.n_features, tree.n_targets)
    out._ensure(tree.n_nodes)

    def rebuild(old: int, new: int) -> None:
        if tree.is_leaf(old) or not keep_split[old]:
            out.left[new] = -1
            out.right[new] = -1
            out.split_cond[new] = (
                tree.split_cond[old] if tree.is_leaf(old)
                else np.float32(tree.base_weight[old] * param.eta))
            out.base_weight[new] = tree.base_weight[old]
            out.sum_hess[new] = tree.sum_hess[old]
            return
        l = out.n_nodes
        r = out.n_nodes + 1
        out._ensure(r + 1)
        out.n_nodes += 2
        out.left[new] = l
        out.right[new] = r
        out.parent[l] = new
        out.parent[r] = new
        out.split_index[new] = tree.split_index[old]
        out.split_cond[new] = tree.split_cond[old]
        out.default_left[new] = tree.default_left[old]
        out.loss_chg[new] = tree.loss_chg[old]
        out.sum_hess[new] = tree.sum_hess[old]
        out.base_weight[new] = tree.base_weight[old]
        out.split_type[new] = tree.split_type[old]
        if old in tree.cat_segments:
            out.cat_segments[new] = tree.cat_segments[old]
        rebuild(int(tree.left[old]), l)
        rebuild(int(tree.right[old]), r)

    rebuild(0, 0)
    return out


def refresh_tree(tree: RegTree, X: np.ndarray, gpair: np.ndarray,
                 param: TrainParam, missing: float = np.nan,
                 refresh_leaf: bool = True) -> None:
    """Recompute node statistics (and optionally leaf values) from new
    gradients, in place (reference TreeRefresher)."""
    n = X.shape[0]
    g = gpair[:, 0].astype(np.float64)
    h = gpair[:, 1].astype(np.float64)
    node_g = np.zeros(tree.n_nodes)
    node_h = np.zeros(tree.n_nodes)
    # route every row through the tree accumulating stats at every node
    pos = np.zeros(n, dtype=np.int64)
    active = np.ones(n, dtype=bool)
    while True:
        np.add.at(node_g, pos[active], g[active])
        np.add.at(node_h, pos[active], h[active])
        inner = active & (tree.left[pos] != -1)
        if not inner.any():
            break
        idx = np.nonzero(inner)[0]
        nid = pos[idx]
        feat = tree.split_index[nid]
        fval = X[idx, feat]
        if np.isnan(missing):
            is_missing = np.isnan(fval)
        else:
            is_missing = (fval == missing) | np.isnan(fval)
        go_left = np.where(is_missing, tree.default_left[nid].astype(bool),
                           fval < tree.split_cond[nid])
        pos[idx] = np.where(go_left, tree.left[nid], tree.right[nid])
        active = inner
    tree.sum_hess[:tree.n_nodes] = node_h[:tree.n_nodes]
    w = calc_weight(node_g, node_h, param)
    tree.base_weight[:tree.n_nodes] = w[:tree.n_nodes]
    if refresh_leaf:
        for nid in range(tree.n_nodes):
            if tree.is_leaf(nid):
                tree.set_leaf(nid, float(w[nid]) * param.eta)
