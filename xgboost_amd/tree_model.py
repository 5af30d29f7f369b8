"""Regression tree model — structure-of-arrays + xgboost-schema JSON IO.

Reference behavior: include/xgboost/tree_model.h:94 (Node),
src/tree/tree_model.cc:1198 (SaveModel JSON keys, via
src/tree/io_utils.h:50 tree_field), categorical split storage
tree_model.cc:976-1080.

Unlike the reference's array-of-16-byte-Node layout, we keep SoA numpy
arrays — that is the layout the HIP predictor kernel consumes directly.
"""
from __future__ import annotations

from typing import Dict, List, Optional

import numpy as np

_INIT_CAP = 8


class RegTree:
    """Binary tree with scalar leaves (vector leaves: see n_targets>1)."""

    def __init__(self, n_features: int, n_targets: int = 1):
        self.n_features = n_features
        self.n_targets = n_targets
        cap = _INIT_CAP
        self.left = np.full(cap, -1, dtype=np.int32)
        self.right = np.full(cap, -1, dtype=np.int32)
        self.parent = np.full(cap, -1, dtype=np.int32)
        self.split_index = np.zeros(cap, dtype=np.int32)
        self.split_cond = np.zeros(cap, dtype=np.float32)  # leaf value if leaf
        self.default_left = np.zeros(cap, dtype=np.uint8)
        self.loss_chg = np.zeros(cap, dtype=np.float32)
        self.sum_hess = np.zeros(cap, dtype=np.float32)
        self.base_weight = np.zeros(cap, dtype=np.float32)
        # categorical splits: node -> slice into self.categories (bin ids
        # that go LEFT is xgboost's "right category list"?  reference: the
        # stored set is categories that go RIGHT? -- we store "go right"
        # matching reference DecisionCat (cats contain -> right? see
        # predict_fn.h: GetCat -> go left if NOT in set).  We store the
        # category values for which the row goes RIGHT... see note in
        # categories_go_right.
        self.split_type = np.zeros(cap, dtype=np.uint8)  # 0 num, 1 cat
        self.cat_segments: Dict[int, np.ndarray] = {}  # node -> sorted int32 cats (go RIGHT)
        self.n_nodes = 1
        # leaf vector for multi-target trees
        self.leaf_values: Optional[np.ndarray] = None  # [n_nodes, n_targets]
        if n_targets > 1:
            self.leaf_values = np.zeros((cap, n_targets), dtype=np.float32)

    # -- growth ---------------------------------------------------------------
    def _ensure(self, n: int) -> None:
        cap = len(self.left)
        if n <= cap:
            return
        new_cap = max(n, cap * 2)
        for name in ("left", "right", "parent", "split_index", "split_cond",
                     "default_left", "loss_chg", "sum_hess", "base_weight",
                     "split_type"):
            arr = getattr(self, name)
            grown = np.zeros(new_cap, dtype=arr.dtype)
            if name in ("left", "right", "parent"):
                grown[:] = -1
            grown[:cap] = arr
            setattr(self, name, grown)
        if self.leaf_values is not None:
            grown = np.zeros((new_cap, self.n_targets), dtype=np.float32)
            grown[:cap] = self.leaf_values
            self.leaf_values = grown

    def add_split(self, nid: int, feature: int, cond: float, default_left: bool,
                  gain: float, base_weight: float, left_weight: float,
                  right_weight: float, sum_hess: float, left_hess: float,
                  right_hess: float, categories_go_right: Optional[np.ndarray] = None) -> tuple:
        l, r = self.n_nodes, self.n_nodes + 1
        self._ensure(r + 1)
        self.n_nodes += 2
        self.left[nid] = l
        self.right[nid] = r
        self.split_index[nid] = feature
        self.split_cond[nid] = cond
        self.default_left[nid] = 1 if default_left else 0
        self.loss_chg[nid] = gain
        self.sum_hess[nid] = sum_hess
        self.base_weight[nid] = base_weight
        for c, w, h in ((l, left_weight, left_hess), (r, right_weight, right_hess)):
            self.parent[c] = nid
            self.left[c] = -1
            self.right[c] = -1
            self.split_cond[c] = w       # leaf value placeholder
            self.base_weight[c] = w
            self.sum_hess[c] = h
        if categories_go_right is not None:
            self.split_type[nid] = 1
            self.cat_segments[nid] = np.asarray(
                sorted(categories_go_right), dtype=np.int32)
        return l, r

    def set_leaf(self, nid: int, value) -> None:
        self.left[nid] = -1
        self.right[nid] = -1
        if self.leaf_values is not None:
            self.leaf_values[nid] = np.asarray(value, dtype=np.float32)
            self.split_cond[nid] = 0.0
        else:
            self.split_cond[nid] = np.float32(value)

    def is_leaf(self, nid: int) -> bool:
        return self.left[nid] == -1

    def leaf_value(self, nid: int):
        if self.leaf_values is not None:
            return self.leaf_values[nid]
        return self.split_cond[nid]

    @property
    def num_nodes(self) -> int:
        return self.n_nodes

    def max_depth(self) -> int:
        depth = np.zeros(self.n_nodes, dtype=np.int32)
        out = 0
        for nid in range(1, self.n_nodes):
            depth[nid] = depth[self.parent[nid]] + 1
            out = max(out, int(depth[nid]))
        return out

    # -- prediction (numpy fallback / oracle) ---------------------------------
    def predict_leaf_np(self, X: np.ndarray, missing: float = np.nan) -> np.ndarray:
        """Vectorized traversal: returns leaf node id per row."""
        n = X.shape[0]
        pos = np.zeros(n, dtype=np.int32)
        active = self.left[pos] != -1
        while active.any():
            idx = np.nonzero(active)[0]
            nid = pos[idx]
            feat = self.split_index[nid]
            fval = X[idx, feat]
            if np.isnan(missing):
                is_missing = np.isnan(fval)
            else:
                is_missing = (fval == missing) | np.isnan(fval)
            go_left = np.where(is_missing,
                               self.default_left[nid].astype(bool),
                               fval < self.split_cond[nid])
            # categorical nodes
            cat_nodes = self.split_type[nid] == 1
            if cat_nodes.any():
                for k in np.nonzero(cat_nodes)[0]:
                    if is_missing[k]:
                        continue
                    cats = self.cat_segments[int(nid[k])]
                    go_left[k] = int(fval[k]) not in cats
            pos[idx] = np.where(go_left, self.left[nid], self.right[nid])
            active[idx] = self.left[pos[idx]] != -1
        return pos

    def predict_leaf_bins(self, gidx_global: "np.ndarray",
                          cuts) -> np.ndarray:
        """Traversal over GLOBAL bin ids (external-memory predict where
        raw values are gone): left iff bin <= split_bin; missing = -1."""
        split_bin = np.zeros(self.n_nodes, dtype=np.int64)
        for nid in range(self.n_nodes):
            if not self.is_leaf(nid):
                f = int(self.split_index[nid])
                fc = cuts.feature_cuts(f)
                b = int(np.searchsorted(fc, self.split_cond[nid], side="left"))
                split_bin[nid] = cuts.ptrs[f] + b
        n = gidx_global.shape[0]
        pos = np.zeros(n, dtype=np.int32)
        active = self.left[pos] != -1
        while active.any():
            idx = np.nonzero(active)[0]
            nid = pos[idx]
            feat = self.split_index[nid]
            b = gidx_global[idx, feat]
            missing = b < 0
            go_left = np.where(missing, self.default_left[nid].astype(bool),
                               b <= split_bin[nid])
            cat_nodes = self.split_type[nid] == 1
            if cat_nodes.any():
                for k in np.nonzero(cat_nodes)[0]:
                    if missing[k]:
                        continue
                    local = int(b[k] - cuts.ptrs[int(self.split_index[nid[k]])])
                    cats = self.cat_segments[int(nid[k])]
                    go_left[k] = local not in cats
            pos[idx] = np.where(go_left, self.left[nid], self.right[nid])
            active[idx] = self.left[pos[idx]] != -1
        return pos

    # -- JSON schema ----------------------------------------------------------
    def to_json(self, tree_id: int = 0) -> dict:
        n = self.n_nodes
        size_leaf_vector = self.n_targets
        out = {
            "tree_param": {
                "num_feature": str(self.n_features),
                "num_nodes": str(n),
                "size_leaf_vector": str(size_leaf_vector),
            },
            "id": tree_id,
            "loss_changes": self.loss_chg[:n].tolist(),
            "sum_hessian": self.sum_hess[:n].tolist(),
            "base_weights": self.base_weight[:n].tolist(),
            "split_indices": self.split_index[:n].tolist(),
            "split_conditions": self.split_cond[:n].astype(float).tolist(),
            "default_left": self.default_left[:n].astype(int).tolist(),
            "left_children": self.left[:n].tolist(),
            "right_children": self.right[:n].tolist(),
            "parents": self.parent[:n].tolist(),
        }
        if self.cat_segments:
            cat_nodes, segs, sizes, cats = [], [], [], []
            off = 0
            for nid in sorted(self.cat_segments):
                c = self.cat_segments[nid]
                cat_nodes.append(int(nid))
                segs.append(off)
                sizes.append(len(c))
                cats.extend(int(x) for x in c)
                off += len(c)
            out["split_type"] = self.split_type[:n].astype(int).tolist()
            out["categories_nodes"] = cat_nodes
            out["categories_segments"] = segs
            out["categories_sizes"] = sizes
            out["categories"] = cats
        if self.leaf_values is not None:
            out["leaf_weights"] = self.leaf_values[:n].reshape(-1).astype(float).tolist()
        return out

    @classmethod
    def from_json(cls, j: dict) -> "RegTree":
        tp = j["tree_param"]
        n = int(tp["num_nodes"])
        n_targets = max(1, int(tp.get("size_leaf_vector", "1")))
        t = cls(int(tp["num_feature"]), n_targets)
        t._ensure(n)
        t.n_nodes = n
        t.left[:n] = np.asarray(j["left_children"], dtype=np.int32)
        t.right[:n] = np.asarray(j["right_children"], dtype=np.int32)
        t.parent[:n] = np.asarray(j["parents"], dtype=np.int32)
        t.split_index[:n] = np.asarray(j["split_indices"], dtype=np.int32)
        t.split_cond[:n] = np.asarray(j["split_conditions"], dtype=np.float32)
        t.default_left[:n] = np.asarray(j["default_left"], dtype=np.uint8)
        t.loss_chg[:n] = np.asarray(j["loss_changes"], dtype=np.float32)
        t.sum_hess[:n] = np.asarray(j["sum_hessian"], dtype=np.float32)
        t.base_weight[:n] = np.asarray(j["base_weights"], dtype=np.float32)
        if "split_type" in j:
            t.split_type[:n] = np.asarray(j["split_type"], dtype=np.uint8)
            cats = np.asarray(j.get("categories", []), dtype=np.int32)
            nodes = j.get("categories_nodes", [])
            segs = j.get("categories_segments", [])
            sizes = j.get("categories_sizes", [])
            for nid, s, sz in zip(nodes, segs, sizes):
                t.cat_segments[int(nid)] = cats[s:s + sz].copy()
        if "leaf_weights" in j and n_targets > 1:
            t.leaf_values = np.asarray(
                j["leaf_weights"], dtype=np.float32).reshape(n, n_targets)
        return t

    # -- dumps ----------------------------------------------------------------
    def dump(self, fmap: Optional[List[str]] = None, with_stats: bool = False,
             format: str = "text") -> str:
        if format == "text":
            return self._dump_text(fmap, with_stats)
        if format == "json":
            import json as _json
            return _json.dumps(self._dump_json_node(0, fmap, with_stats))
        if format == "dot":
            return self._dump_dot(fmap, with_stats)
        raise ValueError(f"unknown dump format: {format}")

    def _fname(self, f: int, fmap) -> str:
        return fmap[f] if fmap and f < len(fmap) else f"f{f}"

    def _dump_text(self, fmap, with_stats) -> str:
        lines: List[str] = []

        def rec(nid: int, depth: int) -> None:
            indent = "\t" * depth
            if self.is_leaf(nid):
                s = f"{indent}{nid}:leaf={self.leaf_value(nid)}"
                if with_stats:
                    s += f",cover={self.sum_hess[nid]}"
            else:
                fn = self._fname(int(self.split_index[nid]), fmap)
                if self.split_type[nid] == 1:
                    cats = ",".join(str(c) for c in self.cat_segments.get(nid, []))
                    cond = f"[{fn}:{{{cats}}}]"
                else:
                    cond = f"[{fn}<{self.split_cond[nid]}]"
                s = (f"{indent}{nid}:{cond} yes={self.left[nid]},"
                     f"no={self.right[nid]},missing="
                     f"{self.left[nid] if self.default_left[nid] else self.right[nid]}")
                if with_stats:
                    s += f",gain={self.loss_chg[nid]},cover={self.sum_hess[nid]}"
            lines.append(s)
            if not self.is_leaf(nid):
                rec(int(self.left[nid]), depth + 1)
                rec(int(self.right[nid]), depth + 1)

        rec(0, 0)
        return "\n".join(lines) + "\n"

    def _dump_json_node(self, nid: int, fmap, with_stats) -> dict:
        if self.is_leaf(nid):
            out = {"nodeid": int(nid), "leaf": float(self.leaf_value(nid))
                   if self.leaf_values is None else self.leaf_value(nid).tolist()}
            if with_stats:
                out["cover"] = float(self.sum_hess[nid])
            return out
        out = {
            "nodeid": int(nid), "depth": 0,
            "split": self._fname(int(self.split_index[nid]), fmap),
            "split_condition": float(self.split_cond[nid]),
            "yes": int(self.left[nid]), "no": int(self.right[nid]),
            "missing": int(self.left[nid] if self.default_left[nid] else self.right[nid]),
        }
        if with_stats:
            out["gain"] = float(self.loss_chg[nid])
            out["cover"] = float(self.sum_hess[nid])
        out["children"] = [self._dump_json_node(int(self.left[nid]), fmap, with_stats),
                           self._dump_json_node(int(self.right[nid]), fmap, with_stats)]
        return out

    def _dump_dot(self, fmap, with_stats) -> str:
        lines = ["digraph {", "    graph [rankdir=TB]"]
        for nid in range(self.n_nodes):
            if self.parent[nid] == -1 and nid != 0:
                continue
            if self.is_leaf(nid):
                lines.append(f'    {nid} [ label="leaf={self.leaf_value(nid)}" ]')
            else:
                fn = self._fname(int(self.split_index[nid]), fmap)
                lines.append(f'    {nid} [ label="{fn}<{self.split_cond[nid]}" ]')
                yes, no = int(self.left[nid]), int(self.right[nid])
                miss = yes if self.default_left[nid] else no
                lines.append(f'    {nid} -> {yes} [label="yes, missing" ]'
                             if miss == yes else f'    {nid} -> {yes} [label="yes" ]')
                lines.append(f'    {nid} -> {no} [label="no, missing" ]'
                             if miss == no else f'    {nid} -> {no} [label="no" ]')
        lines.append("}")
        return "\n".join(lines)
