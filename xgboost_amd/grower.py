"""Histogram-based tree grower — the training hot loop.

Reference behavior: src/tree/updater_gpu_hist.cu:588 (UpdateTree loop),
driver.h (depthwise/lossguide priority queue), updater_gpu_hist.cuh:62
(AssignNodes build-smaller/subtract-larger), hist/evaluate_splits.h,
split_evaluator.h (monotone bounds), common/random.h:74 (ColumnSampler).

Backend-agnostic: all data-touching primitives go through an `ops`
object (CpuOps or GpuOps) so the same driver code runs the torch-CPU
oracle path and the HIP kernel path.
"""
from __future__ import annotations

import dataclasses
import heapq
from typing import Dict, List, Optional, Sequence, Tuple

import numpy as np
import torch

from .backend.cpu import GradQuantizer
from .params import TrainParam
from .splits import SplitEntry, calc_weight
from .tree_model import RegTree


class ColumnSampler:
    """Nested per-tree/level/node feature sampling (reference
    src/common/random.h:74).  Seeded identically on every worker so
    feature sets agree without communication.  With `feature_weights`
    set, sampling is weighted without replacement via the
    Efraimidis-Spirakis key trick (rand^(1/w), top-k — reference
    WeightedSamplingWithoutReplacement, random.h:104)."""

    def __init__(self, n_features: int, param: TrainParam, seed: int,
                 feature_weights: Optional[np.ndarray] = None):
        self.rng = np.random.RandomState(seed & 0x7FFFFFFF)
        self.n_features = n_features
        self.param = param
        self.weights = (np.asarray(feature_weights, dtype=np.float64)
                        if feature_weights is not None else None)
        feats = np.arange(n_features)
        if param.colsample_bytree < 1.0:
            k = max(1, int(round(param.colsample_bytree * n_features)))
            feats = self._sample(feats, k)
        self.tree_set = feats
        self._level_cache: Dict[int, np.ndarray] = {}

    def _sample(self, feats: np.ndarray, k: int) -> np.ndarray:
        if self.weights is None:
            return np.sort(self.rng.choice(feats, size=k, replace=False))
        w = self.weights[feats]
        pos = w > 0
        feats, w = feats[pos], w[pos]
        if feats.size == 0:
            raise ValueError("all sampled feature_weights are zero")
        k = min(k, feats.size)
        keys = self.rng.random_sample(feats.size) ** (1.0 / w)
        return np.sort(feats[np.argsort(-keys)[:k]])

    def level_set(self, depth: int) -> np.ndarray:
        if self.param.colsample_bylevel >= 1.0:
            return self.tree_set
        if depth not in self._level_cache:
            k = max(1, int(round(self.param.colsample_bylevel * len(self.tree_set))))
            self._level_cache[depth] = self._sample(self.tree_set, k)
        return self._level_cache[depth]

    def node_set(self, depth: int) -> Optional[np.ndarray]:
        base = self.level_set(depth)
        if self.param.colsample_bynode >= 1.0:
            if (self.param.colsample_bytree >= 1.0
                    and self.param.colsample_bylevel >= 1.0):
                return None  # all features
            return base
        k = max(1, int(round(self.param.colsample_bynode * len(base))))
        return self._sample(base, k)


class InteractionConstraints:
    """Per-node allowed-feature sets (reference src/tree/constraints.cc:103)."""

    def __init__(self, spec: Sequence[Sequence[int]], n_features: int):
        self.sets = [frozenset(int(f) for f in s) for s in spec]
        self.n_features = n_features
        self.node_path: Dict[int, frozenset] = {0: frozenset()}

    def split(self, nid: int, left: int, right: int, feature: int) -> None:
        path = self.node_path.get(nid, frozenset()) | {int(feature)}
        self.node_path[left] = path
        self.node_path[right] = path

    def allowed(self, nid: int) -> Optional[np.ndarray]:
        path = self.node_path.get(nid, frozenset())
        if not path:
            return None
        allowed = set(path)
        for s in self.sets:
            if path <= s:
                allowed |= s
        return np.array(sorted(allowed), dtype=np.int64)


@dataclasses.dataclass(order=True)
class _QueueEntry:
    sort_key: float
    seq: int
    nid: int = dataclasses.field(compare=False)
    depth: int = dataclasses.field(compare=False)
    split: SplitEntry = dataclasses.field(compare=False)


class TreeGrower:
    """Grows one tree from quantized gradients."""

    def __init__(self, ops, param: TrainParam, quantizer: GradQuantizer,
                 n_rows: int, seed: int = 0,
                 monotone: Optional[np.ndarray] = None,
                 interaction: Optional[Sequence[Sequence[int]]] = None,
                 feature_weights: Optional[np.ndarray] = None):
        self.ops = ops
        self.param = param
        self.quantizer = quantizer
        self.n_rows = n_rows
        self.col_sampler = ColumnSampler(ops.qm.n_features, param, seed,
                                         feature_weights)
        self.monotone = monotone
        self.interaction = (InteractionConstraints(interaction, ops.qm.n_features)
                            if interaction else None)
        cuts = ops.qm.cuts
        if cuts.feature_types is not None:
            self.cat_mask = np.array([t == "c" for t in cuts.feature_types])
            if not self.cat_mask.any():
                self.cat_mask = None
        else:
            self.cat_mask = None

    _monitor = None

    @classmethod
    def monitor(cls):
        if cls._monitor is None:
            from .monitor import Monitor
            cls._monitor = Monitor("TreeGrower")
        return cls._monitor

    def grow(self, qgpair: torch.Tensor, tree: RegTree
             ) -> Tuple[RegTree, torch.Tensor]:
        """Returns (tree, leaf position per row int32)."""
        mon = self.monitor()
        mon.start("grow")
        try:
            native = self._try_native(qgpair, tree)
            if native is not None:
                return native
            return self._grow(qgpair, tree)
        finally:
            mon.stop("grow")

    def _try_native(self, qgpair, tree):
        """C++ level-loop fast path (ops/cpp/driver.hip) for the common
        depthwise numeric config; feature-rich paths stay in Python."""
        p = self.param
        ops = self.ops
        if not hasattr(ops, "grow_tree_native"):
            return None
        if (self.cat_mask is not None or self.interaction is not None
                or p.colsample_bylevel < 1.0 or p.colsample_bynode < 1.0):
            return None
        fmask = None
        if p.colsample_bytree < 1.0:
            # per-TREE sampling is a constant feature set: the native
            # evaluator masks it with one broadcast row (rank-identical:
            # ColumnSampler seeding is deterministic per seed)
            fmask = np.zeros(ops.qm.n_features, np.uint8)
            fmask[self.col_sampler.tree_set] = 1
        needs_replay = p.grow_policy == "lossguide" or p.max_leaves > 0
        if needs_replay:
            if p.max_depth <= 0 or p.max_depth > 14:
                return None  # unbounded-depth lossguide: python driver
            if p.max_leaves > 0 and (1 << p.max_depth) > 16 * p.max_leaves:
                return None  # the full depthwise chain would be wasteful
        # root sums stay on device: their readback piggybacks on the
        # driver's root-eval sync (one fewer host round-trip per tree);
        # the fused gpair path accumulates them inside QuantizeKernel
        rs = getattr(self.quantizer, "root_sums_dev", None)
        if rs is None:
            from . import collective
            rs = qgpair.to(torch.int64).sum(dim=0).contiguous()
            if collective.is_distributed():
                collective.allreduce_sum_(rs)
        out = ops.grow_tree_native(qgpair, tree, p, self.quantizer,
                                   self.monotone, rs, feature_mask=fmask)
        if out is None or not needs_replay:
            return out
        dtree, pos = out
        return self._policy_replay(dtree, pos, p)

    def _policy_replay(self, dt: RegTree, pos: torch.Tensor,
                       param) -> Tuple[RegTree, torch.Tensor]:
        """Re-drive the grow-policy priority queue over the native
        driver's depthwise expansion records (reference Driver,
        src/tree/driver.h:17-51).

        The depthwise chain expands EVERY gain-valid node, a superset
        of what lossguide / max_leaves would expand (a node's candidate
        split depends only on its own histogram, never on which other
        nodes expanded), so the policy can be replayed on the host: pop
        candidates by (-gain, seq) — or by (depth, seq) for depthwise
        with a leaf cap — renumber nodes in pop order exactly like the
        Python driver, truncate at max_leaves, and remap row positions
        from pruned subtrees onto their covering leaf."""
        import heapq
        n = dt.n_nodes
        internal = dt.left[:n] >= 0
        gains = getattr(dt, "_gain64", None)
        if gains is None:
            gains = dt.loss_chg[:n].astype(np.float64)
        nt = RegTree(dt.n_features)
        nt._ensure(n)
        new_of = np.full(n, -1, np.int32)
        new_of[0] = 0
        nt.base_weight[0] = dt.base_weight[0]
        nt.sum_hess[0] = dt.sum_hess[0]
        nt.left[0] = nt.right[0] = nt.parent[0] = -1
        state = {"n_new": 1, "n_leaves": 1, "seq": 0}
        heap: List[Tuple[float, int, int, int]] = []

        def push(dnid: int, depth: int) -> None:
            if not internal[dnid]:
                return
            if param.max_depth > 0 and depth >= param.max_depth:
                return
            if param.max_leaves > 0 and state["n_leaves"] >= param.max_leaves:
                return
            key = (float(depth) if param.grow_policy == "depthwise"
                   else -float(gains[dnid]))
            heapq.heappush(heap, (key, state["seq"], dnid, depth))
            state["seq"] += 1

        push(0, 0)
        while heap:
            batch = [heapq.heappop(heap)]
            if param.grow_policy == "depthwise":
                while heap and heap[0][0] == batch[0][0]:
                    batch.append(heapq.heappop(heap))
            if param.max_leaves > 0:
                batch = batch[:max(0, param.max_leaves - state["n_leaves"])]
            if not batch:
                continue
            for _key, _sq, dnid, _depth in batch:
                nn = int(new_of[dnid])
                l, r = state["n_new"], state["n_new"] + 1
                state["n_new"] += 2
                dl, dr = int(dt.left[dnid]), int(dt.right[dnid])
                new_of[dl], new_of[dr] = l, r
                nt.left[nn], nt.right[nn] = l, r
                nt.parent[l] = nt.parent[r] = nn
                nt.split_index[nn] = dt.split_index[dnid]
                nt.split_cond[nn] = dt.split_cond[dnid]
                nt.default_left[nn] = dt.default_left[dnid]
                nt.loss_chg[nn] = dt.loss_chg[dnid]
                nt.sum_hess[nn] = dt.sum_hess[dnid]
                for c_new, c_d in ((l, dl), (r, dr)):
                    nt.base_weight[c_new] = dt.base_weight[c_d]
                    nt.sum_hess[c_new] = dt.sum_hess[c_d]
                    nt.left[c_new] = nt.right[c_new] = -1
                state["n_leaves"] += 1
            for _key, _sq, dnid, depth in batch:
                push(int(dt.left[dnid]), depth + 1)
                push(int(dt.right[dnid]), depth + 1)
        nt.n_nodes = state["n_new"]
        for i in range(nt.n_nodes):
            if nt.left[i] == -1:
                nt.split_cond[i] = np.float32(
                    float(nt.base_weight[i]) * param.eta)
        # remap positions: every depthwise leaf -> its covering new leaf
        lut = np.zeros(n, np.int32)
        stack = [(0, -1)]
        while stack:
            dnid, cover = stack.pop()
            if cover < 0:
                nn = int(new_of[dnid])
                if nn >= 0 and nt.left[nn] == -1:
                    cover = nn
            lut[dnid] = cover if cover >= 0 else 0
            if internal[dnid]:
                stack.append((int(dt.left[dnid]), cover))
                stack.append((int(dt.right[dnid]), cover))
        lut_t = torch.from_numpy(lut).to(pos.device)
        return nt, lut_t[pos.long()].to(torch.int32)

    def _grow(self, qgpair: torch.Tensor, tree: RegTree
              ) -> Tuple[RegTree, torch.Tensor]:
        param = self.param
        ops = self.ops
        if getattr(self.quantizer, "g_scale", 1.0) is None:
            # the fused GPU path leaves scales on the device; the python
            # driver needs host copies (same formula, one sync)
            ma = self.quantizer.maxabs_dev.cpu()
            mg, mh = float(ma[0]), float(ma[1])
            self.quantizer.g_scale = (1 << 30) / mg if mg > 0 else 1.0
            self.quantizer.h_scale = (1 << 30) / mh if mh > 0 else 1.0
        ops.reset(self.n_rows)
        node_sums: Dict[int, Tuple[int, int]] = {}  # exact int64 (gq, hq)
        node_bounds: Dict[int, Tuple[float, float]] = {0: (-np.inf, np.inf)}
        hists: Dict[int, torch.Tensor] = {}

        # root
        qg, qh = ops.root_sum(qgpair)
        root_g = qg / self.quantizer.g_scale
        root_h = qh / self.quantizer.h_scale
        node_sums[0] = (qg, qh)
        root_w = float(calc_weight(root_g, root_h, param))
        tree.base_weight[0] = root_w
        tree.sum_hess[0] = root_h
        hist = ops.build_hist_nodes(qgpair, [0])
        ops.allreduce_hist(hist)
        hists[0] = hist[0]
        root_entry = self._evaluate([0], node_sums, hists, node_bounds, depth=0)[0]

        n_leaves = 1
        seq = 0
        heap: List[_QueueEntry] = []

        def push(nid, depth, split):
            nonlocal seq
            if not self._expandable(split, depth, n_leaves):
                return
            key = (float(depth) if param.grow_policy == "depthwise"
                   else -split.gain)
            heapq.heappush(heap, _QueueEntry(key, seq, nid, depth, split))
            seq += 1

        push(0, 0, root_entry)

        while heap:
            # pop a batch: depthwise = whole level; lossguide = single best
            batch = [heapq.heappop(heap)]
            if param.grow_policy == "depthwise":
                while heap and heap[0].sort_key == batch[0].sort_key:
                    batch.append(heapq.heappop(heap))
            # re-check leaf budget
            batch = [b for b in batch
                     if self._expandable(b.split, b.depth, n_leaves + 0)]
            if param.max_leaves > 0:
                keep = []
                for b in batch:
                    if n_leaves + len(keep) + 1 <= param.max_leaves:
                        keep.append(b)
                batch = keep
            if not batch:
                continue

            # 1. apply splits to tree
            children = []  # (entry, lnid, rnid)
            for b in batch:
                sp = b.split
                cuts = ops.qm.cuts
                if sp.is_cat:
                    cond = np.nan
                    cats_right = sp.cat_bits
                else:
                    cond = float(cuts.values[sp.split_bin])
                    cats_right = None
                wl = float(calc_weight(sp.left_g, sp.left_h, param))
                wr = float(calc_weight(sp.right_g, sp.right_h, param))
                lo, hi = node_bounds.get(b.nid, (-np.inf, np.inf))
                wl, wr = self._clip_bounds(wl, wr, lo, hi)
                l, r = tree.add_split(
                    b.nid, sp.feature, cond, sp.default_left, sp.gain,
                    float(tree.base_weight[b.nid]), wl, wr,
                    sp.left_h + sp.right_h, sp.left_h, sp.right_h,
                    categories_go_right=cats_right)
                node_sums[l] = (sp.left_gq, sp.left_hq)
                node_sums[r] = (sp.right_gq, sp.right_hq)
                self._propagate_bounds(node_bounds, b.nid, l, r, sp, wl, wr)
                if self.interaction is not None:
                    self.interaction.split(b.nid, l, r, sp.feature)
                children.append((b, l, r))
                n_leaves += 1

            # 2. partition rows
            parents = [b.nid for b, _, _ in children]
            splits = [b.split for b, _, _ in children]
            child_pairs = [(l, r) for _, l, r in children]
            ops.partition_nodes(parents, splits, child_pairs)

            # 3. decide build vs subtract (smaller child built,
            #    reference AssignNodes updater_gpu_hist.cuh:62)
            build_nodes = []
            for (b, l, r) in children:
                expand_more = (b.depth + 1 < (param.max_depth or 10 ** 9)
                               or param.grow_policy == "lossguide")
                if not expand_more and param.max_depth > 0:
                    continue
                # build the smaller sibling by GLOBAL hessian sum:
                # local row counts are rank-dependent, and ranks must
                # build/allreduce the SAME node set (exact int64
                # subtraction makes the choice performance-only)
                if node_sums[l][1] <= node_sums[r][1]:
                    build_nodes.append((l, b.nid, r))
                else:
                    build_nodes.append((r, b.nid, l))
            # 4. build + allreduce + subtract — all into ONE level
            # buffer ([built... | subtracted...]) so evaluation reads a
            # contiguous stack with no concatenation copies
            if build_nodes:
                kb = len(build_nodes)
                stack = (ops.alloc_hist(2 * kb)
                         if hasattr(ops, "alloc_hist") else None)
                bnids = [n for n, _, _ in build_nodes]
                if stack is None:  # extmem / csr ops: no out= support
                    bh = ops.build_hist_nodes(qgpair, bnids)
                else:
                    bh = ops.build_hist_nodes(qgpair, bnids,
                                              out=stack[:kb])
                ops.allreduce_hist(bh)
                parent_stack = torch.stack(
                    [hists[p] for _, p, _ in build_nodes])
                if stack is None:
                    sib_stack = parent_stack - bh
                    hist_stack = torch.cat([bh, sib_stack], dim=0)
                else:
                    torch.sub(parent_stack, bh, out=stack[kb:])
                    sib_stack = stack[kb:]
                    hist_stack = stack
                for i, (n, parent, sib) in enumerate(build_nodes):
                    hists[n] = bh[i]
                    hists[sib] = sib_stack[i]
                    del hists[parent]
                eval_nids = ([n for n, _, _ in build_nodes]
                             + [s for _, _, s in build_nodes])
                depth = children[0][0].depth + 1
                entries = self._evaluate(eval_nids, node_sums, hists,
                                         node_bounds, depth,
                                         hist_stack=hist_stack)
                entry_of = dict(zip(eval_nids, entries))
                # push in (left, right) pair order so node numbering at
                # the next level matches the native C++ driver
                for (b, l, r) in children:
                    if l in entry_of:
                        push(l, depth, entry_of[l])
                    if r in entry_of:
                        push(r, depth, entry_of[r])

        # finalize leaves
        leaf_nids = []
        for nid in range(tree.n_nodes):
            if tree.is_leaf(nid):
                w = float(tree.base_weight[nid])
                tree.set_leaf(nid, w * param.eta)
                leaf_nids.append(nid)
        positions = ops.leaf_positions(leaf_nids)
        return tree, positions

    # ------------------------------------------------------------------
    def _expandable(self, split: SplitEntry, depth: int, n_leaves: int) -> bool:
        param = self.param
        # reference prunes loss_chg < min_split_loss (driver.h:37): a
        # split with gain exactly == gamma IS expanded; the separate
        # gain > 0 validity lives in SplitEntry.is_valid
        if not split.is_valid or split.gain < param.gamma:
            return False
        if param.max_depth > 0 and depth >= param.max_depth:
            return False
        if param.max_leaves > 0 and n_leaves >= param.max_leaves:
            return False
        return True

    def _evaluate(self, nids, node_sums, hists, node_bounds, depth,
                  hist_stack: Optional[torch.Tensor] = None
                  ) -> List[SplitEntry]:
        param = self.param
        hist = (hist_stack if hist_stack is not None
                else torch.stack([hists[n] for n in nids]))
        parent_sums = [node_sums[n] for n in nids]
        feature_sets = None
        node_feats = self.col_sampler.node_set(depth)
        inter_sets = ([self.interaction.allowed(n) for n in nids]
                      if self.interaction is not None else None)
        if node_feats is not None or inter_sets is not None:
            feature_sets = []
            for i, n in enumerate(nids):
                fs = node_feats
                if self.param.colsample_bynode < 1.0:
                    fs = self.col_sampler.node_set(depth)  # fresh draw per node
                if inter_sets is not None and inter_sets[i] is not None:
                    fs = (np.intersect1d(fs, inter_sets[i]) if fs is not None
                          else inter_sets[i])
                feature_sets.append(fs)
        bounds = (np.array([node_bounds.get(n, (-np.inf, np.inf)) for n in nids])
                  if self.monotone is not None else None)
        return self.ops.evaluate_splits(
            hist, self.quantizer,
            [(g, h) for g, h in parent_sums], nids, param,
            feature_sets=feature_sets, monotone=self.monotone,
            cat_mask=self.cat_mask, node_bounds=bounds)

    def _clip_bounds(self, wl, wr, lo, hi):
        return float(np.clip(wl, lo, hi)), float(np.clip(wr, lo, hi))

    def _propagate_bounds(self, node_bounds, nid, l, r, sp, wl, wr) -> None:
        lo, hi = node_bounds.get(nid, (-np.inf, np.inf))
        node_bounds[l] = (lo, hi)
        node_bounds[r] = (lo, hi)
        if self.monotone is None:
            return
        c = int(self.monotone[sp.feature]) if sp.feature < len(self.monotone) else 0
        if c == 0:
            return
        mid = (wl + wr) / 2.0
        if c > 0:
            node_bounds[l] = (lo, min(hi, mid))
            node_bounds[r] = (max(lo, mid), hi)
        else:
            node_bounds[l] = (max(lo, mid), hi)
            node_bounds[r] = (lo, min(hi, mid))


class MultiTargetGrower:
    """Vector-leaf tree grower: ONE tree structure, per-target leaf
    values (reference MultiTargetHistMaker, updater_gpu_hist.cuh:109;
    target-major histograms histogram.cu:256; leaf sums leaf_sum.cuh).

    Histograms are built per target over the same row partition; split
    gain is the sum of per-target gains (multi_evaluate_splits.cu)."""

    def __init__(self, ops, param: TrainParam, quantizers, n_rows: int,
                 seed: int = 0,
                 feature_weights: Optional[np.ndarray] = None):
        self.ops = ops
        self.param = param
        self.quantizers = quantizers  # one GradQuantizer per target
        self.n_rows = n_rows
        self.col_sampler = ColumnSampler(ops.qm.n_features, param, seed,
                                         feature_weights)

    def grow(self, qgpairs, tree: RegTree):
        from .splits import evaluate_splits_multi_np
        param = self.param
        ops = self.ops
        T = len(qgpairs)
        ops.reset(self.n_rows)
        g_scales = np.array([q.g_scale for q in self.quantizers])
        h_scales = np.array([q.h_scale for q in self.quantizers])

        node_sums: Dict[int, np.ndarray] = {}  # nid -> int64 [T, 2]
        hists: Dict[int, torch.Tensor] = {}    # nid -> [T, bins, 2]

        root = np.stack([np.asarray(ops.root_sum(qgpairs[t]), np.int64)
                         for t in range(T)])
        node_sums[0] = root
        w0 = calc_weight(root[:, 0] / g_scales, root[:, 1] / h_scales, param)
        tree.base_weight[0] = float(np.mean(w0))
        tree.sum_hess[0] = float((root[:, 1] / h_scales).sum())

        qg_mt = None
        if hasattr(ops, "build_hist_nodes_mt") and T <= 8:
            qg_mt = torch.stack(qgpairs, dim=1).contiguous()

        def mt_hist(nids):
            """[T, k, bins, 2] — fused single-pass kernel when
            available (gbt_hist_mt), else per-target launches."""
            if qg_mt is not None:
                bh = ops.build_hist_nodes_mt(qg_mt, nids)
                if bh is not None:
                    return bh
            return torch.stack([ops.build_hist_nodes(qgpairs[t], nids)
                                for t in range(T)])

        hist0 = mt_hist([0])[:, 0]
        ops.allreduce_hist(hist0)
        hists[0] = hist0

        def evaluate(nids, depth):
            h = torch.stack([hists[n] for n in nids], dim=1)  # [T,k,b,2]
            parents = np.stack([node_sums[n] for n in nids])  # [k,T,2]
            fs = None
            node_feats = self.col_sampler.node_set(depth)
            if node_feats is not None:
                fs = [node_feats for _ in nids]
            if h.is_cuda:
                out = self._evaluate_gpu(h, parents, g_scales, h_scales,
                                         nids, fs)
                if out is not None:
                    return out
            return evaluate_splits_multi_np(
                h.cpu().numpy(), parents, g_scales, h_scales, nids,
                ops.qm.cuts.ptrs, param, feature_sets=fs)

        root_entry = evaluate([0], 0)[0]
        n_leaves = 1
        seq = 0
        heap: List[_QueueEntry] = []

        def push(nid, depth, split):
            nonlocal seq
            if not split.is_valid or split.gain < param.gamma:
                return
            if param.max_depth > 0 and depth >= param.max_depth:
                return
            if param.max_leaves > 0 and n_leaves >= param.max_leaves:
                return
            key = (float(depth) if param.grow_policy == "depthwise"
                   else -split.gain)
            heapq.heappush(heap, _QueueEntry(key, seq, nid, depth, split))
            seq += 1

        push(0, 0, root_entry)
        while heap:
            batch = [heapq.heappop(heap)]
            if param.grow_policy == "depthwise":
                while heap and heap[0].sort_key == batch[0].sort_key:
                    batch.append(heapq.heappop(heap))
            if param.max_leaves > 0:
                batch = batch[:max(0, param.max_leaves - n_leaves)]
            if not batch:
                continue
            children = []
            for b in batch:
                sp = b.split
                cond = float(ops.qm.cuts.values[sp.split_bin])
                wl = calc_weight(sp.left_q[:, 0] / g_scales,
                                 sp.left_q[:, 1] / h_scales, param)
                wr = calc_weight(sp.right_q[:, 0] / g_scales,
                                 sp.right_q[:, 1] / h_scales, param)
                lh = float((sp.left_q[:, 1] / h_scales).sum())
                rh = float((sp.right_q[:, 1] / h_scales).sum())
                l, r = tree.add_split(
                    b.nid, sp.feature, cond, sp.default_left, sp.gain,
                    float(tree.base_weight[b.nid]),
                    float(np.mean(wl)), float(np.mean(wr)),
                    lh + rh, lh, rh)
                tree.leaf_values[l] = wl.astype(np.float32)
                tree.leaf_values[r] = wr.astype(np.float32)
                node_sums[l] = sp.left_q
                node_sums[r] = sp.right_q
                children.append((b, l, r))
                n_leaves += 1
            parents = [b.nid for b, _, _ in children]
            from .splits import SplitEntry
            scalar_splits = [SplitEntry(
                nid=b.nid, feature=b.split.feature,
                split_bin=b.split.split_bin,
                default_left=b.split.default_left) for b, _, _ in children]
            ops.partition_nodes(parents, scalar_splits,
                                [(l, r) for _, l, r in children])
            build_nodes = []
            for (b, l, r) in children:
                if param.max_depth > 0 and b.depth + 1 >= param.max_depth \
                        and param.grow_policy == "depthwise":
                    continue
                # global hessian-sum choice: rank-identical (see the
                # single-target driver note above)
                if node_sums[l][:, 1].sum() <= node_sums[r][:, 1].sum():
                    build_nodes.append((l, b.nid, r))
                else:
                    build_nodes.append((r, b.nid, l))
            if build_nodes:
                bh = mt_hist([n for n, _, _ in build_nodes])
                bh = bh.contiguous()
                ops.allreduce_hist(bh)
                for i, (n, parent, sib) in enumerate(build_nodes):
                    hists[n] = bh[:, i]
                    hists[sib] = hists[parent] - bh[:, i]
                    del hists[parent]
                eval_nids = ([n for n, _, _ in build_nodes]
                             + [s for _, _, s in build_nodes])
                depth = children[0][0].depth + 1
                entry_of = dict(zip(eval_nids, evaluate(eval_nids, depth)))
                for (b, l, r) in children:
                    if l in entry_of:
                        push(l, depth, entry_of[l])
                    if r in entry_of:
                        push(r, depth, entry_of[r])

        leaf_nids = []
        for nid in range(tree.n_nodes):
            if tree.is_leaf(nid):
                tree.leaf_values[nid] = tree.leaf_values[nid] * param.eta
                tree.set_leaf(nid, tree.leaf_values[nid])
                leaf_nids.append(nid)
        positions = ops.leaf_positions(leaf_nids)
        return tree, positions

    def _evaluate_gpu(self, h, parents_np, g_scales, h_scales, nids, fs):
        """Device MT split evaluation (ops/cpp/mt_evaluate.hip); left
        sums for the winning bin are derived with torch cumsum on device
        (reference ScanHistogramKernel + MT EvaluateSplitsKernel)."""
        from .splits import MultiSplitEntry
        ops = self.ops
        if not hasattr(ops, "lib") or not hasattr(ops.lib, "gbt_mt_evaluate"):
            return None
        hip = ops.hip
        dev = h.device
        T, k, n_bins, _ = h.shape
        f = ops.qm.n_features
        cuts = ops.qm.cuts
        hc = h.contiguous()
        parents_t = torch.from_numpy(
            np.ascontiguousarray(parents_np, np.int64)).to(dev)
        gs_t = torch.from_numpy(np.ascontiguousarray(
            g_scales, np.float64)).to(dev)
        hs_t = torch.from_numpy(np.ascontiguousarray(
            h_scales, np.float64)).to(dev)
        mask_t = None
        if fs is not None:
            m = np.zeros((k, f), np.uint8)
            for i, s in enumerate(fs):
                if s is None:
                    m[i] = 1
                else:
                    m[i, np.asarray(s, np.int64)] = 1
            mask_t = torch.from_numpy(m).to(dev)
        gain = torch.empty((k, f), dtype=torch.float64, device=dev)
        bins = torch.empty((k, f), dtype=torch.int32, device=dev)
        dirs = torch.empty((k, f), dtype=torch.uint8, device=dev)
        p = self.param
        ops.lib.gbt_mt_evaluate(
            hip.ptr(hc), T, k, n_bins, f, hip.ptr(ops.cut_ptrs),
            hip.ptr(parents_t), hip.ptr(gs_t), hip.ptr(hs_t),
            p.reg_lambda, p.reg_alpha, p.max_delta_step, p.min_child_weight,
            hip.ptr(mask_t), hip.ptr(gain), hip.ptr(bins), hip.ptr(dirs),
            hip.stream())
        best_f = torch.argmax(gain, dim=1)
        ar = torch.arange(k, device=dev)
        packed = torch.stack([gain[ar, best_f].view(torch.int64),
                              bins[ar, best_f].to(torch.int64),
                              dirs[ar, best_f].to(torch.int64),
                              best_f], dim=1).cpu().numpy()
        ptrs = cuts.ptrs
        out = []
        for i, nid in enumerate(nids):
            e = MultiSplitEntry(nid=int(nid), n_targets=T)
            gv = packed[i, 0:1].view(np.float64)[0]
            b = int(packed[i, 1])
            if b >= 0 and np.isfinite(gv):
                fi = int(packed[i, 3])
                e.gain = float(gv)
                e.feature = fi
                e.split_bin = b
                e.default_left = bool(packed[i, 2])
                fb0, fb1 = int(ptrs[fi]), int(ptrs[fi + 1])
                seg = hc[:, i, fb0:fb1, :]           # [T, w, 2]
                cum = seg.cumsum(dim=1)
                left = cum[:, b - fb0, :]            # [T, 2]
                if e.default_left:
                    parent_i = parents_t[i]          # [T, 2]
                    feat_tot = cum[:, -1, :]
                    left = left + (parent_i - feat_tot)
                lq = left.cpu().numpy().astype(np.int64)
                e.left_q = lq
                e.right_q = parents_np[i] - lq
            out.append(e)
        return out
