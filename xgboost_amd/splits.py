"""Split evaluation math — gain, weights, constraints, scan.

Reference behavior: src/tree/split_evaluator.h (TreeEvaluator), param.h
CalcGain/CalcWeight, src/tree/hist/evaluate_splits.h:30 and
src/tree/gpu_hist/evaluate_splits.cu (EvaluateSplitAgent).

Determinism: histograms and all left/right sums are int64 fixed-point
(quantized gradients, see backend/cpu.py GradQuantizer) — scans and
sibling subtraction are exact integer math, so the numpy oracle and the
HIP kernel produce identical child sums; only the gain value itself is
computed in float64 after dequantization, with a fixed operation order
shared by both implementations.
"""
from __future__ import annotations

import dataclasses
from typing import List, Optional, Sequence

import numpy as np

from .params import TrainParam


def threshold_l1(g, alpha: float):
    return np.sign(g) * np.maximum(np.abs(g) - alpha, 0.0)


def calc_weight(g, h, param: TrainParam):
    """-ThresholdL1(G)/(H+lambda), optionally clipped by max_delta_step."""
    g = np.asarray(g, dtype=np.float64)
    h = np.asarray(h, dtype=np.float64)
    w = -threshold_l1(g, param.reg_alpha) / (h + param.reg_lambda)
    if param.max_delta_step > 0:
        w = np.clip(w, -param.max_delta_step, param.max_delta_step)
    return w


def calc_gain_given_weight(g, h, w, param: TrainParam):
    """-(2*G*w + (H+lambda)*w^2) (reference param.h CalcGainGivenWeight)."""
    g = np.asarray(g, dtype=np.float64)
    h = np.asarray(h, dtype=np.float64)
    w = np.asarray(w, dtype=np.float64)
    return -(2.0 * g * w + (h + param.reg_lambda) * w * w)


def calc_gain(g, h, param: TrainParam):
    return calc_gain_given_weight(g, h, calc_weight(g, h, param), param)


@dataclasses.dataclass
class SplitEntry:
    nid: int
    gain: float = -np.inf         # loss_chg
    feature: int = -1
    split_bin: int = -1           # global bin index
    default_left: bool = False
    left_gq: int = 0              # exact int64 fixed-point sums
    left_hq: int = 0
    right_gq: int = 0
    right_hq: int = 0
    g_scale: float = 1.0
    h_scale: float = 1.0
    is_cat: bool = False
    cat_bits: Optional[np.ndarray] = None  # local bin ids going RIGHT

    @property
    def left_g(self) -> float:
        return self.left_gq / self.g_scale

    @property
    def left_h(self) -> float:
        return self.left_hq / self.h_scale

    @property
    def right_g(self) -> float:
        return self.right_gq / self.g_scale

    @property
    def right_h(self) -> float:
        return self.right_hq / self.h_scale

    @property
    def is_valid(self) -> bool:
        return self.feature >= 0 and np.isfinite(self.gain) and self.gain > 0


def _sorted_cat_split(e: SplitEntry, node_hist: np.ndarray, pgq: int, phq: int,
                      f: int, cut_ptrs: np.ndarray, param: TrainParam,
                      inv_g: float, inv_h: float, parent_gain: float) -> None:
    """Partition-based categorical split for wide categorical features
    (reference gpu_hist/evaluate_splits.cuh SortHistogram + Partition
    agent): categories sorted by gradient ratio, scanned like a numeric
    feature; the best prefix goes LEFT, the complement is the stored
    go-RIGHT set.  Updates `e` in place if a better split is found."""
    b0, b1 = int(cut_ptrs[f]), int(cut_ptrs[f + 1])
    Gc = node_hist[b0:b1, 0].astype(np.int64)
    Hc = node_hist[b0:b1, 1].astype(np.int64)
    present = Hc != 0
    if present.sum() < 2:
        return
    ratio = np.where(present,
                     (Gc * inv_g) / (Hc * inv_h + param.reg_lambda), np.inf)
    order = np.argsort(ratio, kind="stable")
    order = order[present[order]]          # present categories, sorted
    order = order[:param.max_cat_threshold + 1]
    cg = np.cumsum(Gc[order])
    ch = np.cumsum(Hc[order])
    featG, featH = int(Gc.sum()), int(Hc.sum())
    missG, missH = pgq - featG, phq - featH
    for missing_left in (False, True):
        gl = cg + (missG if missing_left else 0)
        hl = ch + (missH if missing_left else 0)
        gr = pgq - gl
        hr = phq - hl
        glf, hlf = gl * inv_g, hl * inv_h
        grf, hrf = gr * inv_g, hr * inv_h
        wl = calc_weight(glf, hlf, param)
        wr = calc_weight(grf, hrf, param)
        gains = (calc_gain_given_weight(glf, hlf, wl, param)
                 + calc_gain_given_weight(grf, hrf, wr, param) - parent_gain)
        ok = ((hlf >= param.min_child_weight) & (hrf >= param.min_child_weight)
              & (hl > 0) & (hr > 0))
        gains = np.where(ok, gains, -np.inf)
        if not np.isfinite(gains).any():
            continue
        i = int(np.argmax(gains))
        gv = float(gains[i])
        if gv > e.gain:
            e.gain = gv
            e.feature = f
            e.split_bin = b0  # informational; cat_bits carries the split
            e.default_left = missing_left
            e.left_gq = int(gl[i])
            e.left_hq = int(hl[i])
            e.right_gq = pgq - e.left_gq
            e.right_hq = phq - e.left_hq
            e.is_cat = True
            # everything NOT in the chosen prefix goes RIGHT
            left_set = set(int(c) for c in order[:i + 1])
            e.cat_bits = np.array(
                sorted(c for c in range(b1 - b0) if c not in left_set),
                dtype=np.int32)


def evaluate_splits_np(hist_q: np.ndarray,
                       parent_q: Sequence,
                       g_scale: float, h_scale: float,
                       nids: Sequence[int], cut_ptrs: np.ndarray,
                       param: TrainParam,
                       feature_sets: Optional[List[np.ndarray]] = None,
                       monotone: Optional[np.ndarray] = None,
                       cat_mask: Optional[np.ndarray] = None,
                       node_bounds: Optional[np.ndarray] = None,
                       ) -> List[SplitEntry]:
    """Vectorized split evaluation over int64 [n_nodes, n_bins, 2] hists.

    parent_q: [(gq, hq)] exact node totals (int64).
    """
    n_nodes, n_bins, _ = hist_q.shape
    n_features = len(cut_ptrs) - 1
    widths = np.diff(cut_ptrs)
    feat_of_bin = np.repeat(np.arange(n_features), widths)
    seg_start = np.repeat(cut_ptrs[:-1], widths)
    seg_end = np.repeat(cut_ptrs[1:] - 1, widths)

    pgq = np.array([p[0] for p in parent_q], dtype=np.int64)
    phq = np.array([p[1] for p in parent_q], dtype=np.int64)

    Gq = hist_q[:, :, 0]
    Hq = hist_q[:, :, 1]
    cumG = np.cumsum(Gq, axis=1)
    cumH = np.cumsum(Hq, axis=1)
    baseG = np.where(seg_start > 0, cumG[:, np.maximum(seg_start - 1, 0)], 0)
    baseH = np.where(seg_start > 0, cumH[:, np.maximum(seg_start - 1, 0)], 0)
    GLq = cumG - baseG   # int64 exact: left sums including bin b
    HLq = cumH - baseH
    featGq = cumG[:, seg_end] - baseG
    featHq = cumH[:, seg_end] - baseH
    missGq = pgq[:, None] - featGq
    missHq = phq[:, None] - featHq

    inv_g = 1.0 / g_scale
    inv_h = 1.0 / h_scale
    parent_gain = calc_gain(pgq * inv_g, phq * inv_h, param)  # [n_nodes]

    best = [SplitEntry(nid=int(nid), g_scale=g_scale, h_scale=h_scale)
            for nid in nids]

    mono_bins = monotone[feat_of_bin][None, :] if monotone is not None else None

    for missing_left in (False, True):
        glq = GLq + (missGq if missing_left else 0)
        hlq = HLq + (missHq if missing_left else 0)
        grq = pgq[:, None] - glq
        hrq = phq[:, None] - hlq
        gl = glq * inv_g
        hl = hlq * inv_h
        gr = grq * inv_g
        hr = hrq * inv_h
        wl = calc_weight(gl, hl, param)
        wr = calc_weight(gr, hr, param)
        if node_bounds is not None:
            lo = node_bounds[:, 0][:, None]
            hi = node_bounds[:, 1][:, None]
            wl = np.clip(wl, lo, hi)
            wr = np.clip(wr, lo, hi)
        gain = (calc_gain_given_weight(gl, hl, wl, param)
                + calc_gain_given_weight(gr, hr, wr, param)
                - parent_gain[:, None])
        ok = (hl >= param.min_child_weight) & (hr >= param.min_child_weight)
        # degenerate last-bin split (empty right) is rejected by the
        # hessian checks / zero gain; last-bin + missing-right is a VALID
        # "present vs absent" split (essential for one-hot sparse data)
        ok &= (hrq > 0) & (hlq > 0)
        if mono_bins is not None:
            ok &= ((mono_bins == 0) | ((mono_bins > 0) & (wl <= wr))
                   | ((mono_bins < 0) & (wl >= wr)))
        gain = np.where(ok, gain, -np.inf)
        if cat_mask is not None and cat_mask.any():
            # one-hot only below max_cat_to_onehot; wide categorical
            # features use the sorted-partition path below
            narrow_cat = cat_mask & (widths <= param.max_cat_to_onehot)
            wide_bins = (cat_mask & ~narrow_cat)[feat_of_bin]
            gain, cat_lq = _onehot_cat_gains(
                gain, Gq, Hq, pgq, phq, missGq, missHq, feat_of_bin,
                narrow_cat, param, missing_left, parent_gain, node_bounds,
                mono_bins, inv_g, inv_h)
            gain = np.where(wide_bins[None, :], -np.inf, gain)
        else:
            cat_lq = None
        for i in range(n_nodes):
            row = gain[i]
            if feature_sets is not None and feature_sets[i] is not None:
                mask = np.zeros(n_bins, dtype=bool)
                for f in feature_sets[i]:
                    mask[cut_ptrs[f]:cut_ptrs[f + 1]] = True
                row = np.where(mask, row, -np.inf)
            b = int(np.argmax(row))
            gval = float(row[b])
            e = best[i]
            if gval > e.gain and np.isfinite(gval):
                f = int(feat_of_bin[b])
                is_cat = bool(cat_mask is not None and cat_mask[f])
                if is_cat:
                    lgq = int(cat_lq[0][i, b])
                    lhq = int(cat_lq[1][i, b])
                else:
                    lgq = int(glq[i, b])
                    lhq = int(hlq[i, b])
                e.gain = gval
                e.feature = f
                e.split_bin = b
                e.default_left = missing_left
                e.left_gq = lgq
                e.left_hq = lhq
                e.right_gq = int(pgq[i]) - lgq
                e.right_hq = int(phq[i]) - lhq
                e.is_cat = is_cat
                if is_cat:
                    e.cat_bits = np.array([b - int(cut_ptrs[f])], dtype=np.int32)
    # sorted-partition splits for wide categorical features
    if cat_mask is not None and cat_mask.any():
        wide_feats = np.nonzero(cat_mask
                                & (widths > param.max_cat_to_onehot))[0]
        if len(wide_feats):
            for i in range(n_nodes):
                allowed = (set(int(x) for x in feature_sets[i])
                           if feature_sets is not None
                           and feature_sets[i] is not None else None)
                pgain = parent_gain[i]
                for f in wide_feats:
                    if allowed is not None and int(f) not in allowed:
                        continue
                    _sorted_cat_split(best[i], hist_q[i], int(pgq[i]),
                                      int(phq[i]), int(f), cut_ptrs, param,
                                      inv_g, inv_h, float(pgain))
    return best


def _onehot_cat_gains(gain, Gq, Hq, pgq, phq, missGq, missHq,
                      feat_of_bin, cat_mask, param, missing_left,
                      parent_gain, node_bounds, mono_bins, inv_g, inv_h):
    """Categorical one-vs-rest: the chosen category (one bin) goes RIGHT,
    everything else left (reference evaluate_splits.cu OneHot)."""
    cat_bins = cat_mask[feat_of_bin]
    glq = pgq[:, None] - Gq - (0 if missing_left else missGq)
    hlq = phq[:, None] - Hq - (0 if missing_left else missHq)
    gl = glq * inv_g
    hl = hlq * inv_h
    gr = (pgq[:, None] - glq) * inv_g
    hr = (phq[:, None] - hlq) * inv_h
    wl = calc_weight(gl, hl, param)
    wr = calc_weight(gr, hr, param)
    if node_bounds is not None:
        lo = node_bounds[:, 0][:, None]
        hi = node_bounds[:, 1][:, None]
        wl = np.clip(wl, lo, hi)
        wr = np.clip(wr, lo, hi)
    g2 = (calc_gain_given_weight(gl, hl, wl, param)
          + calc_gain_given_weight(gr, hr, wr, param)
          - parent_gain[:, None])
    ok = (hl >= param.min_child_weight) & (hr >= param.min_child_weight)
    if mono_bins is not None:
        ok &= ((mono_bins == 0) | ((mono_bins > 0) & (wl <= wr))
               | ((mono_bins < 0) & (wl >= wr)))
    g2 = np.where(ok, g2, -np.inf)
    return np.where(cat_bins[None, :], g2, gain), (glq, hlq)


def evaluate_splits_multi_np(hists: np.ndarray,
                             parents_q: np.ndarray,
                             g_scales: np.ndarray, h_scales: np.ndarray,
                             nids: Sequence[int], cut_ptrs: np.ndarray,
                             param: TrainParam,
                             feature_sets: Optional[List[np.ndarray]] = None,
                             ) -> List["MultiSplitEntry"]:
    """Vector-leaf split evaluation (reference MultiHistEvaluator,
    src/tree/gpu_hist/multi_evaluate_splits.cuh): gain per bin = sum of
    per-target gains; one shared structure, per-target child sums.

    hists: int64 [T, n_nodes, n_bins, 2]; parents_q: int64 [n_nodes, T, 2].
    """
    T, n_nodes, n_bins, _ = hists.shape
    n_features = len(cut_ptrs) - 1
    widths = np.diff(cut_ptrs)
    feat_of_bin = np.repeat(np.arange(n_features), widths)
    seg_start = np.repeat(cut_ptrs[:-1], widths)
    seg_end = np.repeat(cut_ptrs[1:] - 1, widths)

    inv_g = (1.0 / g_scales).reshape(T, 1, 1)
    inv_h = (1.0 / h_scales).reshape(T, 1, 1)

    Gq = hists[:, :, :, 0]
    Hq = hists[:, :, :, 1]
    cumG = np.cumsum(Gq, axis=2)
    cumH = np.cumsum(Hq, axis=2)
    baseG = np.where(seg_start > 0, cumG[:, :, np.maximum(seg_start - 1, 0)], 0)
    baseH = np.where(seg_start > 0, cumH[:, :, np.maximum(seg_start - 1, 0)], 0)
    GLq = cumG - baseG
    HLq = cumH - baseH
    featGq = cumG[:, :, seg_end] - baseG
    featHq = cumH[:, :, seg_end] - baseH
    pq = parents_q.transpose(1, 0, 2)       # [T, n_nodes, 2]
    pgq = pq[:, :, 0][:, :, None]
    phq = pq[:, :, 1][:, :, None]
    missGq = pgq - featGq
    missHq = phq - featHq

    pg = pgq * inv_g
    ph = phq * inv_h
    parent_gain = calc_gain(pg, ph, param).sum(axis=0)  # [n_nodes, 1]

    best = [MultiSplitEntry(nid=int(nid), n_targets=T) for nid in nids]
    for missing_left in (False, True):
        glq = GLq + (missGq if missing_left else 0)
        hlq = HLq + (missHq if missing_left else 0)
        grq = pgq - glq
        hrq = phq - hlq
        gl = glq * inv_g
        hl = hlq * inv_h
        gr = grq * inv_g
        hr = hrq * inv_h
        wl = calc_weight(gl, hl, param)
        wr = calc_weight(gr, hr, param)
        gain = (calc_gain_given_weight(gl, hl, wl, param)
                + calc_gain_given_weight(gr, hr, wr, param)).sum(axis=0) \
            - parent_gain
        ok = ((hl.sum(axis=0) >= param.min_child_weight)
              & (hr.sum(axis=0) >= param.min_child_weight)
              & (hlq.sum(axis=0) > 0) & (hrq.sum(axis=0) > 0))
        gain = np.where(ok, gain, -np.inf)
        for i in range(n_nodes):
            row = gain[i]
            if feature_sets is not None and feature_sets[i] is not None:
                mask = np.zeros(n_bins, dtype=bool)
                for f in feature_sets[i]:
                    mask[cut_ptrs[f]:cut_ptrs[f + 1]] = True
                row = np.where(mask, row, -np.inf)
            b = int(np.argmax(row))
            gval = float(row[b])
            e = best[i]
            if gval > e.gain and np.isfinite(gval):
                e.gain = gval
                e.feature = int(feat_of_bin[b])
                e.split_bin = b
                e.default_left = missing_left
                e.left_q = np.stack([glq[:, i, b], hlq[:, i, b]], axis=1)
                e.right_q = parents_q[i] - e.left_q
    return best


@dataclasses.dataclass
class MultiSplitEntry:
    nid: int
    n_targets: int = 1
    gain: float = -np.inf
    feature: int = -1
    split_bin: int = -1
    default_left: bool = False
    left_q: Optional[np.ndarray] = None    # int64 [T, 2]
    right_q: Optional[np.ndarray] = None
    is_cat: bool = False
    cat_bits: Optional[np.ndarray] = None

    @property
    def is_valid(self) -> bool:
        return self.feature >= 0 and np.isfinite(self.gain) and self.gain > 0
