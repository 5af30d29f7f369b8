"""Split evaluation math — gain, weights, constraints, scan.

Reference behavior: src/tree/split_evaluator.h (TreeEvaluator), param.h
CalcGain/CalcWeight, src/tree/hist/evaluate_splits.h:30 and
src/tree/gpu_hist/evaluate_splits.cu (EvaluateSplitAgent).

This module holds (a) the scalar gain/weight formulas shared by the
numpy oracle, the HIP kernel's verification tests, and leaf-value
computation, and (b) a vectorized numpy evaluator used by the CPU
backend.  Histogram inputs are deterministic int64 fixed-point sums;
gain math runs in float64 after dequantization.
"""
from __future__ import annotations

import dataclasses
from typing import List, Optional, Sequence

import numpy as np

from .params import TrainParam


def threshold_l1(g: np.ndarray, alpha: float) -> np.ndarray:
    return np.sign(g) * np.maximum(np.abs(g) - alpha, 0.0)


def calc_weight(g, h, param: TrainParam):
    """-ThresholdL1(G)/(H+lambda), optionally clipped by max_delta_step."""
    w = -threshold_l1(np.asarray(g, dtype=np.float64), param.reg_alpha) / (
        np.asarray(h, dtype=np.float64) + param.reg_lambda)
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
    left_g: float = 0.0           # dequantized sums
    left_h: float = 0.0
    right_g: float = 0.0
    right_h: float = 0.0
    is_cat: bool = False
    cat_bits: Optional[np.ndarray] = None  # local bin ids going RIGHT

    @property
    def is_valid(self) -> bool:
        return self.feature >= 0 and np.isfinite(self.gain) and self.gain > 0


def evaluate_splits_np(hist: np.ndarray, parent_g: np.ndarray, parent_h: np.ndarray,
                       nids: Sequence[int], cut_ptrs: np.ndarray,
                       param: TrainParam,
                       feature_sets: Optional[List[np.ndarray]] = None,
                       monotone: Optional[np.ndarray] = None,
                       cat_mask: Optional[np.ndarray] = None,
                       node_bounds: Optional[np.ndarray] = None,
                       ) -> List[SplitEntry]:
    """Vectorized split evaluation over [n_nodes, n_bins, 2] float64 hists.

    hist: dequantized float64 (G, H per global bin)
    parent_g/h: [n_nodes] float64 node totals
    feature_sets: per-node allowed features (colsample / interaction)
    monotone: [n_features] in {-1, 0, +1}
    cat_mask: [n_features] bool, True = categorical (one-hot eval)
    node_bounds: [n_nodes, 2] (lower, upper) weight bounds from monotone
      constraint propagation; leaf weights are clipped into these.
    """
    n_nodes, n_bins, _ = hist.shape
    n_features = len(cut_ptrs) - 1
    widths = np.diff(cut_ptrs)
    feat_of_bin = np.repeat(np.arange(n_features), widths)
    seg_start = np.repeat(cut_ptrs[:-1], widths)

    G = hist[:, :, 0]
    H = hist[:, :, 1]
    lam = param.reg_lambda

    cumG = np.cumsum(G, axis=1)
    cumH = np.cumsum(H, axis=1)
    baseG = np.where(seg_start > 0, cumG[:, np.maximum(seg_start - 1, 0)], 0.0)
    baseH = np.where(seg_start > 0, cumH[:, np.maximum(seg_start - 1, 0)], 0.0)
    GL = cumG - baseG   # left sums including bin b (missing right)
    HL = cumH - baseH

    seg_end = np.repeat(cut_ptrs[1:] - 1, widths)
    featG = cumG[:, seg_end] - baseG   # per-bin: total of its feature
    featH = cumH[:, seg_end] - baseH
    missG = parent_g[:, None] - featG
    missH = parent_h[:, None] - featH

    parent_gain = calc_gain(parent_g, parent_h, param)  # [n_nodes]

    best = [SplitEntry(nid=int(nid)) for nid in nids]

    is_last_bin = np.arange(n_bins) == seg_end  # splitting at last bin: right empty

    for missing_left in (False, True):
        gl = GL + (missG if missing_left else 0.0)
        hl = HL + (missH if missing_left else 0.0)
        gr = parent_g[:, None] - gl
        hr = parent_h[:, None] - hl
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
        ok &= ~is_last_bin[None, :]
        if monotone is not None:
            c = monotone[feat_of_bin][None, :]
            ok &= (c == 0) | ((c > 0) & (wl <= wr)) | ((c < 0) & (wl >= wr))
        gain = np.where(ok, gain, -np.inf)
        if cat_mask is not None and cat_mask.any():
            gain = _onehot_cat_gains(gain, G, H, parent_g, parent_h, missG,
                                     missH, feat_of_bin, cat_mask, param,
                                     missing_left, parent_gain, node_bounds,
                                     monotone)
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
                    # one-hot: chosen category (stored set) goes RIGHT
                    glv = float(parent_g[i]) - float(G[i, b]) - (
                        0.0 if missing_left else float(missG[i, b]))
                    hlv = float(parent_h[i]) - float(H[i, b]) - (
                        0.0 if missing_left else float(missH[i, b]))
                else:
                    glv = float(GL[i, b]) + (float(missG[i, b]) if missing_left else 0.0)
                    hlv = float(HL[i, b]) + (float(missH[i, b]) if missing_left else 0.0)
                e.gain = gval
                e.feature = f
                e.split_bin = b
                e.default_left = missing_left
                e.left_g = glv
                e.left_h = hlv
                e.right_g = float(parent_g[i]) - glv
                e.right_h = float(parent_h[i]) - hlv
                e.is_cat = is_cat
                if is_cat:
                    # one-hot: category == this bin goes LEFT; others right
                    e.cat_bits = np.array([b - int(cut_ptrs[f])], dtype=np.int32)
    return best


def _onehot_cat_gains(gain, G, H, parent_g, parent_h, missG, missH,
                      feat_of_bin, cat_mask, param, missing_left,
                      parent_gain, node_bounds, monotone):
    """For categorical features evaluate one-vs-rest per bin instead of the
    cumulative scan (reference: one-hot split when n_cats is small)."""
    cat_bins = cat_mask[feat_of_bin]
    # chosen category goes RIGHT: left = parent - cat - (miss unless missing_left)
    gl = parent_g[:, None] - G - (0.0 if missing_left else missG)
    hl = parent_h[:, None] - H - (0.0 if missing_left else missH)
    gr = parent_g[:, None] - gl
    hr = parent_h[:, None] - hl
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
    if monotone is not None:
        c = monotone[feat_of_bin][None, :]
        ok &= (c == 0) | ((c > 0) & (wl <= wr)) | ((c < 0) & (wl >= wr))
    g2 = np.where(ok, g2, -np.inf)
    return np.where(cat_bins[None, :], g2, gain)
