"""Evaluation metrics.

Reference behavior: src/metric/*.cc (24 registered metrics, SURVEY.md
§2.4).  Distributed-aware: each metric reduces to (numerator,
denominator) partial sums that are summed across workers with
collective.allreduce_sum_scalars (the reference's GlobalRatio pattern,
src/collective/aggregator.h:50); AUC follows src/metric/auc.cc:125.

Metrics are evaluated on *transformed* predictions except where the
reference evaluates on margins (logloss uses probabilities; we receive
whatever Booster.eval passes, which mirrors the reference: objective
pred_transform applied first except for output_margin-style metrics).
"""
from __future__ import annotations

import math
from typing import Callable, Dict, Tuple

import numpy as np

from . import collective

_REGISTRY: Dict[str, Callable] = {}


def register(name: str):
    def deco(fn):
        _REGISTRY[name] = fn
        return fn
    return deco


def create_metric(name: str):
    """Returns fn(preds, info) -> float.  Supports name@param forms."""
    base, param = name, None
    if "@" in name:
        base, param = name.split("@", 1)
    if base not in _REGISTRY:
        raise ValueError(f"unknown metric: {name}; known: {sorted(_REGISTRY)}")
    fn = _REGISTRY[base]

    def call(preds, info):
        import torch as _torch
        if isinstance(preds, _torch.Tensor) and \
                not getattr(fn, "_torch_ok", False):
            preds = preds.detach().cpu().numpy()
        return fn(preds, info, param)

    call.metric_name = name
    return call


def _torch_ok(fn):
    """Marks a metric as device-resident capable: it accepts torch
    tensors (CPU or HIP) and reduces on the tensor's device, syncing
    one scalar pair to the host (reference device AUC / elementwise
    metrics, src/metric/auc.cu:168, elementwise_metric.cu)."""
    fn._torch_ok = True
    return fn


def _yw(preds, info):
    """(labels, preds, weights) as matching fp64 arrays — torch tensors
    on preds.device when preds is a tensor, else numpy."""
    import torch as _torch
    if isinstance(preds, _torch.Tensor):
        dev = preds.device
        p = preds.detach().double()
        if p.dim() == 1:
            p = p.view(-1, 1)
        y = _torch.as_tensor(np.asarray(info.labels, np.float64),
                             device=dev).view(p.shape[0], -1)
        if y.shape[1] == 1 and p.shape[1] > 1:
            y = y.expand(p.shape)
        w = (_torch.as_tensor(np.asarray(info.weights, np.float64),
                              device=dev) if info.weights is not None
             else _torch.ones(p.shape[0], dtype=_torch.float64, device=dev))
        return y, p, w
    y = np.asarray(info.labels, dtype=np.float64).reshape(preds.shape[0], -1)
    p = np.asarray(preds, dtype=np.float64)
    if p.ndim == 1:
        p = p.reshape(-1, 1)
    if y.shape[1] == 1 and p.shape[1] > 1:
        y = np.broadcast_to(y, p.shape)
    w = (np.asarray(info.weights, dtype=np.float64)
         if info.weights is not None else np.ones(p.shape[0]))
    return y, p, w


def _ratio(num: float, den: float) -> float:
    num, den = collective.allreduce_sum_scalars([num, den])
    return num / max(den, 1e-16)


def _wmean(err, w) -> float:
    # err is [n, k]; weight applies per row (torch or numpy — ONE host
    # sync for the two scalars when err lives on the GPU)
    num = float((err * w[:, None]).sum())
    den = float(w.sum()) * err.shape[1]
    return _ratio(num, den)


@register("rmse")
@_torch_ok
def rmse(preds, info, param=None):
    y, p, w = _yw(preds, info)
    return math.sqrt(_wmean((p - y) ** 2, w))


@register("rmsle")
def rmsle(preds, info, param=None):
    y, p, w = _yw(preds, info)
    return math.sqrt(_wmean((np.log1p(np.clip(p, 0, None)) - np.log1p(y)) ** 2, w))


@register("mae")
@_torch_ok
def mae(preds, info, param=None):
    y, p, w = _yw(preds, info)
    return _wmean(abs(p - y), w)


@register("mape")
def mape(preds, info, param=None):
    y, p, w = _yw(preds, info)
    return _wmean(np.abs((y - p) / np.clip(np.abs(y), 1e-16, None)), w)


@register("mphe")
def mphe(preds, info, param=None):
    slope = float(param) if param else 1.0
    y, p, w = _yw(preds, info)
    z = (p - y) / slope
    return _wmean(slope * slope * (np.sqrt(1 + z * z) - 1), w)


@register("logloss")
@_torch_ok
def logloss(preds, info, param=None):
    import torch as _torch
    y, p, w = _yw(preds, info)
    eps = 1e-16
    if isinstance(p, _torch.Tensor):
        p = p.clamp(eps, 1 - eps)
        ll = -(y * _torch.log(p) + (1 - y) * _torch.log(1 - p))
    else:
        p = np.clip(p, eps, 1 - eps)
        ll = -(y * np.log(p) + (1 - y) * np.log(1 - p))
    return _wmean(ll, w)


@register("error")
@_torch_ok
def error(preds, info, param=None):
    import torch as _torch
    t = float(param) if param else 0.5
    y, p, w = _yw(preds, info)
    if isinstance(p, _torch.Tensor):
        wrong = _torch.where(p > t, y != 1.0, y != 0.0).double()
    else:
        wrong = np.where(p > t, y != 1.0, y != 0.0).astype(np.float64)
    return _wmean(wrong, w)


@register("merror")
@_torch_ok
def merror(preds, info, param=None):
    import torch as _torch
    if isinstance(preds, _torch.Tensor):
        dev = preds.device
        y = _torch.as_tensor(np.asarray(info.labels, np.int64),
                             device=dev).view(-1)
        p = preds.detach().double()
        cls = p.argmax(dim=1) if p.dim() == 2 and p.shape[1] > 1 \
            else p.view(-1)
        w = (_torch.as_tensor(np.asarray(info.weights, np.float64),
                              device=dev) if info.weights is not None
             else _torch.ones(y.shape[0], dtype=_torch.float64, device=dev))
        return _ratio(float(((cls != y) * w).sum()), float(w.sum()))
    y = np.asarray(info.labels, dtype=np.int64).reshape(-1)
    p = np.asarray(preds, dtype=np.float64)
    cls = p.argmax(axis=1) if p.ndim == 2 and p.shape[1] > 1 else p.reshape(-1)
    w = (np.asarray(info.weights, dtype=np.float64)
         if info.weights is not None else np.ones(y.shape[0]))
    return _ratio(float(((cls != y) * w).sum()), float(w.sum()))


@register("mlogloss")
@_torch_ok
def mlogloss(preds, info, param=None):
    import torch as _torch
    if isinstance(preds, _torch.Tensor):
        dev = preds.device
        y = _torch.as_tensor(np.asarray(info.labels, np.int64),
                             device=dev).view(-1)
        p = preds.detach().double().clamp(1e-16, 1 - 1e-16)
        w = (_torch.as_tensor(np.asarray(info.weights, np.float64),
                              device=dev) if info.weights is not None
             else _torch.ones(y.shape[0], dtype=_torch.float64, device=dev))
        ll = -_torch.log(p[_torch.arange(y.numel(), device=dev), y])
        return _ratio(float((ll * w).sum()), float(w.sum()))
    y = np.asarray(info.labels, dtype=np.int64).reshape(-1)
    p = np.asarray(preds, dtype=np.float64)
    eps = 1e-16
    p = np.clip(p, eps, 1 - eps)
    w = (np.asarray(info.weights, dtype=np.float64)
         if info.weights is not None else np.ones(y.shape[0]))
    ll = -np.log(p[np.arange(y.size), y])
    return _ratio(float((ll * w).sum()), float(w.sum()))


@register("poisson-nloglik")
def poisson_nloglik(preds, info, param=None):
    from scipy.special import gammaln
    y, p, w = _yw(preds, info)
    p = np.clip(p, 1e-16, None)
    nll = p - y * np.log(p) + gammaln(y + 1.0)
    return _wmean(nll, w)


@register("gamma-deviance")
def gamma_deviance(preds, info, param=None):
    y, p, w = _yw(preds, info)
    eps = 1e-16
    ratio = np.clip(y, eps, None) / np.clip(p, eps, None)
    dev = 2 * (-np.log(ratio) + ratio - 1)
    num = float((dev * w[:, None]).sum())
    den = float(w.sum() * dev.shape[1])
    num, den = collective.allreduce_sum_scalars([num, den])
    return num / max(den, 1e-16)


@register("gamma-nloglik")
def gamma_nloglik(preds, info, param=None):
    y, p, w = _yw(preds, info)
    psi = 1.0
    theta = -1.0 / np.clip(p, 1e-16, None)
    a = psi
    b = -np.log(-theta)
    nll = -((y * theta - b) / a + (1 / psi) * np.log(np.clip(y, 1e-16, None) / psi)
            - np.log(np.clip(y, 1e-16, None)) - math.lgamma(1 / psi))
    return _wmean(nll, w)


@register("tweedie-nloglik")
def tweedie_nloglik(preds, info, param=None):
    rho = float(param) if param else 1.5
    y, p, w = _yw(preds, info)
    p = np.clip(p, 1e-16, None)
    nll = -y * np.power(p, 1 - rho) / (1 - rho) + np.power(p, 2 - rho) / (2 - rho)
    return _wmean(nll, w)


@register("quantile")
def quantile_loss(preds, info, param=None):
    alpha = float(param) if param else 0.5
    y, p, w = _yw(preds, info)
    d = y - p
    loss = np.where(d >= 0, alpha * d, (alpha - 1) * d)
    return _wmean(loss, w)


@register("expectile")
def expectile_loss(preds, info, param=None):
    alpha = float(param) if param else 0.5
    y, p, w = _yw(preds, info)
    d = p - y
    wgt = np.where(d >= 0, 1 - alpha, alpha)
    return _wmean(wgt * d * d, w)


@register("auc")
@_torch_ok
def auc(preds, info, param=None):
    import torch as _torch
    if isinstance(preds, _torch.Tensor):
        p = preds.detach().double()
        if not (p.dim() == 2 and p.shape[1] > 1) and (
                info.group_ptr is None or len(info.group_ptr) <= 2):
            dev = p.device
            y = _torch.as_tensor(np.asarray(info.labels, np.float64),
                                 device=dev).view(-1)
            w = (_torch.as_tensor(np.asarray(info.weights, np.float64),
                                  device=dev) if info.weights is not None
                 else _torch.ones(y.shape[0], dtype=_torch.float64,
                                  device=dev))
            a, valid = _binary_auc_t(p.view(-1), y, w)
            s_, v_ = collective.allreduce_sum_scalars([a * valid, valid])
            if v_ == 0:
                return _degenerate_auc()
            return s_ / v_
        preds = p.cpu().numpy()
    p = np.asarray(preds, dtype=np.float64)
    if p.ndim == 2 and p.shape[1] > 1:
        return _multi_auc(p, info)
    if info.group_ptr is not None and len(info.group_ptr) > 2:
        return _ranking_auc(p.reshape(-1), info)
    y = np.asarray(info.labels, dtype=np.float64).reshape(-1)
    w = (np.asarray(info.weights, dtype=np.float64)
         if info.weights is not None else np.ones(y.shape[0]))
    a, valid = _binary_auc(p.reshape(-1), y, w)
    # distributed: weighted mean of per-worker AUC (reference auc.cc:125)
    s, v = collective.allreduce_sum_scalars([a * valid, valid])
    if v == 0:
        return _degenerate_auc()
    return s / v


def _degenerate_auc() -> float:
    """reference auc.cc:351: AUC over a single-class dataset is NaN."""
    import warnings
    warnings.warn("Dataset is empty, or contains only positive or "
                  "negative samples.")
    return float("nan")


def _binary_auc_t(p, y, w) -> Tuple[float, float]:
    """Device-resident weighted ROC AUC: sort + cumsum + tie merge, one
    host sync (reference BinaryROCAUC, src/metric/auc.cu:172)."""
    import torch as _torch
    order = _torch.argsort(-p, stable=True)
    p, y, w = p[order], y[order], w[order]
    pos = (w * y).sum()
    neg = (w * (1 - y)).sum()
    tp = _torch.cumsum(w * y, 0)
    fp = _torch.cumsum(w * (1 - y), 0)
    # keep the last index of each distinct prediction value
    n = p.numel()
    keep = _torch.ones(n, dtype=_torch.bool, device=p.device)
    if n > 1:
        keep[:-1] = p[:-1] != p[1:]
    tp, fp = tp[keep], fp[keep]
    z = _torch.zeros(1, dtype=tp.dtype, device=tp.device)
    tp0 = _torch.cat([z, tp[:-1]])
    fp0 = _torch.cat([z, fp[:-1]])
    area = ((fp - fp0) * (tp + tp0) * 0.5).sum()
    stats = _torch.stack([pos, neg, area]).cpu()  # ONE sync
    pos_, neg_, area_ = (float(stats[0]), float(stats[1]), float(stats[2]))
    if pos_ == 0 or neg_ == 0:
        return 0.5, 0.0
    return area_ / (pos_ * neg_), 1.0


def _binary_auc(p, y, w) -> Tuple[float, float]:
    order = np.argsort(-p, kind="stable")
    p, y, w = p[order], y[order], w[order]
    pos = float((w * y).sum())
    neg = float((w * (1 - y)).sum())
    if pos == 0 or neg == 0:
        return 0.5, 0.0
    tp = np.cumsum(w * y)
    fp = np.cumsum(w * (1 - y))
    # merge ties: keep last index of each distinct prediction
    distinct = np.nonzero(np.diff(p))[0]
    idx = np.concatenate([distinct, [p.size - 1]])
    tp, fp = tp[idx], fp[idx]
    tp0 = np.concatenate([[0.0], tp[:-1]])
    fp0 = np.concatenate([[0.0], fp[:-1]])
    area = float(np.sum((fp - fp0) * (tp + tp0) * 0.5))
    return area / (pos * neg), 1.0


def _multi_auc(p, info) -> float:
    y = np.asarray(info.labels, dtype=np.int64).reshape(-1)
    w = (np.asarray(info.weights, dtype=np.float64)
         if info.weights is not None else np.ones(y.shape[0]))
    n_class = p.shape[1]
    aucs = []
    for c in range(n_class):
        a, valid = _binary_auc(p[:, c], (y == c).astype(np.float64), w)
        aucs.append(a if valid else 0.5)
    return float(np.mean(aucs))


def _ranking_auc(p, info) -> float:
    y = np.asarray(info.labels, dtype=np.float64).reshape(-1)
    gp = info.group_ptr
    aucs, valid = [], 0.0
    for i in range(len(gp) - 1):
        s, e = int(gp[i]), int(gp[i + 1])
        yy = y[s:e]
        if yy.min() == yy.max():
            continue
        a, v = _binary_auc(p[s:e], (yy > yy.min()).astype(np.float64),
                           np.ones(e - s))
        if v:
            aucs.append(a)
            valid += 1
    sa, sv = collective.allreduce_sum_scalars([float(np.sum(aucs)), valid])
    return sa / max(sv, 1e-16) if sv else 0.5


@register("aucpr")
def aucpr(preds, info, param=None):
    p = np.asarray(preds, dtype=np.float64).reshape(-1)
    y = np.asarray(info.labels, dtype=np.float64).reshape(-1)
    w = (np.asarray(info.weights, dtype=np.float64)
         if info.weights is not None else np.ones(y.shape[0]))
    order = np.argsort(-p, kind="stable")
    y, w = y[order], w[order]
    tp = np.cumsum(w * y)
    fp = np.cumsum(w * (1 - y))
    total_pos = tp[-1]
    if total_pos == 0:
        return _degenerate_auc()
    prec = tp / np.clip(tp + fp, 1e-16, None)
    rec = tp / total_pos
    rec0 = np.concatenate([[0.0], rec[:-1]])
    area = float(np.sum((rec - rec0) * prec))
    s, v = collective.allreduce_sum_scalars([area, 1.0])
    return s / v


# -- ranking metrics ---------------------------------------------------------

def _groups(info, n):
    gp = info.group_ptr
    if gp is None:
        return np.array([0, n], dtype=np.int64)
    return np.asarray(gp, dtype=np.int64)


@register("ndcg")
def ndcg(preds, info, param=None):
    topn = int(param.rstrip("-")) if param else 2 ** 31 - 1
    minus = bool(param and param.endswith("-"))
    p = np.asarray(preds, dtype=np.float64).reshape(-1)
    y = np.asarray(info.labels, dtype=np.float64).reshape(-1)
    gp = _groups(info, p.size)
    scores, nvalid = [], 0.0
    for i in range(len(gp) - 1):
        s, e = int(gp[i]), int(gp[i + 1])
        yy, pp = y[s:e], p[s:e]
        k = min(topn, e - s)
        if yy.sum() == 0:
            # reference rank_metric.cc:385: no-relevance groups score
            # minus ? 0 : 1 and STAY in the denominator
            scores.append(0.0 if minus else 1.0)
            nvalid += 1
            continue
        order = np.argsort(-pp, kind="stable")
        gains = (2.0 ** yy - 1.0)
        disc = 1.0 / np.log2(np.arange(2, e - s + 2))
        dcg = float((gains[order] * disc)[:k].sum())
        ideal = float((np.sort(gains)[::-1] * disc)[:k].sum())
        scores.append(dcg / ideal if ideal > 0 else (0.0 if minus else 1.0))
        nvalid += 1
    sa, sv = collective.allreduce_sum_scalars([float(np.sum(scores)), nvalid])
    return sa / max(sv, 1e-16)


@register("map")
def map_metric(preds, info, param=None):
    topn = int(param.rstrip("-")) if param else 2 ** 31 - 1
    minus = bool(param and param.endswith("-"))
    p = np.asarray(preds, dtype=np.float64).reshape(-1)
    y = np.asarray(info.labels, dtype=np.float64).reshape(-1)
    gp = _groups(info, p.size)
    scores, nvalid = [], 0.0
    for i in range(len(gp) - 1):
        s, e = int(gp[i]), int(gp[i + 1])
        order = np.argsort(-p[s:e], kind="stable")
        rel = (y[s:e][order] > 0).astype(np.float64)
        k = min(topn, e - s)
        hits = np.cumsum(rel)
        prec_at = rel[:k] * (hits[:k] / np.arange(1, k + 1))
        npos = rel.sum()
        # reference rank_metric.cc:446: no-relevance groups score
        # minus ? 0 : 1 and stay in the denominator
        scores.append(float(prec_at.sum() / npos) if npos > 0
                      else (0.0 if minus else 1.0))
        nvalid += 1
    sa, sv = collective.allreduce_sum_scalars([float(np.sum(scores)), nvalid])
    return sa / max(sv, 1e-16)


@register("pre")
def precision_at(preds, info, param=None):
    topn = int(param) if param else 2 ** 31 - 1
    p = np.asarray(preds, dtype=np.float64).reshape(-1)
    y = np.asarray(info.labels, dtype=np.float64).reshape(-1)
    gp = _groups(info, p.size)
    num, den = 0.0, 0.0
    for i in range(len(gp) - 1):
        s, e = int(gp[i]), int(gp[i + 1])
        k = min(topn, e - s)
        order = np.argsort(-p[s:e], kind="stable")[:k]
        num += float((y[s:e][order] > 0).sum())
        den += k
    return _ratio(num, den)


@register("ams")
def ams(preds, info, param=None):
    ratio = float(param) if param else 0.5
    p = np.asarray(preds, dtype=np.float64).reshape(-1)
    y = np.asarray(info.labels, dtype=np.float64).reshape(-1)
    w = (np.asarray(info.weights, dtype=np.float64)
         if info.weights is not None else np.ones(y.shape[0]))
    order = np.argsort(-p, kind="stable")
    y, w = y[order], w[order]
    ntop = max(1, int(ratio * p.size))
    s = float((w[:ntop] * y[:ntop]).sum())
    b = float((w[:ntop] * (1 - y[:ntop])).sum())
    br = 10.0
    return math.sqrt(2 * ((s + b + br) * math.log(1 + s / (b + br)) - s))


@register("cox-nloglik")
def cox_nloglik(preds, info, param=None):
    p = np.asarray(preds, dtype=np.float64).reshape(-1)
    y = np.asarray(info.labels, dtype=np.float64).reshape(-1)
    order = np.argsort(np.abs(y))
    p = p[order]
    yy = y[order]
    exp_p = np.exp(p - p.max())
    rev_cum = np.cumsum(exp_p[::-1])[::-1]
    event = yy > 0
    nll = -(np.log(np.clip(exp_p[event], 1e-30, None))
            - np.log(np.clip(rev_cum[event], 1e-30, None))).sum()
    return _ratio(float(nll), float(event.sum()))


@register("aft-nloglik")
def aft_nloglik(preds, info, param=None):
    from .objectives import AFT
    import torch
    obj = AFT({})
    p = torch.as_tensor(np.asarray(preds, np.float32).reshape(-1))
    lo, hi = obj._bounds(info, p.device)
    s = obj.sigma
    eps = 1e-12
    logp = torch.log(torch.clamp(p, min=eps))
    z_lo = (torch.log(torch.clamp(lo, min=eps)) - logp) / s
    z_hi = (torch.log(torch.clamp(hi, min=eps)) - logp) / s
    pdf_l, cdf_l, _, _ = obj._pdf_cdf(z_lo)
    pdf_u, cdf_u, _, _ = obj._pdf_cdf(z_hi)
    uncensored = torch.isfinite(hi) & (lo == hi)
    cdf_u = torch.where(torch.isfinite(hi), cdf_u, torch.ones_like(cdf_u))
    cdf_l = torch.where(lo > 0, cdf_l, torch.zeros_like(cdf_l))
    lik_unc = pdf_l / (s * torch.clamp(lo, min=eps))
    lik_cen = torch.clamp(cdf_u - cdf_l, min=eps)
    nll = -torch.log(torch.clamp(torch.where(uncensored, lik_unc, lik_cen),
                                 min=1e-30))
    w = (np.asarray(info.weights, dtype=np.float64)
         if info.weights is not None else np.ones(nll.shape[0]))
    return _ratio(float((nll.numpy() * w).sum()), float(w.sum()))


@register("interval-regression-accuracy")
def interval_accuracy(preds, info, param=None):
    p = np.asarray(preds, dtype=np.float64).reshape(-1)
    lo = (np.asarray(info.label_lower_bound, np.float64)
          if info.label_lower_bound is not None
          else np.asarray(info.labels, np.float64).reshape(-1))
    hi = (np.asarray(info.label_upper_bound, np.float64)
          if info.label_upper_bound is not None else lo)
    ok = ((p >= lo) | ~np.isfinite(lo)) & ((p <= hi) | ~np.isfinite(hi))
    return _ratio(float(ok.sum()), float(p.size))
