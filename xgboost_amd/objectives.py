"""Objective functions (losses).

Reference behavior: src/objective/*.cc (22 registered objectives, see
SURVEY.md §2.4), interface include/xgboost/objective.h:28.

MI355X-native design: gradients are computed with torch elementwise ops
on whatever device holds the predictions — this is the analog of the
reference's common::Transform dual-dispatch (src/common/transform.h:62);
on ROCm these lower to HIP elementwise kernels, on CPU to vectorized
loops.  Each objective provides:
  get_gradient(preds, info, it) -> (grad, hess)   [n, n_out]
  pred_transform(margin) -> user-facing prediction
  prob_to_margin(base_score) -> margin-space intercept
  init_estimation(info) -> base_score (stored space), one Newton step
  (reference FitStump, src/tree/fit_stump.cc:92)
"""
from __future__ import annotations

import math
from typing import Callable, Dict, Optional, Tuple

import numpy as np
import torch

_REGISTRY: Dict[str, Callable[..., "Objective"]] = {}


def register(name: str):
    def deco(cls):
        _REGISTRY[name] = cls
        cls.name = name
        return cls
    return deco


def create_objective(name: str, params: Optional[dict] = None) -> "Objective":
    params = params or {}
    if name == "reg:linear":  # deprecated alias
        name = "reg:squarederror"
    if name == "binary:logitraw":
        pass
    if name not in _REGISTRY:
        raise ValueError(f"unknown objective: {name}; known: {sorted(_REGISTRY)}")
    return _REGISTRY[name](params)


def _weights(info, t: torch.Tensor) -> Optional[torch.Tensor]:
    if info.weights is None:
        return None
    return torch.as_tensor(info.weights, dtype=torch.float32, device=t.device)


def _labels(info, t: torch.Tensor) -> torch.Tensor:
    return torch.as_tensor(info.labels, dtype=torch.float32, device=t.device)




def _parse_alphas(alpha):
    """Accept float, list, numpy array, or the stringified forms the
    config writer emits ("0.5" / "[0.1, 0.5]")."""
    if isinstance(alpha, str):
        alpha = alpha.strip()
        if alpha.startswith("["):
            return [float(x) for x in alpha.strip("[] ").split(",") if x.strip()]
        return [float(alpha)]
    if isinstance(alpha, (list, tuple, np.ndarray)):
        return [float(a) for a in alpha]
    return [float(alpha)]


class Objective:
    name = "base"
    n_class = 1
    default_metric = "rmse"
    task = "regression"

    def __init__(self, params: Optional[dict] = None):
        self.params = params or {}

    def n_outputs(self, n_targets: int = 1) -> int:
        return n_targets

    def get_gradient(self, preds: torch.Tensor, info, iteration: int
                     ) -> Tuple[torch.Tensor, torch.Tensor]:
        raise NotImplementedError

    def pred_transform(self, margin: torch.Tensor) -> torch.Tensor:
        return margin

    def eval_transform(self, margin: torch.Tensor) -> torch.Tensor:
        """Transform used for metric evaluation (reference
        ObjFunction::EvalTransform — differs from PredTransform for
        multi:softmax, which argmaxes for prediction but evaluates
        mlogloss on probabilities)."""
        return self.pred_transform(margin)

    def prob_to_margin(self, base_score: float) -> float:
        return base_score

    def init_estimation(self, info) -> float:
        """One Newton step at margin 0 (reference FitStump + PredTransform).
        Multi-target labels produce the mean of per-target estimates
        (scalar base_score, like the reference's ParamArray mean).

        Distributed: the G/H sums are allreduced so the intercept is the
        GLOBAL -sum(G)/sum(H), matching the reference's GlobalSum inside
        FitStump (src/tree/fit_stump.cu:46-49) — not a local estimate."""
        from . import collective
        n = info.num_row
        n_out = 1
        if info.labels is not None and info.labels.ndim == 2:
            n_out = info.labels.shape[1]
        preds = torch.zeros((n, n_out), dtype=torch.float32)
        g, h = self.get_gradient(preds, info, 0)
        gs = g.double().sum(dim=0)
        hs = h.double().sum(dim=0)
        if collective.is_distributed():
            k = gs.numel()
            glob = collective.allreduce_sum_scalars(
                gs.cpu().tolist() + hs.cpu().tolist())
            gs = torch.tensor(glob[:k], dtype=torch.float64)
            hs = torch.tensor(glob[k:], dtype=torch.float64)
        hs = hs.clamp(min=1e-16)
        margin = float((-gs / hs).mean())
        out = self.pred_transform(torch.tensor([margin])).item()
        return out

    def save_config(self) -> dict:
        return {"name": self.name}

    def _apply_weight(self, g, h, info):
        w = _weights(info, g)
        if w is not None:
            w = w.view(-1, *([1] * (g.dim() - 1)))
            g = g * w
            h = h * w
        return g, h


# ---------------------------------------------------------------------------
# regression


@register("reg:squarederror")
class SquaredError(Objective):
    default_metric = "rmse"

    def get_gradient(self, preds, info, it):
        y = _labels(info, preds).view(preds.shape)
        g = preds - y
        h = torch.ones_like(preds)
        return self._apply_weight(g, h, info)


@register("reg:squaredlogerror")
class SquaredLogError(Objective):
    default_metric = "rmsle"

    def get_gradient(self, preds, info, it):
        y = _labels(info, preds).view(preds.shape)
        p1 = preds + 1.0
        lg = torch.log1p(preds) - torch.log1p(y)
        g = lg / p1
        h = torch.clamp((1.0 - lg) / (p1 * p1), min=1e-6)
        return self._apply_weight(g, h, info)


@register("reg:pseudohubererror")
class PseudoHuber(Objective):
    default_metric = "mphe"

    def get_gradient(self, preds, info, it):
        slope = float(self.params.get("huber_slope", 1.0))
        y = _labels(info, preds).view(preds.shape)
        z = preds - y
        scale = 1.0 + (z / slope) ** 2
        sq = torch.sqrt(scale)
        g = z / sq
        h = 1.0 / (scale * sq)
        return self._apply_weight(g, h, info)

    def save_config(self):
        return {"name": self.name,
                "pseudo_huber_param": {"huber_slope": str(self.params.get("huber_slope", 1.0))}}


@register("reg:absoluteerror")
class AbsoluteError(Objective):
    """MAE with smoothed curvature (reference absolute_error_obj.cc:44-90):
    scale = (Σ w·sqrt(|r|) / Σ w)²; g = w·r·scale/hypot(scale, r);
    h = w·scale/hypot(scale, r).  Scale statistics are global sums
    (distributed: allreduced via collective.allreduce_sum)."""
    default_metric = "mae"

    def get_gradient(self, preds, info, it):
        from . import collective
        y = _labels(info, preds).view(preds.shape)
        r = y - preds
        w = _weights(info, preds)
        if w is None:
            w_sum = float(r.numel())
            sw = torch.sqrt(torch.abs(r)).double().sum()
        else:
            wv = w.view(-1, *([1] * (r.dim() - 1)))
            w_sum = float(wv.double().sum()) * r.shape[-1]
            sw = (wv * torch.sqrt(torch.abs(r))).double().sum()
        stats = collective.allreduce_sum_scalars([float(sw), w_sum])
        scale = (stats[0] / max(stats[1], 1e-16)) ** 2
        scale = max(scale, 1e-16)
        hyp = torch.sqrt(r * r + scale * scale)
        g = -r * scale / hyp
        h = torch.full_like(preds, scale) / hyp
        if w is not None:
            wv = w.view(-1, *([1] * (g.dim() - 1)))
            g, h = g * wv, h * wv
        return g, h

    def init_estimation(self, info) -> float:
        y = np.asarray(info.labels, dtype=np.float64).reshape(info.num_row, -1)
        if info.weights is not None:
            return float(_weighted_quantile(y[:, 0], info.weights, 0.5))
        return float(np.median(y[:, 0]))


@register("reg:quantileerror")
class QuantileError(Objective):
    default_metric = "quantile"

    def __init__(self, params=None):
        super().__init__(params)
        alpha = self.params.get("quantile_alpha", 0.5)
        self.alphas = _parse_alphas(alpha)
        for a in self.alphas:
            if not 0.0 < a < 1.0:
                raise ValueError("quantile_alpha must be in (0, 1)")

    def n_outputs(self, n_targets: int = 1) -> int:
        return len(self.alphas)

    def get_gradient(self, preds, info, it):
        y = _labels(info, preds)
        if y.dim() == 1 or y.shape[-1] == 1:
            y = y.view(-1, 1).expand(-1, len(self.alphas))
        d = preds - y  # >0 means over-prediction
        alphas = torch.tensor(self.alphas, device=preds.device).view(1, -1)
        g = torch.where(d >= 0, 1.0 - alphas, -alphas)
        h = torch.ones_like(preds)
        return self._apply_weight(g, h, info)

    def init_estimation(self, info) -> float:
        y = np.asarray(info.labels, dtype=np.float64).reshape(-1)
        w = info.weights
        vals = [(_weighted_quantile(y, w, a) if w is not None
                 else np.quantile(y, a)) for a in self.alphas]
        return float(np.mean(vals))

    def save_config(self):
        return {"name": self.name,
                "quantile_loss_param": {"quantile_alpha": str(self.alphas)}}


@register("reg:expectileerror")
class ExpectileError(Objective):
    default_metric = "expectile"

    def __init__(self, params=None):
        super().__init__(params)
        alpha = self.params.get("expectile_alpha", 0.5)
        self.alphas = _parse_alphas(alpha)

    def n_outputs(self, n_targets: int = 1) -> int:
        return len(self.alphas)

    def get_gradient(self, preds, info, it):
        y = _labels(info, preds)
        if y.dim() == 1 or y.shape[-1] == 1:
            y = y.view(-1, 1).expand(-1, len(self.alphas))
        d = preds - y
        alphas = torch.tensor(self.alphas, device=preds.device).view(1, -1)
        wgt = torch.where(d >= 0, 1.0 - alphas, alphas)
        g = 2.0 * wgt * d
        h = 2.0 * wgt
        return self._apply_weight(g, h, info)


@register("reg:gamma")
class GammaDeviance(Objective):
    """Gamma regression with log link (reference regression_obj:
    GammaRegression)."""
    default_metric = "gamma-nloglik"

    def get_gradient(self, preds, info, it):
        y = _labels(info, preds).view(preds.shape)
        e = torch.exp(-preds)
        g = 1.0 - y * e
        h = y * e
        return self._apply_weight(g, h, info)

    def pred_transform(self, margin):
        return torch.exp(margin)

    def prob_to_margin(self, base_score):
        return math.log(max(base_score, 1e-16))


@register("count:poisson")
class Poisson(Objective):
    default_metric = "poisson-nloglik"

    def get_gradient(self, preds, info, it):
        max_delta = float(self.params.get("max_delta_step", 0.7))
        y = _labels(info, preds).view(preds.shape)
        e = torch.exp(preds)
        g = e - y
        h = torch.exp(preds + max_delta)
        return self._apply_weight(g, h, info)

    def pred_transform(self, margin):
        return torch.exp(margin)

    def prob_to_margin(self, base_score):
        return math.log(max(base_score, 1e-16))


@register("reg:tweedie")
class Tweedie(Objective):
    def __init__(self, params=None):
        super().__init__(params)
        self.rho = float(self.params.get("tweedie_variance_power", 1.5))
        if not 1.0 <= self.rho < 2.0:
            raise ValueError("tweedie_variance_power must be in [1, 2)")

    @property
    def default_metric(self):
        return f"tweedie-nloglik@{self.rho}"

    def get_gradient(self, preds, info, it):
        y = _labels(info, preds).view(preds.shape)
        rho = self.rho
        g = -y * torch.exp((1 - rho) * preds) + torch.exp((2 - rho) * preds)
        h = (-y * (1 - rho) * torch.exp((1 - rho) * preds)
             + (2 - rho) * torch.exp((2 - rho) * preds))
        return self._apply_weight(g, h, info)

    def pred_transform(self, margin):
        return torch.exp(margin)

    def prob_to_margin(self, base_score):
        return math.log(max(base_score, 1e-16))

    def save_config(self):
        return {"name": self.name,
                "tweedie_regression_param": {"tweedie_variance_power": str(self.rho)}}


# ---------------------------------------------------------------------------
# binary classification


@register("binary:logistic")
class BinaryLogistic(Objective):
    default_metric = "logloss"
    task = "binary"

    def get_gradient(self, preds, info, it):
        y = _labels(info, preds).view(preds.shape)
        p = torch.sigmoid(preds)
        g = p - y
        h = torch.clamp(p * (1.0 - p), min=1e-16)
        g, h = self._scale_pos(g, h, y)
        return self._apply_weight(g, h, info)

    def _scale_pos(self, g, h, y):
        spw = float(self.params.get("scale_pos_weight", 1.0))
        if spw != 1.0:
            w = torch.where(y == 1.0, spw, 1.0)
            g, h = g * w, h * w
        return g, h

    def pred_transform(self, margin):
        return torch.sigmoid(margin)

    def prob_to_margin(self, base_score):
        if not 0.0 < base_score < 1.0:
            raise ValueError("base_score must be in (0,1) for logistic")
        return math.log(base_score / (1.0 - base_score))


@register("binary:logitraw")
class BinaryLogitRaw(BinaryLogistic):
    default_metric = "logloss"

    def pred_transform(self, margin):
        return margin

    def init_estimation(self, info) -> float:
        return 0.0

    def prob_to_margin(self, base_score):
        return base_score


@register("binary:hinge")
class BinaryHinge(Objective):
    default_metric = "error"
    task = "binary"

    def get_gradient(self, preds, info, it):
        y = _labels(info, preds).view(preds.shape) * 2.0 - 1.0  # {0,1}->{-1,1}
        margin = preds * y
        g = torch.where(margin < 1.0, -y, torch.zeros_like(y))
        h = torch.where(margin < 1.0, torch.ones_like(y),
                        torch.full_like(y, 1e-16))
        return self._apply_weight(g, h, info)

    def pred_transform(self, margin):
        return (margin > 0.0).to(margin.dtype)

    def init_estimation(self, info) -> float:
        return 0.0


# ---------------------------------------------------------------------------
# multiclass


@register("multi:softmax")
class SoftmaxMulti(Objective):
    default_metric = "mlogloss"
    task = "multiclass"

    def __init__(self, params=None):
        super().__init__(params)
        self.num_class = int(self.params.get("num_class", 0))
        if self.num_class < 2:
            raise ValueError("multi:softmax requires num_class >= 2")
        self.output_prob = False

    def n_outputs(self, n_targets: int = 1) -> int:
        return self.num_class

    def get_gradient(self, preds, info, it):
        y = _labels(info, preds).view(-1).long()
        p = torch.softmax(preds, dim=1)
        g = p.clone()
        g[torch.arange(g.shape[0], device=g.device), y] -= 1.0
        h = torch.clamp(2.0 * p * (1.0 - p), min=1e-16)
        return self._apply_weight(g, h, info)

    def pred_transform(self, margin):
        if self.output_prob:
            return torch.softmax(margin, dim=1)
        return torch.argmax(margin, dim=1).to(torch.float32)

    def eval_transform(self, margin):
        return torch.softmax(margin, dim=1)

    def init_estimation(self, info) -> float:
        return 0.5

    def prob_to_margin(self, base_score):
        return base_score  # reference keeps raw margins for multiclass

    def save_config(self):
        return {"name": self.name,
                "softmax_multiclass_param": {"num_class": str(self.num_class)}}


@register("multi:softprob")
class SoftprobMulti(SoftmaxMulti):
    def __init__(self, params=None):
        super().__init__(params)
        self.output_prob = True


# ---------------------------------------------------------------------------
# survival


@register("survival:cox")
class CoxPH(Objective):
    default_metric = "cox-nloglik"

    def get_gradient(self, preds, info, it):
        # labels: abs = time, sign: positive = event, negative = censored
        y = _labels(info, preds).view(-1)
        p = preds.view(-1)
        order = torch.argsort(torch.abs(y))
        exp_p = torch.exp(p - p.max())
        # cumulative sum of exp(pred) over risk sets (sorted by time desc)
        sorted_exp = exp_p[order]
        rev_cum = torch.flip(torch.cumsum(torch.flip(sorted_exp, [0]), 0), [0])
        # for each event, accumulate 1/risk over events with time <= t_i
        event = (y[order] > 0).to(torch.float32)
        inv_risk = torch.where(rev_cum > 0, event / rev_cum, torch.zeros_like(rev_cum))
        cum_inv = torch.cumsum(inv_risk, 0)
        cum_inv_sq = torch.cumsum(inv_risk / torch.clamp(rev_cum, min=1e-16), 0)
        g_s = sorted_exp * cum_inv - event
        h_s = sorted_exp * cum_inv - (sorted_exp ** 2) * cum_inv_sq
        g = torch.empty_like(g_s)
        h = torch.empty_like(h_s)
        g[order] = g_s
        h[order] = torch.clamp(h_s, min=1e-16)
        return self._apply_weight(g.view(preds.shape), h.view(preds.shape), info)

    def pred_transform(self, margin):
        return torch.exp(margin)

    def init_estimation(self, info) -> float:
        return 1.0

    def prob_to_margin(self, base_score):
        return math.log(max(base_score, 1e-16))


@register("survival:aft")
class AFT(Objective):
    """Accelerated failure time (reference aft_obj.cc, survival_util.h).
    Supports uncensored / left / right / interval censoring via
    label_lower_bound / label_upper_bound."""
    default_metric = "aft-nloglik"

    def __init__(self, params=None):
        super().__init__(params)
        self.dist = str(self.params.get("aft_loss_distribution", "normal"))
        self.sigma = float(self.params.get("aft_loss_distribution_scale", 1.0))
        if self.dist not in ("normal", "logistic", "extreme"):
            raise ValueError(f"unknown aft_loss_distribution: {self.dist}")

    def _bounds(self, info, device):
        lo = info.label_lower_bound
        hi = info.label_upper_bound
        if lo is None:
            lo = info.labels.reshape(-1)
        if hi is None:
            hi = lo
        lo_t = torch.as_tensor(np.asarray(lo, np.float32), device=device)
        hi_t = torch.as_tensor(np.asarray(hi, np.float32), device=device)
        return lo_t, hi_t

    def _pdf_cdf(self, z):
        """Returns (pdf, cdf, dlogpdf/dz, -d2logpdf/dz2) — closed forms
        per distribution (reference probability_distribution.h:35)."""
        if self.dist == "normal":
            pdf = torch.exp(-0.5 * z * z) / math.sqrt(2 * math.pi)
            cdf = 0.5 * (1 + torch.erf(z / math.sqrt(2)))
            grad_pdf = -z
            curv = torch.ones_like(z)
        elif self.dist == "logistic":
            ez = torch.exp(z)
            pdf = ez / (1 + ez) ** 2
            cdf = ez / (1 + ez)
            grad_pdf = (1 - ez) / (1 + ez)
            curv = 2 * ez / (1 + ez) ** 2
        else:  # extreme (Gumbel)
            ez = torch.exp(z)
            pdf = ez * torch.exp(-ez)
            cdf = 1 - torch.exp(-ez)
            grad_pdf = 1 - ez
            curv = ez
        return pdf, cdf, grad_pdf, curv

    def get_gradient(self, preds, info, it):
        lo, hi = self._bounds(info, preds.device)
        p = preds.view(-1)
        s = self.sigma
        eps = 1e-12
        z_lo = (torch.log(torch.clamp(lo, min=eps)) - p) / s
        z_hi = (torch.log(torch.clamp(hi, min=eps)) - p) / s
        uncensored = torch.isfinite(hi) & (lo == hi)
        pdf_l, cdf_l, glp_l, curv_l = self._pdf_cdf(z_lo)
        pdf_u, cdf_u, _, _ = self._pdf_cdf(z_hi)
        cdf_u = torch.where(torch.isfinite(hi), cdf_u, torch.ones_like(cdf_u))
        pdf_u = torch.where(torch.isfinite(hi), pdf_u, torch.zeros_like(pdf_u))
        cdf_l = torch.where(lo > 0, cdf_l, torch.zeros_like(cdf_l))
        pdf_l = torch.where(lo > 0, pdf_l, torch.zeros_like(pdf_l))
        # uncensored: -log pdf(z)/ (s t); censored: -log(cdf_u - cdf_l)
        g_unc = glp_l / s  # d/dp of -log pdf(z_lo): -(dlogpdf/dz)(dz/dp)= glp/s
        h_unc = torch.clamp(curv_l, min=1e-6) / (s * s)
        denom = torch.clamp(cdf_u - cdf_l, min=eps)
        g_cen = (pdf_u - pdf_l) / (s * denom)
        h_cen = torch.clamp(g_cen * g_cen, min=1e-16) + 1e-6
        g = torch.where(uncensored, g_unc, g_cen)
        h = torch.where(uncensored, h_unc, h_cen)
        return self._apply_weight(g.view(preds.shape), h.view(preds.shape), info)

    def pred_transform(self, margin):
        return torch.exp(margin)

    def init_estimation(self, info) -> float:
        lo = info.label_lower_bound if info.label_lower_bound is not None \
            else info.labels.reshape(-1)
        v = np.asarray(lo, np.float64)
        v = v[np.isfinite(v) & (v > 0)]
        return float(np.exp(np.mean(np.log(v)))) if v.size else 1.0

    def prob_to_margin(self, base_score):
        return math.log(max(base_score, 1e-16))

    def save_config(self):
        return {"name": self.name, "aft_loss_param": {
            "aft_loss_distribution": self.dist,
            "aft_loss_distribution_scale": str(self.sigma)}}


def _weighted_quantile(v: np.ndarray, w: Optional[np.ndarray], q: float) -> float:
    if w is None:
        return float(np.quantile(v, q))
    order = np.argsort(v)
    v = v[order]
    cw = np.cumsum(np.asarray(w, np.float64)[order])
    t = q * cw[-1]
    i = int(np.searchsorted(cw, t))
    return float(v[min(i, len(v) - 1)])


# ---------------------------------------------------------------------------
# learning to rank (reference: src/objective/lambdarank_obj.{cc,cu},
# pair generation lambdarank_obj.cuh:76, NDCG deltas ranking_utils.h)


class _LambdaRankBase(Objective):
    task = "ranking"

    def __init__(self, params=None):
        super().__init__(params)
        self.num_pair = int(self.params.get("lambdarank_num_pair_per_sample",
                                            0) or 0)
        self.pair_method = str(self.params.get("lambdarank_pair_method",
                                               "topk"))
        self.normalize = bool(self.params.get("lambdarank_normalization",
                                              True))

    def _groups(self, info):
        if info.group_ptr is None:
            return np.array([0, info.num_row], dtype=np.int64)
        return np.asarray(info.group_ptr, dtype=np.int64)

    def _delta(self, y_sorted, ranks_i, ranks_j, i_idx, j_idx, inv_idcg):
        """|delta metric| for swapping documents at ranks_i/ranks_j."""
        raise NotImplementedError

    def get_gradient(self, preds, info, it):
        p = preds.detach().cpu().numpy().reshape(-1).astype(np.float64)
        y = np.asarray(info.labels, np.float64).reshape(-1)
        gp = self._groups(info)
        g = np.zeros_like(p)
        h = np.zeros_like(p)
        rng = np.random.RandomState(1234 + it)
        for gi in range(len(gp) - 1):
            s, e = int(gp[gi]), int(gp[gi + 1])
            if e - s < 2:
                continue
            self._group_gradient(p[s:e], y[s:e], g[s:e], h[s:e], rng)
        gt = torch.as_tensor(g.astype(np.float32),
                             device=preds.device).view(preds.shape)
        ht = torch.as_tensor(np.maximum(h, 1e-16).astype(np.float32),
                             device=preds.device).view(preds.shape)
        w = _weights(info, preds)
        if w is not None and info.weights.shape[0] == len(gp) - 1:
            # per-group weights
            wr = np.repeat(np.asarray(info.weights, np.float32),
                           np.diff(gp))
            wt = torch.as_tensor(wr, device=preds.device).view(preds.shape)
            gt, ht = gt * wt, ht * wt
        elif w is not None:
            gt, ht = gt * w.view(preds.shape), ht * w.view(preds.shape)
        return gt, ht

    def _make_pairs(self, n, y, order, rng):
        """Yield (i, j) index pairs (into the group) with y[i] > y[j]."""
        if self.pair_method == "mean" and self.num_pair > 0:
            k = self.num_pair
            pairs = []
            for i in range(n):
                js = rng.randint(0, n, size=k)
                for j in js:
                    if y[i] > y[j]:
                        pairs.append((i, j))
                    elif y[j] > y[i]:
                        pairs.append((j, i))
            return pairs
        # topk/full: all label-discordant pairs (n<=512 full, else truncate)
        pairs = []
        cap = 512
        idx = order[:cap]
        for a in range(len(idx)):
            for b in range(a + 1, len(idx)):
                i, j = idx[a], idx[b]
                if y[i] > y[j]:
                    pairs.append((i, j))
                elif y[j] > y[i]:
                    pairs.append((j, i))
        return pairs

    def _group_gradient(self, p, y, g, h, rng):
        n = len(p)
        order = np.argsort(-p, kind="stable")
        ranks = np.empty(n, dtype=np.int64)
        ranks[order] = np.arange(n)  # 0-based rank by prediction
        inv_idcg = self._inv_idcg(y)
        pairs = self._make_pairs(n, y, order, rng)
        if not pairs:
            return
        total_lambda = 0.0
        for i, j in pairs:
            delta = self._delta(y, ranks[i], ranks[j], i, j, inv_idcg)
            sij = p[i] - p[j]
            rho = 1.0 / (1.0 + np.exp(sij))  # d/ds of log(1+e^-s)
            lam = -rho * delta
            hess = max(rho * (1.0 - rho) * delta, 1e-16)
            g[i] += lam
            g[j] -= lam
            h[i] += hess
            h[j] += hess
            total_lambda += abs(lam)
        if self.normalize and total_lambda > 0:
            norm = np.log2(1.0 + total_lambda) / total_lambda
            g *= norm
            h *= norm

    def _inv_idcg(self, y):
        gains = np.sort(2.0 ** y - 1.0)[::-1]
        disc = 1.0 / np.log2(np.arange(2, len(y) + 2))
        idcg = float((gains * disc).sum())
        return 1.0 / idcg if idcg > 0 else 0.0

    def init_estimation(self, info) -> float:
        return 0.5

    def prob_to_margin(self, base_score):
        return base_score


@register("rank:ndcg")
class LambdaRankNDCG(_LambdaRankBase):
    @property
    def default_metric(self):
        return "ndcg"

    def _delta(self, y, rank_i, rank_j, i, j, inv_idcg):
        gain_i = 2.0 ** y[i] - 1.0
        gain_j = 2.0 ** y[j] - 1.0
        disc_i = 1.0 / np.log2(rank_i + 2.0)
        disc_j = 1.0 / np.log2(rank_j + 2.0)
        return abs((gain_i - gain_j) * (disc_i - disc_j)) * inv_idcg


@register("rank:map")
class LambdaRankMAP(_LambdaRankBase):
    @property
    def default_metric(self):
        return "map"

    def _delta(self, y, rank_i, rank_j, i, j, inv_idcg):
        # MAP delta approximated by reciprocal-rank difference on binary rel
        ri, rj = min(rank_i, rank_j), max(rank_i, rank_j)
        return abs(1.0 / (ri + 1.0) - 1.0 / (rj + 1.0))


@register("rank:pairwise")
class LambdaRankPairwise(_LambdaRankBase):
    @property
    def default_metric(self):
        return "map"

    def _delta(self, y, rank_i, rank_j, i, j, inv_idcg):
        return 1.0
