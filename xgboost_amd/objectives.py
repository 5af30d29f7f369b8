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


def _pbool(v, default: bool) -> bool:
    """Parse a boolean objective parameter: model JSON stores them as
    "0"/"1" strings, so bool("0") must not read as True."""
    if v is None:
        return default
    if isinstance(v, str):
        return v.strip().lower() in ("1", "true", "yes")
    return bool(v)



class _LambdaRankBase(Objective):
    """LambdaMART gradients, fully vectorized torch ops (device-resident
    when predictions live on the GPU).

    Reference math reproduced exactly (src/objective/lambdarank_obj.h):
    - LambdaGrad (:93): sigma = sigmoid(s_high - s_low),
      lambda = (sigma - 1) * |delta|, hessian = max(sigma(1-sigma), eps)
      * |delta| * 2; optional score normalization |delta| /= (|ds|+0.01)
      when the group is not degenerate (lambdarank_score_normalization,
      default true).
    - MakePairs (:225): topk = truncation pairs (i<min(k,n), j>i over the
      model-ranked list, k default 32); mean = per-doc sampled opponents
      outside the label bucket (num_pair default 1).
    - Normalization (lambdarank_obj.cc:230): topk
      log2(1+sum_lambda)/sum_lambda per group; mean 1/num_pair; group
      weights scaled by n_groups/sum(weights) (ranking_utils.cc:44).
    """

    task = "ranking"

    def __init__(self, params=None):
        super().__init__(params)
        self.num_pair = int(self.params.get("lambdarank_num_pair_per_sample",
                                            0) or 0)
        self.pair_method = str(self.params.get("lambdarank_pair_method",
                                               "topk"))
        self.normalize = _pbool(
            self.params.get("lambdarank_normalization"), True)
        self.score_norm = _pbool(
            self.params.get("lambdarank_score_normalization"), True)
        self.exp_gain = _pbool(self.params.get("ndcg_exp_gain"), True)
        self.unbiased = _pbool(self.params.get("lambdarank_unbiased"),
                               False)
        self.bias_norm = float(self.params.get("lambdarank_bias_norm", 1.0))
        # Unbiased LambdaMART position-bias ratios ti+ / tj- (reference
        # lambdarank_obj.cc:133-136), carried across boosting iterations.
        self._ti = None
        self._tj = None

    def _k(self) -> int:
        """reference LambdaRankParam::NumPair (ranking_utils.h:102)."""
        if self.num_pair > 0:
            return self.num_pair
        return 32 if self.pair_method == "topk" else 1

    def _max_position(self, cnt: torch.Tensor) -> int:
        """reference RankingCache::MaxPositionSize (ranking_utils.h:224):
        truncation level when pair_method=topk, else min(max group, 32)."""
        if self.pair_method == "topk":
            return self._k()
        mx = int(cnt.max()) if cnt.numel() else 0
        return min(mx, 32)

    def _groups(self, info):
        if info.group_ptr is None:
            return np.array([0, info.num_row], dtype=np.int64)
        return np.asarray(info.group_ptr, dtype=np.int64)

    # -- pair construction -------------------------------------------------
    @staticmethod
    def _topk_pairs(cnt: torch.Tensor, gp: torch.Tensor, k: int):
        """All (i, j) rank-position pairs with i < min(k, cnt), i < j < cnt
        (reference MakePairs truncation branch).  Returns flat rank-list
        positions (a, b) and the pair's group id."""
        dev = cnt.device
        G = cnt.numel()
        kk = torch.minimum(cnt, torch.tensor(k, device=dev))
        m = kk * (cnt - 1) - kk * (kk - 1) // 2  # pairs per group
        M = int(m.sum())
        if M == 0:
            z = torch.zeros(0, dtype=torch.long, device=dev)
            return z, z, z
        pg = torch.repeat_interleave(torch.arange(G, device=dev), m)
        off = torch.zeros(G + 1, dtype=torch.long, device=dev)
        torch.cumsum(m, 0, out=off[1:])
        t = torch.arange(M, device=dev) - off[pg]
        c = cnt[pg]
        # i = largest i0 with prefix(i0) <= t, prefix(i) = i(c-1) - i(i-1)/2
        b = (2 * c - 1).double()
        i = ((b - (b * b - 8.0 * t.double()).clamp(min=0).sqrt()) / 2
             ).floor().long().clamp(min=0)
        for _ in range(2):  # fp fix-up
            pref = i * (c - 1) - i * (i - 1) // 2
            i = (i - (pref > t).long()).clamp(min=0)
            nxt = (i + 1) * (c - 1) - (i + 1) * i // 2
            i = i + ((nxt <= t) & (i + 1 < c)).long()
        pref = i * (c - 1) - i * (i - 1) // 2
        j = i + 1 + (t - pref)
        return gp[pg] + i, gp[pg] + j, pg

    def _mean_pairs(self, y_by_rank: torch.Tensor, cnt: torch.Tensor,
                    gp: torch.Tensor, n_samples: int, seed: int):
        """Sampled pairs: for each doc, n_samples opponents drawn from
        outside its label bucket on the label-sorted list (reference
        MakePairs sampling branch)."""
        dev = y_by_rank.device
        G = cnt.numel()
        N = y_by_rank.numel()
        gid = torch.repeat_interleave(torch.arange(G, device=dev), cnt)
        # label-desc order over rank positions, grouped
        o1 = torch.argsort(-y_by_rank, stable=True)
        o2 = torch.argsort(gid[o1], stable=True)
        ls = o1[o2]  # ls[q] = rank position of q-th label-sorted doc
        ys = y_by_rank[ls]
        # bucket boundaries within groups
        start = torch.zeros(N, dtype=torch.bool, device=dev)
        start[gp[:-1][cnt > 0]] = True
        if N > 1:
            start[1:] |= (ys[1:] != ys[:-1]) & (gid[1:] == gid[:-1])
        bucket = torch.cumsum(start.long(), 0) - 1
        b_first = torch.nonzero(start, as_tuple=True)[0]
        b_last = torch.empty_like(b_first)
        if b_first.numel() > 1:
            b_last[:-1] = b_first[1:] - 1
        b_last[-1] = N - 1
        i_of = (b_first - gp[gid[b_first]])[bucket]   # bucket start (i)
        j_of = (b_last - gp[gid[b_last]] + 1)[bucket]  # bucket end (j)
        c = cnt[gid]
        n_lefts = i_of
        n_rights = c - j_of
        tot = n_lefts + n_rights
        valid = tot > 0
        # deterministic sampling (CPU RNG, then moved to the device)
        rng = np.random.RandomState(seed)
        r = torch.as_tensor(
            rng.randint(0, 2 ** 31 - 1, size=(n_samples, N)), device=dev)
        out_a, out_b, out_g = [], [], []
        for sidx in range(n_samples):
            ridx = r[sidx] % tot.clamp(min=1)
            ridx = torch.where(ridx >= n_lefts, ridx - i_of + j_of, ridx)
            bpos = (gp[gid] + ridx).clamp(max=N - 1)
            out_a.append(ls[valid])
            out_b.append(ls[bpos][valid])
            out_g.append(gid[valid])
        return (torch.cat(out_a), torch.cat(out_b), torch.cat(out_g))

    # -- delta hooks -------------------------------------------------------
    def _prepare(self, y_by_rank, cnt, gp, pos):
        """Per-iteration group statistics for the delta (subclasses)."""
        return None

    def _delta_vec(self, stats, y_high, y_low, rank_high, rank_low, g):
        raise NotImplementedError

    # -- main --------------------------------------------------------------
    def get_gradient(self, preds, info, it):
        dev = preds.device
        p = preds.detach().double().reshape(-1)
        N = p.numel()
        y = torch.as_tensor(np.asarray(info.labels, np.float64).reshape(-1),
                            device=dev)
        gp_np = self._groups(info)
        gp = torch.as_tensor(gp_np, device=dev)
        cnt = gp[1:] - gp[:-1]
        G = cnt.numel()
        gid = torch.repeat_interleave(torch.arange(G, device=dev), cnt)
        # model-rank order (pred desc, stable) within each group
        o1 = torch.argsort(-p, stable=True)
        o2 = torch.argsort(gid[o1], stable=True)
        order = o1[o2]           # order[gp[g]+r] = doc index at rank r
        pos = torch.arange(N, device=dev) - gp[gid]
        y_by_rank = y[order]
        stats = self._prepare(y_by_rank, cnt, gp, pos)

        if self.pair_method == "topk":
            a_flat, b_flat, pg = self._topk_pairs(cnt, gp, self._k())
        else:
            a_flat, b_flat, pg = self._mean_pairs(
                y_by_rank, cnt, gp, self._k(), seed=1234 + it)
        g_out = torch.zeros(N, dtype=torch.float64, device=dev)
        h_out = torch.zeros(N, dtype=torch.float64, device=dev)
        if a_flat.numel():
            da = order[a_flat]
            db = order[b_flat]
            ya, yb = y[da], y[db]
            keep = ya != yb
            da, db, ya, yb = da[keep], db[keep], ya[keep], yb[keep]
            af = a_flat[keep]
            bf = b_flat[keep]
            ra = af - gp[pg][keep]
            rb = bf - gp[pg][keep]
            pgk = pg[keep]
            swap = ya < yb
            idx_high = torch.where(swap, db, da)
            idx_low = torch.where(swap, da, db)
            y_high = torch.where(swap, yb, ya)
            y_low = torch.where(swap, ya, yb)
            rank_high = torch.where(swap, rb, ra)
            rank_low = torch.where(swap, ra, rb)
            s_high = p[idx_high]
            s_low = p[idx_low]
            sig = torch.sigmoid(s_high - s_low)
            delta = self._delta_vec(stats, y_high, y_low, rank_high,
                                    rank_low, pgk).abs()
            if self.score_norm:
                # skip for degenerate groups (best score == worst score,
                # reference LambdaGrad best/worst check)
                best = p[order[gp[:-1].clamp(max=max(N - 1, 0))]]
                worst = p[order[(gp[1:] - 1).clamp(min=0)]]
                ok = (best != worst)[pgk]
                delta = torch.where(
                    ok, delta / ((s_high - s_low).abs() + 0.01), delta)
            lam = (sig - 1.0) * delta
            hess = torch.clamp(sig * (1.0 - sig), min=1e-16) * delta * 2.0
            li_pos = lj_pos = None
            if self.unbiased:
                # Unbiased LambdaMART (reference lambdarank_obj.h:128-147
                # and lambdarank_obj.cc:205-221): divide the pair gradient
                # by ti+[idx_high] * tj-[idx_low] (positions on the ORIGINAL
                # label-sorted list) and accumulate the pair cost
                # log(1/(1-sigma)) * |delta| into per-position sums.
                k_pos = max(self._max_position(cnt), 1)
                if (self._ti is None or self._ti.numel() != k_pos
                        or self._ti.device != dev):
                    self._ti = torch.ones(k_pos, dtype=torch.float64,
                                          device=dev)
                    self._tj = torch.ones(k_pos, dtype=torch.float64,
                                          device=dev)
                eps64 = 1e-16
                ih_pos = idx_high - gp[pgk]
                il_pos = idx_low - gp[pgk]
                inside = (ih_pos < k_pos) & (il_pos < k_pos)
                ih_c = ih_pos.clamp(max=k_pos - 1)
                il_c = il_pos.clamp(max=k_pos - 1)
                t_hi = self._ti[ih_c]
                t_lo = self._tj[il_c]
                ok_t = inside & (t_hi >= eps64) & (t_lo >= eps64)
                scale = torch.where(ok_t, 1.0 / (t_hi * t_lo),
                                    torch.ones_like(t_hi))
                lam = lam * scale
                hess = hess * scale
                # cost = log(1/(1-sigma)) * delta_metric (eq. 30/31 input)
                cost = -torch.log1p(-sig) * delta
                zero = torch.zeros((), dtype=torch.float64, device=dev)
                li_add = torch.where(inside & (t_lo >= eps64),
                                     cost / t_lo.clamp(min=eps64), zero)
                lj_add = torch.where(inside & (t_hi >= eps64),
                                     cost / t_hi.clamp(min=eps64), zero)

                def possum(keys, vals):
                    # deterministic per-position sum (k_pos bins):
                    # sort + cumsum + searchsorted, no fp64 atomics
                    ks, pi = torch.sort(keys)
                    cs = torch.zeros(vals.numel() + 1, dtype=torch.float64,
                                     device=dev)
                    torch.cumsum(vals[pi], 0, out=cs[1:])
                    bnd = torch.searchsorted(
                        ks, torch.arange(k_pos + 1, device=dev))
                    return cs[bnd[1:]] - cs[bnd[:-1]]

                li_pos = possum(ih_c, li_add)
                lj_pos = possum(il_c, lj_add)
            # Scatter WITHOUT atomics (fp64 atomics CAS-loop on ROCm and
            # the hot top-ranked docs make them ~100-way contended):
            # accumulate by RANK-LIST POSITION with sort + cumsum +
            # searchsorted segment sums, then write through `order` as a
            # pure permutation.  Also makes gradients accumulation-order
            # deterministic.
            idxN = torch.arange(N + 1, device=dev)

            def segsum(keys_sorted, vals):
                cs = torch.zeros(vals.numel() + 1, dtype=torch.float64,
                                 device=dev)
                torch.cumsum(vals, 0, out=cs[1:])
                bnd = torch.searchsorted(keys_sorted, idxN)
                return cs[bnd[1:]] - cs[bnd[:-1]]

            val_a = torch.where(swap, -lam, lam)  # pair grad at side a
            ka, pa = torch.sort(af)
            kb, pb = torch.sort(bf)
            g_r = segsum(ka, val_a[pa]) + segsum(kb, -val_a[pb])
            h_r = segsum(ka, hess[pa]) + segsum(kb, hess[pb])
            g_out[order] = g_r
            h_out[order] = h_r
            if self.normalize:
                if self.pair_method == "topk":
                    # per-group sum WITHOUT atomics: pgk is sorted
                    # (repeat_interleave order survives the keep mask),
                    # and ~pairs-per-group-way contended fp64 atomics
                    # serialize into a CAS loop on ROCm (measured 380 ms
                    # of a 395 ms round); cumsum + searchsorted is
                    # bandwidth-bound and deterministic
                    cs = torch.zeros(pgk.numel() + 1, dtype=torch.float64,
                                     device=dev)
                    torch.cumsum(-2.0 * lam, 0, out=cs[1:])
                    grange = torch.arange(G, device=dev)
                    lo = torch.searchsorted(pgk, grange, side="left")
                    hi = torch.searchsorted(pgk, grange, side="right")
                    s_g = cs[hi] - cs[lo]
                    norm_g = torch.where(
                        s_g > 0,
                        torch.log2(1.0 + s_g) / s_g.clamp(min=1e-300),
                        torch.ones_like(s_g))
                    g_out *= norm_g[gid]
                    h_out *= norm_g[gid]
                else:
                    g_out /= self._k()
                    h_out /= self._k()
            if self.unbiased and li_pos is not None:
                # UpdatePositionBias (reference lambdarank_obj.cc:40-86):
                # ti+(i) = (li(i)/li(0))^(1/(1+bias_norm)), fresh each
                # iteration (the accumulators reset; ti carries over).
                reg = 1.0 / (1.0 + self.bias_norm)
                if float(li_pos[0]) >= 1e-16:
                    self._ti = (li_pos / li_pos[0]).pow(reg)
                if float(lj_pos[0]) >= 1e-16:
                    self._tj = (lj_pos / lj_pos[0]).pow(reg)
        # group weights * weight_norm (ranking_utils.cc:44)
        if info.weights is not None:
            w_np = np.asarray(info.weights, np.float64).reshape(-1)
            if w_np.shape[0] == G:
                wn = G / max(float(w_np.sum()), 1e-300)
                w_doc = torch.as_tensor(w_np, device=dev)[gid] * wn
            else:
                w_doc = torch.as_tensor(w_np, device=dev)
            g_out *= w_doc
            h_out *= w_doc
        gt = g_out.float().view(preds.shape)
        ht = torch.clamp(h_out, min=1e-16).float().view(preds.shape)
        return gt, ht

    def init_estimation(self, info) -> float:
        return 0.5

    def prob_to_margin(self, base_score):
        return base_score

    def save_config(self):
        return {"name": self.name, "lambdarank_param": {
            "lambdarank_pair_method": self.pair_method,
            "lambdarank_num_pair_per_sample": str(self._k()),
            "lambdarank_normalization": str(int(self.normalize)),
            "lambdarank_score_normalization": str(int(self.score_norm)),
            "ndcg_exp_gain": str(int(self.exp_gain)),
            "lambdarank_unbiased": str(int(self.unbiased)),
            "lambdarank_bias_norm": str(self.bias_norm),
        }}


@register("rank:ndcg")
class LambdaRankNDCG(_LambdaRankBase):
    @property
    def default_metric(self):
        return "ndcg"

    def _prepare(self, y_by_rank, cnt, gp, pos):
        """inv_IDCG per group, truncated at TopK like NDCGCache
        (ranking_utils.cc:96-108)."""
        dev = y_by_rank.device
        G = cnt.numel()
        N = y_by_rank.numel()
        gid = torch.repeat_interleave(torch.arange(G, device=dev), cnt)
        # label-desc within group
        o1 = torch.argsort(-y_by_rank, stable=True)
        o2 = torch.argsort(gid[o1], stable=True)
        ysort = y_by_rank[o1[o2]]
        gain = (2.0 ** ysort - 1.0) if self.exp_gain else ysort
        disc = 1.0 / torch.log2(pos.double() + 2.0)
        topk = self._k() if self.pair_method == "topk" else N + 1
        mask = pos < topk
        idcg = torch.zeros(G, dtype=torch.float64, device=dev)
        idcg.index_add_(0, gid[mask], (gain * disc)[mask])
        inv_idcg = torch.where(idcg > 0, 1.0 / idcg.clamp(min=1e-300),
                               torch.zeros_like(idcg))
        return {"inv_idcg": inv_idcg}

    def _delta_vec(self, stats, y_high, y_low, rank_high, rank_low, g):
        """DeltaNDCG (lambdarank_obj.h:42)."""
        if self.exp_gain:
            gain_h = 2.0 ** y_high - 1.0
            gain_l = 2.0 ** y_low - 1.0
        else:
            gain_h, gain_l = y_high, y_low
        disc_h = 1.0 / torch.log2(rank_high.double() + 2.0)
        disc_l = 1.0 / torch.log2(rank_low.double() + 2.0)
        return (gain_h - gain_l) * (disc_h - disc_l) * stats["inv_idcg"][g]


@register("rank:map")
class LambdaRankMAP(_LambdaRankBase):
    @property
    def default_metric(self):
        return "map"

    def _prepare(self, y_by_rank, cnt, gp, pos):
        """MAPStat (lambdarank_obj.cc): n_rel = running count of relevant
        docs down the ranked list, acc = running sum of label/rank."""
        if not bool(((y_by_rank == 0) | (y_by_rank == 1)).all()):
            raise ValueError("rank:map requires binary labels")
        dev = y_by_rank.device
        G = cnt.numel()
        gid = torch.repeat_interleave(torch.arange(G, device=dev), cnt)
        nz = cnt > 0
        # segmented cumsums via global cumsum minus per-group base
        cw = torch.cumsum(y_by_rank, 0)
        base = torch.zeros(G, dtype=torch.float64, device=dev)
        base[nz] = cw[gp[:-1][nz]] - y_by_rank[gp[:-1][nz]]
        n_rel = cw - base[gid]
        a = y_by_rank / (pos.double() + 1.0)
        ca = torch.cumsum(a, 0)
        base_a = torch.zeros(G, dtype=torch.float64, device=dev)
        base_a[nz] = ca[gp[:-1][nz]] - a[gp[:-1][nz]]
        acc = ca - base_a[gid]
        last = (gp[1:] - 1).clamp(min=0)
        n_total = torch.where(nz, n_rel[last],
                              torch.zeros(G, dtype=torch.float64, device=dev))
        return {"n_rel": n_rel, "acc": acc, "n_total": n_total, "gp": gp}

    def _delta_vec(self, stats, y_high, y_low, rank_high, rank_low, g):
        """DeltaMAP (lambdarank_obj.h:62) with the rank min/max swap the
        reference's caller applies (lambdarank_obj.cu:484-491): positions
        are ordered while the labels keep their high/low roles."""
        gp = stats["gp"]
        rh = torch.minimum(rank_high, rank_low)
        rl = torch.maximum(rank_high, rank_low)
        base = gp[g]
        n_rel = stats["n_rel"]
        acc = stats["acc"]
        n_tot = stats["n_total"][g].clamp(min=1e-300)
        r_h = rh.double() + 1.0
        r_l = rl.double() + 1.0
        m = n_rel[base + rl]
        n = n_rel[base + rh]
        b = acc[base + rl - 1] - acc[base + rh]
        swapped = y_high < y_low  # the relevant doc already ranks higher
        a1 = m / r_l - (n + 1.0) / r_h   # y_high < y_low branch
        a2 = n / r_h - m / r_l           # y_high > y_low branch
        return torch.where(swapped, a1 - b, a2 + b) / n_tot


@register("rank:pairwise")
class LambdaRankPairwise(_LambdaRankBase):
    @property
    def default_metric(self):
        return "map"

    def _delta_vec(self, stats, y_high, y_low, rank_high, rank_low, g):
        return torch.ones_like(y_high)
