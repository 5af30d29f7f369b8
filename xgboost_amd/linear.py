"""Linear booster (gblinear) with elastic-net coordinate descent.

Reference behavior: src/gbm/gblinear.cc, src/linear/updater_coordinate.cc
("coord_descent"), updater_shotgun.cc ("shotgun"), coordinate_common.h
(CoordinateDelta/CoordinateDeltaBias, feature selectors).

Weights: [n_features + 1, n_out], bias in the last row.  The GPU path
runs the same torch ops on device (X is dense; the per-feature reduces
are torch matvec slices) — the reference's gpu_coord_descent analog.
"""
from __future__ import annotations

from typing import Optional

import numpy as np
import torch

from .params import canonicalize


def _threshold_l1(w: torch.Tensor, alpha: float) -> torch.Tensor:
    return torch.sign(w) * torch.clamp(torch.abs(w) - alpha, min=0.0)


class GBLinearModel:
    def __init__(self, n_features: int, n_out: int, params: dict,
                 device: torch.device):
        p = canonicalize(dict(params))
        self.n_features = n_features
        self.n_out = n_out
        self.device = device
        self.weights = torch.zeros((n_features + 1, n_out),
                                   dtype=torch.float32, device=device)
        self.reg_lambda = float(p.get("lambda", 0.0))
        self.reg_alpha = float(p.get("alpha", 0.0))
        self.eta = float(p.get("eta", 0.5))
        self.feature_selector = str(p.get("feature_selector", "cyclic"))
        self.top_k = int(p.get("top_k", 0))
        self.updater = str(p.get("updater", "coord_descent")
                           if p.get("updater") else "coord_descent")
        if self.updater == "shotgun" and self.feature_selector not in (
                "cyclic", "shuffle"):
            # reference updater_shotgun.cc:20: parallel shotgun steps
            # cannot honor gradient-ordered selectors
            raise ValueError(
                "Unsupported feature selector for shotgun updater. "
                "Supported options are: {cyclic, shuffle}")

    def predict_margin(self, X: torch.Tensor) -> torch.Tensor:
        return X @ self.weights[:-1] + self.weights[-1]

    def _feature_order(self, it: int, grad2: Optional[np.ndarray]) -> np.ndarray:
        n = self.n_features
        if self.feature_selector == "cyclic":
            order = np.arange(n)
        elif self.feature_selector == "shuffle":
            order = np.random.RandomState(it).permutation(n)
        elif self.feature_selector == "random":
            order = np.random.RandomState(it).randint(0, n, size=n)
        elif self.feature_selector in ("greedy", "thrifty"):
            order = (np.argsort(-grad2) if grad2 is not None
                     else np.arange(n))
        else:
            raise ValueError(
                f"unknown feature_selector: {self.feature_selector}")
        if self.top_k > 0:
            order = order[:self.top_k]
        return order

    def update(self, X: torch.Tensor, grad: torch.Tensor,
               hess: torch.Tensor, iteration: int,
               sum_instance_weight: float) -> None:
        """One boosting round of coordinate descent.

        grad/hess: [n, n_out]; mutated in place as weights move
        (reference UpdateResidualParallel)."""
        lam = self.reg_lambda * sum_instance_weight
        alp = self.reg_alpha * sum_instance_weight
        Xsq = X * X
        for k in range(self.n_out):
            g = grad[:, k]
            h = hess[:, k]
            # bias first (no regularization; reference CoordinateDeltaBias)
            sg = float(g.sum())
            sh = float(h.sum())
            if sh > 1e-16:
                dbias = self.eta * (-sg / sh)
                self.weights[-1, k] += dbias
                g += h * dbias
            if self.updater == "shotgun":
                # Parallel coordinate step (reference updater_shotgun.cc):
                # all deltas computed against the same residual in two
                # matvecs, then applied together.  The reference races
                # Hogwild-style across threads; computing the whole batch
                # from one residual snapshot is the deterministic
                # equivalent and maps to GEMV instead of n_features
                # host-synced dot products.
                sum_grad = X.t() @ g + lam * self.weights[:-1, k]
                sum_hess = Xsq.t() @ h
                dw = -_threshold_l1(sum_grad, alp) / (sum_hess + lam)
                dw = self.eta * torch.where(
                    sum_hess > 1e-16, dw, torch.zeros_like(dw))
                if self.top_k > 0:
                    keep = torch.zeros_like(dw)
                    sel = torch.as_tensor(
                        np.ascontiguousarray(
                            self._feature_order(iteration, None)),
                        device=dw.device, dtype=torch.long)
                    keep[sel] = dw[sel]
                    dw = keep
                self.weights[:-1, k] += dw
                g += h * (X @ dw)
                continue
            if self.feature_selector in ("greedy", "thrifty"):
                gf = (X * g.view(-1, 1)).sum(dim=0).abs().cpu().numpy()
            else:
                gf = None
            for j in self._feature_order(iteration, gf):
                xj = X[:, j]
                sum_grad = float((g * xj).sum())
                sum_hess = float((h * Xsq[:, j]).sum())
                if sum_hess < 1e-16:
                    continue
                w = float(self.weights[j, k])
                dw = self.eta * (
                    -float(_threshold_l1(
                        torch.tensor(sum_grad + lam * w), alp))
                    / (sum_hess + lam))
                if dw == 0.0:
                    continue
                self.weights[j, k] += dw
                g += h * xj * dw

    def to_json(self) -> dict:
        return {
            "model": {
                "weights": self.weights.cpu().numpy()
                .reshape(-1).astype(float).tolist(),
                "boosted_rounds": 0,
            },
            "name": "gblinear",
        }

    @classmethod
    def from_json(cls, j: dict, n_features: int, n_out: int, params: dict,
                  device) -> "GBLinearModel":
        m = cls(n_features, n_out, params, device)
        w = np.asarray(j["model"]["weights"], dtype=np.float32)
        m.weights = torch.from_numpy(
            w.reshape(n_features + 1, n_out)).to(device)
        return m
