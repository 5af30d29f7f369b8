"""Interpretability helpers (reference:
python-package/xgboost/interpret.py — `shap_values` returns the
per-feature contributions with the bias term split off)."""
from __future__ import annotations

from typing import Optional, Tuple, Union

import numpy as np

from .core import Booster
from .data import DMatrix


def _as_booster(model: object) -> Booster:
    if isinstance(model, Booster):
        return model
    get_booster = getattr(model, "get_booster", None)
    if not callable(get_booster):
        raise TypeError(
            "`model` must be a Booster or an object with get_booster().")
    booster = get_booster()
    if not isinstance(booster, Booster):
        raise TypeError("`model.get_booster()` must return a Booster.")
    return booster


def shap_values(model: object, X: Union[DMatrix, np.ndarray], *,
                X_background=None, output_margin: bool = False,
                iteration_range: Optional[Tuple[int, int]] = None,
                missing: Optional[float] = None,
                validate_features: bool = True
                ) -> Tuple[np.ndarray, np.ndarray]:
    """SHAP values for a model: ``(values, bias)`` where ``values`` is the
    contributions array without the bias column and ``bias`` the separated
    bias term (reference interpret.py:54)."""
    if X_background is not None:
        raise NotImplementedError("`X_background` is not yet supported.")
    _ = output_margin  # contributions are in margin space, as upstream
    booster = _as_booster(model)
    if isinstance(X, DMatrix):
        if missing is not None:
            raise ValueError(
                "`missing` must not be specified when `X` is a DMatrix.")
        data = X
    else:
        data = DMatrix(X, missing=missing if missing is not None
                       else getattr(model, "missing", np.nan))
    contribs = booster.predict(
        data, pred_contribs=True,
        iteration_range=iteration_range or (0, 0),
        validate_features=validate_features)
    values = contribs[..., :-1]
    bias = contribs[..., -1]
    return values, bias


__all__ = ["shap_values"]
