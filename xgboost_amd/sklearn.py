"""scikit-learn estimator wrappers.

Reference behavior: python-package/xgboost/sklearn.py (XGBModel :868,
XGBClassifier :1728, XGBRegressor :2023, XGBRanker :2165, RF variants).
"""
from __future__ import annotations

from typing import Any, Dict, Optional, Sequence, Tuple, Union

import numpy as np

from .callback import TrainingCallback
from .core import Booster
from .data import DMatrix
from .training import train as _train

# Inherit scikit-learn's estimator protocol when it is installed
# (reference sklearn.py XGBModelBase extends BaseEstimator through the
# compat shim): modern sklearn (>=1.6) meta-estimators call
# `__sklearn_tags__`, which only BaseEstimator + the mixins provide.
try:
    from sklearn.base import (BaseEstimator as _SklBaseEstimator,
                              ClassifierMixin as _SklClassifierMixin,
                              RegressorMixin as _SklRegressorMixin)
except ImportError:  # sklearn not installed: plain classes
    class _SklBaseEstimator:  # type: ignore[no-redef]
        pass

    class _SklClassifierMixin:  # type: ignore[no-redef]
        pass

    class _SklRegressorMixin:  # type: ignore[no-redef]
        pass

_PARAM_NAMES = [
    "max_depth", "max_leaves", "max_bin", "grow_policy", "learning_rate",
    "n_estimators", "verbosity", "objective", "booster", "tree_method",
    "gamma", "min_child_weight", "max_delta_step", "subsample",
    "sampling_method", "colsample_bytree", "colsample_bylevel",
    "colsample_bynode", "reg_alpha", "reg_lambda", "scale_pos_weight",
    "base_score", "random_state", "missing", "num_parallel_tree",
    "monotone_constraints", "interaction_constraints", "importance_type",
    "device", "validate_parameters", "enable_categorical",
    "feature_types", "max_cat_to_onehot", "max_cat_threshold",
    "multi_strategy", "eval_metric", "early_stopping_rounds", "callbacks",
    "n_jobs", "seed",
]


class XGBModel(_SklBaseEstimator):
    """Base sklearn-style estimator (reference sklearn.py:868)."""

    _estimator_type = "regressor"

    def __init__(self, max_depth: Optional[int] = None,
                 max_leaves: Optional[int] = None,
                 max_bin: Optional[int] = None,
                 grow_policy: Optional[str] = None,
                 learning_rate: Optional[float] = None,
                 n_estimators: Optional[int] = None,
                 verbosity: Optional[int] = None,
                 objective: Optional[str] = None,
                 booster: Optional[str] = None,
                 tree_method: Optional[str] = None,
                 n_jobs: Optional[int] = None,
                 gamma: Optional[float] = None,
                 min_child_weight: Optional[float] = None,
                 max_delta_step: Optional[float] = None,
                 subsample: Optional[float] = None,
                 sampling_method: Optional[str] = None,
                 colsample_bytree: Optional[float] = None,
                 colsample_bylevel: Optional[float] = None,
                 colsample_bynode: Optional[float] = None,
                 reg_alpha: Optional[float] = None,
                 reg_lambda: Optional[float] = None,
                 scale_pos_weight: Optional[float] = None,
                 base_score: Optional[float] = None,
                 random_state: Optional[int] = None,
                 missing: float = np.nan,
                 num_parallel_tree: Optional[int] = None,
                 monotone_constraints=None, interaction_constraints=None,
                 importance_type: Optional[str] = None,
                 device: Optional[str] = None,
                 validate_parameters: Optional[bool] = None,
                 enable_categorical: bool = False,
                 feature_types=None,
                 max_cat_to_onehot: Optional[int] = None,
                 max_cat_threshold: Optional[int] = None,
                 multi_strategy: Optional[str] = None,
                 eval_metric=None,
                 early_stopping_rounds: Optional[int] = None,
                 callbacks: Optional[Sequence[TrainingCallback]] = None,
                 seed: Optional[int] = None,
                 **kwargs):
        args = locals()
        for name in _PARAM_NAMES:
            if name in args:
                setattr(self, name, args[name])
        self.kwargs = kwargs
        self._Booster: Optional[Booster] = None
        self.evals_result_: Dict = {}

    # -- sklearn protocol ---------------------------------------------------
    def get_params(self, deep: bool = True) -> Dict[str, Any]:
        out = {k: getattr(self, k) for k in _PARAM_NAMES if hasattr(self, k)}
        out.update(self.kwargs)
        return out

    def set_params(self, **params) -> "XGBModel":
        for k, v in params.items():
            if k in _PARAM_NAMES:
                setattr(self, k, v)
            else:
                self.kwargs[k] = v
        return self

    def get_xgb_params(self) -> Dict[str, Any]:
        params = {}
        skip = {"n_estimators", "missing", "importance_type",
                "enable_categorical", "feature_types", "eval_metric",
                "early_stopping_rounds", "callbacks", "random_state",
                "learning_rate", "n_jobs"}
        for k in _PARAM_NAMES:
            v = getattr(self, k, None)
            if v is None or k in skip:
                continue
            params[k] = v
        if getattr(self, "learning_rate", None) is not None:
            params["eta"] = self.learning_rate
        if getattr(self, "random_state", None) is not None:
            params["seed"] = self.random_state
        if getattr(self, "n_jobs", None) is not None:
            params["nthread"] = self.n_jobs
        if getattr(self, "eval_metric", None) is not None:
            params["eval_metric"] = self.eval_metric
        params.update(self.kwargs)
        return params

    @property
    def n_estimators_(self) -> int:
        return self.n_estimators if self.n_estimators is not None else 100

    def get_booster(self) -> Booster:
        if self._Booster is None:
            raise ValueError("need to call fit or load_model beforehand")
        return self._Booster

    def _make_dmatrix(self, X, y=None, sample_weight=None, base_margin=None,
                      group=None, qid=None) -> DMatrix:
        return DMatrix(X, label=y, weight=sample_weight,
                       base_margin=base_margin, missing=self.missing,
                       feature_types=getattr(self, "feature_types", None),
                       group=group, qid=qid,
                       enable_categorical=getattr(self, "enable_categorical",
                                                  False))

    def fit(self, X, y, *, sample_weight=None, base_margin=None,
            eval_set=None, sample_weight_eval_set=None,
            base_margin_eval_set=None, verbose: Union[bool, int] = True,
            xgb_model=None, feature_weights=None) -> "XGBModel":
        dtrain = self._make_dmatrix(X, y, sample_weight, base_margin)
        if feature_weights is not None:
            dtrain.set_info(feature_weights=feature_weights)
        evals = []
        if eval_set:
            for i, (ex, ey) in enumerate(eval_set):
                sw = (sample_weight_eval_set[i]
                      if sample_weight_eval_set else None)
                bm = (base_margin_eval_set[i]
                      if base_margin_eval_set else None)
                evals.append((self._make_dmatrix(ex, ey, sw, bm),
                              f"validation_{i}"))
        params = self.get_xgb_params()
        obj = None
        if callable(self.objective):
            # reference _objective_decorator (sklearn.py): a callable
            # objective takes (y_true, y_pred) -> (grad, hess); train's
            # fobj convention is (preds, dtrain)
            user_fn = self.objective
            params["objective"] = self._default_objective()

            def obj(preds, dmat):  # noqa: ANN001
                return user_fn(dmat.get_label(), preds)
        elif self.objective is None:
            params.setdefault("objective", self._default_objective())
        custom_metric = None
        if callable(getattr(self, "eval_metric", None)):
            # reference _metric_decorator: sklearn-style metrics take
            # (y_true, y_pred); predictions arrive transformed for
            # builtin objectives
            metric_fn = self.eval_metric
            metric_name = getattr(metric_fn, "__name__", "custom_metric")
            params.pop("eval_metric", None)

            def custom_metric(preds, dmat):  # noqa: ANN001
                return metric_name, float(metric_fn(dmat.get_label(), preds))
        self.evals_result_ = {}
        self._Booster = _train(
            params, dtrain, self.n_estimators_, evals=evals, obj=obj,
            custom_metric=custom_metric,
            early_stopping_rounds=getattr(self, "early_stopping_rounds", None),
            evals_result=self.evals_result_, verbose_eval=verbose,
            xgb_model=xgb_model,
            callbacks=list(self.callbacks) if getattr(self, "callbacks", None)
            else None)
        self._set_fitted_attrs(dtrain)
        return self

    def _default_objective(self) -> str:
        return "reg:squarederror"

    def _set_fitted_attrs(self, dtrain: DMatrix) -> None:
        # best_iteration/best_score/feature_names_in_ are properties
        # derived from the booster
        self.n_features_in_ = dtrain.num_col()

    def _iteration_range(self) -> Tuple[int, int]:
        b = self.get_booster()
        if getattr(self, "early_stopping_rounds", None) and \
                b.best_iteration is not None:
            return (0, b.best_iteration + 1)
        return (0, 0)

    def predict(self, X, *, output_margin=False, validate_features=True,
                base_margin=None, iteration_range=None):
        d = self._make_dmatrix(X, base_margin=base_margin)
        return self.get_booster().predict(
            d, output_margin=output_margin,
            validate_features=validate_features,
            iteration_range=iteration_range or self._iteration_range())

    def apply(self, X, iteration_range=None):
        d = self._make_dmatrix(X)
        return self.get_booster().predict(
            d, pred_leaf=True,
            iteration_range=iteration_range or self._iteration_range())

    def evals_result(self) -> Dict:
        return self.evals_result_

    @property
    def feature_importances_(self) -> np.ndarray:
        b = self.get_booster()
        imp_type = getattr(self, "importance_type", None) or "weight"
        score = b.get_score(importance_type=imp_type)
        n = self.n_features_in_
        out = np.zeros(n, dtype=np.float32)
        names = b.feature_names or [f"f{i}" for i in range(n)]
        for i, name in enumerate(names):
            out[i] = score.get(name, 0.0)
        total = out.sum()
        return out / total if total > 0 else out

    @property
    def intercept_(self) -> np.ndarray:
        return np.array([self.get_booster().base_score], dtype=np.float32)

    @property
    def coef_(self) -> np.ndarray:
        """Linear coefficients (gblinear boosters only, like upstream)."""
        if getattr(self, "booster", None) not in ("gblinear",):
            raise AttributeError(
                "coef_ is only defined for booster=gblinear")
        b = self.get_booster()
        j = b._model_to_json()
        w = np.array(j["learner"]["gradient_booster"]["model"]["weights"],
                     dtype=np.float32)
        n = self.n_features_in_
        w = w.reshape(n + 1, -1)[:n]
        return w[:, 0] if w.shape[1] == 1 else w.T

    @property
    def best_iteration(self) -> int:
        bi = self.get_booster().best_iteration
        if bi is None:
            raise AttributeError(
                "best_iteration is only defined when early stopping is used")
        return bi

    @property
    def best_score(self) -> float:
        bs = self.get_booster().best_score
        if bs is None:
            raise AttributeError(
                "best_score is only defined when early stopping is used")
        return bs

    @property
    def feature_names_in_(self) -> np.ndarray:
        names = self.get_booster().feature_names
        if names is None:
            raise AttributeError("feature names are not available")
        return np.array(names, dtype=object)

    def get_num_boosting_rounds(self) -> int:
        return self.n_estimators_

    def save_model(self, fname: str) -> None:
        self.get_booster().save_model(fname)

    def load_model(self, fname) -> None:
        self._Booster = Booster({
            k: v for k, v in self.get_xgb_params().items()
            if k not in ("objective",)})
        self._Booster.load_model(fname)
        self.n_features_in_ = self._Booster.num_features()

    # __sklearn_tags__ comes from sklearn's BaseEstimator + the
    # Classifier/Regressor mixins (sklearn >= 1.6 tags protocol)


class XGBRegressor(_SklRegressorMixin, XGBModel):
    _estimator_type = "regressor"

    def _default_objective(self) -> str:
        return "reg:squarederror"

    def score(self, X, y, sample_weight=None) -> float:
        from sklearn.metrics import r2_score
        return r2_score(y, self.predict(X), sample_weight=sample_weight)


class XGBClassifier(_SklClassifierMixin, XGBModel):
    _estimator_type = "classifier"

    def _default_objective(self) -> str:
        return "binary:logistic"

    def fit(self, X, y, **kwargs) -> "XGBClassifier":
        y = np.asarray(y)
        self.classes_ = np.unique(y)
        self.n_classes_ = len(self.classes_)
        y_enc = np.searchsorted(self.classes_, y).astype(np.float32)
        if self.n_classes_ > 2:
            if self.objective is None or str(self.objective).startswith("binary"):
                self.objective = "multi:softprob"
            self.kwargs.setdefault("num_class", self.n_classes_)
        super().fit(X, y_enc, **kwargs)
        return self

    def predict(self, X, *, validate_features=True, base_margin=None,
                iteration_range=None):
        proba = self.predict_proba(X, validate_features=validate_features,
                                   base_margin=base_margin,
                                   iteration_range=iteration_range)
        if proba.ndim == 1:
            idx = (proba > 0.5).astype(np.int64)
        else:
            idx = proba.argmax(axis=1)
        return self.classes_[idx]

    def predict_proba(self, X, *, validate_features=True, base_margin=None,
                      iteration_range=None) -> np.ndarray:
        d = self._make_dmatrix(X, base_margin=base_margin)
        raw = self.get_booster().predict(
            d, validate_features=validate_features,
            iteration_range=iteration_range or self._iteration_range())
        if raw.ndim == 1:
            if getattr(self, "n_classes_", 2) == 2:
                return np.stack([1 - raw, raw], axis=1)
            return raw
        return raw

    def score(self, X, y, sample_weight=None) -> float:
        pred = self.predict(X)
        return float(np.average(pred == np.asarray(y), weights=sample_weight))


class XGBRanker(XGBModel):
    _estimator_type = "ranker"

    def _default_objective(self) -> str:
        return "rank:ndcg"

    def fit(self, X, y, *, group=None, qid=None, sample_weight=None,
            base_margin=None, eval_set=None, eval_group=None, eval_qid=None,
            sample_weight_eval_set=None, base_margin_eval_set=None,
            verbose=False, xgb_model=None) -> "XGBRanker":
        if group is None and qid is None:
            raise ValueError("ranker requires group or qid")
        dtrain = self._make_dmatrix(X, y, sample_weight, base_margin,
                                    group=group, qid=qid)
        evals = []
        if eval_set:
            for i, (ex, ey) in enumerate(eval_set):
                g = eval_group[i] if eval_group else None
                q = eval_qid[i] if eval_qid else None
                evals.append((self._make_dmatrix(ex, ey, group=g, qid=q),
                              f"validation_{i}"))
        params = self.get_xgb_params()
        params.setdefault("objective", self._default_objective())
        self.evals_result_ = {}
        self._Booster = _train(
            params, dtrain, self.n_estimators_, evals=evals,
            early_stopping_rounds=getattr(self, "early_stopping_rounds", None),
            evals_result=self.evals_result_, verbose_eval=verbose,
            xgb_model=xgb_model)
        self._set_fitted_attrs(dtrain)
        return self

    def score(self, X, y, qid=None):
        """Mean NDCG over query groups (reference sklearn.py
        XGBRanker.score semantics)."""
        from .metrics import create_metric
        d = self._make_dmatrix(X, y, qid=qid) if qid is not None \
            else self._make_dmatrix(X, y)
        margin = self.get_booster().predict(d, output_margin=True)
        return float(create_metric("ndcg")(margin, d.info))


class XGBRFRegressor(XGBRegressor):
    """Random-forest-style: one boosting round of num_parallel_tree trees
    (reference sklearn.py:2047)."""

    def __init__(self, *, learning_rate=1.0, subsample=0.8,
                 colsample_bynode=0.8, reg_lambda=1e-5, **kwargs):
        super().__init__(learning_rate=learning_rate, subsample=subsample,
                         colsample_bynode=colsample_bynode,
                         reg_lambda=reg_lambda, **kwargs)

    def fit(self, X, y, **kwargs):
        n = self.n_estimators
        self.n_estimators = 1
        try:
            self.kwargs["num_parallel_tree"] = n or 100
            super().fit(X, y, **kwargs)
        finally:
            self.n_estimators = n
        return self


class XGBRFClassifier(XGBClassifier):
    def __init__(self, *, learning_rate=1.0, subsample=0.8,
                 colsample_bynode=0.8, reg_lambda=1e-5, **kwargs):
        super().__init__(learning_rate=learning_rate, subsample=subsample,
                         colsample_bynode=colsample_bynode,
                         reg_lambda=reg_lambda, **kwargs)

    def fit(self, X, y, **kwargs):
        n = self.n_estimators
        self.n_estimators = 1
        try:
            self.kwargs["num_parallel_tree"] = n or 100
            super().fit(X, y, **kwargs)
        finally:
            self.n_estimators = n
        return self
