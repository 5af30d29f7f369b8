"""Training parameters — parsing, aliases, validation.

Re-provides the behavioral surface of the reference's TrainParam /
LearnerTrainParam (reference: src/tree/param.h, src/learner.cc:154) as a
plain Python dataclass.  All parameter names and defaults match the
reference's documented semantics.
"""
from __future__ import annotations

import dataclasses
from typing import Any, Dict, List, Optional, Sequence

ALIASES = {
    "reg_lambda": "lambda",
    "reg_alpha": "alpha",
    "learning_rate": "eta",
    "min_split_loss": "gamma",
    "colsample": "colsample_bytree",
}

_DEPRECATED_TREE_METHODS = {"gpu_hist": "hist", "gpu_exact": "exact"}


@dataclasses.dataclass
class TrainParam:
    """Tree-building hyperparameters (reference: src/tree/param.h:517)."""

    eta: float = 0.3
    gamma: float = 0.0  # min_split_loss
    max_depth: int = 6
    max_leaves: int = 0
    grow_policy: str = "depthwise"  # or "lossguide"
    min_child_weight: float = 1.0
    reg_lambda: float = 1.0
    reg_alpha: float = 0.0
    subsample: float = 1.0
    sampling_method: str = "uniform"  # or "gradient_based"
    colsample_bytree: float = 1.0
    colsample_bylevel: float = 1.0
    colsample_bynode: float = 1.0
    max_bin: int = 256
    sparse_threshold: float = 0.2
    max_delta_step: float = 0.0
    monotone_constraints: Optional[Sequence[int]] = None
    interaction_constraints: Optional[Sequence[Sequence[int]]] = None
    max_cat_to_onehot: int = 4
    max_cat_threshold: int = 64
    num_parallel_tree: int = 1
    tree_method: str = "auto"  # auto|hist|approx|exact
    max_cached_hist_node: int = 65536
    process_type: str = "default"  # or "update"
    refresh_leaf: bool = True
    updater: Optional[str] = None
    debug_synchronize: bool = False

    @property
    def min_split_loss(self) -> float:
        return self.gamma

    @property
    def learning_rate(self) -> float:
        return self.eta

    def max_nodes(self) -> int:
        if self.grow_policy == "depthwise":
            depth = self.max_depth if self.max_depth > 0 else 31
            return (1 << (depth + 1)) - 1
        n_leaves = self.max_leaves if self.max_leaves > 0 else (1 << 31)
        return 2 * n_leaves - 1


def canonicalize(params: Dict[str, Any]) -> Dict[str, Any]:
    """Resolve aliases into canonical names (last setting wins)."""
    out: Dict[str, Any] = {}
    for k, v in params.items():
        out[ALIASES.get(k, k)] = v
    return out


_BOOL_PARAMS = {"refresh_leaf", "debug_synchronize"}


def make_train_param(params: Dict[str, Any]) -> TrainParam:
    params = canonicalize(params)
    tp = TrainParam()
    for f in dataclasses.fields(TrainParam):
        key = "lambda" if f.name == "reg_lambda" else (
            "alpha" if f.name == "reg_alpha" else f.name)
        if key in params and params[key] is not None:
            v = params[key]
            if f.type in ("float", float):
                v = float(v)
            elif f.type in ("int", int):
                v = int(v)
            elif f.name in _BOOL_PARAMS:
                v = _to_bool(v)
            setattr(tp, f.name, v)
    if tp.tree_method in _DEPRECATED_TREE_METHODS:
        tp.tree_method = _DEPRECATED_TREE_METHODS[tp.tree_method]
    if tp.updater and tp.process_type == "default" \
            and tp.tree_method == "auto":
        # explicit updater sequences select the growth algorithm
        # (reference gbtree.cc specified_updater_): prune/refresh/sync
        # suffixes need no separate pass here — gamma gating during
        # growth subsumes the prune updater's loss_chg test
        _grow_updaters = {"grow_colmaker": "exact",
                          "grow_histmaker": "approx",
                          "grow_quantile_histmaker": "hist",
                          "grow_gpu_hist": "hist",
                          "grow_gpu_approx": "approx"}
        for u in str(tp.updater).split(","):
            u = u.strip()
            if u in _grow_updaters:
                tp.tree_method = _grow_updaters[u]
                break
    _validate(tp)
    return tp


def _to_bool(v: Any) -> bool:
    if isinstance(v, str):
        return v.lower() in ("1", "true", "yes")
    return bool(v)


def _validate(tp: TrainParam) -> None:
    if tp.grow_policy not in ("depthwise", "lossguide"):
        raise ValueError(f"unknown grow_policy: {tp.grow_policy}")
    if tp.tree_method not in ("auto", "hist", "approx", "exact"):
        raise ValueError(f"unknown tree_method: {tp.tree_method}")
    if tp.max_bin < 2:
        raise ValueError("max_bin must be >= 2")
    if tp.eta < 0:
        raise ValueError("eta (learning_rate) must be >= 0 "
                         "(reference param.h:83 lower bound)")
    if tp.max_depth < 0:
        raise ValueError("max_depth must be >= 0 "
                         "(0 = unbounded; reference param.h:90)")
    if not (0.0 < tp.subsample <= 1.0):
        raise ValueError("subsample must be in (0, 1]")
    for name in ("colsample_bytree", "colsample_bylevel", "colsample_bynode"):
        v = getattr(tp, name)
        if not (0.0 < v <= 1.0):
            raise ValueError(f"{name} must be in (0, 1]")
    if tp.max_depth == 0 and tp.max_leaves == 0 and tp.grow_policy == "depthwise":
        raise ValueError("max_depth and max_leaves cannot both be 0")
    if tp.sampling_method not in ("uniform", "gradient_based"):
        raise ValueError(f"unknown sampling_method: {tp.sampling_method}")


# Learner-level parameter names that are consumed outside TrainParam.
LEARNER_PARAMS = {
    "objective", "base_score", "num_class", "num_target", "eval_metric",
    "seed", "seed_per_iteration", "nthread", "n_jobs", "device", "verbosity",
    "booster", "validate_parameters", "disable_default_eval_metric",
    "multi_strategy", "scale_pos_weight", "max_delta_step",
    # objective-specific
    "tweedie_variance_power", "huber_slope", "quantile_alpha",
    "aft_loss_distribution", "aft_loss_distribution_scale",
    "lambdarank_pair_method", "lambdarank_num_pair_per_sample",
    "lambdarank_normalization", "lambdarank_score_normalization",
    "lambdarank_unbiased", "lambdarank_bias_norm", "ndcg_exp_gain",
    "expectile_alpha",
    # DART dropout (gbtree-level, reference gbm/gbtree.h DartTrainParam)
    "rate_drop", "one_drop", "skip_drop", "sample_type", "normalize_type",
    # deprecated device spelling (reference maps gpu_id -> device)
    "gpu_id",
}

_KNOWN = {f.name for f in dataclasses.fields(TrainParam)} | {"lambda", "alpha"} | LEARNER_PARAMS | set(ALIASES)


def check_unknown(params: Dict[str, Any], validate: bool) -> List[str]:
    unknown = [k for k in params if k not in _KNOWN]
    if unknown and validate:
        raise ValueError(f"unknown parameters: {unknown}")
    return unknown
