"""xgboost_amd — an MI355X-native gradient-boosted tree framework.

A from-scratch re-implementation of the capabilities of dmlc/xgboost,
designed for AMD Instinct MI355X (gfx950/CDNA4): hand-written HIP
kernels for the training hot path (quantize, histogram build, split
evaluation, partition, predict), torch tensors for memory management,
and torch.distributed (RCCL over xGMI) for multi-GPU data parallelism.

Public API mirrors the xgboost Python package: DMatrix, QuantileDMatrix,
Booster, train, cv, callbacks, sklearn wrappers.
"""
from .collective import init as collective_init  # noqa: F401
from .config import config_context, get_config, set_config  # noqa: F401
from .core import Booster  # noqa: F401
from .data import DMatrix, QuantileDMatrix  # noqa: F401
from .plotting import plot_importance, plot_tree, to_graphviz  # noqa: F401
from .training import cv, train  # noqa: F401
from . import callback  # noqa: F401
from . import objective  # noqa: F401  (experimental class objectives)
from . import collective  # noqa: F401

__version__ = "0.1.0"


def build_info():
    """Build/runtime info (reference: xgboost.build_info)."""
    import torch
    from . import ops as _ops
    return {
        "version": __version__,
        "rocm_torch": torch.version.hip or "",
        "gfx_arch": "gfx950",
        "native_kernels": _ops.available(),
        "USE_RCCL": True,
    }

__all__ = [
    "Booster", "DMatrix", "QuantileDMatrix", "DataIter",
    "ExtMemQuantileDMatrix", "train", "cv", "callback",
    "collective", "config_context", "set_config", "get_config",
    "plot_importance", "plot_tree", "to_graphviz", "interpret",
    "RabitTracker", "XGBModel", "XGBRegressor", "XGBClassifier",
    "XGBRanker", "XGBRFRegressor", "XGBRFClassifier", "build_info",
]


def _lazy(name):
    if name in ("XGBRegressor", "XGBClassifier", "XGBRanker", "XGBRFRegressor",
                "XGBRFClassifier", "XGBModel"):
        from . import sklearn as _sk
        return getattr(_sk, name)
    if name in ("DataIter", "ExtMemQuantileDMatrix"):
        from . import extmem as _em
        return getattr(_em, name)
    if name == "RabitTracker":
        from .tracker import RabitTracker
        return RabitTracker
    if name == "interpret":
        import importlib
        return importlib.import_module(".interpret", __name__)
    raise AttributeError(name)


def __getattr__(name):
    return _lazy(name)
