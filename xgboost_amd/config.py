"""Global configuration (reference: include/xgboost/global_config.h:16,
python-package/xgboost/config.py — config_context / set_config /
get_config)."""
from __future__ import annotations

import contextlib
import threading
from typing import Any, Dict

_DEFAULTS = {"verbosity": 1, "use_rmm": False, "nthread": 0}
_local = threading.local()


def _state() -> Dict[str, Any]:
    if not hasattr(_local, "cfg"):
        _local.cfg = dict(_DEFAULTS)
    return _local.cfg


def set_config(**kwargs) -> None:
    cfg = _state()
    for k, v in kwargs.items():
        if k not in _DEFAULTS:
            raise ValueError(f"unknown global config key: {k}")
        cfg[k] = v


def get_config() -> Dict[str, Any]:
    return dict(_state())


@contextlib.contextmanager
def config_context(**kwargs):
    saved = get_config()
    set_config(**kwargs)
    try:
        yield
    finally:
        _state().update(saved)


def verbosity() -> int:
    return int(_state()["verbosity"])
