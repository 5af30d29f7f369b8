"""ctypes loader for the HIP kernel library (libgbt_hip.so).

The kernels are plain HIP compiled with hipcc for gfx950 — no
compatibility layers.  Tensors are passed as raw device pointers plus
torch's current HIP stream, so kernel launches land on the same stream
as surrounding torch ops (no extra synchronization).

On a GPU machine a missing/unbuildable library is a HARD error — the
HIP path must never silently fall back to eager torch.
"""
from __future__ import annotations

import ctypes
import os
from typing import Optional

import torch

_HERE = os.path.dirname(os.path.abspath(__file__))
_LIB_PATH = os.path.join(_HERE, "libgbt_hip.so")
_lib: Optional[ctypes.CDLL] = None

_c = ctypes
_p = _c.c_void_p
_i = _c.c_int
_i64 = _c.c_int64
_f = _c.c_float
_d = _c.c_double

_SIGS = {
    "gbt_hist": [_p, _p, _i, _p, _p, _p, _i, _p, _i, _p, _p, _i, _i, _p, _i, _p, _p],
    "gbt_partition": [_p, _p, _i, _p, _p, _i64, _p, _p, _p, _i, _p, _p, _p,
                      _p, _p, _p, _p, _p],
    "gbt_evaluate": [_p, _i, _i, _i, _p, _p, _p, _d, _d, _d, _d, _d, _d,
                     _p, _p, _p, _p, _p, _p, _p, _p, _p, _i, _p],
    "gbt_compress": [_p, _i64, _i, _p, _p, _p, _f, _i, _p, _p, _p],
    "gbt_predict": [_p, _i64, _i, _f, _i, _p, _p, _p, _p, _p, _p, _p, _p,
                    _p, _p, _i, _i, _p, _p, _p],
    "gbt_leaf_partition": [_p, _p, _i, _p, _p, _p],
    "gbt_leaf_decide": [_p, _p, _i, _p, _p, _i64, _p, _p, _i, _p, _p, _p,
                        _p, _p, _p, _p],
    "gbt_copy_ranges": [_p, _p, _p, _i, _p],
    "gbt_select_best": [_p, _p, _p, _p, _i, _i, _p, _p, _p],
    "gbt_mt_evaluate": [_p, _i, _i, _i, _i, _p, _p, _p, _p,
                        _d, _d, _d, _d, _p, _p, _p, _p, _p],
    "gbt_gpair_fused": [_i, _p, _p, _p, _f, _i64, _p, _p, _p],
    "gbt_quantize": [_p, _i64, _d, _d, _p, _p, _p, _p],
    "gbt_margin_add": [_p, _p, _p, _i64, _i, _i, _p],
    "gbt_hist_cpu": [_p, _p, _i, _p, _p, _p, _p, _i, _p, _p, _i],
    "gbt_partition_cpu": [_p, _p, _i, _p, _i64, _i64, _i, _i, _i, _p, _i,
                          _i, _p],
    "gbt_evaluate_cpu": [_p, _i, _i, _i, _p, _p, _d, _d, _d, _d, _d, _d,
                         _p, _p, _p, _p],
    "gbt_hist_csr": [_p, _p, _p, _p, _p, _i, _p, _i, _p],
    "gbt_partition_csr": [_p, _p, _p, _p, _p, _i, _p, _p, _p, _p, _p, _p],
    # native level-loop driver
    "gbt_driver_create": [],
    "gbt_driver_destroy": [_p],
    "gbt_grow_tree": [
        _p,                      # ctx
        _p, _p, _i,              # gidx8/16, n_features
        _p, _p,                  # feature-major gidx copies (or null)
        _i64, _p,                # n_rows, qgpair
        _p, _p, _p, _p,          # cut_ptrs_dev, cut_values_host, cut_ptrs_host, n_bins_feat_dev
        _p, _p, _i, _i, _i, _i,  # groups, n_groups, max_group_bins, use_shared, n_bins
        _p, _p, _p, _p,          # ridx, ridx_out, pool_a, pool_b
        _p, _p, _p, _p, _p, _p,  # eval_gain/bin/dir/lsum/best, pos_out
        _i,                      # max_nodes_level
        _p, _p, _i, _p,          # part_counters, hist_tasks_dev, cap, tg_scratch
        _p, _i64, _i,            # wt_ws, wt_ws_bytes, wt_max_ptasks
        _p,                      # root_sums_dev [2] int64
        _p, _p,                  # maxabs_dev [2] f32 | out_scales [2] f64
        _d, _d,                  # scales
        _d, _d, _d, _d, _d, _d,  # lambda, alpha, mds, mcw, gamma, eta
        _i,                      # max_depth
        _p, _p,                  # monotone dev/host
        _p,                      # colsample_bytree feature mask (dev)
        _p,                      # allreduce callback
        _p, _p, _p, _p, _p, _p, _p, _p, _p,  # tree out arrays
        _p,                      # stream
    ],
    "gbt_shap": [_p, _i64, _i, _f, _i, _p, _p, _p, _p, _p, _p, _p, _p, _p,
                 _p, _p, _i, _i, _i, _p, _p, _p],
    "gbt_shap_paths": [_p, _i64, _i, _f, _i, _p, _p, _p, _p, _p, _p,
                       _p, _p, _p, _i64, _i, _i, _p, _p],
    "gbt_shap_paths16": [_p, _i64, _i, _f, _i, _p, _p, _p, _p, _p, _p,
                         _p, _p, _p, _i64, _i, _i, _p, _p],
    "gbt_shap_ix": [_p, _i64, _i, _f, _i, _p, _p, _p, _p, _p, _p, _p, _p,
                    _p, _i64, _i, _i, _p, _p],
    "gbt_hist_mt": [_p, _p, _i, _p, _i, _p, _p, _i, _p, _i, _i,
                    _p, _p, _i, _i, _p, _p],
}


def load() -> ctypes.CDLL:
    global _lib
    if _lib is not None:
        return _lib
    if not os.path.exists(_LIB_PATH):
        try:
            from .build_ext import build
            build()
        except Exception as e:  # noqa: BLE001
            raise RuntimeError(
                f"HIP kernel library missing and build failed: {e}. "
                f"Run `python {os.path.join(_HERE, 'build_ext.py')}`."
            ) from e
    _lib = ctypes.CDLL(_LIB_PATH)
    for name, argtypes in _SIGS.items():
        try:
            fn = getattr(_lib, name)
        except AttributeError:
            continue  # optional kernels (shap) may not be built yet
        fn.argtypes = argtypes
        fn.restype = None
    if hasattr(_lib, "gbt_driver_create"):
        _lib.gbt_driver_create.restype = _p
        _lib.gbt_grow_tree.restype = _c.c_int
    if hasattr(_lib, "gbt_partition_cpu"):
        _lib.gbt_partition_cpu.restype = _c.c_longlong
    return _lib


ALLREDUCE_FN = _c.CFUNCTYPE(None, _c.POINTER(_c.c_longlong), _c.c_longlong)


def available() -> bool:
    try:
        load()
        return True
    except RuntimeError:
        return False


def ptr(t: Optional[torch.Tensor]):
    if t is None:
        return None
    assert t.is_contiguous(), "kernel arg tensors must be contiguous"
    return _c.c_void_p(t.data_ptr())


def stream() -> _c.c_void_p:
    return _c.c_void_p(torch.cuda.current_stream().cuda_stream)
