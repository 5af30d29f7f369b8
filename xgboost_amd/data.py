"""DMatrix family and MetaInfo.

Reference behavior: include/xgboost/data.h:65 (MetaInfo), data.h:549
(DMatrix), src/data/simple_dmatrix.h, src/data/iterative_dmatrix.h.

Design (MI355X-native): the in-core representation is a dense torch
float32 tensor plus a quantized bin matrix ("Ellpack") built lazily per
max_bin.  With 288 GB HBM3E per GPU the quantized matrix for even very
large datasets is device-resident; the raw float matrix is only needed
for sketching and can stay on host.  Quantized bins are stored row-major
as u8/u16 local (per-feature-relative) bin ids — the reference's
"dense compressed" Ellpack mode (src/data/ellpack_page.cuh:26) — so the
histogram kernel reads 1-2 bytes per (row, feature).
"""
from __future__ import annotations

import dataclasses
from typing import Any, Dict, List, Optional, Sequence

import numpy as np
import torch

from .quantile import HistogramCuts, make_cuts, search_bins


@dataclasses.dataclass
class MetaInfo:
    """Per-dataset metadata (reference: include/xgboost/data.h:65)."""

    num_row: int = 0
    num_col: int = 0
    labels: Optional[np.ndarray] = None          # [n] or [n, n_targets]
    weights: Optional[np.ndarray] = None         # [n]
    base_margin: Optional[np.ndarray] = None     # [n] or [n, n_out]
    group_ptr: Optional[np.ndarray] = None       # [n_groups+1] for ranking
    label_lower_bound: Optional[np.ndarray] = None  # survival
    label_upper_bound: Optional[np.ndarray] = None
    feature_names: Optional[List[str]] = None
    feature_types: Optional[List[str]] = None
    feature_weights: Optional[np.ndarray] = None  # [n_col] colsample weights

    def validate(self) -> None:
        if self.labels is not None and self.labels.shape[0] != self.num_row:
            raise ValueError(
                f"label length {self.labels.shape[0]} != num_row {self.num_row}")
        if self.weights is not None:
            if self.weights.shape[0] != self.num_row and (
                    self.group_ptr is None
                    or self.weights.shape[0] != len(self.group_ptr) - 1):
                raise ValueError("weight length mismatch")
            if np.any(self.weights < 0):
                raise ValueError("weights must be non-negative")


@dataclasses.dataclass
class QuantizedMatrix:
    """Dense quantized bin matrix (the Ellpack analog).

    gidx: [n_rows, n_features] u8/u16/i32 of local bin ids; rows with a
    missing value carry the per-feature sentinel `n_bins(f)` (only when
    has_missing).
    """

    gidx: torch.Tensor
    cuts: HistogramCuts
    has_missing: bool

    @property
    def n_rows(self) -> int:
        return self.gidx.shape[0]

    @property
    def n_features(self) -> int:
        return self.gidx.shape[1]

    @property
    def device(self) -> torch.device:
        return self.gidx.device

    def to(self, device) -> "QuantizedMatrix":
        if torch.device(device) == self.gidx.device:
            return self
        return QuantizedMatrix(self.gidx.to(device), self.cuts, self.has_missing)

    def global_gidx(self) -> torch.Tensor:
        """int64 [n, f] global bin ids, -1 for missing (CPU-oracle form)."""
        ptrs = torch.from_numpy(self.cuts.ptrs).to(self.gidx.device)
        n_bins = (ptrs[1:] - ptrs[:-1])
        g = self.gidx.long()
        missing = g >= n_bins.unsqueeze(0)
        g = g + ptrs[:-1].unsqueeze(0)
        g[missing] = -1
        return g


def _pick_bin_dtype(max_local: int) -> torch.dtype:
    if max_local <= 255:
        return torch.uint8
    if max_local <= 65535:
        return torch.int16  # stored as i16; kernels reinterpret as u16
    return torch.int32


def quantize_dense(X: np.ndarray, cuts: HistogramCuts,
                   missing: float = np.nan) -> QuantizedMatrix:
    """Host-side quantization: numeric bin = #cuts<=v, categorical bin = v."""
    gidx_global = search_bins(X, cuts, missing)  # int32, -1 missing
    has_missing = bool((gidx_global < 0).any())
    offsets = cuts.ptrs[:-1].astype(np.int32)
    n_bins = np.diff(cuts.ptrs).astype(np.int32)
    local = gidx_global - offsets[None, :]
    if has_missing:
        local = np.where(gidx_global < 0, n_bins[None, :], local)
        max_local = int(n_bins.max())
    else:
        max_local = int(n_bins.max()) - 1
    dtype = _pick_bin_dtype(max_local)
    t = torch.from_numpy(local.astype(_np_dtype(dtype), copy=False))
    return QuantizedMatrix(gidx=t, cuts=cuts, has_missing=has_missing)


def _np_dtype(td: torch.dtype):
    return {torch.uint8: np.uint8, torch.int16: np.int16, torch.int32: np.int32}[td]


def make_cuts_device(Xd: torch.Tensor, max_bin: int, missing: float,
                     feature_types=None, weights=None) -> HistogramCuts:
    """Cuts from a device-resident matrix via the weighted device
    quantile sketch (gpu_sketch.DeviceSketch — the SketchContainer
    equivalent, reference src/common/quantile.cuh:41): exact weighted
    ranks + bounded-error prune, rank-merged when distributed."""
    from .gpu_sketch import device_cuts
    return device_cuts(Xd, max_bin, missing=missing, weights=weights,
                       feature_types=(list(feature_types)
                                      if feature_types else None))


def quantize_dense_device(Xd: torch.Tensor, cuts: HistogramCuts,
                          missing: float) -> QuantizedMatrix:
    """GPU compression via the HIP kernel (gbt_compress)."""
    from . import ops as hip_ops
    lib = hip_ops.load()
    n, f = Xd.shape
    dev = Xd.device
    has_missing = bool(torch.isnan(Xd).any().item()) if np.isnan(missing) \
        else bool(((Xd == missing) | torch.isnan(Xd)).any().item())
    n_bins = np.diff(cuts.ptrs).astype(np.int64)
    max_local = int(n_bins.max()) - (0 if has_missing else 1)
    dtype = _pick_bin_dtype(max_local)
    cut_vals = torch.from_numpy(cuts.values).to(dev)
    cut_ptrs = torch.from_numpy(cuts.ptrs.astype(np.int32)).to(dev)
    cat_t = None
    if cuts.feature_types is not None:
        cat_t = torch.tensor([1 if t == "c" else 0
                              for t in cuts.feature_types],
                             dtype=torch.uint8, device=dev)
    out = torch.empty((n, f), dtype=dtype, device=dev)
    missing_is_nan = 1 if np.isnan(missing) else 0
    Xc = Xd.contiguous()
    lib.gbt_compress(
        hip_ops.ptr(Xc), n, f, hip_ops.ptr(cut_vals), hip_ops.ptr(cut_ptrs),
        hip_ops.ptr(cat_t), float(0.0 if missing_is_nan else missing),
        missing_is_nan,
        hip_ops.ptr(out) if dtype == torch.uint8 else None,
        hip_ops.ptr(out) if dtype != torch.uint8 else None,
        hip_ops.stream())
    return QuantizedMatrix(gidx=out, cuts=cuts, has_missing=has_missing)


class DMatrix:
    """User-facing data holder (reference: python Booster/DMatrix API).

    Accepts numpy 2-D arrays, torch tensors, scipy CSR/CSC, pandas
    DataFrames and (nested) lists.  Internally dense float32.
    """

    def __init__(self, data: Any, label: Any = None, *, weight: Any = None,
                 base_margin: Any = None, missing: float = np.nan,
                 feature_names: Optional[Sequence[str]] = None,
                 feature_types: Optional[Sequence[str]] = None,
                 group: Any = None, qid: Any = None,
                 label_lower_bound: Any = None, label_upper_bound: Any = None,
                 nthread: Optional[int] = None, enable_categorical: bool = False,
                 feature_weights: Any = None, silent: bool = False):
        self.missing = float("nan") if missing is None else float(missing)
        self._device_data: Optional[torch.Tensor] = None
        if isinstance(data, torch.Tensor) and data.is_cuda:
            # zero-copy device ingestion (reference CupyAdapter,
            # src/data/device_adapter.cuh); sketch+compress run on GPU
            self._device_data = data.to(torch.float32).contiguous()
            data = None
        elif hasattr(data, "__cuda_array_interface__"):
            self._device_data = torch.as_tensor(data).to(
                torch.float32).contiguous()
            data = None
        inferred_names = inferred_types = None
        if isinstance(data, str) or hasattr(data, "__fspath__"):
            # text file (libsvm format; deprecated upstream, data.cc:930)
            import os as _os
            from .libsvm import load_svmlight
            path = _os.fspath(data).split("?", 1)[0]
            data, file_labels, file_qid = load_svmlight(path)
            if label is None:
                label = file_labels
            if qid is None and file_qid is not None:
                qid = file_qid
        self._sparse_data = None
        if (hasattr(data, "tocsr") and hasattr(data, "nnz")
                and self._device_data is None):
            # scipy sparse: keep CSR — absent entries are MISSING values
            # (reference SparsePage semantics), never densified
            self._sparse_data = data.tocsr().astype(np.float32)
            self._data = None
            n_row, n_col = self._sparse_data.shape
            inferred_names = inferred_types = None
        elif self._device_data is None:
            (X, inferred_names, inferred_types,
             self.categories_) = _ingest(data, enable_categorical)
            self._data = X  # np.float32 [n, f]
            n_row, n_col = X.shape
        else:
            self._data = None
            n_row, n_col = self._device_data.shape
        self.info = MetaInfo(num_row=n_row, num_col=n_col)
        if label is not None:
            self.info.labels = _checked_labels(label)
        if weight is not None:
            self.info.weights = _as_float_array(weight).reshape(-1)
        if base_margin is not None:
            self.info.base_margin = _as_float_array(base_margin)
        if label_lower_bound is not None:
            self.info.label_lower_bound = _as_float_array(label_lower_bound).reshape(-1)
        if label_upper_bound is not None:
            self.info.label_upper_bound = _as_float_array(label_upper_bound).reshape(-1)
        if group is not None:
            g = np.asarray(group, dtype=np.int64).reshape(-1)
            self.info.group_ptr = np.concatenate(
                [[0], np.cumsum(g)]).astype(np.int64)
            if int(self.info.group_ptr[-1]) != self.info.num_row:
                # reference metainfo.h ValidateQueryGroup
                raise ValueError(
                    "Invalid group structure: group sizes sum to "
                    f"{int(self.info.group_ptr[-1])} but the data has "
                    f"{self.info.num_row} rows")
        elif qid is not None:
            q = np.asarray(qid).reshape(-1)
            if np.any(q[1:] < q[:-1]):
                raise ValueError("qid must be sorted in non-decreasing order")
            boundaries = np.nonzero(np.diff(q))[0] + 1
            self.info.group_ptr = np.concatenate(
                [[0], boundaries, [q.size]]).astype(np.int64)
        self.info.feature_names = (list(feature_names) if feature_names
                                   else inferred_names)
        self.info.feature_types = (list(feature_types) if feature_types
                                   else inferred_types)
        if self.info.feature_types is not None and len(self.info.feature_types) != X.shape[1]:
            raise ValueError("feature_types length mismatch")
        if feature_weights is not None:
            self.set_info(feature_weights=feature_weights)
        self.info.validate()
        self._quantized: Dict[int, QuantizedMatrix] = {}
        self._ref_cuts: Optional[HistogramCuts] = None

    # -- reference API surface ------------------------------------------------
    def num_row(self) -> int:
        return self.info.num_row

    def num_col(self) -> int:
        return self.info.num_col

    def get_label(self) -> np.ndarray:
        return (self.info.labels if self.info.labels is not None
                else np.zeros(self.num_row(), dtype=np.float32))

    def get_weight(self) -> np.ndarray:
        return (self.info.weights if self.info.weights is not None
                else np.ones(self.num_row(), dtype=np.float32))

    def get_base_margin(self) -> Optional[np.ndarray]:
        return self.info.base_margin

    def set_info(self, *, label=None, weight=None, base_margin=None,
                 group=None, qid=None, feature_names=None, feature_types=None,
                 label_lower_bound=None, label_upper_bound=None,
                 feature_weights=None) -> None:
        if label is not None:
            self.info.labels = _checked_labels(label)
        if weight is not None:
            self.info.weights = _as_float_array(weight).reshape(-1)
        if base_margin is not None:
            self.info.base_margin = _as_float_array(base_margin)
        if label_lower_bound is not None:
            self.info.label_lower_bound = _as_float_array(label_lower_bound).reshape(-1)
        if label_upper_bound is not None:
            self.info.label_upper_bound = _as_float_array(label_upper_bound).reshape(-1)
        if group is not None:
            g = np.asarray(group, dtype=np.int64).reshape(-1)
            self.info.group_ptr = np.concatenate(
                [[0], np.cumsum(g)]).astype(np.int64)
            if int(self.info.group_ptr[-1]) != self.info.num_row:
                # reference metainfo.h ValidateQueryGroup
                raise ValueError(
                    "Invalid group structure: group sizes sum to "
                    f"{int(self.info.group_ptr[-1])} but the data has "
                    f"{self.info.num_row} rows")
        if qid is not None:
            q = np.asarray(qid).reshape(-1)
            if q.size > 1 and not (q[1:] >= q[:-1]).all():
                # reference data.cc:621
                raise ValueError("`qid` must be sorted in non-decreasing "
                                 "order along with data.")
            boundaries = np.nonzero(np.diff(q))[0] + 1
            self.info.group_ptr = np.concatenate([[0], boundaries, [q.size]]).astype(np.int64)
        if feature_names is not None:
            self.info.feature_names = list(feature_names)
        if feature_types is not None:
            self.info.feature_types = list(feature_types)
        if feature_weights is not None:
            fw = _as_float_array(feature_weights).reshape(-1)
            if fw.size != self.info.num_col:
                raise ValueError(
                    f"feature_weights length {fw.size} != num_col "
                    f"{self.info.num_col}")
            if (fw < 0).any():
                raise ValueError("feature_weights must be non-negative")
            self.info.feature_weights = fw
        self.info.validate()

    # generic info accessors (reference core.py get_float_info/...)
    _FLOAT_FIELDS = {"label": "labels", "weight": "weights",
                     "base_margin": "base_margin",
                     "label_lower_bound": "label_lower_bound",
                     "label_upper_bound": "label_upper_bound"}

    # deprecated per-field setters kept for API compatibility
    # (reference core.py set_label/set_weight/set_base_margin/set_group)
    def set_label(self, label) -> None:
        self.set_info(label=label)

    def set_weight(self, weight) -> None:
        self.set_info(weight=weight)

    def set_base_margin(self, margin) -> None:
        self.set_info(base_margin=margin)

    def set_group(self, group) -> None:
        self.set_info(group=group)

    def get_float_info(self, field: str) -> np.ndarray:
        if field in self._FLOAT_FIELDS:
            v = getattr(self.info, self._FLOAT_FIELDS[field])
            return (np.array([], dtype=np.float32) if v is None
                    else np.asarray(v, dtype=np.float32).reshape(-1))
        raise ValueError(f"unknown float field: {field}")

    def set_float_info(self, field: str, data) -> None:
        if field in self._FLOAT_FIELDS:
            self.set_info(**{field: data})
            return
        raise ValueError(f"unknown float field: {field}")

    set_float_info_npy2d = set_float_info

    def get_uint_info(self, field: str) -> np.ndarray:
        if field in ("group_ptr", "group"):
            g = self.info.group_ptr
            return (np.array([], dtype=np.uint32) if g is None
                    else np.asarray(g, dtype=np.uint32))
        raise ValueError(f"unknown uint field: {field}")

    def set_uint_info(self, field: str, data) -> None:
        if field == "group":
            self.set_info(group=data)
            return
        raise ValueError(f"unknown uint field: {field}")

    def get_group(self) -> np.ndarray:
        g = self.info.group_ptr
        return (np.array([], dtype=np.int64) if g is None
                else np.diff(np.asarray(g, dtype=np.int64)))

    def get_data(self):
        """The feature matrix as scipy CSR (reference DMatrix.get_data);
        missing entries are absent from the CSR."""
        from scipy import sparse as sp
        if self._sparse_data is not None:
            return self._sparse_data.copy()
        if self._data is None:
            raise RuntimeError("no host feature data available")
        X = self._data
        mask = ~np.isnan(X) if np.isnan(self.missing) else X != self.missing
        return _dense_to_csr(X, mask)

    def num_nonmissing(self) -> int:
        if self._sparse_data is not None:
            return int(self._sparse_data.nnz)
        if self._data is None:
            return self.num_row() * self.num_col()
        X = self._data
        mask = ~np.isnan(X) if np.isnan(self.missing) else X != self.missing
        return int(mask.sum())

    def get_quantile_cut(self):
        """(indptr, values) of the quantile cuts (reference
        DMatrix.get_quantile_cut); requires a quantized matrix."""
        cuts = self.cached_cuts()
        if cuts is None:
            if not self._quantized:
                raise RuntimeError(
                    "no quantile cuts yet: train or quantize first")
            cuts = next(iter(self._quantized.values())).cuts
        return (np.asarray(cuts.ptrs, dtype=np.uint64),
                np.asarray(cuts.values, dtype=np.float32))

    def get_categories(self, export_to_arrow: bool = False):
        if export_to_arrow:
            raise NotImplementedError("arrow export is not supported")
        cats = getattr(self, "categories_", None)
        if not cats:
            return None
        names = self.info.feature_names
        return {(names[f] if names else f"f{f}"): list(v)
                for f, v in cats.items()}

    set_label = lambda self, label: self.set_info(label=label)
    set_weight = lambda self, weight: self.set_info(weight=weight)
    set_base_margin = lambda self, m: self.set_info(base_margin=m)
    set_group = lambda self, group: self.set_info(group=group)

    @property
    def feature_names(self):
        return self.info.feature_names

    @property
    def feature_types(self):
        return self.info.feature_types

    def save_binary(self, fname: str, silent: bool = True) -> None:
        """Binary DMatrix cache (reference XGDMatrixSaveBinary); ours is
        an npz with the dense matrix + metadata."""
        payload = {"X": self.raw_data(), "missing": self.missing}
        for k in ("labels", "weights", "base_margin", "group_ptr"):
            v = getattr(self.info, k)
            if v is not None:
                payload[k] = v
        if self.info.feature_names:
            payload["feature_names"] = np.asarray(self.info.feature_names)
        if self.info.feature_types:
            payload["feature_types"] = np.asarray(self.info.feature_types)
        np.savez_compressed(fname, **payload)

    @classmethod
    def load_binary(cls, fname: str) -> "DMatrix":
        z = np.load(fname, allow_pickle=False)
        d = cls(z["X"], missing=float(z["missing"]))
        if "labels" in z:
            d.info.labels = z["labels"]
        if "weights" in z:
            d.info.weights = z["weights"]
        if "base_margin" in z:
            d.info.base_margin = z["base_margin"]
        if "group_ptr" in z:
            d.info.group_ptr = z["group_ptr"]
        if "feature_names" in z:
            d.info.feature_names = [str(x) for x in z["feature_names"]]
        if "feature_types" in z:
            d.info.feature_types = [str(x) for x in z["feature_types"]]
        return d

    def slice(self, rindex: Sequence[int]) -> "DMatrix":
        idx = np.asarray(rindex, dtype=np.int64)
        out = DMatrix(self._data[idx], missing=self.missing,
                      feature_names=self.info.feature_names,
                      feature_types=self.info.feature_types)
        if self.info.labels is not None:
            out.info.labels = self.info.labels[idx]
        if self.info.weights is not None:
            out.info.weights = self.info.weights[idx]
        if self.info.base_margin is not None:
            out.info.base_margin = self.info.base_margin[idx]
        return out

    # -- internal --------------------------------------------------------------
    def raw_data(self) -> np.ndarray:
        if self._data is None and self._device_data is not None:
            self._data = np.ascontiguousarray(
                self._device_data.cpu().numpy(), dtype=np.float32)
        return self._data

    def device_data(self) -> Optional[torch.Tensor]:
        return self._device_data

    def sparse_data(self):
        return self._sparse_data

    def sparse_quantized(self, max_bin: int):
        """Quantized CSR (sparse path; see sparse.py)."""
        key = ("sparse", max_bin)
        sqm = self._quantized.get(key)
        if sqm is None:
            from .sparse import quantize_csr, sketch_csr
            cuts = self._ref_cuts or sketch_csr(self._sparse_data, max_bin)
            sqm = quantize_csr(self._sparse_data, cuts)
            self._quantized[key] = sqm
        return sqm

    def set_ref_cuts(self, cuts: HistogramCuts) -> None:
        """Bin this matrix with cut points from a training DMatrix
        (reference: GetCutsFromRef, src/data/quantile_dmatrix.cc:19)."""
        self._ref_cuts = cuts
        self._quantized.clear()

    def quantized(self, max_bin: int, sketch_fn=None) -> QuantizedMatrix:
        """Lazily build (and cache) the quantized matrix for max_bin."""
        qm = self._quantized.get(max_bin)
        if qm is None:
            if self._device_data is not None:
                qm = self._quantize_on_device(max_bin)
            else:
                if self._ref_cuts is not None:
                    cuts = self._ref_cuts
                elif sketch_fn is not None:
                    cuts = sketch_fn(self, max_bin)
                else:
                    cuts = make_cuts(self._data, max_bin,
                                     weights=None,
                                     feature_types=self.info.feature_types,
                                     missing=self.missing)
                qm = quantize_dense(self._data, cuts, self.missing)
            self._quantized[max_bin] = qm
        return qm

    def _quantize_on_device(self, max_bin: int) -> QuantizedMatrix:
        """GPU sketch + compress for device-resident input (reference
        AdapterDeviceSketch hist_util.cuh:323 + CompressBinEllpackKernel)."""
        cuts = self._ref_cuts or make_cuts_device(
            self._device_data, max_bin, self.missing,
            self.info.feature_types)
        return quantize_dense_device(self._device_data, cuts, self.missing)

    def cached_cuts(self) -> Optional[HistogramCuts]:
        for qm in self._quantized.values():
            return qm.cuts
        return self._ref_cuts


class QuantileDMatrix(DMatrix):
    """Quantized-only DMatrix (reference: src/data/iterative_dmatrix.h:34).

    Builds the bin matrix at construction; `ref` shares cut points with a
    training matrix so validation data is binned identically.
    """

    def __init__(self, data: Any, label: Any = None, *, max_bin: int = 256,
                 ref: Optional[DMatrix] = None, **kwargs):
        from .extmem import DataIter, _drive
        if isinstance(data, DataIter):
            # in-core QuantileDMatrix from an iterator: concatenate
            # batches (reference IterativeDMatrix two-pass; with 288 GB
            # HBM in-core concat is the right default)
            Xs, ys, ws = [], [], []
            for batch in _drive(data):
                Xs.append(np.ascontiguousarray(batch["data"], np.float32))
                if batch.get("label") is not None:
                    ys.append(np.asarray(batch["label"], np.float32))
                if batch.get("weight") is not None:
                    ws.append(np.asarray(batch["weight"], np.float32))
            data = np.concatenate(Xs)
            if label is None and ys:
                label = np.concatenate(ys)
            if ws and "weight" not in kwargs:
                kwargs["weight"] = np.concatenate(ws)
        super().__init__(data, label, **kwargs)
        self.max_bin = max_bin
        if ref is not None:
            cuts = ref.cached_cuts()
            if cuts is None:
                cuts = ref.quantized(max_bin).cuts
            self.set_ref_cuts(cuts)
        self.quantized(max_bin)


def _as_float_array(v: Any) -> np.ndarray:
    if isinstance(v, torch.Tensor):
        return v.detach().cpu().numpy().astype(np.float32, copy=False)
    return np.asarray(v, dtype=np.float32)



def _checked_labels(label) -> np.ndarray:
    """reference data.cc:566 LabelsCheck: labels must be finite (AFT
    censoring bounds live in label_lower/upper_bound, not here)."""
    lab = _as_float_array(label)
    if lab.size and not np.isfinite(lab).all():
        raise ValueError("Label contains NaN, infinity or a value too large.")
    return lab


def _ingest(data: Any, enable_categorical: bool):
    """Normalize input to dense np.float32 [n, f]; returns (X, names, types)."""
    names = None
    types = None
    if isinstance(data, DMatrix):
        raise TypeError("cannot construct a DMatrix from a DMatrix")
    if isinstance(data, torch.Tensor):
        X = data.detach().cpu().numpy()
    elif hasattr(data, "toarray") and hasattr(data, "tocsr"):  # scipy sparse
        X = data.toarray()
    elif _is_pandas(data):
        X, names, types, cats = _from_pandas(data, enable_categorical)
        if X.ndim == 1:
            X = X.reshape(-1, 1)
        return np.ascontiguousarray(X, dtype=np.float32), names, types, cats
    elif _is_arrow(data):
        # Arrow table / record batch (reference _from_arrow_table,
        # python-package/xgboost/data.py:836): columnar ingestion with
        # dictionary columns as categoricals
        return _from_arrow(data, enable_categorical)
    else:
        X = np.asarray(data)
    if X.ndim == 1:
        X = X.reshape(-1, 1)
    if X.ndim != 2:
        raise ValueError(f"expected 2-D data, got shape {X.shape}")
    X = np.ascontiguousarray(X, dtype=np.float32)
    return X, names, types, {}


def _is_pandas(data: Any) -> bool:
    return type(data).__module__.startswith("pandas") and hasattr(data, "dtypes")


def _is_arrow(data: Any) -> bool:
    mod = type(data).__module__
    return mod.startswith("pyarrow") and hasattr(data, "column_names")


def _from_arrow(table, enable_categorical: bool):
    """pyarrow Table/RecordBatch -> dense float32 + names/types/categories.

    Dictionary-encoded columns become categoricals (codes as values,
    like the pandas path); nulls become NaN (missing)."""
    import pyarrow as pa
    names = [str(c) for c in table.column_names]
    types: List[str] = []
    cols = []
    categories: Dict[int, list] = {}
    for j, name in enumerate(table.column_names):
        col = table.column(name)
        if isinstance(col, pa.ChunkedArray):
            col = col.combine_chunks()
        if pa.types.is_dictionary(col.type):
            if not enable_categorical:
                raise ValueError(
                    f"categorical (dictionary) column {name!r} needs "
                    f"enable_categorical=True")
            categories[j] = col.dictionary.to_pylist()
            codes = col.indices.to_numpy(zero_copy_only=False)
            codes = codes.astype(np.float32)
            null_mask = ~np.asarray(col.is_valid())
            codes[null_mask] = np.nan
            cols.append(codes)
            types.append("c")
        else:
            arr = col.to_numpy(zero_copy_only=False).astype(np.float32)
            null_mask = ~np.asarray(col.is_valid())
            if null_mask.any():
                arr = arr.copy()
                arr[null_mask] = np.nan
            cols.append(arr)
            types.append("float")
    X = np.ascontiguousarray(np.stack(cols, axis=1), dtype=np.float32)
    return X, names, types, categories


def _from_pandas(df, enable_categorical: bool):
    import pandas as pd
    names = [str(c) for c in df.columns]
    types: List[str] = []
    cols = []
    categories: Dict[int, list] = {}
    for j, c in enumerate(df.columns):
        s = df[c]
        if isinstance(s.dtype, pd.CategoricalDtype):
            if not enable_categorical:
                raise ValueError(
                    f"categorical column {c!r} needs enable_categorical=True")
            codes = s.cat.codes.to_numpy(np.float32)
            codes = np.where(codes < 0, np.nan, codes)  # NaN category
            cols.append(codes)
            categories[j] = list(s.cat.categories)
            types.append("c")
        else:
            import pandas.api.types as pdt
            # accepts numpy AND pandas nullable extension dtypes
            # (Int64/Float64/boolean, NA -> NaN); anything non-numeric
            # is rejected like the reference data adapter
            if pdt.is_bool_dtype(s.dtype):
                kind = "int"
            elif pdt.is_integer_dtype(s.dtype):
                kind = "int"
            elif pdt.is_float_dtype(s.dtype):
                kind = "float"
            else:
                raise ValueError(
                    "DataFrame.dtypes for data must be int, float, bool "
                    f"or category; column {c!r} is {s.dtype}")
            v = s.to_numpy(dtype=np.float32, na_value=np.nan)
            cols.append(v)
            types.append(kind)
    X = np.stack(cols, axis=1)
    if "c" not in types:
        types_out = None
    else:
        types_out = ["c" if t == "c" else "q" for t in types]
    return X, names, types_out, categories


def align_categories(X: np.ndarray, pred_cats: Dict[int, list],
                     train_cats: Dict[int, list]) -> np.ndarray:
    """Re-code categorical codes from a prediction frame's dictionary to
    the training dictionary (reference: src/encoder/ordinal.h:349 Recode,
    data/cat_container.h).  Unseen categories become missing (NaN)."""
    X = X.copy()
    for j, cats in pred_cats.items():
        tcats = train_cats.get(j)
        if tcats is None or list(cats) == list(tcats):
            continue
        lut = {c: i for i, c in enumerate(tcats)}
        remap = np.full(len(cats), np.nan, dtype=np.float32)
        for i, c in enumerate(cats):
            if c in lut:
                remap[i] = lut[c]
        col = X[:, j]
        valid = ~np.isnan(col)
        idx = col[valid].astype(np.int64)
        idx = np.clip(idx, 0, len(cats) - 1)
        col[valid] = remap[idx]
        X[:, j] = col
    return X


# deprecated alias kept for API compatibility (reference: core.py)
DeviceQuantileDMatrix = QuantileDMatrix


def _dense_to_csr(X: np.ndarray, mask: np.ndarray):
    from scipy import sparse as sp
    indptr = np.zeros(X.shape[0] + 1, dtype=np.int64)
    np.cumsum(mask.sum(axis=1), out=indptr[1:])
    rows, cols = np.nonzero(mask)
    return sp.csr_matrix((X[rows, cols], cols, indptr), shape=X.shape)


class _ProxyMatrix:
    """Zero-copy inplace-predict proxy (reference ProxyDMatrix,
    src/data/proxy_dmatrix.h): wraps a user array without building a
    DMatrix.  Quacks like DMatrix for the predictor paths only
    (num_row/num_col/raw_data/device_data/missing/info)."""

    def __init__(self):
        self.info = MetaInfo()
        self.missing = float("nan")
        self._device_data: Optional[torch.Tensor] = None
        self._host: Optional[np.ndarray] = None
        self.feature_names = None
        self.feature_types = None

    @staticmethod
    def wrap(data, missing, base_margin, device) -> Optional["_ProxyMatrix"]:
        p = _ProxyMatrix()
        p.missing = float(missing)
        if isinstance(data, torch.Tensor):
            if data.dim() != 2:
                return None
            t = data.to(torch.float32)
            if not t.is_contiguous():
                t = t.contiguous()
            if t.is_cuda:
                p._device_data = t
            else:
                p._host = t.numpy()
        elif hasattr(data, "__cuda_array_interface__"):
            t = torch.as_tensor(data, device=device)
            if t.dim() != 2:
                return None
            t = t.to(torch.float32)
            p._device_data = t if t.is_contiguous() else t.contiguous()
        elif isinstance(data, np.ndarray):
            if data.ndim != 2:
                return None
            p._host = np.ascontiguousarray(data, np.float32)
        else:
            return None  # DataFrame / CSR etc: full DMatrix path
        n, c = (p._device_data.shape if p._device_data is not None
                else p._host.shape)
        p.info.num_row, p.info.num_col = int(n), int(c)
        if base_margin is not None:
            p.info.base_margin = np.asarray(base_margin, np.float32)
        return p

    def num_row(self) -> int:
        return self.info.num_row

    def num_col(self) -> int:
        return self.info.num_col

    def raw_data(self) -> np.ndarray:
        if self._host is None and self._device_data is not None:
            self._host = self._device_data.cpu().numpy()
        return self._host

    def device_data(self) -> Optional[torch.Tensor]:
        return self._device_data
