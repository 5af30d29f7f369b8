"""Sparse (CSR) quantized data path.

Reference behavior: sparse SparsePage/EllpackPage training where absent
entries are missing values routed by the learned default direction
(include/xgboost/data.h SparsePage; EllpackPage sparse mode with
binary-search row access, src/data/ellpack_page.cuh:103).

MI355X design: instead of the reference's fixed-row-stride bit-packed
ELLPACK we keep a quantized CSR — row_ptr + per-nonzero GLOBAL bin id
(sorted within a row because column indices are sorted).  The histogram
kernel walks each row's nonzeros; the partition kernel binary-searches
the split feature's bin range inside the row slice.  Same int64
fixed-point determinism as the dense path; the split-evaluation kernel
is shared unchanged.
"""
from __future__ import annotations

import dataclasses
from typing import List, Tuple

import numpy as np
import torch

from .quantile import HistogramCuts
from .splits import SplitEntry, evaluate_splits_np


@dataclasses.dataclass
class SparseQuantizedMatrix:
    """CSR of global bin ids (sorted per row)."""

    row_ptr: torch.Tensor   # int64 [n+1]
    bin_idx: torch.Tensor   # int32 [nnz], global bins
    cuts: HistogramCuts
    n_features: int

    @property
    def n_rows(self) -> int:
        return self.row_ptr.shape[0] - 1

    @property
    def device(self):
        return self.bin_idx.device

    def to(self, device) -> "SparseQuantizedMatrix":
        if torch.device(device) == self.bin_idx.device:
            return self
        return SparseQuantizedMatrix(self.row_ptr.to(device),
                                     self.bin_idx.to(device), self.cuts,
                                     self.n_features)


def sketch_csr(X_csr, max_bin: int) -> HistogramCuts:
    """Cuts from the nonzero values of each column (absent = missing).
    Distributed: per-rank column summaries are allgathered and merged so
    every rank quantizes with IDENTICAL cuts (same guarantee as the
    dense sketch; rank-local cuts would break the histogram allreduce).
    """
    from . import collective
    csc = X_csr.tocsc()
    n_features = csc.shape[1]
    distributed = collective.get_world_size() > 1
    # fast path: single-valued data (one-hot / binary indicator matrices
    # at Criteo scale have 1e6 columns — the per-column loop would
    # dominate).  Every column's cuts = [v + |v| + 1e-5] (one bin).
    # Distributed: the single-value test must be GLOBAL.
    lo = float(csc.data.min()) if csc.nnz else np.inf
    hi = float(csc.data.max()) if csc.nnz else -np.inf
    if distributed:
        lo = -collective.allreduce_max_scalars([-lo])[0]
        (hi,) = collective.allreduce_max_scalars([hi])
    if csc.nnz and csc.data.size and lo == hi:
        v = float(csc.data.flat[0])
        sentinel = np.float32(v + (abs(v) + 1e-5))
        values = np.full(n_features, sentinel, dtype=np.float32)
        ptrs = np.arange(n_features + 1, dtype=np.int64)
        min_vals = np.full(n_features, np.float32(v), dtype=np.float32)
        return HistogramCuts(values=values, ptrs=ptrs, min_vals=min_vals)
    if distributed:
        # per-column quantile summaries, merged like the dense path
        from .sketch import cuts_from_summaries
        K = max(64, 8 * max_bin)
        qs = (np.arange(K) + 0.5) / K
        local = []
        for f in range(n_features):
            vals = csc.data[csc.indptr[f]:csc.indptr[f + 1]].astype(
                np.float32)
            vals = vals[~np.isnan(vals)]
            if vals.size == 0:
                local.append(("q", np.zeros(0, np.float32), 0.0))
            else:
                pts = np.quantile(vals, qs).astype(np.float32)
                local.append(("q", np.concatenate(
                    [[np.float32(vals.min())], pts,
                     [np.float32(vals.max())]]), float(vals.size)))
        gathered = collective.allgather_obj(local)
        return cuts_from_summaries(gathered, max_bin, n_features, None)
    all_cuts: List[np.ndarray] = []
    min_vals = np.zeros(n_features, dtype=np.float32)
    from .quantile import _cuts_for_column
    for f in range(n_features):
        vals = csc.data[csc.indptr[f]:csc.indptr[f + 1]].astype(np.float32)
        vals = vals[~np.isnan(vals)]
        cuts = _cuts_for_column(vals, None, max_bin)
        min_vals[f] = float(vals.min()) if vals.size else 0.0
        all_cuts.append(cuts)
    ptrs = np.zeros(n_features + 1, dtype=np.int64)
    np.cumsum([c.size for c in all_cuts], out=ptrs[1:])
    return HistogramCuts(values=np.concatenate(all_cuts).astype(np.float32),
                         ptrs=ptrs, min_vals=min_vals)


def quantize_csr(X_csr, cuts: HistogramCuts) -> SparseQuantizedMatrix:
    csr = X_csr.tocsr()
    csr.sort_indices()
    n, f = csr.shape
    # fast path: one bin per feature -> global bin == column index
    if cuts.total_bins == f and np.array_equal(
            cuts.ptrs, np.arange(f + 1, dtype=cuts.ptrs.dtype)):
        return SparseQuantizedMatrix(
            row_ptr=torch.from_numpy(csr.indptr.astype(np.int64)),
            bin_idx=torch.from_numpy(csr.indices.astype(np.int32)),
            cuts=cuts, n_features=f)
    bins = np.empty(csr.nnz, dtype=np.int32)
    # per column quantization via CSC, then map back by position
    csc = csr.tocsc()
    csc_bins = np.empty(csc.nnz, dtype=np.int32)
    for j in range(f):
        s, e = csc.indptr[j], csc.indptr[j + 1]
        if e > s:
            fc = cuts.feature_cuts(j)
            local = np.searchsorted(fc, csc.data[s:e], side="right")
            np.clip(local, 0, fc.size - 1, out=local)
            csc_bins[s:e] = local + int(cuts.ptrs[j])
    # convert csc ordering back to csr ordering
    import scipy.sparse as sp
    tmp = sp.csc_matrix((csc_bins.astype(np.float64), csc.indices,
                         csc.indptr), shape=(n, f)).tocsr()
    tmp.sort_indices()
    bins = tmp.data.astype(np.int32)
    return SparseQuantizedMatrix(
        row_ptr=torch.from_numpy(csr.indptr.astype(np.int64)),
        bin_idx=torch.from_numpy(bins), cuts=cuts, n_features=f)


class _SparseQMView:
    """Duck-typed stand-in for QuantizedMatrix attributes the grower
    reads (cuts / n_features)."""

    def __init__(self, sqm: SparseQuantizedMatrix):
        self.cuts = sqm.cuts
        self.n_features = sqm.n_features
        self.has_missing = True


class CsrCpuOps:
    """CPU ops over the quantized CSR (numpy oracle + CPU training)."""

    device = torch.device("cpu")

    def __init__(self, sqm: SparseQuantizedMatrix):
        self.sqm = sqm
        self.qm = _SparseQMView(sqm)
        self.n_bins = sqm.cuts.total_bins
        self._indptr = sqm.row_ptr.numpy()
        self._bins = sqm.bin_idx.numpy()

    # -- stateful interface ------------------------------------------------
    def reset(self, n_rows: int) -> None:
        self.ridx = np.arange(n_rows, dtype=np.int64)
        self.segments = {0: (0, n_rows)}
        self._n_rows = n_rows

    def node_size(self, nid: int) -> int:
        s, e = self.segments[nid]
        return e - s

    def root_sum(self, qgpair: torch.Tensor) -> Tuple[int, int]:
        from . import collective
        s = qgpair.to(torch.int64).sum(dim=0)
        collective.allreduce_sum_(s)
        return int(s[0]), int(s[1])

    def build_hist_nodes(self, qgpair: torch.Tensor, nids) -> torch.Tensor:
        q = qgpair.cpu().numpy()
        out = np.zeros((len(nids), self.n_bins, 2), dtype=np.int64)
        for i, nid in enumerate(nids):
            s, e = self.segments[nid]
            rows = self.ridx[s:e]
            # gather all nnz of these rows
            starts = self._indptr[rows]
            ends = self._indptr[rows + 1]
            lens = ends - starts
            total = int(lens.sum())
            if total == 0:
                continue
            idx = np.repeat(starts - np.cumsum(lens) + lens, lens) \
                + np.arange(total)
            bins = self._bins[idx]
            row_of = np.repeat(rows, lens)
            np.add.at(out[i, :, 0], bins, q[row_of, 0].astype(np.int64))
            np.add.at(out[i, :, 1], bins, q[row_of, 1].astype(np.int64))
        return torch.from_numpy(out)

    def allreduce_hist(self, hist: torch.Tensor) -> torch.Tensor:
        from . import collective
        collective.allreduce_sum_(hist)
        return hist

    def evaluate_splits(self, hist, quantizer, parent_sums, nids, param,
                        feature_sets=None, monotone=None, cat_mask=None,
                        node_bounds=None) -> List[SplitEntry]:
        return evaluate_splits_np(hist.cpu().numpy(), parent_sums,
                                  quantizer.g_scale, quantizer.h_scale,
                                  nids, self.sqm.cuts.ptrs, param,
                                  feature_sets=feature_sets,
                                  monotone=monotone, cat_mask=cat_mask,
                                  node_bounds=node_bounds)

    def _row_bin(self, rows: np.ndarray, feature: int) -> np.ndarray:
        """Global bin of `feature` for each row, or -1 when absent."""
        lo = int(self.sqm.cuts.ptrs[feature])
        hi = int(self.sqm.cuts.ptrs[feature + 1])
        out = np.full(rows.shape, -1, dtype=np.int64)
        for k, r in enumerate(rows):
            s, e = self._indptr[r], self._indptr[r + 1]
            seg = self._bins[s:e]
            j = np.searchsorted(seg, lo, side="left")
            if j < seg.size and lo <= seg[j] < hi:
                out[k] = seg[j]
        return out

    def partition_nodes(self, parents, splits, children) -> None:
        for pnid, sp, (l, r) in zip(parents, splits, children):
            s, e = self.segments[pnid]
            rows = self.ridx[s:e]
            bins = self._row_bin(rows, sp.feature)
            missing = bins < 0
            go_left = np.where(missing, sp.default_left, bins <= sp.split_bin)
            left_rows = rows[go_left]
            right_rows = rows[~go_left]
            nl = len(left_rows)
            self.ridx[s:s + nl] = left_rows
            self.ridx[s + nl:e] = right_rows
            self.segments[l] = (s, s + nl)
            self.segments[r] = (s + nl, e)

    def leaf_positions(self, leaf_nids) -> torch.Tensor:
        pos = np.zeros(self._n_rows, dtype=np.int32)
        for nid in leaf_nids:
            if nid in self.segments:
                s, e = self.segments[nid]
                pos[self.ridx[s:e]] = nid
        return torch.from_numpy(pos)


class CsrGpuOps(CsrCpuOps):
    """GPU ops over quantized CSR — HIP kernels gbt_hist_csr /
    gbt_partition_csr; shares the dense split-evaluation kernel."""

    def __init__(self, sqm: SparseQuantizedMatrix, device):
        from . import ops as hip_ops
        self.lib = hip_ops.load()
        self.hip = hip_ops
        sqm = sqm.to(device)
        self.sqm = sqm
        self.qm = _SparseQMView(sqm)
        self.device = torch.device(device)
        self.n_bins = sqm.cuts.total_bins
        dev = self.device
        self.cut_ptrs = torch.from_numpy(
            sqm.cuts.ptrs.astype(np.int32)).to(dev)
        from .backend.gpu import _PinnedStager
        self.stager = _PinnedStager(dev)

    def reset(self, n_rows: int) -> None:
        self.ridx = torch.arange(n_rows, dtype=torch.int32,
                                 device=self.device)
        self._ridx_out = torch.empty_like(self.ridx)
        self.segments = {0: (0, n_rows)}
        self._n_rows = n_rows

    def root_sum(self, qgpair: torch.Tensor) -> Tuple[int, int]:
        from . import collective
        s = qgpair.to(torch.int64).sum(dim=0)
        collective.allreduce_sum_(s)
        h = s.cpu()
        return int(h[0]), int(h[1])

    def build_hist_nodes(self, qgpair: torch.Tensor, nids) -> torch.Tensor:
        from .backend.gpu import _chunk_tasks
        segs = [self.segments[n] for n in nids]
        out = torch.zeros((len(nids), self.n_bins, 2), dtype=torch.int64,
                          device=self.device)
        tasks_np = _chunk_tasks(segs)
        (tasks,) = self.stager.upload([tasks_np])
        self.lib.gbt_hist_csr(
            self.hip.ptr(self.sqm.row_ptr), self.hip.ptr(self.sqm.bin_idx),
            self.hip.ptr(qgpair), self.hip.ptr(self.ridx),
            self.hip.ptr(tasks), len(tasks_np), self.hip.ptr(out),
            self.n_bins, self.hip.stream())
        return out

    def evaluate_splits(self, hist, quantizer, parent_sums, nids, param,
                        feature_sets=None, monotone=None, cat_mask=None,
                        node_bounds=None):
        from .backend.gpu import GpuOps
        # reuse the dense GPU evaluator via a tiny shim object
        return GpuOps.evaluate_splits(
            self, hist, quantizer, parent_sums, nids, param,
            feature_sets=feature_sets, monotone=monotone, cat_mask=None,
            node_bounds=node_bounds)

    @property
    def cat_feature(self):
        return None

    def partition_nodes(self, parents, splits, children) -> None:
        from .backend.gpu import _chunk_tasks
        k = len(parents)
        segs = [self.segments[p] for p in parents]
        feat = np.array([sp.feature for sp in splits], np.int32)
        sbin = np.array([sp.split_bin for sp in splits], np.int32)  # GLOBAL
        dleft = np.array([1 if sp.default_left else 0 for sp in splits],
                         np.uint8)
        counters = np.array(segs, np.int32)
        tasks_np = _chunk_tasks(segs)
        tasks, feat_t, sbin_t, dleft_t, cnt_t = self.stager.upload(
            [tasks_np, feat, sbin, dleft, counters])
        self.lib.gbt_partition_csr(
            self.hip.ptr(self.sqm.row_ptr), self.hip.ptr(self.sqm.bin_idx),
            self.hip.ptr(self.ridx), self.hip.ptr(self._ridx_out),
            self.hip.ptr(tasks), len(tasks_np), self.hip.ptr(feat_t),
            self.hip.ptr(sbin_t), self.hip.ptr(dleft_t),
            self.hip.ptr(self.cut_ptrs), self.hip.ptr(cnt_t),
            self.hip.stream())
        self.lib.gbt_copy_ranges(
            self.hip.ptr(self._ridx_out), self.hip.ptr(self.ridx),
            self.hip.ptr(tasks), len(tasks_np), self.hip.stream())
        final = cnt_t.cpu().numpy()
        for i, (p, (l, r)) in enumerate(zip(parents, children)):
            s, e = segs[i]
            mid = int(final[i, 0])
            self.segments[l] = (s, mid)
            self.segments[r] = (mid, e)

    def leaf_positions(self, leaf_nids) -> torch.Tensor:
        from .backend.gpu import _chunk_tasks
        pos = torch.zeros(self._n_rows, dtype=torch.int32,
                          device=self.device)
        segs = [(nid, *self.segments[nid]) for nid in leaf_nids
                if nid in self.segments]
        if not segs:
            return pos
        tasks_np = _chunk_tasks([(s, e) for _, s, e in segs])
        leaf_np = np.asarray([nid for nid, _, _ in segs], np.int32)
        tasks, leaf_ids = self.stager.upload([tasks_np, leaf_np])
        self.lib.gbt_leaf_partition(
            self.hip.ptr(self.ridx), self.hip.ptr(tasks), len(tasks_np),
            self.hip.ptr(leaf_ids), self.hip.ptr(pos), self.hip.stream())
        return pos
