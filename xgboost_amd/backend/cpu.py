"""CPU backend ops — torch/numpy implementations of the tree-building
primitives.

This is both the production CPU training path (reference analog:
grow_quantile_histmaker, src/tree/updater_quantile_hist.cc:664) and the
fp64 numerics oracle the HIP kernels are tested against (SURVEY.md §4:
CPU<->GPU cross-check is the primary kernel oracle).

Determinism: gradients are quantized to int32 fixed point and histograms
accumulated in int64, the same scheme the HIP histogram kernel uses
(reference: src/tree/gpu_hist/quantiser.cuh:52) — so CPU and GPU
histograms are bit-identical regardless of accumulation order.
"""
from __future__ import annotations

from typing import List, Sequence, Tuple

import numpy as np
import torch

from .. import collective
from ..data import QuantizedMatrix
from ..params import TrainParam
from ..splits import SplitEntry, evaluate_splits_np

QSHIFT = 30  # fixed-point fraction bits; sums of 2^30-scaled int32 fit int64


class GradQuantizer:
    """Fixed-point gradient quantizer (reference quantiser.cuh:52).

    scale = 2^30 / global_max_abs (allreduced), per g and h component.
    """

    def __init__(self, gpair: torch.Tensor):
        if gpair.numel():
            m = gpair.abs().amax(dim=0)  # one reduce, one D2H sync
            if collective.is_distributed():
                collective.allreduce_max_(m)
            if gpair.is_cuda:
                # keep the (allreduced) device max-abs: the native
                # whole-tree driver derives the SAME scales on device
                # (bit-identical formula), which unlocks the one-sync
                # chain for every non-fused objective
                self.maxabs_dev = m.contiguous()
            mh = m.cpu()
            max_g, max_h = float(mh[0]), float(mh[1])
        else:
            max_g, max_h = collective.allreduce_max_scalars([0.0, 0.0])
        self.g_scale = (1 << QSHIFT) / max_g if max_g > 0 else 1.0
        self.h_scale = (1 << QSHIFT) / max_h if max_h > 0 else 1.0

    def quantize(self, gpair: torch.Tensor) -> torch.Tensor:
        """float32 [n, 2] -> int32 [n, 2] (round half away from zero)."""
        scale = torch.tensor([self.g_scale, self.h_scale],
                             dtype=torch.float64, device=gpair.device)
        return torch.round(gpair.double() * scale).to(torch.int32)

    def dequantize_pair(self, qg: int, qh: int) -> Tuple[float, float]:
        return qg / self.g_scale, qh / self.h_scale

    def dequantize_hist(self, hist: torch.Tensor) -> np.ndarray:
        """int64 [..., 2] -> float64 numpy."""
        h = hist.to("cpu").numpy().astype(np.float64)
        h[..., 0] /= self.g_scale
        h[..., 1] /= self.h_scale
        return h


class SegmentedOpsMixin:
    """Stateful node-segment bookkeeping shared by the in-core backends.

    The grower talks to this interface only (reset / build_hist_nodes /
    partition_nodes / leaf_positions / node_size); the external-memory
    backend re-implements it with per-page state."""

    def reset(self, n_rows: int) -> None:
        self.ridx = self.make_ridx(n_rows)
        self.segments = {0: (0, n_rows)}
        self._n_rows = n_rows

    def node_size(self, nid: int) -> int:
        s, e = self.segments[nid]
        return e - s

    def build_hist_nodes(self, qgpair: torch.Tensor, nids,
                         out: "torch.Tensor" = None) -> torch.Tensor:
        return self.build_hist(qgpair, self.ridx,
                               [self.segments[n] for n in nids], out=out)

    def alloc_hist(self, k: int) -> torch.Tensor:
        """Uninitialized [k, n_bins, 2] on this backend's device — the
        level driver builds/subtracts into slices of one buffer instead
        of concatenating per-node histograms."""
        dev = getattr(self, "device", None)
        return torch.empty((k, self.n_bins, 2), dtype=torch.int64,
                           device=dev if dev is not None else "cpu")

    def partition_nodes(self, parents, splits, children) -> None:
        """parents: [nid], splits: [SplitEntry], children: [(l, r)].
        Updates internal segments for the child nodes."""
        segs = [self.segments[p] for p in parents]
        new_segs = self.partition(self.ridx, segs, splits)
        for (l, r), (ls, rs) in zip(children, new_segs):
            self.segments[l] = ls
            self.segments[r] = rs

    def leaf_positions(self, leaf_nids) -> torch.Tensor:
        segs = [(nid, *self.segments[nid]) for nid in leaf_nids
                if nid in self.segments]
        return self.leaf_partition(self.ridx, segs, self._n_rows)


class CpuOps(SegmentedOpsMixin):
    """Tree-building primitive ops on CPU tensors.

    Two implementations behind one interface:
    - native C (OpenMP) kernels in libgbt_hip.so (gbt_hist_cpu /
      gbt_partition_cpu) — the production CPU path, same int64
      fixed-point scheme => bit-identical results;
    - pure torch/numpy (`use_native=False`) — the independent oracle the
      HIP and C kernels are tested against.
    """

    device = torch.device("cpu")

    def __init__(self, qm: QuantizedMatrix, use_native: bool = True):
        self.qm = qm
        self.gidx_global = qm.global_gidx()  # int64 [n, f], -1 missing
        self.n_bins = qm.cuts.total_bins
        self.lib = None
        if use_native:
            try:
                from .. import ops as hip_ops
                lib = hip_ops.load()
                if hasattr(lib, "gbt_hist_cpu"):
                    self.lib = lib
                    self.hip = hip_ops
            except RuntimeError:
                pass
        self._gidx_np = np.ascontiguousarray(qm.gidx.numpy())
        self._cut_ptrs_np = np.ascontiguousarray(qm.cuts.ptrs, np.int32)
        self._scratch = None

    def swap_gidx(self, gidx: torch.Tensor) -> None:
        """Point the kernels at a different (streamed-in) quantized page
        with the same cuts (external-memory disk path)."""
        self.qm = QuantizedMatrix(gidx, self.qm.cuts, self.qm.has_missing)
        self.gidx_global = self.qm.global_gidx()
        self._gidx_np = np.ascontiguousarray(gidx.numpy())

    def _gidx_c_ptrs(self):
        import ctypes
        p = self._gidx_np.ctypes.data_as(ctypes.c_void_p)
        if self._gidx_np.dtype == np.uint8:
            return p, None
        return None, p

    def make_ridx(self, n_rows: int) -> torch.Tensor:
        return torch.arange(n_rows, dtype=torch.int64)

    def root_sum(self, qgpair: torch.Tensor) -> Tuple[int, int]:
        s = qgpair.to(torch.int64).sum(dim=0)
        t = s.clone()
        collective.allreduce_sum_(t)
        return int(t[0]), int(t[1])

    def build_hist(self, qgpair: torch.Tensor, ridx: torch.Tensor,
                   segments: Sequence[Tuple[int, int]],
                   out: Optional[torch.Tensor] = None) -> torch.Tensor:
        """-> int64 [len(segments), n_bins, 2]."""
        k = len(segments)
        if out is None:
            out = torch.zeros((k, self.n_bins, 2), dtype=torch.int64)
        else:
            assert out.is_contiguous()
            out.zero_()
        if self.lib is not None:
            import ctypes
            # the quantized gradient tensor is identical across every
            # call within a tree: cache the int32 view by data pointer
            key = (qgpair.data_ptr(), qgpair._version)
            if getattr(self, "_qnp_key", None) != key:
                self._qnp_key = key
                self._qnp = np.ascontiguousarray(
                    qgpair.cpu().numpy(), np.int32)
            q_np = self._qnp
            r_np = np.ascontiguousarray(ridx.numpy(), np.int64)
            sb = np.ascontiguousarray([s for s, _ in segments], np.int64)
            se = np.ascontiguousarray([e for _, e in segments], np.int64)
            o_np = out.numpy()
            p8, p16 = self._gidx_c_ptrs()
            self.lib.gbt_hist_cpu(
                p8, p16, self.qm.n_features,
                q_np.ctypes.data_as(ctypes.c_void_p),
                r_np.ctypes.data_as(ctypes.c_void_p),
                sb.ctypes.data_as(ctypes.c_void_p),
                se.ctypes.data_as(ctypes.c_void_p), k,
                self._cut_ptrs_np.ctypes.data_as(ctypes.c_void_p),
                o_np.ctypes.data_as(ctypes.c_void_p), self.n_bins)
            return out
        for i, (s, e) in enumerate(segments):
            rows = ridx[s:e]
            g = self.gidx_global[rows]            # [m, f]
            valid = g >= 0
            flat = g[valid]
            qg = qgpair[rows, 0].to(torch.int64).unsqueeze(1).expand_as(g)[valid]
            qh = qgpair[rows, 1].to(torch.int64).unsqueeze(1).expand_as(g)[valid]
            out[i, :, 0].index_add_(0, flat, qg)
            out[i, :, 1].index_add_(0, flat, qh)
        return out

    def allreduce_hist(self, hist: torch.Tensor) -> torch.Tensor:
        collective.allreduce_sum_(hist)
        return hist

    def evaluate_splits(self, hist: torch.Tensor, quantizer: GradQuantizer,
                        parent_sums: Sequence[Tuple[int, int]],
                        nids: Sequence[int], param: TrainParam,
                        feature_sets=None, monotone=None, cat_mask=None,
                        node_bounds=None) -> List[SplitEntry]:
        if (self.lib is not None and cat_mask is None
                and hasattr(self.lib, "gbt_evaluate_cpu")):
            return self._evaluate_native(hist, quantizer, parent_sums, nids,
                                         param, feature_sets, monotone,
                                         node_bounds)
        # parent_sums are exact int64 (gq, hq) pairs
        return evaluate_splits_np(hist.cpu().numpy(), parent_sums,
                                  quantizer.g_scale, quantizer.h_scale,
                                  nids, self.qm.cuts.ptrs, param,
                                  feature_sets=feature_sets, monotone=monotone,
                                  cat_mask=cat_mask, node_bounds=node_bounds)

    def _evaluate_native(self, hist, quantizer, parent_sums, nids, param,
                         feature_sets, monotone, node_bounds):
        """C evaluator (gbt_evaluate_cpu): same semantics/tie rules as
        numpy and the HIP kernel."""
        import ctypes

        def cp(a):
            return a.ctypes.data_as(ctypes.c_void_p) if a is not None else None

        k = len(nids)
        f = self.qm.n_features
        h_np = np.ascontiguousarray(hist.numpy())
        parents = np.ascontiguousarray(parent_sums, np.int64).reshape(k, 2)
        mono = (np.ascontiguousarray(monotone, np.int8)
                if monotone is not None else None)
        bounds = (np.ascontiguousarray(node_bounds, np.float64)
                  if node_bounds is not None else None)
        mask = None
        if feature_sets is not None and any(
                fs is not None for fs in feature_sets):
            mask = np.zeros((k, f), np.uint8)
            for i, fs in enumerate(feature_sets):
                if fs is None:
                    mask[i] = 1
                else:
                    mask[i, np.asarray(fs, np.int64)] = 1
        out = np.zeros((k, 6), np.int64)
        self.lib.gbt_evaluate_cpu(
            cp(h_np), k, self.n_bins, f, cp(self._cut_ptrs_np), cp(parents),
            quantizer.g_scale, quantizer.h_scale, param.reg_lambda,
            param.reg_alpha, param.max_delta_step, param.min_child_weight,
            cp(mono), cp(bounds), cp(mask), cp(out))
        entries = []
        for i, nid in enumerate(nids):
            e = SplitEntry(nid=int(nid), g_scale=quantizer.g_scale,
                           h_scale=quantizer.h_scale)
            if out[i, 1] >= 0 and out[i, 5] >= 0:
                gain = out[i, 0:1].view(np.float64)[0]
                if np.isfinite(gain):
                    e.gain = float(gain)
                    e.split_bin = int(out[i, 1])
                    e.default_left = bool(out[i, 2])
                    e.left_gq = int(out[i, 3])
                    e.left_hq = int(out[i, 4])
                    e.feature = int(out[i, 5])
                    e.right_gq = int(parents[i, 0]) - e.left_gq
                    e.right_hq = int(parents[i, 1]) - e.left_hq
            entries.append(e)
        return entries

    def partition(self, ridx: torch.Tensor,
                  segments: Sequence[Tuple[int, int]],
                  splits: Sequence[SplitEntry]
                  ) -> List[Tuple[Tuple[int, int], Tuple[int, int]]]:
        """Reorder ridx within each segment into [left | right].

        Returns [(left_seg, right_seg), ...].  Stable on CPU (the GPU
        kernel is unstable; histogram sums don't depend on order).
        """
        if self.lib is not None:
            return self._partition_native(ridx, segments, splits)
        out = []
        for (s, e), sp in zip(segments, splits):
            rows = ridx[s:e]
            bins = self.gidx_global[rows, sp.feature]
            missing = bins < 0
            if sp.is_cat:
                fstart = int(self.qm.cuts.ptrs[sp.feature])
                local = bins - fstart
                go_right = torch.zeros_like(missing)
                cats = torch.as_tensor(np.asarray(sp.cat_bits, np.int64))
                go_right = torch.isin(local, cats)
                go_left = ~go_right
            else:
                go_left = bins <= sp.split_bin
            go_left = torch.where(missing,
                                  torch.tensor(bool(sp.default_left)), go_left)
            left_rows = rows[go_left]          # advanced indexing: copies
            right_rows = rows[~go_left]        # copy BEFORE writing into ridx
            nl = int(left_rows.numel())
            ridx[s:s + nl] = left_rows
            ridx[s + nl:e] = right_rows
            out.append(((s, s + nl), (s + nl, e)))
        return out

    def _partition_native(self, ridx, segments, splits):
        import ctypes
        r_np = np.ascontiguousarray(ridx.numpy(), np.int64)
        assert r_np.base is not None or r_np.ctypes.data == \
            ridx.data_ptr(), "ridx must share memory"
        max_seg = max((e - s for s, e in segments), default=0)
        if self._scratch is None or self._scratch.size < max_seg:
            self._scratch = np.empty(max(max_seg, 1024), np.int64)
        cuts = self.qm.cuts
        out = []
        p8, p16 = self._gidx_c_ptrs()
        for (s, e), sp in zip(segments, splits):
            fbins = int(cuts.ptrs[sp.feature + 1] - cuts.ptrs[sp.feature])
            if sp.is_cat:
                nw = (fbins + 31) // 32
                w = np.zeros(nw, np.uint32)
                for c in sp.cat_bits:
                    w[c >> 5] |= np.uint32(1 << (c & 31))
                cat_ptr = w.ctypes.data_as(ctypes.c_void_p)
                cat_words = nw
                sbin = -1
            else:
                cat_ptr = None
                cat_words = 0
                sbin = sp.split_bin - int(cuts.ptrs[sp.feature])
            nl = self.lib.gbt_partition_cpu(
                p8, p16, self.qm.n_features,
                r_np.ctypes.data_as(ctypes.c_void_p), s, e,
                sp.feature, sbin, 1 if sp.default_left else 0,
                cat_ptr, cat_words, fbins,
                self._scratch.ctypes.data_as(ctypes.c_void_p))
            out.append(((s, s + int(nl)), (s + int(nl), e)))
        return out

    def leaf_partition(self, ridx: torch.Tensor,
                       leaf_segments: Sequence[Tuple[int, int, int]],
                       n_rows: int) -> torch.Tensor:
        """-> int32 [n_rows] leaf node id per row (for prediction cache).

        One contiguous fill per leaf + a single scatter: per-leaf fancy
        indexing cost ~0.25 ms x n_leaves in torch."""
        vals = torch.zeros(n_rows, dtype=torch.int32)
        for nid, s, e in leaf_segments:
            vals[s:e] = nid  # positional (contiguous) fill
        pos = torch.zeros(n_rows, dtype=torch.int32)
        pos[ridx.long()] = vals
        return pos
