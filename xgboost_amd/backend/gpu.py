"""GPU backend ops — drives the HIP/CDNA4 kernels in ops/cpp.

Same interface as backend/cpu.CpuOps; the grower is backend-agnostic.
All launches go on torch's current HIP stream.  These python-side ops
back the feature-rich driver paths (categorical, sampling, lossguide,
external memory); the hot path is the native C++ driver
(ops/cpp/driver.hip) wrapped by grow_tree_native below, which in
whole-tree mode runs with ONE host sync per tree.
"""
from __future__ import annotations

from typing import List, Optional, Sequence, Tuple

import numpy as np
import torch

from .. import collective
from ..data import QuantizedMatrix
from ..params import TrainParam
from ..splits import SplitEntry
from .cpu import GradQuantizer

# LDS budget for the histogram kernel: 8192 bins * 16 B = 128 KiB
# (160 KiB/CU on MI355X; leave headroom for occupancy)
LDS_MAX_GROUP_BINS = 8192
TARGET_HIST_TASKS = 2048  # fill 256 CUs x 8 blocks
MIN_ROWS_PER_TASK = 1024


def _chunk_tasks(segments: Sequence[Tuple[int, int]],
                 slots: Optional[Sequence[int]] = None,
                 target_tasks: int = TARGET_HIST_TASKS,
                 min_rows: int = MIN_ROWS_PER_TASK) -> np.ndarray:
    """Build BlockTask array [(slot, row_begin, row_end, 0)] chunked so the
    grid fills the chip; hist callers pass fewer/bigger tasks so the LDS
    zero+flush cost amortizes."""
    total = sum(e - s for s, e in segments)
    rows_per_task = max(min_rows,
                        (total + target_tasks - 1) // max(target_tasks, 1))
    tasks = []
    for i, (s, e) in enumerate(segments):
        slot = slots[i] if slots is not None else i
        b = s
        while b < e:
            tasks.append((slot, b, min(b + rows_per_task, e), 0))
            b += rows_per_task
        if s == e:  # empty segment still needs no task
            pass
    if not tasks:
        tasks.append((0, 0, 0, 0))
    return np.asarray(tasks, dtype=np.int32)


class _PinnedStager:
    """Reusable pinned-host staging buffer: pack many small numpy arrays
    into one H2D copy per level (profiling: dozens of tiny pageable
    uploads per level cost more than the kernels)."""

    ALIGN = 8
    NSLOTS = 4  # ring: a slot is never reused before >=2 stream syncs

    def __init__(self, device):
        self.device = device
        self.slots = [[torch.empty(1 << 16, dtype=torch.uint8,
                                   pin_memory=True),
                       torch.empty(1 << 16, dtype=torch.uint8, device=device)]
                      for _ in range(self.NSLOTS)]
        self.cur = 0

    def upload(self, arrays):
        """arrays: list of (np.ndarray | None); returns device tensors."""
        slot = self.slots[self.cur]
        self.cur = (self.cur + 1) % self.NSLOTS
        offs = []
        total = 0
        for a in arrays:
            if a is None:
                offs.append((None, 0, 0))
                continue
            a = np.ascontiguousarray(a)
            nbytes = a.nbytes
            total = (total + self.ALIGN - 1) & ~(self.ALIGN - 1)
            offs.append((a, total, nbytes))
            total += nbytes
        if total > slot[0].numel():
            cap = 1 << max(int(total).bit_length(), 16)
            slot[0] = torch.empty(cap, dtype=torch.uint8, pin_memory=True)
            slot[1] = torch.empty(cap, dtype=torch.uint8, device=self.device)
        host = slot[0].numpy()
        for a, off, nbytes in offs:
            if a is not None:
                host[off:off + nbytes] = a.view(np.uint8).reshape(-1)
        slot[1][:total].copy_(slot[0][:total], non_blocking=True)
        out = []
        for a, off, nbytes in offs:
            if a is None:
                out.append(None)
                continue
            t = slot[1][off:off + nbytes]
            td = t.view(_np_to_torch(a.dtype)).view(*a.shape)
            out.append(td)
        return out


def _np_to_torch(dt: np.dtype) -> torch.dtype:
    return {np.dtype(np.int32): torch.int32, np.dtype(np.uint8): torch.uint8,
            np.dtype(np.int8): torch.int8, np.dtype(np.int64): torch.int64,
            np.dtype(np.float64): torch.float64,
            np.dtype(np.float32): torch.float32}[dt]


from .cpu import SegmentedOpsMixin


class GpuOps(SegmentedOpsMixin):
    def __init__(self, qm: QuantizedMatrix, col_copy: bool = True):
        from .. import ops as hip_ops
        if not torch.cuda.is_available():
            raise RuntimeError("GpuOps requires a GPU")
        self.lib = hip_ops.load()
        self.hip = hip_ops
        self.qm = qm
        assert qm.gidx.is_cuda, "QuantizedMatrix must be on the GPU"
        self.device = qm.gidx.device
        cuts = qm.cuts
        self.n_bins = cuts.total_bins
        dev = self.device
        self.cut_ptrs = torch.from_numpy(
            cuts.ptrs.astype(np.int32)).to(dev)
        self.n_bins_feat = torch.from_numpy(
            np.diff(cuts.ptrs).astype(np.int32)).to(dev)
        self.cut_values = torch.from_numpy(cuts.values).to(dev)
        # feature groups for LDS privatization
        widths = np.diff(cuts.ptrs)
        groups_f = [0]
        groups_b = [0]
        acc = 0
        for f, w in enumerate(widths):
            if acc + w > LDS_MAX_GROUP_BINS and acc > 0:
                groups_f.append(f)
                groups_b.append(int(cuts.ptrs[f]))
                acc = 0
            acc += int(w)
        groups_f.append(len(widths))
        groups_b.append(int(cuts.ptrs[-1]))
        self.feat_group_start = torch.tensor(groups_f, dtype=torch.int32,
                                             device=dev)
        self.bin_group_start = torch.tensor(groups_b, dtype=torch.int32,
                                            device=dev)
        self.n_groups = len(groups_f) - 1
        self.max_group_bins = int(np.max(np.diff(groups_b)))
        self.use_shared = 1 if self.max_group_bins <= LDS_MAX_GROUP_BINS else 0
        # hist-kernel mode upgrade (see hist.hip launcher): 2 = the
        # register-metadata kernel (every group <= 32 features, u8
        # bins), 3 = + dword-packed bin loads (4-aligned stride/groups)
        esz = qm.gidx.element_size()
        if (self.use_shared and esz <= 2
                and max(groups_f[i + 1] - groups_f[i]
                        for i in range(self.n_groups)) <= 32):
            lanes = 4 // esz  # features per 32-bit load
            aligned = (qm.n_features % lanes == 0
                       and all(f % lanes == 0 for f in groups_f[:-1])
                       and qm.gidx.data_ptr() % 4 == 0)
            self.use_shared = 3 if aligned else 2
        self._ridx_out: Optional[torch.Tensor] = None
        if cuts.feature_types is not None:
            self.cat_feature = torch.tensor(
                [1 if t == "c" else 0 for t in cuts.feature_types],
                dtype=torch.uint8, device=dev)
        else:
            self.cat_feature = None
        g8 = qm.gidx.dtype == torch.uint8
        self._gidx8 = qm.gidx if g8 else None
        self._gidx16 = None if g8 else qm.gidx
        # feature-major copy for the partition/leaf-decide kernels:
        # they read ONE feature per node, so the column layout keeps
        # their gathers inside a single cache-resident column instead
        # of touching a 64-byte line per row across the whole matrix.
        # (External-memory page caches skip it: doubling every cached
        # page would blow the device budget for a secondary kernel.)
        self._gidx_T = qm.gidx.t().contiguous() if col_copy else None
        self.stager = _PinnedStager(dev)

    # ------------------------------------------------------------------
    def swap_gidx(self, gidx: torch.Tensor) -> None:
        """Point the kernels at a different (streamed-in) quantized page
        with the same cuts (external-memory path)."""
        g8 = gidx.dtype == torch.uint8
        self._gidx8 = gidx if g8 else None
        self._gidx16 = None if g8 else gidx
        self._gidx_T = None  # streamed pages: transposing every sweep
        # would cost more than the column gathers save
        self.qm = QuantizedMatrix(gidx, self.qm.cuts, self.qm.has_missing)

    def _gidx_ptrs(self):
        return self.hip.ptr(self._gidx8), self.hip.ptr(self._gidx16)

    def _gidx_col_args(self):
        """(col8_ptr, col16_ptr, leading dim) for the kernels that take
        the optional feature-major copy."""
        t = self._gidx_T
        if t is None:
            return None, None, 0
        if t.dtype == torch.uint8:
            return self.hip.ptr(t), None, t.shape[1]
        return None, self.hip.ptr(t), t.shape[1]

    def make_ridx(self, n_rows: int) -> torch.Tensor:
        self._ridx_out = torch.empty(n_rows, dtype=torch.int32,
                                     device=self.device)
        return torch.arange(n_rows, dtype=torch.int32, device=self.device)

    def root_sum(self, qgpair: torch.Tensor) -> Tuple[int, int]:
        s = qgpair.to(torch.int64).sum(dim=0)
        collective.allreduce_sum_(s)
        host = s.cpu()
        return int(host[0]), int(host[1])

    def build_hist(self, qgpair: torch.Tensor, ridx: torch.Tensor,
                   segments: Sequence[Tuple[int, int]],
                   out: Optional[torch.Tensor] = None) -> torch.Tensor:
        k = len(segments)
        if out is not None:
            assert out.is_contiguous()
            out.zero_()
            return self._build_hist_into(qgpair, ridx, segments, out)
        out = torch.zeros((k, self.n_bins, 2), dtype=torch.int64,
                          device=self.device)
        return self._build_hist_into(qgpair, ridx, segments, out)

    def _build_hist_into(self, qgpair, ridx, segments, out):
        tasks_np = _chunk_tasks(segments, target_tasks=512)
        (tasks,) = self.stager.upload([tasks_np])
        p8, p16 = self._gidx_ptrs()
        self.lib.gbt_hist(
            p8, p16, self.qm.n_features, self.hip.ptr(qgpair),
            self.hip.ptr(ridx), self.hip.ptr(tasks), len(tasks_np),
            self.hip.ptr(out), self.n_bins,
            self.hip.ptr(self.feat_group_start),
            self.hip.ptr(self.bin_group_start), self.n_groups,
            self.max_group_bins, self.hip.ptr(self.cut_ptrs),
            self.use_shared, None, self.hip.stream())
        return out

    def allreduce_hist(self, hist: torch.Tensor) -> torch.Tensor:
        collective.allreduce_sum_(hist)
        return hist

    def _mt_grouping(self, T: int):
        """Feature groups sized so T histograms fit the LDS budget
        (reference MtHistKernel AllocateBlocks, histogram.cu:309)."""
        cache = self.__dict__.setdefault("_mt_group_cache", {})
        if T in cache:
            return cache[T]
        budget = LDS_MAX_GROUP_BINS // T
        widths = np.diff(self.qm.cuts.ptrs)
        if int(widths.max()) > budget:
            cache[T] = None  # a single feature exceeds the MT budget
            return None
        groups_f, groups_b, acc = [0], [0], 0
        for f, w in enumerate(widths):
            if acc + w > budget and acc > 0:
                groups_f.append(f)
                groups_b.append(int(self.qm.cuts.ptrs[f]))
                acc = 0
            acc += int(w)
        groups_f.append(len(widths))
        groups_b.append(int(self.qm.cuts.ptrs[-1]))
        dev = self.device
        out = (torch.tensor(groups_f, dtype=torch.int32, device=dev),
               torch.tensor(groups_b, dtype=torch.int32, device=dev),
               len(groups_f) - 1,
               int(np.max(np.diff(groups_b))))
        cache[T] = out
        return out

    def build_hist_nodes_mt(self, qg_mt: torch.Tensor, nids):
        """Fused multi-target histogram: ONE pass over the bin matrix
        accumulates all targets (gbt_hist_mt; reference MtHistKernel).
        qg_mt: [n, T, 2] int32 contiguous.  Returns [T, k, n_bins, 2]
        or None when the shape is unsupported (caller falls back to the
        per-target loop)."""
        T = int(qg_mt.shape[1])
        if T > 8 or not hasattr(self.lib, "gbt_hist_mt"):
            return None
        grouping = self._mt_grouping(T)
        if grouping is None:
            return None
        fg, bg, n_groups, max_gb = grouping
        segs = [self.segments[n] for n in nids]
        tasks_np = _chunk_tasks(segs, target_tasks=512)
        (tasks,) = self.stager.upload([tasks_np])
        k = len(nids)
        out = torch.zeros((T, k, self.n_bins, 2), dtype=torch.int64,
                          device=self.device)
        p8, p16 = self._gidx_ptrs()
        self.lib.gbt_hist_mt(
            p8, p16, self.qm.n_features, self.hip.ptr(qg_mt), T,
            self.hip.ptr(self.ridx), self.hip.ptr(tasks), len(tasks_np),
            self.hip.ptr(out), self.n_bins, k,
            self.hip.ptr(fg), self.hip.ptr(bg), n_groups, max_gb,
            self.hip.ptr(self.cut_ptrs), self.hip.stream())
        return out

    def evaluate_splits(self, hist: torch.Tensor, quantizer: GradQuantizer,
                        parent_sums: Sequence[Tuple[int, int]],
                        nids: Sequence[int], param: TrainParam,
                        feature_sets=None, monotone=None, cat_mask=None,
                        node_bounds=None) -> List[SplitEntry]:
        k = len(nids)
        f = self.qm.n_features
        dev = self.device
        gain = torch.empty((k, f), dtype=torch.float64, device=dev)
        bins = torch.empty((k, f), dtype=torch.int32, device=dev)
        dirs = torch.empty((k, f), dtype=torch.uint8, device=dev)
        lsum = torch.empty((k, f, 2), dtype=torch.int64, device=dev)
        parents_np = np.asarray(parent_sums, dtype=np.int64).reshape(k, 2)
        mono_np = (np.asarray(monotone, np.int8)
                   if monotone is not None else None)
        bounds_np = (np.ascontiguousarray(node_bounds, np.float64)
                     if node_bounds is not None else None)
        mask_np = None
        if feature_sets is not None and any(fs is not None for fs in feature_sets):
            mask_np = np.zeros((k, f), dtype=np.uint8)
            for i, fs in enumerate(feature_sets):
                if fs is None:
                    mask_np[i] = 1
                else:
                    mask_np[i, np.asarray(fs, np.int64)] = 1
        # wide categorical features (cardinality > max_cat_to_onehot) use
        # sorted-partition splits, evaluated HOST-side with the SAME
        # routine as the CPU oracle (splits._sorted_cat_split) so GPU and
        # CPU trees stay identical; mask them out of the kernel's
        # one-hot/numeric scan
        wide_feats = None
        if cat_mask is not None:
            widths = np.diff(self.qm.cuts.ptrs)
            wm = np.asarray(cat_mask, bool) & (
                widths > param.max_cat_to_onehot)
            if wm.any():
                wide_feats = np.nonzero(wm)[0]
                if mask_np is None:
                    mask_np = np.ones((k, f), dtype=np.uint8)
                mask_np[:, wide_feats] = 0
        parents, mono_t, bounds_t, mask_t = self.stager.upload(
            [parents_np, mono_np, bounds_np, mask_np])
        self.lib.gbt_evaluate(
            self.hip.ptr(hist), k, self.n_bins, f,
            self.hip.ptr(self.cut_ptrs), self.hip.ptr(parents), None,
            quantizer.g_scale, quantizer.h_scale,
            param.reg_lambda, param.reg_alpha, param.max_delta_step,
            param.min_child_weight, self.hip.ptr(mono_t),
            self.hip.ptr(bounds_t), self.hip.ptr(mask_t),
            self.hip.ptr(self.cat_feature),
            self.hip.ptr(gain), self.hip.ptr(bins), self.hip.ptr(dirs),
            self.hip.ptr(lsum), None,
            8 if f > 4096 else 0,  # scalar eval for narrow features on
                                   # one-hot-scale data (see evaluate.hip)
            self.hip.stream())
        # device-side per-node argmax + packing -> ONE D2H sync
        out_best = torch.empty((k, 6), dtype=torch.int64, device=dev)
        self.lib.gbt_select_best(
            self.hip.ptr(gain), self.hip.ptr(bins), self.hip.ptr(dirs),
            self.hip.ptr(lsum), k, f, self.hip.ptr(out_best), None,
            self.hip.stream())
        host = out_best.cpu().numpy()
        sel_gain = host[:, 0].view(np.float64)
        sel_bin = host[:, 1]
        sel_dir = host[:, 2]
        sel_lsum = host[:, 3:5]
        best_f = host[:, 5]
        cuts = self.qm.cuts
        out = []
        for i, nid in enumerate(nids):
            e = SplitEntry(nid=int(nid), g_scale=quantizer.g_scale,
                           h_scale=quantizer.h_scale)
            if np.isfinite(sel_gain[i]) and sel_bin[i] >= 0:
                fidx = int(best_f[i])
                e.gain = float(sel_gain[i])
                e.feature = fidx
                e.split_bin = int(sel_bin[i])
                e.default_left = bool(sel_dir[i])
                e.left_gq = int(sel_lsum[i, 0])
                e.left_hq = int(sel_lsum[i, 1])
                e.right_gq = int(parent_sums[i][0]) - e.left_gq
                e.right_hq = int(parent_sums[i][1]) - e.left_hq
                e.is_cat = bool(cat_mask is not None and cat_mask[fidx])
                if e.is_cat:
                    e.cat_bits = np.array(
                        [e.split_bin - int(cuts.ptrs[fidx])], dtype=np.int32)
            out.append(e)
        if wide_feats is not None:
            self._host_sorted_cat(out, hist, parents_np, quantizer, param,
                                  wide_feats, feature_sets)
        return out

    def _host_sorted_cat(self, out, hist, parents_np, quantizer, param,
                         wide_feats, feature_sets) -> None:
        """Sorted-partition categorical splits (reference
        gpu_hist/evaluate_splits.cuh SortHistogram): the per-category
        ratio sort, present-compaction and prefix cumsums run ON DEVICE;
        only the small (k x width) cumsum/order tables come to the host,
        where the final gain math is the SAME numpy ops as the CPU
        oracle (splits._sorted_cat_split), batched over nodes — so GPU
        and CPU trees stay identical."""
        from ..splits import calc_gain_given_weight, calc_weight
        cuts = self.qm.cuts
        k = len(out)
        inv_g = 1.0 / quantizer.g_scale
        inv_h = 1.0 / quantizer.h_scale
        pgq_v = parents_np[:, 0].astype(np.int64)
        phq_v = parents_np[:, 1].astype(np.int64)
        pw_v = calc_weight(pgq_v * inv_g, phq_v * inv_h, param)
        pgain_v = calc_gain_given_weight(pgq_v * inv_g, phq_v * inv_h,
                                         pw_v, param)
        allowed = [None] * k
        if feature_sets is not None:
            for i, fs in enumerate(feature_sets):
                if fs is not None:
                    allowed[i] = set(int(x) for x in fs)
        mct = param.max_cat_threshold + 1
        for fi in wide_feats:
            b0, b1 = int(cuts.ptrs[fi]), int(cuts.ptrs[fi + 1])
            w = b1 - b0
            Gc = hist[:, b0:b1, 0]                      # int64 [k, w]
            Hc = hist[:, b0:b1, 1]
            present = Hc != 0
            ratio = torch.where(
                present,
                (Gc.double() * inv_g) / (Hc.double() * inv_h
                                         + param.reg_lambda),
                torch.full_like(Gc, float("inf"), dtype=torch.float64))
            order = torch.argsort(ratio, dim=1, stable=True)  # absent last
            Gs = Gc.gather(1, order)
            Hs = Hc.gather(1, order)
            cg_t = torch.cumsum(Gs, 1)
            ch_t = torch.cumsum(Hs, 1)
            npres_t = present.sum(1)
            # ONE compact download per feature: [k, w] cumsums + order
            cg = cg_t.cpu().numpy()
            ch = ch_t.cpu().numpy()
            order_h = order.cpu().numpy()
            npres = npres_t.cpu().numpy()
            featG = cg[:, -1]
            featH = ch[:, -1]
            missG = pgq_v - featG
            missH = phq_v - featH
            # positions beyond the present prefix / threshold are invalid
            pos = np.arange(w)[None, :]
            limit = np.minimum(npres, mct)[:, None]
            pos_ok = pos < limit
            for missing_left in (False, True):
                gl = cg + (missG[:, None] if missing_left else 0)
                hl = ch + (missH[:, None] if missing_left else 0)
                gr = pgq_v[:, None] - gl
                hr = phq_v[:, None] - hl
                glf, hlf = gl * inv_g, hl * inv_h
                grf, hrf = gr * inv_g, hr * inv_h
                wl = calc_weight(glf, hlf, param)
                wr = calc_weight(grf, hrf, param)
                gains = (calc_gain_given_weight(glf, hlf, wl, param)
                         + calc_gain_given_weight(grf, hrf, wr, param)
                         - pgain_v[:, None])
                ok = (pos_ok & (hlf >= param.min_child_weight)
                      & (hrf >= param.min_child_weight)
                      & (hl > 0) & (hr > 0))
                gains = np.where(ok, gains, -np.inf)
                for i, e in enumerate(out):
                    if npres[i] < 2:
                        continue
                    if allowed[i] is not None and int(fi) not in allowed[i]:
                        continue
                    row = gains[i]
                    if not np.isfinite(row).any():
                        continue
                    j = int(np.argmax(row))
                    gv = float(row[j])
                    if gv > e.gain:
                        e.gain = gv
                        e.feature = int(fi)
                        e.split_bin = b0
                        e.default_left = missing_left
                        e.left_gq = int(gl[i, j])
                        e.left_hq = int(hl[i, j])
                        e.right_gq = int(pgq_v[i]) - e.left_gq
                        e.right_hq = int(phq_v[i]) - e.left_hq
                        e.is_cat = True
                        left_set = set(int(c) for c in order_h[i, :j + 1])
                        e.cat_bits = np.array(
                            sorted(c for c in range(w)
                                   if c not in left_set), dtype=np.int32)

    def partition(self, ridx: torch.Tensor,
                  segments: Sequence[Tuple[int, int]],
                  splits: Sequence[SplitEntry]
                  ) -> List[Tuple[Tuple[int, int], Tuple[int, int]]]:
        k = len(segments)
        dev = self.device
        cuts = self.qm.cuts
        feat = np.empty(k, np.int32)
        sbin = np.empty(k, np.int32)
        dleft = np.empty(k, np.uint8)
        counters = np.empty((k, 2), np.int32)
        cat_words: List[np.ndarray] = []
        cat_offsets = np.zeros(k + 1, np.int32)
        any_cat = False
        for i, ((s, e), sp) in enumerate(zip(segments, splits)):
            feat[i] = sp.feature
            dleft[i] = 1 if sp.default_left else 0
            counters[i] = (s, e)
            if sp.is_cat:
                any_cat = True
                sbin[i] = -1
                nw = (int(self.qm.cuts.ptrs[sp.feature + 1]
                          - self.qm.cuts.ptrs[sp.feature]) + 31) // 32
                w = np.zeros(nw, np.uint32)
                for c in sp.cat_bits:
                    w[c >> 5] |= np.uint32(1 << (c & 31))
                cat_words.append(w)
                cat_offsets[i + 1] = cat_offsets[i] + nw
            else:
                sbin[i] = sp.split_bin - int(cuts.ptrs[sp.feature])
                cat_words.append(np.zeros(0, np.uint32))
                cat_offsets[i + 1] = cat_offsets[i]
        tasks_np = _chunk_tasks(segments)
        cat_bits_np = (np.concatenate(cat_words).view(np.int32)
                       if any_cat else None)
        cat_off_np = cat_offsets if any_cat else None
        tasks, feat_t, sbin_t, dleft_t, cnt_t, cat_bits_t, cat_off_t = \
            self.stager.upload([tasks_np, feat, sbin, dleft, counters,
                                cat_bits_np, cat_off_np])
        p8, p16 = self._gidx_ptrs()
        c8, c16, cld = self._gidx_col_args()
        self.lib.gbt_partition(
            p8, p16, self.qm.n_features, c8, c16, cld, self.hip.ptr(ridx),
            self.hip.ptr(self._ridx_out), self.hip.ptr(tasks), len(tasks_np),
            self.hip.ptr(feat_t), self.hip.ptr(sbin_t), self.hip.ptr(dleft_t),
            self.hip.ptr(cat_bits_t), self.hip.ptr(cat_off_t),
            self.hip.ptr(self.n_bins_feat), self.hip.ptr(cnt_t),
            self.hip.stream())
        # copy partitioned ranges back into the primary buffer (1 launch)
        self.lib.gbt_copy_ranges(
            self.hip.ptr(self._ridx_out), self.hip.ptr(ridx),
            self.hip.ptr(tasks), len(tasks_np), self.hip.stream())
        final = cnt_t.cpu().numpy()  # sync
        out = []
        for i, (s, e) in enumerate(segments):
            mid = int(final[i, 0])
            assert mid == int(final[i, 1]), (
                f"partition counters disagree: {final[i]}")
            out.append(((s, mid), (mid, e)))
        return out

    # -- native C++ level-loop driver (driver.hip) ----------------------
    def grow_tree_native(self, qgpair: torch.Tensor, tree, param,
                         quantizer, monotone: Optional[np.ndarray],
                         root_sums: torch.Tensor,
                         feature_mask: Optional[np.ndarray] = None):
        """Run the whole per-tree loop in C++ (gbt_grow_tree).  Returns
        (tree, positions) or None when the config is unsupported (the
        Python driver handles those)."""
        import ctypes
        if not hasattr(self.lib, "gbt_grow_tree"):
            return None
        # lossguide / max_leaves configs run the SAME depthwise chain:
        # the grower replays the priority-queue policy on the recorded
        # candidates afterwards (TreeGrower._policy_replay)
        if param.max_depth <= 0 or param.max_depth > 14:
            return None
        n_rows = qgpair.shape[0]
        max_build = max(1, 1 << max(param.max_depth - 2, 0))
        pool_rows = 2 * max_build
        pool_bytes = pool_rows * self.n_bins * 16
        if pool_bytes > (32 << 30):  # fall back rather than blow memory
            return None
        dev = self.device
        ws = getattr(self, "_native_ws", None)
        if ws is None or ws["pool_rows"] < pool_rows or ws["n"] < n_rows:
            ws = {
                "pool_rows": pool_rows, "n": n_rows,
                "ridx": torch.empty(n_rows, dtype=torch.int32, device=dev),
                "ridx_out": torch.empty(n_rows, dtype=torch.int32,
                                        device=dev),
                "pool_a": torch.empty((pool_rows, self.n_bins, 2),
                                      dtype=torch.int64, device=dev),
                "pool_b": torch.empty((pool_rows, self.n_bins, 2),
                                      dtype=torch.int64, device=dev),
                "eval_gain": torch.empty((pool_rows, self.qm.n_features),
                                         dtype=torch.float64, device=dev),
                "eval_bin": torch.empty((pool_rows, self.qm.n_features),
                                        dtype=torch.int32, device=dev),
                "eval_dir": torch.empty((pool_rows, self.qm.n_features),
                                        dtype=torch.uint8, device=dev),
                "eval_lsum": torch.empty((pool_rows, self.qm.n_features, 2),
                                         dtype=torch.int64, device=dev),
                "eval_best": torch.empty((pool_rows, 6), dtype=torch.int64,
                                         device=dev),
                "pos": torch.zeros(n_rows, dtype=torch.int32, device=dev),
                # 1-sync driver: persistent partition counters (expand
                # nodes per level <= 2^(max_depth-1)), device-generated
                # hist task buffer + task-gen scratch
                "counters": torch.empty(max(4, 1 << param.max_depth),
                                        dtype=torch.int32, device=dev),
                "hist_tasks_cap": 16384 + max_build + 8,
                "hist_tasks": torch.empty((16384 + max_build + 8, 4),
                                          dtype=torch.int32, device=dev),
                "tg_scratch": torch.empty(3 * max_build + 8,
                                          dtype=torch.int32, device=dev),
                # pointer-stable staging for the per-round inputs: the
                # driver's hipGraph replay bakes device addresses, so
                # the fresh-per-round tensors (quantized gradients, root
                # sums, max-abs, colsample mask) are copied into these
                # persistent buffers instead of being passed directly
                "qg_stage": torch.empty((n_rows, 2), dtype=torch.int32,
                                        device=dev),
                "rs_stage": torch.empty(2, dtype=torch.int64, device=dev),
                "ma_stage": torch.empty(2, dtype=torch.float32,
                                        device=dev),
                "fm_stage": torch.empty(self.qm.n_features,
                                        dtype=torch.uint8, device=dev),
                "driver": self.lib.gbt_driver_create(),
            }
            # whole-tree mode arena (single sync per tree): per-level
            # best/segment records + device-side partition/task args
            pool = 2 * max_build
            if pool <= 2048:
                rec = 1 + param.max_depth * pool
                mp = n_rows // 1024 + pool + 2
                # + 2 fp64 bound ping-pong buffers (monotone whole-tree)
                # + per-level built-is-left mode records
                wt_bytes = (48 * rec + 8 * rec + 8 * (param.max_depth + 2)
                            + 96 * pool + 32 * pool + 16 * mp
                            + param.max_depth * pool + 32 * 1024)
                ws["wt_max_ptasks"] = mp
                ws["wt_ws"] = torch.empty(wt_bytes, dtype=torch.uint8,
                                          device=dev)
            else:
                ws["wt_max_ptasks"] = 0
                ws["wt_ws"] = None
            self._native_ws = ws
        cap = 1 << (param.max_depth + 1)
        host = {name: np.zeros(cap, dt) for name, dt in [
            ("left", np.int32), ("right", np.int32), ("parent", np.int32),
            ("split_index", np.int32), ("split_cond", np.float32),
            ("default_left", np.uint8), ("loss_chg", np.float64),
            ("sum_hess", np.float32), ("base_weight", np.float32)]}
        fmask_ptr = None
        if feature_mask is not None:
            fm = np.ascontiguousarray(feature_mask, np.uint8)
            ws["fm_stage"].copy_(
                torch.from_numpy(fm), non_blocking=True)
            fmask_ptr = self.hip.ptr(ws["fm_stage"])
        mono_dev = mono_host = None
        if monotone is not None:
            m8 = np.ascontiguousarray(monotone, np.int8)
            key = m8.tobytes()
            if getattr(self, "_mono_key", None) != key:
                self._mono_dev_t = torch.from_numpy(m8).to(dev)
                self._mono_host = m8
                self._mono_key = key
            mono_dev = self.hip.ptr(self._mono_dev_t)
            mono_host = self._mono_host.ctypes.data_as(ctypes.c_void_p)
        cuts = self.qm.cuts
        cut_ptrs_host = np.ascontiguousarray(cuts.ptrs, np.int32)
        cut_values_host = np.ascontiguousarray(cuts.values, np.float32)
        cb = None
        if collective.is_distributed():
            # the driver hands back raw device pointers into the hist
            # pools (per-level + whole-tree hist reduce) or the wt arena
            # (whole-tree pair-sum reduce); resolve to the owning torch
            # tensor so torch.distributed handles stream ordering
            regions = [(t.data_ptr(), t.numel() * t.element_size(), t)
                       for t in (ws["pool_a"], ws["pool_b"])]
            if ws["wt_ws"] is not None:
                w = ws["wt_ws"]
                regions.append((w.data_ptr(), w.numel(), w))

            def _allreduce(ptr, n_elems):
                addr = ctypes.addressof(ptr.contents)
                for base, nbytes, t in regions:
                    if base <= addr < base + nbytes:
                        off = addr - base
                        i64 = t.view(torch.uint8).view(-1)[
                            off:off + 8 * n_elems].view(torch.int64)
                        collective.allreduce_sum_(i64)
                        return
                raise RuntimeError("allreduce pointer outside workspaces")

            cb = self.hip.ALLREDUCE_FN(_allreduce)
        p8, p16 = self._gidx_ptrs()
        c8, c16, cld = self._gidx_col_args()
        ma = getattr(quantizer, "maxabs_dev", None)
        out_scales = np.zeros(2, dtype=np.float64)
        gsc = quantizer.g_scale if ma is None else 0.0
        hsc = quantizer.h_scale if ma is None else 0.0
        # stage per-round tensors at stable addresses (hipGraph replay)
        qg_st = ws["qg_stage"][:n_rows]
        qg_st.copy_(qgpair.view(n_rows, 2))
        ws["rs_stage"].copy_(root_sums.view(-1)[:2])
        if ma is not None:
            ws["ma_stage"].copy_(ma.view(-1)[:2])
        rc = self.lib.gbt_grow_tree(
            ws["driver"], p8, p16, self.qm.n_features, c8, c16, n_rows,
            self.hip.ptr(qg_st),
            self.hip.ptr(self.cut_ptrs),
            cut_values_host.ctypes.data_as(ctypes.c_void_p),
            cut_ptrs_host.ctypes.data_as(ctypes.c_void_p),
            self.hip.ptr(self.n_bins_feat),
            self.hip.ptr(self.feat_group_start),
            self.hip.ptr(self.bin_group_start), self.n_groups,
            self.max_group_bins, self.use_shared, self.n_bins,
            self.hip.ptr(ws["ridx"]), self.hip.ptr(ws["ridx_out"]),
            self.hip.ptr(ws["pool_a"]), self.hip.ptr(ws["pool_b"]),
            self.hip.ptr(ws["eval_gain"]), self.hip.ptr(ws["eval_bin"]),
            self.hip.ptr(ws["eval_dir"]), self.hip.ptr(ws["eval_lsum"]),
            self.hip.ptr(ws["eval_best"]), self.hip.ptr(ws["pos"]),
            max_build,
            self.hip.ptr(ws["counters"]), self.hip.ptr(ws["hist_tasks"]),
            ws["hist_tasks_cap"], self.hip.ptr(ws["tg_scratch"]),
            self.hip.ptr(ws["wt_ws"]),
            0 if ws["wt_ws"] is None else ws["wt_ws"].numel(),
            ws["wt_max_ptasks"],
            self.hip.ptr(ws["rs_stage"]),
            self.hip.ptr(ws["ma_stage"]) if ma is not None else None,
            out_scales.ctypes.data_as(ctypes.c_void_p),
            gsc, hsc,
            param.reg_lambda, param.reg_alpha, param.max_delta_step,
            param.min_child_weight, param.gamma, param.eta, param.max_depth,
            mono_dev, mono_host, fmask_ptr, cb,
            *[host[k].ctypes.data_as(ctypes.c_void_p) for k in (
                "left", "right", "parent", "split_index", "split_cond",
                "default_left", "loss_chg", "sum_hess", "base_weight")],
            self.hip.stream())
        if rc in (-9998, -9999):
            return None  # capacity guard tripped: python driver handles it
        if rc <= 0:
            raise RuntimeError(f"gbt_grow_tree failed: rc={rc}")
        if ma is not None:
            # host copies of the device-derived scales (bit-identical
            # formula), for any later host-side use of this quantizer
            quantizer.g_scale = float(out_scales[0])
            quantizer.h_scale = float(out_scales[1])
        n = rc
        tree._ensure(n)
        tree.n_nodes = n
        tree.left[:n] = host["left"][:n]
        tree.right[:n] = host["right"][:n]
        tree.parent[:n] = host["parent"][:n]
        tree.split_index[:n] = host["split_index"][:n]
        tree.split_cond[:n] = host["split_cond"][:n]
        tree.default_left[:n] = host["default_left"][:n]
        tree.loss_chg[:n] = host["loss_chg"][:n]
        # exact fp64 gains: the grow-policy replay's heap keys must
        # match the Python driver's fp64 ordering bit-for-bit
        tree._gain64 = host["loss_chg"][:n].copy()
        tree.sum_hess[:n] = host["sum_hess"][:n]
        tree.base_weight[:n] = host["base_weight"][:n]
        return tree, ws["pos"]

    def leaf_partition(self, ridx: torch.Tensor,
                       leaf_segments: Sequence[Tuple[int, int, int]],
                       n_rows: int) -> torch.Tensor:
        pos = torch.zeros(n_rows, dtype=torch.int32, device=self.device)
        if not leaf_segments:
            return pos
        segs = [(s, e) for _, s, e in leaf_segments]
        tasks_np = _chunk_tasks(segs)
        leaf_np = np.asarray([nid for nid, _, _ in leaf_segments], np.int32)
        tasks, leaf_ids = self.stager.upload([tasks_np, leaf_np])
        self.lib.gbt_leaf_partition(
            self.hip.ptr(ridx), self.hip.ptr(tasks), len(tasks_np),
            self.hip.ptr(leaf_ids), self.hip.ptr(pos), self.hip.stream())
        return pos


# ---------------------------------------------------------------------------
# prediction


class _ForestArrays:
    """Device SoA for a tree range [lo, hi) of a booster.  DART tree
    weights are folded into the LEAF values (margin = sum w_i*tree_i),
    so the predict kernel needs no weight array."""

    def __init__(self, booster, lo: int, hi: int, device, idxs=None,
                 fold_weights: bool = False):
        if idxs is None:
            idxs = range(lo, hi)
        idxs = list(idxs)
        trees = [booster.trees[i] for i in idxs]
        offs = np.zeros(len(trees) + 1, np.int32)
        for i, t in enumerate(trees):
            offs[i + 1] = offs[i] + t.n_nodes
        total = int(offs[-1])
        left = np.empty(total, np.int32)
        right = np.empty(total, np.int32)
        sidx = np.empty(total, np.int32)
        cond = np.empty(total, np.float32)
        dft = np.empty(total, np.uint8)
        stype = np.empty(total, np.uint8)
        hess = np.empty(total, np.float32)
        cat_off = np.zeros(total + 1, np.int32)
        cat_bits: List[np.ndarray] = []
        pos = 0
        for i, t in enumerate(trees):
            n = t.n_nodes
            o = offs[i]
            left[o:o + n] = t.left[:n]
            right[o:o + n] = t.right[:n]
            sidx[o:o + n] = t.split_index[:n]
            cond[o:o + n] = t.split_cond[:n]
            w = (booster._tw(idxs[i])
                 if fold_weights and hasattr(booster, "_tw") else 1.0)
            if w != 1.0:
                leaf_mask = t.left[:n] == -1
                cond[o:o + n][leaf_mask] *= np.float32(w)
            dft[o:o + n] = t.default_left[:n]
            stype[o:o + n] = t.split_type[:n]
            hess[o:o + n] = t.sum_hess[:n]
            for nid in range(n):
                if t.split_type[nid] == 1 and nid in t.cat_segments:
                    cats = t.cat_segments[nid]
                    nw = (int(cats.max()) >> 5) + 1 if len(cats) else 0
                    w = np.zeros(nw, np.uint32)
                    for c in cats:
                        w[c >> 5] |= np.uint32(1 << (c & 31))
                    cat_bits.append(w)
                    pos += nw
                cat_off[o + nid + 1] = pos
        self.tree_offsets = torch.from_numpy(offs).to(device)
        self.left = torch.from_numpy(left).to(device)
        self.right = torch.from_numpy(right).to(device)
        self.split_index = torch.from_numpy(sidx).to(device)
        self.split_cond = torch.from_numpy(cond).to(device)
        self.default_left = torch.from_numpy(dft).to(device)
        self.split_type = torch.from_numpy(stype).to(device)
        self.sum_hess = torch.from_numpy(hess).to(device)
        self.cat_offsets = torch.from_numpy(cat_off).to(device)
        if cat_bits:
            self.cat_bits = torch.from_numpy(
                np.concatenate(cat_bits).view(np.int32)).to(device)
        else:
            self.cat_bits = torch.zeros(1, dtype=torch.int32, device=device)
        self.tree_group = torch.tensor(
            [booster.tree_info[t] for t in idxs],
            dtype=torch.int32, device=device)
        self.n_trees = len(trees)


def shap_gpu(booster, dmat, lo: int, hi: int, phi: np.ndarray) -> np.ndarray:
    """GPU pred_contribs.  Numeric forests use the path-table kernel
    (shap_paths.hip: host-precomputed path decomposition, register
    Extend/Unwind, ~2 orders of magnitude faster than the old scratch
    DFS); categorical forests fall back to the DFS kernel."""
    from .. import ops as hip_ops
    from ..shap import _expected_value
    lib = hip_ops.load()
    has_cat = any(
        t.split_type[:t.n_nodes].any() for t in booster.trees[lo:hi]
        if hasattr(t, "split_type"))
    if not has_cat and hasattr(lib, "gbt_shap_paths"):
        return _shap_gpu_paths(booster, dmat, lo, hi, phi)
    if not hasattr(lib, "gbt_shap"):
        raise ImportError("gbt_shap kernel not built")
    max_depth = max((t.max_depth() for t in booster.trees[lo:hi]), default=0)
    n, n_groups, n_cols = phi.shape
    if max_depth > 16 or n_cols > 129:
        raise ImportError("GPU SHAP limits exceeded; falling back to CPU")
    device = booster.device
    fa = _ForestArrays(booster, lo, hi, device)
    dd = dmat.device_data() if hasattr(dmat, 'device_data') else None
    X = dd if dd is not None else torch.from_numpy(dmat.raw_data()).to(device)
    expected = torch.tensor(
        [_expected_value(booster.trees[t]) for t in range(lo, hi)],
        dtype=torch.float64, device=device)
    out = torch.from_numpy(
        np.ascontiguousarray(phi, np.float32)).to(device)
    missing = dmat.missing
    missing_is_nan = 1 if np.isnan(missing) else 0
    lib.gbt_shap(
        hip_ops.ptr(X), n, dmat.num_col(),
        float(0.0 if missing_is_nan else missing), missing_is_nan,
        hip_ops.ptr(fa.tree_offsets), hip_ops.ptr(fa.left),
        hip_ops.ptr(fa.right), hip_ops.ptr(fa.split_index),
        hip_ops.ptr(fa.split_cond), hip_ops.ptr(fa.default_left),
        hip_ops.ptr(fa.split_type), hip_ops.ptr(fa.cat_offsets),
        hip_ops.ptr(fa.cat_bits), hip_ops.ptr(fa.sum_hess),
        hip_ops.ptr(fa.tree_group), fa.n_trees, n_groups, n_cols,
        hip_ops.ptr(expected), hip_ops.ptr(out), hip_ops.stream())
    res = out.cpu().numpy()
    if n_groups == 1:
        return res[:, 0, :]
    return res


def _shap_gpu_paths(booster, dmat, lo: int, hi: int,
                    phi: np.ndarray) -> np.ndarray:
    from .. import ops as hip_ops
    from ..shap_paths import build_path_table
    lib = hip_ops.load()
    n, n_groups, n_cols = phi.shape
    device = booster.device
    pp, pg, ef, elo, ehi, emiss, ez, pv, bias = build_path_table(
        booster.trees[lo:hi], booster.tree_info[lo:hi])
    max_elems = int(np.diff(pp).max()) if len(pg) else 0
    if max_elems > 16:
        raise ImportError("GPU SHAP path length > 16; CPU fallback")
    dd = dmat.device_data() if hasattr(dmat, "device_data") else None
    X = dd if dd is not None else torch.from_numpy(dmat.raw_data()).to(device)
    X = X.t().contiguous()  # [F, n]: coalesced per-feature gathers
    # phi transposed to [groups*cols, n] so the accumulation is coalesced
    out = torch.from_numpy(
        np.ascontiguousarray(phi, np.float64)).to(device)
    for grp in range(n_groups):
        out[:, grp, n_cols - 1] += float(bias[grp]) if grp < len(bias) else 0.0
    out = out.permute(1, 2, 0).contiguous()  # [groups, cols, n]
    rz = np.where(ez > 0, 1.0 / np.maximum(ez, 1e-300), 0.0)
    t = {}
    for name, arr in (("pp", pp), ("pg", pg), ("ef", ef), ("elo", elo),
                      ("ehi", ehi), ("emiss", emiss), ("ez", ez),
                      ("rz", rz), ("pv", pv)):
        t[name] = torch.from_numpy(np.ascontiguousarray(arr)).to(device)
    missing = dmat.missing
    missing_is_nan = 1 if np.isnan(missing) else 0
    fn = lib.gbt_shap_paths if max_elems <= 8 else lib.gbt_shap_paths16
    fn(hip_ops.ptr(X), n, dmat.num_col(),
       float(0.0 if missing_is_nan else missing), missing_is_nan,
       hip_ops.ptr(t["pp"]), hip_ops.ptr(t["pg"]), hip_ops.ptr(t["ef"]),
       hip_ops.ptr(t["elo"]), hip_ops.ptr(t["ehi"]),
       hip_ops.ptr(t["emiss"]), hip_ops.ptr(t["ez"]), hip_ops.ptr(t["rz"]),
       hip_ops.ptr(t["pv"]),
       len(pg), n_groups, n_cols, hip_ops.ptr(out), hip_ops.stream())
    res = out.permute(2, 0, 1).contiguous().cpu().numpy()
    if n_groups == 1:
        return res[:, 0, :]
    return res


def shap_interactions_gpu(booster, dmat, lo: int, hi: int,
                          iteration_range=(0, 0)) -> np.ndarray:
    """GPU pred_interactions via the path-table decomposition
    (shap_ix.hip; reference QuadratureShapInteractionTaskKernel,
    src/predictor/interpretability/shap.cu:1068).

    Per (row, path): extend once, unwind each element b, and accumulate
    0.5*v*(one_b-z_b)*(one_a-z_a)*UnwoundSum_a over the reduced path —
    the exact conditional-TreeSHAP pair terms (validated against the
    CPU implementation by tests/test_shap.py).  Diagonal/bias cells are
    completed from the contribution vector like the CPU path."""
    from .. import ops as hip_ops
    from ..shap import shap_values
    from ..shap_paths import build_path_table
    lib = hip_ops.load()
    if not hasattr(lib, "gbt_shap_ix"):
        raise ImportError("gbt_shap_ix kernel not built")
    has_cat = any(
        t.split_type[:t.n_nodes].any() for t in booster.trees[lo:hi]
        if hasattr(t, "split_type"))
    if has_cat:
        raise ImportError("GPU SHAP interactions: categorical fallback")
    n = dmat.num_row()
    f = dmat.num_col()
    n_groups = booster.n_outputs
    C = f + 1
    device = booster.device
    pp, pg, ef, elo, ehi, emiss, ez, pv, bias = build_path_table(
        booster.trees[lo:hi], booster.tree_info[lo:hi])
    m = np.diff(pp)
    if len(m) and int(m.max()) > 16:
        raise ImportError("GPU SHAP interactions: path length > 16")
    out_bytes = n * n_groups * C * C * 8
    free, _total = torch.cuda.mem_get_info()
    if out_bytes > free - (2 << 30):
        raise ImportError("GPU SHAP interactions: output exceeds HBM")
    rz = np.where(ez > 0, 1.0 / np.maximum(ez, 1e-300), 0.0)
    t = {}
    for name, arr in (("pp", pp), ("pg", pg), ("ef", ef), ("elo", elo),
                      ("ehi", ehi), ("emiss", emiss), ("ez", ez),
                      ("rz", rz), ("pv", pv)):
        t[name] = torch.from_numpy(np.ascontiguousarray(arr)).to(device)
    dd = dmat.device_data() if hasattr(dmat, "device_data") else None
    X = dd if dd is not None else torch.from_numpy(dmat.raw_data()).to(device)
    X = X.t().contiguous()  # [F, n] coalesced gathers
    out = torch.zeros((n_groups, C, C, n), dtype=torch.float64,
                      device=device)
    missing = dmat.missing
    missing_is_nan = 1 if np.isnan(missing) else 0
    lib.gbt_shap_ix(
        hip_ops.ptr(X), n, f, float(0.0 if missing_is_nan else missing),
        missing_is_nan, hip_ops.ptr(t["pp"]), hip_ops.ptr(t["pg"]),
        hip_ops.ptr(t["ef"]), hip_ops.ptr(t["elo"]), hip_ops.ptr(t["ehi"]),
        hip_ops.ptr(t["emiss"]), hip_ops.ptr(t["ez"]), hip_ops.ptr(t["rz"]),
        hip_ops.ptr(t["pv"]), len(pg), n_groups, C, hip_ops.ptr(out),
        hip_ops.stream())
    # diagonal completion from the (GPU) contribution vector
    base = shap_values(booster, dmat, iteration_range)
    base_t = torch.as_tensor(np.ascontiguousarray(base, np.float64),
                             device=device)
    if n_groups == 1:
        base_t = base_t[:, None, :]
    ix = out.permute(3, 0, 1, 2)  # [n, g, C, C] view
    off_sum = ix.sum(dim=-1) - torch.diagonal(ix, dim1=-2, dim2=-1)
    diag = base_t - off_sum
    res = ix.clone()
    res.diagonal(dim1=-2, dim2=-1).copy_(diag)
    res = res.cpu().numpy()
    if n_groups == 1:
        return res[:, 0].astype(np.float32)
    return res.astype(np.float32)


def _cached_forest(booster, lo: int, hi: int, device) -> _ForestArrays:
    """Device forest SoA cached across predict calls (serving path:
    repeated inplace_predict uploads nothing but the rows)."""
    wd = getattr(booster, "weight_drop", None)
    key = (lo, hi, len(booster.trees), str(device),
           None if not wd else tuple(wd))
    fc = booster.__dict__.setdefault("_forest_dev_cache", {})
    fa = fc.get(key)
    if fa is None:
        fc.clear()  # model changed or different range: drop stale SoA
        fa = _ForestArrays(booster, lo, hi, device, fold_weights=True)
        fc[key] = fa
    return fa


def predict_subset_gpu(booster, dmat, idxs, out: torch.Tensor
                       ) -> torch.Tensor:
    """DART dropped-trees contribution sum(w_i * tree_i(X)) on device
    (weights folded into the subset forest's leaf values)."""
    from .. import ops as hip_ops
    lib = hip_ops.load()
    device = out.device
    fa = _ForestArrays(booster, 0, 0, device, idxs=idxs,
                       fold_weights=True)
    dd = dmat.device_data() if hasattr(dmat, "device_data") else None
    X = dd if dd is not None else torch.from_numpy(dmat.raw_data()).to(device)
    n = dmat.num_row()
    missing = dmat.missing
    missing_is_nan = 1 if np.isnan(missing) else 0
    lib.gbt_predict(
        hip_ops.ptr(X), n, dmat.num_col(),
        float(0.0 if missing_is_nan else missing), missing_is_nan,
        hip_ops.ptr(fa.tree_offsets), hip_ops.ptr(fa.left),
        hip_ops.ptr(fa.right), hip_ops.ptr(fa.split_index),
        hip_ops.ptr(fa.split_cond), hip_ops.ptr(fa.default_left),
        hip_ops.ptr(fa.split_type), hip_ops.ptr(fa.cat_offsets),
        hip_ops.ptr(fa.cat_bits), hip_ops.ptr(fa.tree_group), fa.n_trees,
        out.shape[1], hip_ops.ptr(out), None, hip_ops.stream())
    return out


def predict_margin_gpu(booster, dmat, out_margin: torch.Tensor,
                       lo: int, hi: int,
                       out_leaf: Optional[torch.Tensor] = None) -> torch.Tensor:
    from .. import ops as hip_ops
    lib = hip_ops.load()
    device = out_margin.device
    fa = _cached_forest(booster, lo, hi, device)
    dd = dmat.device_data() if hasattr(dmat, "device_data") else None
    X = dd if dd is not None else torch.from_numpy(dmat.raw_data()).to(device)
    n = dmat.num_row()
    missing = dmat.missing
    missing_is_nan = 1 if np.isnan(missing) else 0
    lib.gbt_predict(
        hip_ops.ptr(X), n, dmat.num_col(), float(0.0 if missing_is_nan else missing),
        missing_is_nan, hip_ops.ptr(fa.tree_offsets), hip_ops.ptr(fa.left),
        hip_ops.ptr(fa.right), hip_ops.ptr(fa.split_index),
        hip_ops.ptr(fa.split_cond), hip_ops.ptr(fa.default_left),
        hip_ops.ptr(fa.split_type), hip_ops.ptr(fa.cat_offsets),
        hip_ops.ptr(fa.cat_bits), hip_ops.ptr(fa.tree_group), fa.n_trees,
        out_margin.shape[1], hip_ops.ptr(out_margin), hip_ops.ptr(out_leaf),
        hip_ops.stream())
    return out_margin
