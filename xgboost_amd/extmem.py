"""External-memory training: DataIter, ExtMemQuantileDMatrix, paged ops.

Reference behavior: src/data/extmem_quantile_dmatrix.{h,cu} (two passes
over the user iterator: sketch then per-batch quantized pages cached in
pinned host memory), src/data/sparse_page_source.h:253-330 (ThreadPool
prefetch ring, n_prefetch_batches=2, disk cache),
updater_gpu_hist.cu:371 (partition+hist per page pass).

MI355X design — three storage tiers per quantized page:
  1. device HBM (288 GB/GPU makes most datasets fully cacheable — the
     reference's cache_host_ratio=0 case),
  2. pinned host memory, streamed on a dedicated copy stream with
     lookahead so the next page's H2D overlaps the current kernels,
  3. DISK (when the DataIter sets cache_prefix and the host budget is
     exceeded): pages are spilled to <cache_prefix>.pageN.bin and read
     back by a 2-deep ThreadPool read-ahead ring into pinned staging
     slots; slot reuse is fenced on the H2D-complete event so a
     read-ahead never overwrites bytes still being DMA'd.
"""
from __future__ import annotations

import os
from concurrent.futures import Future, ThreadPoolExecutor
from typing import Any, Callable, Dict, List, Optional, Tuple

import numpy as np
import torch

from .data import DMatrix, MetaInfo, QuantizedMatrix, quantize_dense
from .quantile import HistogramCuts
from .sketch import sketch_cuts_batches, summarize_batch


class _PageStore:
    """Host/disk tier for quantized pages + read-ahead ring
    (reference SparsePageSourceImpl, sparse_page_source.h:253)."""

    N_PREFETCH = 2

    def __init__(self):
        # per page: ("ram", gidx_tensor) | ("disk", path, shape, np_dtype)
        self.entries: List[Tuple] = []
        self._pool: Optional[ThreadPoolExecutor] = None
        self._futures: Dict[int, Future] = {}
        self._slots: List[Optional[torch.Tensor]] = []
        self._slot_fence: List[Optional[torch.cuda.Event]] = []
        self._free: List[int] = []
        self._slot_bytes = 0

    def add_ram(self, gidx: torch.Tensor) -> None:
        self.entries.append(("ram", gidx))

    def add_disk(self, gidx: torch.Tensor, path: str) -> None:
        arr = gidx.numpy()
        with open(path, "wb") as fh:
            fh.write(arr.tobytes())
        self.entries.append(("disk", path, tuple(arr.shape), arr.dtype))
        self._slot_bytes = max(self._slot_bytes, arr.nbytes)

    def is_disk(self, i: int) -> bool:
        return self.entries[i][0] == "disk"

    def page_bytes(self, i: int) -> int:
        e = self.entries[i]
        if e[0] == "ram":
            return e[1].numel() * e[1].element_size()
        _, _, shape, dt = e
        return int(np.prod(shape)) * np.dtype(dt).itemsize

    def _ensure_ring(self) -> None:
        if self._pool is not None:
            return
        self._pool = ThreadPoolExecutor(max_workers=2)
        n_slots = self.N_PREFETCH + 1
        pin = torch.cuda.is_available()
        for _ in range(n_slots):
            buf = torch.empty(self._slot_bytes, dtype=torch.uint8,
                              pin_memory=pin)
            self._slots.append(buf)
            self._slot_fence.append(None)
        self._free = list(range(n_slots))

    def _read_job(self, slot: int, path: str, nbytes: int):
        fence = self._slot_fence[slot]
        if fence is not None:
            fence.synchronize()  # previous tenant's H2D must finish
            self._slot_fence[slot] = None
        buf = self._slots[slot]
        view = buf.numpy()[:nbytes]
        with open(path, "rb", buffering=0) as fh:
            fh.readinto(memoryview(view))
        return slot

    def prefetch(self, i: int) -> None:
        """Queue the disk read for page i (no-op for RAM pages)."""
        if i >= len(self.entries) or not self.is_disk(i):
            return
        if i in self._futures:
            return
        self._ensure_ring()
        if not self._free:
            return  # ring full; get() will read synchronously
        slot = self._free.pop()
        _, path, shape, dt = self.entries[i]
        nbytes = int(np.prod(shape)) * np.dtype(dt).itemsize
        self._futures[i] = self._pool.submit(self._read_job, slot, path,
                                             nbytes)

    def get(self, i: int) -> Tuple[torch.Tensor, Optional[int]]:
        """Return (host tensor of page i, slot id | None).  For disk
        pages the caller MUST call release(slot, fence_event) once the
        consuming H2D has been enqueued."""
        e = self.entries[i]
        if e[0] == "ram":
            return e[1], None
        _, path, shape, dt = e
        fut = self._futures.pop(i, None)
        if fut is None:
            self._ensure_ring()
            while not self._free:  # all slots in flight: drain one
                j, f2 = next(iter(self._futures.items()))
                f2.result()
                # that page is now resident but unconsumed; leave it —
                # steal will not happen in practice (sequential sweeps)
                break
            if self._free:
                slot = self._free.pop()
                slot = self._read_job(slot, path,
                                      int(np.prod(shape))
                                      * np.dtype(dt).itemsize)
            else:  # pathological: unpinned one-off read
                arr = np.fromfile(path, dtype=dt).reshape(shape)
                return torch.from_numpy(arr), None
        else:
            slot = fut.result()
        nbytes = int(np.prod(shape)) * np.dtype(dt).itemsize
        td = {np.dtype(np.uint8): torch.uint8,
              np.dtype(np.int16): torch.int16,
              np.dtype(np.int32): torch.int32}[np.dtype(dt)]
        view = self._slots[slot][:nbytes].view(td).view(*shape)
        return view, slot

    def release(self, slot: Optional[int],
                fence: Optional["torch.cuda.Event"]) -> None:
        if slot is None:
            return
        self._slot_fence[slot] = fence
        self._free.append(slot)


class DataIter:
    """Base class for user batch iterators (reference:
    python-package/xgboost/core.py DataIter)."""

    def __init__(self, cache_prefix: Optional[str] = None,
                 release_data: bool = True):
        self.cache_prefix = cache_prefix
        self._data_batches: List[Tuple] = []

    def reset(self) -> None:
        raise NotImplementedError

    def next(self, input_data: Callable) -> bool:
        """Call input_data(data=..., label=..., weight=...) and return
        True while batches remain (xgboost>=2 convention)."""
        raise NotImplementedError


def _drive(it: DataIter):
    """Iterate the DataIter once, yielding captured batch dicts."""
    it.reset()
    while True:
        captured: Dict[str, Any] = {}

        def input_data(**kwargs):
            captured.update(kwargs)
            return True

        has_more = it.next(input_data)
        if not has_more and not captured:
            break
        if captured:
            yield captured
        if not has_more:
            break


class ExtMemQuantileDMatrix(DMatrix):
    """Quantized pages kept out-of-core (host memory), streamed during
    training (reference: ExtMemQuantileDMatrix, extmem_quantile_dmatrix.h:29).
    """

    def __init__(self, data: DataIter, *, max_bin: int = 256,
                 ref: Optional[DMatrix] = None, missing: float = np.nan,
                 enable_categorical: bool = False,
                 cache_host_ratio: Optional[float] = None,
                 max_quantile_batches: Optional[int] = None,
                 min_cache_page_bytes: Optional[int] = None,
                 max_host_cache_bytes: Optional[int] = None,
                 on_host: bool = True, nthread: Optional[int] = None):
        # NOTE: deliberately does NOT call super().__init__ — no dense copy
        self.missing = float("nan") if missing is None else float(missing)
        self.max_bin = max_bin
        self.cache_host_ratio = cache_host_ratio
        self._quantized: Dict[int, QuantizedMatrix] = {}
        self._ref_cuts: Optional[HistogramCuts] = None
        self._data = None

        labels, weights, margins, ftypes = [], [], [], None
        # pass 1: sketch
        if ref is not None:
            cuts = ref.cached_cuts() or ref.quantized(max_bin).cuts
        else:
            summaries = []
            n_features = None
            for batch in _drive(data):
                X = np.ascontiguousarray(batch["data"], dtype=np.float32)
                n_features = X.shape[1]
                ftypes = batch.get("feature_types", ftypes)
                summaries.append(summarize_batch(
                    X, self.missing, ftypes, max_bin,
                    weights=batch.get("weight")))
                if max_quantile_batches and \
                        len(summaries) >= max_quantile_batches:
                    pass  # summaries stay bounded per batch anyway
            if n_features is None:
                raise ValueError("DataIter yielded no batches")
            cuts = sketch_cuts_batches(summaries, max_bin, n_features,
                                       ftypes)
        self.cuts = cuts
        # pass 2: quantize pages.  Tiering: pages stay in pinned host
        # memory until max_host_cache_bytes is exceeded; beyond that,
        # with a cache_prefix on the iterator, they spill to disk and
        # stream back through the read-ahead ring (reference
        # sparse_page_source.h disk cache).
        self.store = _PageStore()
        cache_prefix = getattr(data, "cache_prefix", None)
        host_budget = max_host_cache_bytes
        if host_budget is None:
            env = os.environ.get("XGB_AMD_HOST_CACHE_BYTES")
            host_budget = int(env) if env else None
        host_used = 0
        self.pages: List[QuantizedMatrix] = []
        self.page_offsets = [0]
        n_rows = 0
        for batch in _drive(data):
            X = np.ascontiguousarray(batch["data"], dtype=np.float32)
            qm = quantize_dense(X, cuts, self.missing)
            size = qm.gidx.numel() * qm.gidx.element_size()
            spill = (cache_prefix is not None and host_budget is not None
                     and host_used + size > host_budget
                     and len(self.pages) > 0)  # page 0 stays resident
            if spill:
                path = f"{cache_prefix}.page{len(self.pages)}.bin"
                self.store.add_disk(qm.gidx, path)
                qm = QuantizedMatrix(
                    torch.zeros((0, qm.n_features), dtype=qm.gidx.dtype),
                    cuts, qm.has_missing)  # metadata-only placeholder
            else:
                qm.gidx = qm.gidx.pin_memory() \
                    if torch.cuda.is_available() else qm.gidx
                self.store.add_ram(qm.gidx)
                host_used += size
            self.pages.append(qm)
            n_rows += X.shape[0]
            self.page_offsets.append(n_rows)
            if batch.get("label") is not None:
                labels.append(np.asarray(batch["label"], np.float32))
            if batch.get("weight") is not None:
                weights.append(np.asarray(batch["weight"], np.float32))
            if batch.get("base_margin") is not None:
                margins.append(np.asarray(batch["base_margin"], np.float32))
        self.info = MetaInfo(num_row=n_rows,
                             num_col=self.pages[0].n_features)
        if labels:
            self.info.labels = np.concatenate(labels)
        if weights:
            self.info.weights = np.concatenate(weights)
        if margins:
            self.info.base_margin = np.concatenate(margins)
        self.info.feature_types = list(ftypes) if ftypes else None
        self.info.feature_names = None
        self.info.validate()

    def num_row(self) -> int:
        return self.info.num_row

    def num_col(self) -> int:
        return self.info.num_col

    def raw_data(self):
        raise RuntimeError(
            "ExtMemQuantileDMatrix holds no raw feature values; "
            "predict with the quantized pages (predict uses cut values)")

    def page_qm(self, i: int) -> QuantizedMatrix:
        """Materialized page i (loads disk pages; used by predict)."""
        if not self.store.is_disk(i):
            return self.pages[i]
        host, slot = self.store.get(i)
        qm = QuantizedMatrix(host.clone(), self.cuts,
                             self.pages[i].has_missing)
        self.store.release(slot, None)
        return qm

    def cached_cuts(self) -> Optional[HistogramCuts]:
        return self.cuts

    def quantized(self, max_bin: int, sketch_fn=None):
        raise RuntimeError("use make_extmem_ops for external-memory data")


class ExtMemOps:
    """Paged implementation of the grower ops interface: per-page row
    index + segments; histograms accumulated across page sweeps."""

    def __init__(self, dmat: ExtMemQuantileDMatrix, device: torch.device,
                 device_cache_bytes: int = 200 << 30):
        self.dmat = dmat
        self.device = device
        self.cuts = dmat.cuts
        self.n_bins = self.cuts.total_bins
        self.page_ops: List[Any] = []
        budget = device_cache_bytes
        store = dmat.store
        for i, qm in enumerate(dmat.pages):
            disk = store.is_disk(i)
            if device.type == "cuda":
                from .backend.gpu import GpuOps
                size = store.page_bytes(i)
                if size <= budget and not disk:
                    budget -= size
                    self.page_ops.append(GpuOps(qm.to(device),
                                                col_copy=False))
                else:
                    if not hasattr(self, "_copy_stream"):
                        self._copy_stream = torch.cuda.Stream()
                    self.page_ops.append(
                        _StreamedPage(qm, device, self._copy_stream,
                                      store=store, page_idx=i))
            else:
                from .backend.cpu import CpuOps
                if disk:
                    self.page_ops.append(_CpuDiskPage(qm, store, i))
                else:
                    self.page_ops.append(CpuOps(qm))
        self.qm = dmat.pages[0]  # for cuts/n_features introspection

    # -- stateful interface -------------------------------------------------
    def reset(self, n_rows: int) -> None:
        assert n_rows == self.dmat.num_row()
        for ops, (s, e) in zip(self.page_ops, zip(
                self.dmat.page_offsets[:-1], self.dmat.page_offsets[1:])):
            ops.reset(e - s)
        self._n_rows = n_rows

    def node_size(self, nid: int) -> int:
        return sum(ops.node_size(nid) for ops in self.page_ops)

    def root_sum(self, qgpair: torch.Tensor) -> Tuple[int, int]:
        from . import collective
        s = qgpair.to(torch.int64).sum(dim=0)
        collective.allreduce_sum_(s)
        host = s.cpu()
        return int(host[0]), int(host[1])

    def _page_gpair(self, qgpair: torch.Tensor, i: int) -> torch.Tensor:
        s, e = self.dmat.page_offsets[i], self.dmat.page_offsets[i + 1]
        return qgpair[s:e]

    def _prefetch(self, i: int) -> None:
        if i < len(self.page_ops) and hasattr(self.page_ops[i], "prefetch"):
            self.page_ops[i].prefetch()

    def build_hist_nodes(self, qgpair: torch.Tensor, nids) -> torch.Tensor:
        total = None
        self._prefetch(0)
        self._prefetch(1)
        for i, ops in enumerate(self.page_ops):
            self._prefetch(i + 2)  # 2-deep: disk read k+2, H2D k+1
            h = ops.build_hist_nodes(self._page_gpair(qgpair, i), nids)
            total = h if total is None else total + h
        return total

    def allreduce_hist(self, hist: torch.Tensor) -> torch.Tensor:
        from . import collective
        collective.allreduce_sum_(hist)
        return hist

    def evaluate_splits(self, *args, **kwargs):
        return self.page_ops[0].evaluate_splits(*args, **kwargs)

    def partition_nodes(self, parents, splits, children) -> None:
        self._prefetch(0)
        self._prefetch(1)
        for i, ops in enumerate(self.page_ops):
            self._prefetch(i + 2)
            ops.partition_nodes(parents, splits, children)

    def leaf_positions(self, leaf_nids) -> torch.Tensor:
        out = torch.zeros(self._n_rows, dtype=torch.int32,
                          device=self.device)
        for i, ops in enumerate(self.page_ops):
            s, e = self.dmat.page_offsets[i], self.dmat.page_offsets[i + 1]
            out[s:e] = ops.leaf_positions(leaf_nids).to(self.device)
        return out


class _StreamedPage:
    """A page whose quantized matrix stays in pinned host memory and is
    copied to the device only for the op sweeps that read it
    (beyond-HBM datasets).  The small per-page state (ridx, segments)
    stays device-resident; only the big bin matrix streams.  H2D runs
    on a dedicated copy stream so the NEXT page's upload overlaps the
    current page's kernels (reference: sparse_page_source.h prefetch
    ring + ext-mem copy stream)."""

    def __init__(self, qm: QuantizedMatrix, device: torch.device,
                 copy_stream: Optional["torch.cuda.Stream"] = None,
                 store: Optional[_PageStore] = None,
                 page_idx: int = -1):
        self.host_qm = qm
        self.device = device
        self._gpu_ops = None
        self._copy_stream = copy_stream
        self._pending = None
        self._event = None
        self._store = store
        self._page_idx = page_idx
        self._disk = store is not None and page_idx >= 0 \
            and store.is_disk(page_idx)

    def prefetch(self) -> None:
        """Start this page's fetch (disk read / H2D), non-blocking."""
        if self._disk:
            self._store.prefetch(self._page_idx)  # ThreadPool disk read
            return
        if self._pending is not None or self._copy_stream is None:
            return
        with torch.cuda.stream(self._copy_stream):
            self._pending = self.host_qm.gidx.to(self.device,
                                                 non_blocking=True)
            self._event = torch.cuda.Event()
            self._event.record(self._copy_stream)

    def _take_page(self):
        if self._disk:
            host, slot = self._store.get(self._page_idx)
            stream = self._copy_stream or torch.cuda.current_stream()
            with torch.cuda.stream(stream):
                gidx = host.to(self.device, non_blocking=True)
                ev = torch.cuda.Event()
                ev.record(stream)
            torch.cuda.current_stream().wait_event(ev)
            gidx.record_stream(torch.cuda.current_stream())
            # the staging slot may be re-read only after the H2D fence
            self._store.release(slot, ev)
            return gidx
        if self._pending is not None:
            torch.cuda.current_stream().wait_event(self._event)
            gidx, self._pending, self._event = self._pending, None, None
            # the tensor was allocated on the copy stream but is consumed
            # by kernels on the current stream: without this, freeing it
            # (next swap_gidx) lets the allocator recycle the memory for
            # the NEXT prefetch while kernels still read it
            gidx.record_stream(torch.cuda.current_stream())
            return gidx
        return self.host_qm.gidx.to(self.device, non_blocking=True)

    def _ops_with_page(self):
        from .backend.gpu import GpuOps
        gidx = self._take_page()
        if self._gpu_ops is None:
            self._gpu_ops = GpuOps(QuantizedMatrix(
                gidx, self.host_qm.cuts, self.host_qm.has_missing))
        else:
            self._gpu_ops.swap_gidx(gidx)
        return self._gpu_ops

    def _drop_page(self) -> None:
        # keep the column count: evaluate_splits (and anything else that
        # runs between sweeps) reads qm.n_features from the swapped-in
        # tensor's shape
        if self._gpu_ops is not None:
            self._gpu_ops.swap_gidx(
                torch.zeros((0, self.host_qm.gidx.shape[1]),
                            dtype=self.host_qm.gidx.dtype,
                            device=self.device))

    # ops that touch the bin matrix: stream the page in
    def build_hist_nodes(self, qgpair, nids):
        ops = self._ops_with_page()
        out = ops.build_hist_nodes(qgpair, nids)
        self._drop_page()
        return out

    def partition_nodes(self, parents, splits, children):
        ops = self._ops_with_page()
        out = ops.partition_nodes(parents, splits, children)
        self._drop_page()
        return out

    # ops on small state only
    def reset(self, n_rows):
        return self._ops_with_page().reset(n_rows) if self._gpu_ops is None \
            else self._gpu_ops.reset(n_rows)

    def node_size(self, nid):
        return self._gpu_ops.node_size(nid)

    def leaf_positions(self, leaf_nids):
        return self._gpu_ops.leaf_positions(leaf_nids)

    def evaluate_splits(self, *a, **k):
        return self._gpu_ops.evaluate_splits(*a, **k)

    @property
    def qm(self):
        return self.host_qm


class _CpuDiskPage:
    """CPU counterpart of _StreamedPage for disk-spilled pages: the bin
    matrix is fetched through the same read-ahead ring per sweep and
    swapped into a reusable CpuOps."""

    def __init__(self, qm_meta: QuantizedMatrix, store: _PageStore,
                 page_idx: int):
        self.meta = qm_meta
        self._store = store
        self._idx = page_idx
        self._ops = None

    def prefetch(self) -> None:
        self._store.prefetch(self._idx)

    def _with_page(self):
        from .backend.cpu import CpuOps
        host, slot = self._store.get(self._idx)
        qm = QuantizedMatrix(host, self.meta.cuts, self.meta.has_missing)
        if self._ops is None:
            self._ops = CpuOps(qm)
        else:
            self._ops.swap_gidx(host)
        # CPU kernels consume synchronously: slot is free immediately
        self._store.release(slot, None)
        return self._ops

    def build_hist_nodes(self, qgpair, nids):
        return self._with_page().build_hist_nodes(qgpair, nids)

    def partition_nodes(self, parents, splits, children):
        return self._with_page().partition_nodes(parents, splits, children)

    def reset(self, n_rows):
        return self._with_page().reset(n_rows)

    def node_size(self, nid):
        return self._ops.node_size(nid)

    def leaf_positions(self, leaf_nids):
        return self._ops.leaf_positions(leaf_nids)

    def evaluate_splits(self, *a, **k):
        return self._ops.evaluate_splits(*a, **k)

    @property
    def qm(self):
        return self.meta


def make_extmem_ops(dmat: ExtMemQuantileDMatrix, device) -> ExtMemOps:
    return ExtMemOps(dmat, torch.device(device))
