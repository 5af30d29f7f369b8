"""Performance monitor (reference: src/common/timer.h:45 common::Monitor
— named start/stop accumulators printed at verbosity>=3, with NVTX/roctx
range integration).  Our ranges go to rocprofv3 via roctx when
available (rocTracer python bindings are absent in this image, so
ranges are a no-op unless librocprofiler-sdk-roctx exposes C hooks)."""
from __future__ import annotations

import atexit
import ctypes
import time
from collections import defaultdict
from typing import Dict

from .config import verbosity

_roctx = None


def _load_roctx():
    global _roctx
    if _roctx is None:
        try:
            lib = ctypes.CDLL("librocprofiler-sdk-roctx.so")
            lib.roctxRangePushA.argtypes = [ctypes.c_char_p]
            lib.roctxRangePop.argtypes = []
            _roctx = lib
        except OSError:
            _roctx = False
    return _roctx


class Monitor:
    """Accumulates wall time per named section; prints at verbosity>=3."""

    _instances = []

    def __init__(self, label: str):
        self.label = label
        self.totals: Dict[str, float] = defaultdict(float)
        self.counts: Dict[str, int] = defaultdict(int)
        self._starts: Dict[str, float] = {}
        Monitor._instances.append(self)

    def start(self, name: str) -> None:
        self._starts[name] = time.perf_counter()
        lib = _load_roctx()
        if lib:
            lib.roctxRangePushA(f"{self.label}:{name}".encode())

    def stop(self, name: str) -> None:
        t0 = self._starts.pop(name, None)
        if t0 is not None:
            self.totals[name] += time.perf_counter() - t0
            self.counts[name] += 1
        lib = _load_roctx()
        if lib:
            lib.roctxRangePop()

    def report(self) -> str:
        lines = [f"======== Monitor ({self.label}) ========"]
        for name in sorted(self.totals, key=self.totals.get, reverse=True):
            lines.append(f"  {name}: {self.totals[name]*1e3:.3f} ms "
                         f"({self.counts[name]} calls)")
        return "\n".join(lines)

    def maybe_print(self) -> None:
        if verbosity() >= 3 and self.totals:
            print(self.report(), flush=True)


@atexit.register
def _report_all():
    for m in Monitor._instances:
        m.maybe_print()


class TrainingObserver:
    """Debug observer (reference src/common/observer.h:38
    TrainingObserver): dumps per-iteration gradient statistics and the
    freshly committed tree when enabled.  The reference gates it at
    compile time; here it is runtime-gated by the XGB_AMD_OBSERVER env
    var or config_context(observer=True), so production pays a single
    attribute check."""

    _enabled = None

    @classmethod
    def enabled(cls) -> bool:
        if cls._enabled is None:
            import os
            cls._enabled = os.environ.get("XGB_AMD_OBSERVER", "0") not in (
                "0", "", "false")
        return cls._enabled

    @classmethod
    def observe_gradient(cls, iteration: int, grad, hess) -> None:
        if not cls.enabled():
            return
        g = grad.detach()
        h = hess.detach()
        print(f"[observer] it={iteration} grad: n={g.numel()} "
              f"mean={float(g.mean()):+.6g} absmax={float(g.abs().max()):.6g}"
              f" | hess: mean={float(h.mean()):+.6g} "
              f"min={float(h.min()):.6g}", flush=True)

    @classmethod
    def observe_tree(cls, iteration: int, tree) -> None:
        if not cls.enabled():
            return
        n = tree.n_nodes
        leaves = int((tree.left[:n] == -1).sum())
        print(f"[observer] it={iteration} tree: nodes={n} leaves={leaves} "
              f"max_gain={float(tree.loss_chg[:n].max()):.6g} "
              f"root_hess={float(tree.sum_hess[0]):.6g}", flush=True)

    @classmethod
    def observe_predictions(cls, iteration: int, margin) -> None:
        if not cls.enabled():
            return
        m = margin.detach()
        print(f"[observer] it={iteration} margin: mean="
              f"{float(m.mean()):+.6g} std={float(m.std()):.6g}",
              flush=True)
