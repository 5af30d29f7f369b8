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
