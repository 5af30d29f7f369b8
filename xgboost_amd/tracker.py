"""Rendezvous coordinator (reference: python-package/xgboost/tracker.py
RabitTracker).

The MI355X-native stack uses torch.distributed for all collectives, so
the "tracker" is simply the rendezvous endpoint (MASTER_ADDR /
MASTER_PORT) that every rank's :func:`xgboost_amd.collective.init`
connects to.  This class keeps the reference's API shape — construct,
``start()``, ``worker_args()``, ``wait_for()`` — while delegating the
actual bootstrap to a ``torch.distributed`` TCPStore, which is what
backend "nccl" (RCCL on ROCm) rendezvouses through.
"""
from __future__ import annotations

import socket
from typing import Dict, Optional, Union


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return int(s.getsockname()[1])


class RabitTracker:
    """Coordinator for multi-worker training (reference tracker.py:17).

    With torch.distributed there is no separate tracker protocol: the
    master rank's TCPStore IS the rendezvous.  ``worker_args()`` returns
    the env-style settings each worker should apply before calling
    :func:`xgboost_amd.collective.init` (or that ``torchrun`` sets
    automatically)."""

    def __init__(self, n_workers: int, host_ip: Optional[str] = None,
                 port: int = 0, sortby: str = "host",
                 timeout: int = 0) -> None:
        if n_workers < 1:
            raise ValueError("n_workers must be >= 1")
        self.n_workers = n_workers
        self.host_ip = host_ip or "127.0.0.1"
        self.port = port or _free_port()
        self.timeout = timeout
        self._store = None

    def start(self) -> None:
        """Open the rendezvous store (master side)."""
        import datetime

        from torch.distributed import TCPStore
        kw = {}
        if self.timeout:
            kw["timeout"] = datetime.timedelta(seconds=self.timeout)
        self._store = TCPStore(self.host_ip, self.port,
                               is_master=True, **kw)

    def worker_args(self) -> Dict[str, Union[str, int]]:
        """Rendezvous settings for each worker (maps the reference's
        DMLC_TRACKER_URI/PORT onto MASTER_ADDR/MASTER_PORT)."""
        return {"DMLC_TRACKER_URI": self.host_ip,
                "DMLC_TRACKER_PORT": self.port,
                "MASTER_ADDR": self.host_ip,
                "MASTER_PORT": self.port,
                "WORLD_SIZE": self.n_workers}

    def wait_for(self, timeout: Optional[int] = None) -> None:
        """The torch.distributed store needs no join protocol; kept for
        API compatibility (workers own their process lifetimes)."""
        _ = timeout

    def free(self) -> None:
        self._store = None


__all__ = ["RabitTracker"]
