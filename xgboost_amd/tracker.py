"""Rendezvous coordinator (reference: src/collective/tracker.cc:143-330
RabitTracker + python-package/xgboost/tracker.py).

MI355X-native design: the tracker OWNS a torch.distributed TCPStore —
the same store RCCL process groups bootstrap through — instead of the
reference's bespoke JSON-over-TCP protocol (protocol.h).  It performs
the reference tracker's actual duties:

- rank assignment: workers that arrive without a RANK claim one from
  an atomic counter on the store (tracker.cc assigns ranks on connect);
- join tracking: every worker registers itself, and ``wait_for()``
  blocks until all workers have signalled completion (XGTrackerWaitFor);
- error propagation: a failing worker posts to an error key that
  every other worker's next collective-init/watchdog can observe
  (the reference's dedicated error socket, comm.h:44).

Workers consume ``worker_args()`` (DMLC_TRACKER_URI/PORT compatible)
and call :func:`xgboost_amd.collective.init`, which connects a client
TCPStore and builds the process group on it.
"""
from __future__ import annotations

import socket
import time
from typing import Dict, Optional, Union

_JOIN_KEY = "xgb_amd/joined"
_DONE_KEY = "xgb_amd/done"
_RANK_KEY = "xgb_amd/next_rank"
_ERR_KEY = "xgb_amd/error"


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return int(s.getsockname()[1])


class RabitTracker:
    """Coordinator for multi-worker training (reference tracker.cc:143)."""

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
        """Open the rendezvous store (tracker side) and initialize the
        rank counter / join / done / error state."""
        import datetime

        from torch.distributed import TCPStore
        kw = {}
        if self.timeout:
            kw["timeout"] = datetime.timedelta(seconds=self.timeout)
        self._store = TCPStore(self.host_ip, self.port,
                               is_master=True, **kw)
        self._store.set(_RANK_KEY, "0")
        self._store.set(_JOIN_KEY, "0")
        self._store.set(_DONE_KEY, "0")

    def worker_args(self) -> Dict[str, Union[str, int]]:
        """Rendezvous settings for each worker (the reference's
        DMLC_TRACKER_URI/PORT, plus MASTER_* for torchrun interop)."""
        return {"DMLC_TRACKER_URI": self.host_ip,
                "DMLC_TRACKER_PORT": self.port,
                "MASTER_ADDR": self.host_ip,
                "MASTER_PORT": self.port,
                "WORLD_SIZE": self.n_workers}

    def error(self) -> Optional[str]:
        """The first error any worker posted, or None."""
        if self._store is None:
            return None
        try:
            if self._store.check([_ERR_KEY]):
                return self._store.get(_ERR_KEY).decode()
        except Exception:  # noqa: BLE001 — store torn down
            return None
        return None

    def wait_for(self, timeout: Optional[int] = None) -> None:
        """Block until every worker has signalled completion (reference
        XGTrackerWaitFor, coll_c_api.cc); raises if a worker posted an
        error or the deadline passes."""
        if self._store is None:
            return
        deadline = time.monotonic() + (timeout or self.timeout or 86400)
        while time.monotonic() < deadline:
            err = self.error()
            if err is not None:
                raise RuntimeError(f"tracker: worker error: {err}")
            done = int(self._store.get(_DONE_KEY).decode())
            if done >= self.n_workers:
                return
            time.sleep(0.05)
        raise TimeoutError(
            f"tracker: {self.n_workers} workers did not finish in time")

    def free(self) -> None:
        self._store = None


def connect_tracker(uri: str, port: int, world_size: int,
                    rank: Optional[int], timeout_s: float):
    """Worker side: join the tracker's store, claiming a rank when the
    caller has none (reference: rank assignment on tracker connect).
    Returns (store, rank)."""
    import datetime

    from torch.distributed import TCPStore
    store = TCPStore(uri, int(port), is_master=False,
                     timeout=datetime.timedelta(seconds=timeout_s))
    if rank is None:
        rank = store.add(_RANK_KEY + "/counter", 1) - 1
    store.add(_JOIN_KEY + "/counter", 1)
    return store, int(rank)


def post_done(store) -> None:
    try:
        store.add(_DONE_KEY + "/counter", 1)
        cnt = store.add(_DONE_KEY + "/counter", 0)
        store.set(_DONE_KEY, str(cnt))
    except Exception:  # noqa: BLE001
        pass


def post_error(store, msg: str) -> None:
    try:
        store.set(_ERR_KEY, msg[:4096])
    except Exception:  # noqa: BLE001
        pass


__all__ = ["RabitTracker"]
