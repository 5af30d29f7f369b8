"""Distributed collectives over torch.distributed.

Reference behavior: src/collective/ (Rabit TCP comm + NCCL, SURVEY.md
§2.3).  MI355X-native design: instead of re-implementing Rabit sockets +
an NCCL dlopen stub, we run one process per GPU under
torch.distributed — backend "nccl" IS RCCL on ROCm, carrying histogram
allreduce over xGMI; backend "gloo" covers CPU training and CPU-only
tests.  All call sites from the reference's training path (SURVEY.md
§2.3 table) route through these helpers; every helper is a no-op when
the process group is not initialized, so single-process training pays
nothing.
"""
from __future__ import annotations

import os
from typing import List, Optional, Sequence

import torch
import torch.distributed as dist


def is_distributed() -> bool:
    return dist.is_available() and dist.is_initialized()


def get_rank() -> int:
    return dist.get_rank() if is_distributed() else 0


def get_world_size() -> int:
    return dist.get_world_size() if is_distributed() else 1


_TRACKER_STORE = None  # worker-side handle for done/error posting


def init(backend: Optional[str] = None, timeout: Optional[float] = None,
         **kwargs) -> None:
    """Initialize the communicator.

    Rendezvous: torchrun-style env (RANK/WORLD_SIZE/MASTER_*), or a
    builder-owned RabitTracker via DMLC_TRACKER_URI/PORT — in the
    tracker case the worker connects the tracker's TCPStore, claims a
    rank if it has none (reference tracker.cc rank assignment), and
    builds the process group on that store.

    Watchdog: `timeout` (seconds; env XGB_AMD_COLL_TIMEOUT; default
    1800 like the reference's collective timeout) bounds every
    collective.  For RCCL this arms torch's NCCL watchdog (async error
    handling + comm abort on timeout — the reference's AsyncLaunch
    watchdog + ncclCommAbort, src/collective/coll.cu:93-182); for gloo
    the ops raise directly."""
    global _TRACKER_STORE
    if is_distributed():
        return
    import datetime
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if timeout is None:
        t = (kwargs.get("dmlc_timeout") or kwargs.get("DMLC_TIMEOUT")
             or os.environ.get("XGB_AMD_COLL_TIMEOUT"))
        timeout = float(t) if t else 1800.0
    td = datetime.timedelta(seconds=float(timeout))
    if backend == "nccl":
        # abort the communicator (and surface the error) on watchdog
        # timeout instead of hanging the job
        os.environ.setdefault("TORCH_NCCL_ASYNC_ERROR_HANDLING", "1")
    tracker_uri = (kwargs.get("dmlc_tracker_uri")
                   or os.environ.get("DMLC_TRACKER_URI"))
    extra = {k: v for k, v in kwargs.items()
             if not k.lower().startswith(("dmlc_", "xgboost_"))}
    if tracker_uri and "store" not in extra:
        from .tracker import connect_tracker
        port = int(kwargs.get("dmlc_tracker_port")
                   or os.environ.get("DMLC_TRACKER_PORT"))
        world = int(kwargs.get("dmlc_num_worker")
                    or os.environ.get("WORLD_SIZE", "0"))
        rank_env = os.environ.get("RANK")
        store, rank = connect_tracker(
            tracker_uri, port, world,
            int(rank_env) if rank_env is not None else None,
            float(timeout))
        _TRACKER_STORE = store
        dist.init_process_group(backend=backend, store=store, rank=rank,
                                world_size=world, timeout=td)
        return
    dist.init_process_group(backend=backend, timeout=td, **extra)


def finalize() -> None:
    global _TRACKER_STORE
    if _TRACKER_STORE is not None:
        from .tracker import post_done
        post_done(_TRACKER_STORE)
        _TRACKER_STORE = None
    if is_distributed():
        dist.destroy_process_group()


def _comm_device() -> torch.device:
    """Device collectives must use: nccl/rccl wants the local GPU."""
    if is_distributed() and dist.get_backend() == "nccl":
        return torch.device("cuda", torch.cuda.current_device())
    return torch.device("cpu")


def allreduce_sum_(t: torch.Tensor) -> torch.Tensor:
    """In-place sum-allreduce on the tensor's own device when the backend
    supports it (rccl for cuda tensors), else round-trips via comm device."""
    if not is_distributed():
        return t
    backend = dist.get_backend()
    if backend == "nccl" and not t.is_cuda:
        dev = _comm_device()
        tmp = t.to(dev)
        dist.all_reduce(tmp, op=dist.ReduceOp.SUM)
        t.copy_(tmp.to(t.device))
    elif backend == "gloo" and t.is_cuda:
        tmp = t.cpu()
        dist.all_reduce(tmp, op=dist.ReduceOp.SUM)
        t.copy_(tmp.to(t.device))
    else:
        dist.all_reduce(t, op=dist.ReduceOp.SUM)
    return t


def allreduce_max_(t: torch.Tensor) -> torch.Tensor:
    if not is_distributed():
        return t
    backend = dist.get_backend()
    if (backend == "nccl") != t.is_cuda:
        dev = _comm_device()
        tmp = t.to(dev)
        dist.all_reduce(tmp, op=dist.ReduceOp.MAX)
        t.copy_(tmp.to(t.device))
    else:
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
    return t


def allreduce_sum_scalars(vals: Sequence[float]) -> List[float]:
    if not is_distributed():
        return list(vals)
    t = torch.tensor(list(vals), dtype=torch.float64, device=_comm_device())
    dist.all_reduce(t, op=dist.ReduceOp.SUM)
    return t.cpu().tolist()


def allreduce_max_scalars(vals: Sequence[float]) -> List[float]:
    if not is_distributed():
        return list(vals)
    t = torch.tensor(list(vals), dtype=torch.float64, device=_comm_device())
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    return t.cpu().tolist()


def broadcast_obj(obj, src: int = 0):
    if not is_distributed():
        return obj
    holder = [obj]
    dist.broadcast_object_list(holder, src=src)
    return holder[0]


def allgather_obj(obj) -> list:
    if not is_distributed():
        return [obj]
    out = [None] * get_world_size()
    dist.all_gather_object(out, obj)
    return out


def barrier() -> None:
    if is_distributed():
        dist.barrier()


def check_synchronized(payload: bytes, what: str = "model") -> None:
    """reference CheckTreesSynchronized (src/tree/hist_param.cc:18):
    broadcast rank0's serialized payload and CHECK equality."""
    if not is_distributed():
        return
    ref = broadcast_obj(payload, 0)
    if ref != payload:
        raise RuntimeError(f"{what} differs across workers "
                           f"(rank {get_rank()})")


# ---------------------------------------------------------------------------
# Reference-API compatibility surface (python-package/xgboost/collective.py):
# the names user code calls directly. All map onto torch.distributed.

import dataclasses
from enum import IntEnum
from typing import Any, Optional as _Optional


@dataclasses.dataclass
class Config:
    """Rendezvous configuration (reference collective.py:25). With
    torch.distributed the retry/tracker fields are handled by the
    store; kept for call-site compatibility."""

    retry: _Optional[int] = None
    timeout: _Optional[int] = None
    tracker_host_ip: _Optional[str] = None
    tracker_port: _Optional[int] = None
    tracker_task_id: _Optional[str] = None


class Op(IntEnum):
    """Reduce operation for :func:`allreduce` (reference collective.py:264)."""

    MAX = 0
    MIN = 1
    SUM = 2
    BITWISE_AND = 3
    BITWISE_OR = 4
    BITWISE_XOR = 5


_TORCH_OPS = None


def _torch_op(op: "Op"):
    global _TORCH_OPS
    if _TORCH_OPS is None:
        _TORCH_OPS = {
            Op.MAX: dist.ReduceOp.MAX, Op.MIN: dist.ReduceOp.MIN,
            Op.SUM: dist.ReduceOp.SUM, Op.BITWISE_AND: dist.ReduceOp.BAND,
            Op.BITWISE_OR: dist.ReduceOp.BOR,
            Op.BITWISE_XOR: dist.ReduceOp.BXOR,
        }
    return _TORCH_OPS[op]


def allreduce(data, op: "Op"):
    """In-place allreduce of a numpy array; returns it (reference
    collective.py:275).

    Routed through the backend's communication device like
    allreduce_sum_ (rccl wants GPU tensors); bitwise ops are not
    provided by either nccl/rccl or gloo and raise loudly instead of
    hanging."""
    import numpy as _np
    data = _np.asarray(data)
    if not is_distributed():
        return data
    if op in (Op.BITWISE_AND, Op.BITWISE_OR, Op.BITWISE_XOR):
        raise NotImplementedError(
            f"bitwise allreduce ({op.name}) is not supported by the "
            f"{dist.get_backend()} backend")
    t = torch.from_numpy(data)
    dev = _comm_device()
    if dev.type != "cpu":
        tmp = t.to(dev)
        dist.all_reduce(tmp, op=_torch_op(op))
        t.copy_(tmp.to(t.device))
    else:
        dist.all_reduce(t, op=_torch_op(op))
    return data


def broadcast(data, root: int):
    """Broadcast any picklable object from `root` (reference
    collective.py:190)."""
    return broadcast_obj(data, root)


def communicator_print(msg: Any) -> None:
    """Print with rank prefix (reference collective.py:156)."""
    print(f"[{get_rank()}] {msg}", flush=True)


def get_processor_name() -> str:
    import socket
    return socket.gethostname()


def signal_error(msg: str = "worker error") -> None:
    """Abort the process group after an unrecoverable worker error and
    post it to the tracker (reference SignalError, comm.h:44-101)."""
    if _TRACKER_STORE is not None:
        from .tracker import post_error
        post_error(_TRACKER_STORE, msg)
    if is_distributed():
        dist.destroy_process_group()
    raise RuntimeError(f"collective worker signalled an error: {msg}")


class CommunicatorContext:
    """`with CommunicatorContext(**args):` init/finalize wrapper
    (reference collective.py:350)."""

    def __init__(self, **args: Any) -> None:
        self.args = args

    def __enter__(self) -> dict:
        init(**self.args)
        return self.args

    def __exit__(self, *exc) -> None:
        finalize()
