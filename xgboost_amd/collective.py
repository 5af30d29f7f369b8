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


def init(backend: Optional[str] = None, **kwargs) -> None:
    """Initialize from torchrun-style env vars (RANK/WORLD_SIZE/MASTER_*)."""
    if is_distributed():
        return
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    dist.init_process_group(backend=backend, **kwargs)


def finalize() -> None:
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
