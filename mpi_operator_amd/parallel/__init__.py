"""Horovod-equivalent data-parallel engine over RCCL/xGMI.

The reference operator delegates this entire layer to external Horovod+NCCL
images (reference examples/v2beta1/horovod/tensorflow_mnist.py:90,133,143;
SURVEY.md §2.3 N2-N5). Here it is first-class and MI355X-native:

- ``init()`` bootstraps one process per GPU from the launcher environment
  (amdrun/torchrun/mpirun-style env vars) into a ``torch.distributed``
  process group whose "nccl" backend IS RCCL on ROCm — collectives run over
  the node's point-to-point xGMI links.
- ``DistributedOptimizer`` overlaps bucketized grad allreduce with backward
  (Horovod tensor-fusion equivalent), with bucket sizes chosen for xGMI.
- ``broadcast_parameters`` / ``broadcast_object`` mirror Horovod's
  BroadcastGlobalVariablesHook(0) (reference tensorflow_mnist.py:143).
"""
from __future__ import annotations

import datetime
import os

import torch
import torch.distributed as dist

from .distributed_optimizer import DistributedOptimizer  # noqa: F401

_initialized = False


def _env_int(*names, default=None):
    for n in names:
        v = os.environ.get(n)
        if v is not None:
            return int(v)
    return default


def init(backend: str | None = None, timeout_s: int = 300) -> None:
    """MPI-style rendezvous from the launcher environment.

    Accepts the env conventions of (a) our amdrun launcher / torchrun
    (RANK/WORLD_SIZE/LOCAL_RANK/MASTER_ADDR/MASTER_PORT) and (b) OpenMPI's
    mpirun (OMPI_COMM_WORLD_*), since MPIJob launchers exec `mpirun`-style
    commands (reference mpi_job_controller.go:181-215).
    Single-process (no env) ⇒ a world of 1 without a process group.
    """
    global _initialized
    if _initialized:
        return
    rank = _env_int("RANK", "OMPI_COMM_WORLD_RANK", "PMI_RANK")
    world = _env_int("WORLD_SIZE", "OMPI_COMM_WORLD_SIZE", "PMI_SIZE")
    if rank is None or world is None or world == 1:
        _initialized = True
        return
    os.environ.setdefault("RANK", str(rank))
    os.environ.setdefault("WORLD_SIZE", str(world))
    lr = _env_int("LOCAL_RANK", "OMPI_COMM_WORLD_LOCAL_RANK", default=None)
    if lr is None:
        lr = rank % max(torch.cuda.device_count(), 1) if torch.cuda.is_available() else 0
    os.environ.setdefault("LOCAL_RANK", str(lr))
    os.environ.setdefault("MASTER_ADDR", os.environ.get("MPIAMD_MASTER_ADDR", "127.0.0.1"))
    os.environ.setdefault("MASTER_PORT", os.environ.get("MPIAMD_MASTER_PORT", "29500"))
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    kw = {}
    if torch.cuda.is_available():
        torch.cuda.set_device(local_rank())
        if backend == "nccl":
            # bind the device at init: eager RCCL communicator creation, and
            # a bound PG is what makes collectives hipGraph-capturable
            kw["device_id"] = torch.device(f"cuda:{local_rank()}")
    dist.init_process_group(backend=backend, init_method="env://",
                            timeout=datetime.timedelta(seconds=timeout_s), **kw)
    _initialized = True


def is_initialized() -> bool:
    return _initialized


def shutdown() -> None:
    global _initialized
    if dist.is_available() and dist.is_initialized():
        dist.destroy_process_group()
    _initialized = False


def rank() -> int:
    return dist.get_rank() if dist.is_initialized() else 0


def size() -> int:
    return dist.get_world_size() if dist.is_initialized() else 1


def local_rank() -> int:
    return _env_int("LOCAL_RANK", "OMPI_COMM_WORLD_LOCAL_RANK", default=0)


def barrier() -> None:
    if dist.is_initialized():
        dist.barrier()


def allreduce_(tensor: torch.Tensor, average: bool = True) -> torch.Tensor:
    """In-place synchronous allreduce (Horovod hvd.allreduce_ equivalent)."""
    if dist.is_initialized():
        dist.all_reduce(tensor)
        if average:
            tensor.div_(size())
    return tensor


def broadcast_parameters(module: torch.nn.Module, root_rank: int = 0) -> None:
    """One-time weight broadcast from root — the RCCL equivalent of
    Horovod's BroadcastGlobalVariablesHook(0)."""
    if not dist.is_initialized():
        return
    tensors = [p.data for p in module.parameters()] + list(module.buffers())
    for t in tensors:
        dist.broadcast(t, src=root_rank)


def broadcast_object(obj, root_rank: int = 0):
    if not dist.is_initialized():
        return obj
    box = [obj]
    dist.broadcast_object_list(box, src=root_rank)
    return box[0]
