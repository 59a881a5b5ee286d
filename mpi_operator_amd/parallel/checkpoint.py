"""Workload-side checkpointing — the reference delegates this to the
workload image (rank-0-only checkpoint_dir + MonitoredTrainingSession
restore, reference examples/v2beta1/horovod/tensorflow_mnist.py:157-165;
SURVEY.md §5.4). Here it is a first-class API on the data plane:

- ``save_checkpoint``: rank 0 writes model/optimizer/extra state atomically
  (tmp file + rename so a killed pod never leaves a torn checkpoint), other
  ranks no-op; everyone meets at a barrier so the file is durable before any
  rank advances past the save point.
- ``load_checkpoint``: rank 0 reads the file, then the state is broadcast to
  every rank (broadcast-on-restore) — workers never need the checkpoint
  volume mounted, matching the reference's rank-0-only checkpoint_dir.

Both compose with elastic re-formation: after a host-set change, the runner
re-execs the script, which calls ``load_checkpoint`` and resumes.
"""
from __future__ import annotations

import os
import tempfile

import torch

from . import barrier, broadcast_object, broadcast_parameters, rank


def save_checkpoint(path: str, model: torch.nn.Module, optimizer=None,
                    **extra) -> None:
    """Rank-0-only atomic save of model + optimizer + user state."""
    if rank() == 0:
        state = {
            "model": model.state_dict(),
            "optimizer": optimizer.state_dict() if optimizer is not None else None,
            "extra": extra,
        }
        d = os.path.dirname(os.path.abspath(path))
        os.makedirs(d, exist_ok=True)
        fd, tmp = tempfile.mkstemp(dir=d, suffix=".tmp")
        try:
            with os.fdopen(fd, "wb") as f:
                torch.save(state, f)
            os.replace(tmp, path)
        finally:
            if os.path.exists(tmp):
                os.unlink(tmp)
    barrier()


def load_checkpoint(path: str, model: torch.nn.Module, optimizer=None,
                    map_location="cpu") -> dict:
    """Rank 0 reads, everyone restores via broadcast. Returns the extra
    dict ({} if the file does not exist — fresh start)."""
    state = None
    if rank() == 0 and os.path.exists(path):
        state = torch.load(path, map_location=map_location, weights_only=False)
    found = broadcast_object(state is not None)
    if not found:
        return {}
    if rank() == 0:
        model.load_state_dict(state["model"])
    broadcast_parameters(model)
    if optimizer is not None:
        # Broadcast the CPU-resident dict straight from the file (rank 0
        # loaded it with map_location="cpu").  Broadcasting the live
        # optimizer's state_dict() instead would pickle rank 0's CUDA
        # tensors, and every receiving rank would unpickle the full fp32
        # master + momentum state onto cuda:0 — a multi-GB spike on GPU 0
        # during an 8-rank restore.  load_state_dict re-homes CPU tensors
        # to each rank's own device.
        sd = state["optimizer"] if rank() == 0 else None
        sd = broadcast_object(sd)
        if sd is not None:
            optimizer.load_state_dict(sd)
    return broadcast_object(state["extra"] if rank() == 0 else None) or {}
