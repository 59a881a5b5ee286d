"""Workload-side elastic state helpers — pairs with runtime.elastic's
re-formation: on (re)start every rank calls `resume_state`, which broadcasts
rank 0's model/optimizer state so training continues seamlessly after the
host set changed (the Horovod-elastic state-sync equivalent, reference
proposals/elastic-horovod.md:20-29)."""
from __future__ import annotations

import os

import torch

from . import broadcast_object, broadcast_parameters, rank


def is_elastic() -> bool:
    return os.environ.get("MPIAMD_ELASTIC") == "1"


def restart_count() -> int:
    return int(os.environ.get("MPIAMD_RESTART_COUNT", "0"))


class ElasticState:
    """Holds the replicated training state: model params/buffers, optimizer
    state, and a user dict (epoch/step counters)."""

    def __init__(self, model: torch.nn.Module, optimizer=None, **extra):
        self.model = model
        self.optimizer = optimizer
        self.extra = extra

    def sync(self, root_rank: int = 0) -> dict:
        """Broadcast rank-0 state to all ranks; returns the extra dict."""
        broadcast_parameters(self.model, root_rank)
        if self.optimizer is not None:
            sd = self.optimizer.state_dict() if rank() == root_rank else None
            sd = broadcast_object(sd, root_rank)
            if rank() != root_rank and sd is not None:
                self.optimizer.load_state_dict(sd)
        self.extra = broadcast_object(self.extra, root_rank)
        return self.extra
