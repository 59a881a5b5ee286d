"""FusedSGD — momentum SGD with fp32 master weights and a single fused
multi-tensor HIP kernel per step on GPU (SURVEY.md §2.3 N7 'fused SGD-momentum
update'; the reference's workload used tf SGD inside the external image).

bf16 parameters keep an fp32 master + fp32 momentum in optimizer state; the
kernel updates master & momentum and refreshes the bf16 working copy in one
pass over HBM."""
from __future__ import annotations

import torch

from .ops import sgd_momentum_step
from .ops import functional as _Fx


class FusedSGD(torch.optim.Optimizer):
    def __init__(self, params, lr: float, momentum: float = 0.9,
                 weight_decay: float = 0.0, nesterov: bool = False):
        defaults = dict(lr=lr, momentum=momentum, weight_decay=weight_decay, nesterov=nesterov)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        # side-stream wgrad mode: grads may still be in flight on the side
        # stream — order them before the fused update (no-op otherwise)
        _Fx.join_wgrad_stream()
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            masters, grads, momenta, outs = [], [], [], []
            for p in group["params"]:
                if p.grad is None:
                    continue
                st = self.state[p]
                if "momentum_buffer" not in st:
                    st["momentum_buffer"] = torch.zeros_like(p, dtype=torch.float32)
                    if p.dtype != torch.float32:
                        st["master"] = p.detach().float().clone()
                if p.dtype != torch.float32:
                    masters.append(st["master"])
                    outs.append(p.data)
                else:
                    masters.append(p.data)
                    outs.append(None)
                grads.append(p.grad)
                momenta.append(st["momentum_buffer"])
            if masters:
                sgd_momentum_step(masters, grads, momenta, outs, group["lr"],
                                  group["momentum"], group["weight_decay"], group["nesterov"])
        return loss
