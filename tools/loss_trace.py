#!/usr/bin/env python3
"""Per-step loss trace of the ResNet101 bench config (fixed synthetic batch,
eager) — separates kernel-correctness questions from plain training-dynamics
oscillation when comparing GEMM path configs."""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    steps = int(sys.argv[1]) if len(sys.argv) > 1 else 60
    from mpi_operator_amd import models, parallel as hvd
    from mpi_operator_amd.trainer import SyntheticImageData, make_trainer, train_step

    hvd.init()
    torch.cuda.set_device("cuda:0")
    model = models.to_mi355x(models.resnet101(), "cuda:0")
    model.train()
    data = SyntheticImageData(64, 224, 1000, device="cuda:0",
                              dtype=torch.bfloat16, channels_last=True)
    dopt = make_trainer(model)
    it = iter(data)
    for i in range(steps):
        x, y = next(it)
        loss = train_step(model, dopt, x, y)
        print(f"step {i:3d} loss {float(loss.detach()):.4f}", flush=True)


if __name__ == "__main__":
    main()
