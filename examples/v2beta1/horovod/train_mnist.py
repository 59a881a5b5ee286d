#!/usr/bin/env python3
"""Elastic-capable data-parallel MNIST-scale training on the MI355X stack —
the counterpart of the reference's examples/v2beta1/horovod/tensorflow_mnist.py
(Horovod+TF), rebuilt on mpi_operator_amd.parallel (RCCL over xGMI).

Run under the MPIJob launcher:
    amdrun -np 8 --hostfile /etc/mpi/hostfile -- python3 train_mnist.py
or elastically (reference proposals/elastic-horovod.md path):
    amdrun --elastic --discover /etc/mpi/discover_hosts.sh \
           --min-np 2 --max-np 8 -- python3 train_mnist.py
"""
import argparse
import os
import sys

import torch

# allow running straight from a source checkout (python examples/.../x.py)
sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), "..", "..", "..")))

from mpi_operator_amd import parallel as hvd
from mpi_operator_amd.models import SimpleCNN
from mpi_operator_amd.parallel import elastic


def synthetic_mnist(batch, device, bf16):
    x = torch.randn(batch, 1, 28, 28, device=device)
    if bf16:
        x = x.to(torch.bfloat16).contiguous(memory_format=torch.channels_last)
    y = torch.randint(0, 10, (batch,), device=device)
    return x, y


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=200)
    ap.add_argument("--batch", type=int, default=64)
    ap.add_argument("--lr", type=float, default=0.001)
    args = ap.parse_args()

    hvd.init()
    use_cuda = torch.cuda.is_available()
    device = f"cuda:{hvd.local_rank()}" if use_cuda else "cpu"

    model = SimpleCNN(in_ch=1, num_classes=10)
    if use_cuda:
        # MI355X layout/dtype contract: bf16 conv/linear weights,
        # channels-last memory, fp32 BN stats
        from mpi_operator_amd.models import to_mi355x
        model = to_mi355x(model, device)
    else:
        model = model.to(device)
    # lr scaled by world size, as in the reference example (lr * hvd.size())
    opt = torch.optim.Adam(model.parameters(), lr=args.lr * hvd.size())
    opt = hvd.DistributedOptimizer(opt, model.named_parameters())

    # BroadcastGlobalVariablesHook(0) equivalent; under elastic restarts this
    # also re-syncs optimizer state and the step counter from rank 0.
    state = elastic.ElasticState(model, opt.optimizer, step=0)
    extra = state.sync(root_rank=0)
    start_step = extra.get("step", 0)

    loss_fn = torch.nn.CrossEntropyLoss()
    model.train()
    # StopAtStepHook equivalent: total steps divided across the world
    for step in range(start_step, args.steps // hvd.size()):
        x, y = synthetic_mnist(args.batch, device, use_cuda)
        opt.zero_grad()
        loss = loss_fn(model(x), y)
        loss.backward()
        opt.step()
        state.extra["step"] = step + 1
        if hvd.rank() == 0 and step % 10 == 0:
            print(f"step {step} loss {loss.item():.4f}", flush=True)

    if hvd.rank() == 0:
        # rank-0-only checkpoint, as in the reference (tensorflow_mnist.py:159)
        torch.save(model.state_dict(), "/tmp/mnist.pt")
        print("done", flush=True)


if __name__ == "__main__":
    main()
