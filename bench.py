#!/usr/bin/env python3
"""Flagship benchmark: ResNet101 synthetic-ImageNet training images/sec —
the reference's headline metric (tf_cnn_benchmarks ResNet101,
bs=64/GPU, synthetic data, SGD, Horovod allreduce; 308.27 images/sec on two
2019-class GPUs).

Run (driver contract):
    python bench.py --gpus N --steps K --warmup W
For N>1 the driver launches this under torch.distributed.run with one rank
per GPU; ranks rendezvous over env:// and allreduce over RCCL/xGMI.

Prints ONE JSON line from rank 0.
"""
from __future__ import annotations

import argparse
import json
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

BASELINE_TOTAL_IMAGES_PER_SEC = 308.27  # reference README.md:211-213 (2 GPUs)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    # >=100 graph-replayed steps: a multi-second timed region that coarse
    # SMI samplers can't miss (the 20-step default produced a 0.38 s window)
    ap.add_argument("--steps", type=int, default=100)
    ap.add_argument("--warmup", type=int, default=10)
    ap.add_argument("--model", default="resnet101", choices=["resnet101", "resnet50"])
    ap.add_argument("--batch", type=int, default=64, help="per-GPU batch")
    ap.add_argument("--image", type=int, default=224)
    ap.add_argument("--impl", default="hip", choices=["hip", "torch"],
                    help="hip: hand-written CDNA4 kernels (default, the judged path); "
                         "torch: stock PyTorch-ROCm ops for A/B comparison")
    ap.add_argument("--bucket-mb", type=int, default=28)
    args = ap.parse_args()

    from mpi_operator_amd import parallel as hvd
    from mpi_operator_amd import models
    from mpi_operator_amd.trainer import SyntheticImageData, make_trainer, timed_steps

    hvd.init()
    use_cuda = torch.cuda.is_available()
    # --gpus N>1 REQUIRES an N-rank process group (the driver's torchrun
    # contract). A single process claiming N GPUs would multiply
    # global_batch by N while computing on one GPU — refuse to inflate.
    if use_cuda and args.gpus > 1 and hvd.size() != args.gpus:
        raise SystemExit(
            f"bench.py --gpus {args.gpus} requires world_size={args.gpus} "
            f"(got {hvd.size()}); launch via torch.distributed.run")
    n_gpus = hvd.size() if hvd.size() > 1 else args.gpus

    if args.impl == "torch":
        os.environ["MPIAMD_IMPL"] = "torch"
        from mpi_operator_amd.models import resnet_torch
        model_fn = getattr(resnet_torch, args.model)
    else:
        model_fn = getattr(models, args.model)

    batch, image = args.batch, args.image
    if use_cuda:
        device = f"cuda:{hvd.local_rank()}"
        torch.cuda.set_device(device)
        model = model_fn()
        if args.impl == "hip":
            model = models.to_mi355x(model, device)
        else:
            # stock-PyTorch baseline: fp32 params + bf16 autocast (MIOpen path)
            model = model.to(device=device, memory_format=torch.channels_last)
            fwd = model.forward

            def autocast_fwd(x, _fwd=fwd):
                with torch.autocast("cuda", torch.bfloat16):
                    return _fwd(x)
            model.forward = autocast_fwd
        dtype = torch.bfloat16
    else:
        # CPU smoke mode (no GPU in the build container): tiny shapes, same path
        device, dtype = "cpu", torch.float32
        batch, image = 2, 64
        model = model_fn()
    model.train()

    p2p_busbw = None
    if use_cuda and hvd.size() > 1:
        # same-node P2P guard: fail loudly if RCCL fell back
        # to TCP for ranks sharing a node (multi-pod config 4)
        from mpi_operator_amd.parallel.p2p import verify_p2p
        p2p_busbw = verify_p2p().get("busbw_gb")

    data = SyntheticImageData(batch, image, 1000, device=device, dtype=dtype,
                              channels_last=use_cuda)
    dopt = make_trainer(model, bucket_bytes=args.bucket_mb * 1024 * 1024)
    res = timed_steps(model, dopt, data, args.steps, args.warmup, device)

    global_batch = batch * n_gpus
    images_per_sec = global_batch * args.steps / res["elapsed"]
    ms_per_step = res["elapsed"] / args.steps * 1000.0

    if hvd.rank() == 0:
        out = {
            "metric": "images/sec",
            "value": round(images_per_sec, 2),
            "unit": "images/sec",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": round(images_per_sec / BASELINE_TOTAL_IMAGES_PER_SEC, 3),
            "dtype": "bf16" if use_cuda else "fp32",
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": global_batch,
                "image": image,
                "parallelism": f"dp{n_gpus}",
                "impl": args.impl,
                "graph": res.get("graph", False),
                "loss": res["loss"],
                "p2p_busbw_gb": p2p_busbw,
            },
        }
        print(json.dumps(out), flush=True)


if __name__ == "__main__":
    main()
