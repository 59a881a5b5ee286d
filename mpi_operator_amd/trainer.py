"""Training-loop driver: the in-repo equivalent of the reference's
tf_cnn_benchmarks/Horovod step loop (reference README.md:96-143,
tensorflow_mnist.py:165-171) — synthetic-data ResNet training with
images/sec accounting, DistributedOptimizer overlap, and rank-0 logging."""
from __future__ import annotations

import time

import torch

from . import parallel as hvd
from .optim import FusedSGD
from .parallel import DistributedOptimizer


class SyntheticImageData:
    """Synthetic ImageNet-shaped data, random-init (no network access for
    datasets — mirrors tf_cnn_benchmarks --data_name=synthetic)."""

    def __init__(self, batch: int, image: int = 224, classes: int = 1000,
                 device="cpu", dtype=torch.float32, channels_last=False, seed: int = 1234):
        g = torch.Generator(device="cpu").manual_seed(seed + hvd.rank())
        x = torch.rand(batch, 3, image, image, generator=g) * 2 - 1
        if channels_last:
            # input pipeline emits NHWC8: RGB zero-padded to 8 channels once,
            # so the per-step stem pad disappears from the hot loop
            x = torch.nn.functional.pad(x, (0, 0, 0, 0, 0, 5))
        y = torch.randint(0, classes, (batch,), generator=g)
        self.x = x.to(device=device, dtype=dtype)
        if channels_last:
            self.x = self.x.contiguous(memory_format=torch.channels_last)
        self.y = y.to(device)

    def __iter__(self):
        while True:
            yield self.x, self.y


def make_trainer(model, lr: float = 0.1, momentum: float = 0.9, weight_decay: float = 1e-4,
                 bucket_bytes: int | None = None):
    """LR scaled by world size, as the Horovod example does
    (reference tensorflow_mnist.py:123-130)."""
    opt = FusedSGD(model.parameters(), lr=lr * hvd.size(), momentum=momentum,
                   weight_decay=weight_decay)
    kw = {}
    if bucket_bytes is not None:
        kw["bucket_bytes"] = bucket_bytes
    dopt = DistributedOptimizer(opt, model.named_parameters(), **kw)
    hvd.broadcast_parameters(model, root_rank=0)
    return dopt


def train_step(model, dopt, x, y):
    logits = model(x)
    loss = model.loss(logits, y)
    dopt.zero_grad()
    loss.backward()
    dopt.step()
    return loss


def _all_ranks_ok(local_ok: bool, use_cuda: bool) -> bool:
    """World-level AND of a local success flag (no-op in a world of 1)."""
    if hvd.size() <= 1:
        return local_ok
    import torch.distributed as dist
    flag = torch.tensor([1.0 if local_ok else 0.0],
                        device="cuda" if use_cuda else "cpu")
    dist.all_reduce(flag, op=dist.ReduceOp.MIN)
    return flag.item() > 0


def timed_steps(model, dopt, data_iter, steps: int, warmup: int, device,
                graph: bool | None = None) -> dict:
    """Run warmup+steps; barrier+sync bracketed timing of exactly `steps`.

    On GPU the whole training step (fwd+bwd+allreduce+optimizer) is captured
    into ONE hipGraph after warmup and replayed per step: ~1.8k kernel
    launches/step collapse their ~1.2-1.9 µs boundaries and all host launch
    overhead into a single graph replay. Every replay does the full step —
    nothing leaves the timed region. Set MPIAMD_GRAPH=0 (or graph=False) to
    run eager."""
    import os
    it = iter(data_iter)
    use_cuda = torch.cuda.is_available() and str(device).startswith("cuda")
    if graph is None:
        graph = use_cuda and os.environ.get("MPIAMD_GRAPH", "1") == "1"
    for _ in range(warmup):
        x, y = next(it)
        train_step(model, dopt, x, y)

    g = loss_static = None
    if graph and use_cuda:
        x, y = next(it)  # synthetic data: same tensors every step
        gr = err = None
        try:
            torch.cuda.synchronize()
            gr = torch.cuda.CUDAGraph()
            with torch.cuda.graph(gr):
                loss_static = train_step(model, dopt, x, y)
        except Exception as e:  # capture unsupported (e.g. some comm paths)
            gr, err = None, e
        # Replays contain the bucket allreduces, so graph-vs-eager must be
        # a WORLD-level decision: if any rank failed to capture, all ranks
        # run eager (a split would mismatch collectives and deadlock).
        if _all_ranks_ok(gr is not None, use_cuda):
            gr.replay()  # one verification replay (counts as warmup)
            torch.cuda.synchronize()
            lv = float(loss_static.detach())
            if _all_ranks_ok(lv == lv, use_cuda):
                g = gr
            else:
                err = err or RuntimeError("non-finite loss after replay")
        if g is None and hvd.rank() == 0:
            print(f"# hipGraph capture unavailable ({err}); running eager",
                  flush=True)

    hvd.barrier()
    if use_cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    loss = None
    if g is not None:
        for _ in range(steps):
            g.replay()
        loss = loss_static
    else:
        for _ in range(steps):
            x, y = next(it)
            loss = train_step(model, dopt, x, y)
    hvd.barrier()
    if use_cuda:
        torch.cuda.synchronize()
    t1 = time.perf_counter()
    elapsed = t1 - t0
    # MAX over ranks (slowest rank defines the job's step time)
    t = torch.tensor([elapsed], dtype=torch.float64)
    if hvd.size() > 1:
        import torch.distributed as dist
        t = t.to("cuda") if use_cuda else t
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
    return {"elapsed": float(t.item()), "loss": float(loss.detach().item()),
            "steps": steps, "graph": g is not None}
