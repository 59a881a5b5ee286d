#!/usr/bin/env python3
"""BERT-Large seq/s microbench (GPU box): the config-4 workload of
BASELINE.json measured standalone. Usage:
    python tools/bert_bench.py [--model bert_large] [--batch 32] [--seq 128]
                               [--steps 30] [--warmup 5]
"""
import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="bert_large",
                    choices=["bert_large", "bert_base"])
    ap.add_argument("--batch", type=int, default=32)
    ap.add_argument("--seq", type=int, default=128)
    ap.add_argument("--steps", type=int, default=30)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--graph", type=int, default=1)
    args = ap.parse_args()

    from mpi_operator_amd.models import bert as B
    from mpi_operator_amd.optim import FusedSGD

    torch.manual_seed(0)
    m = B.to_mi355x_bert(getattr(B, args.model)(), "cuda")
    m.train()
    opt = FusedSGD(m.parameters(), lr=1e-3, momentum=0.9)
    ids = torch.randint(0, m.cfg.vocab_size, (args.batch, args.seq), device="cuda")
    mlm_labels = ids.clone()
    nsp = torch.randint(0, 2, (args.batch,), device="cuda")

    def step():
        opt.zero_grad()
        # labels-in-forward: fused MLM head (no unpadded vocab logits)
        loss = m(ids, mlm_labels=mlm_labels, nsp_labels=nsp)
        loss.backward()
        opt.step()
        return loss

    for _ in range(args.warmup):
        loss = step()
    # drop the warmup autograd graph before capture: a live loss keeps
    # AccumulateGrad nodes from the eager stream alive, and capture_end
    # segfaults on the stream mismatch
    loss = None
    g = None
    if args.graph:
        try:
            torch.cuda.synchronize()
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                loss = step()
            g.replay()
        except Exception as e:
            print(f"# graph capture failed ({e}); eager", flush=True)
            g = None
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        if g is not None:
            g.replay()
        else:
            loss = step()
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(f"{args.model} bs{args.batch} seq{args.seq}: "
          f"{args.batch * args.steps / dt:.1f} seq/s "
          f"({dt / args.steps * 1e3:.2f} ms/step, graph={g is not None}, "
          f"loss={float(loss.detach()):.4f})")


if __name__ == "__main__":
    main()
