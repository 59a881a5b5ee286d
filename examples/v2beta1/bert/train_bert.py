#!/usr/bin/env python3
"""BERT-Large pretraining on the MI355X stack — BASELINE.json config 4:
2 worker pods × 4 GPUs each on one node (SSH hostfile + multi-pod
rendezvous; RCCL stays on xGMI across pods because both pods share
/dev/kfd). Synthetic MLM+NSP data, bf16, sequences/sec reported."""
import argparse
import os
import sys
import time

import torch

# allow running straight from a source checkout (python examples/.../x.py)
sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), "..", "..", "..")))

from mpi_operator_amd import parallel as hvd
from mpi_operator_amd.models.bert import bert_large, bert_base, to_mi355x_bert
from mpi_operator_amd.optim import FusedSGD


def synthetic_batch(batch, seq, vocab, device):
    ids = torch.randint(0, vocab, (batch, seq), device=device)
    type_ids = torch.zeros_like(ids)
    mlm_labels = torch.full_like(ids, -100)
    mask = torch.rand(batch, seq, device=device) < 0.15
    mlm_labels[mask] = ids[mask]
    nsp = torch.randint(0, 2, (batch,), device=device)
    return ids, type_ids, mlm_labels, nsp


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="bert_large", choices=["bert_large", "bert_base"])
    ap.add_argument("--steps", type=int, default=50)
    ap.add_argument("--warmup", type=int, default=10)
    ap.add_argument("--batch", type=int, default=8, help="per-GPU sequences")
    ap.add_argument("--seq", type=int, default=128)
    args = ap.parse_args()

    hvd.init()
    use_cuda = torch.cuda.is_available()
    device = f"cuda:{hvd.local_rank()}" if use_cuda else "cpu"

    model = bert_large() if args.model == "bert_large" else bert_base()
    if use_cuda:
        model = to_mi355x_bert(model, device)
    model.train()
    opt = FusedSGD(model.parameters(), lr=1e-4, momentum=0.9)
    opt = hvd.DistributedOptimizer(opt, model.named_parameters())
    hvd.broadcast_parameters(model, 0)

    vocab = model.cfg.vocab_size

    def step():
        ids, type_ids, mlm_labels, nsp = synthetic_batch(args.batch, args.seq, vocab, device)
        opt.zero_grad()
        mlm_logits, nsp_logits = model(ids, type_ids)
        loss = model.loss(mlm_logits, nsp_logits, mlm_labels, nsp)
        loss.backward()
        opt.step()
        return loss

    for _ in range(args.warmup):
        step()
    if use_cuda:
        torch.cuda.synchronize()
    hvd.barrier()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        loss = step()
    if use_cuda:
        torch.cuda.synchronize()
    hvd.barrier()
    dt = time.perf_counter() - t0

    seq_per_sec = args.batch * hvd.size() * args.steps / dt
    if hvd.rank() == 0:
        print(f"bert: {seq_per_sec:.1f} sequences/sec over {hvd.size()} GPUs "
              f"({dt / args.steps * 1000:.1f} ms/step, loss {float(loss):.3f})", flush=True)


if __name__ == "__main__":
    main()
