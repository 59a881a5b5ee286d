#!/usr/bin/env python3
"""Isolate the stride-2 1x1 dgrad: run ONLY that op N times (for rocprof
--stats attribution) and print wall time."""
import sys
import os
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from mpi_operator_amd.ops import hip_ext  # noqa: E402

ext = hip_ext()
N, C, H, W, Kout, st = 64, 256, 56, 56, 128, 2
HO = H // st
x = (torch.rand(N, C, H, W, device="cuda") * 2 - 1).to(torch.bfloat16) \
    .contiguous(memory_format=torch.channels_last)
w = ((torch.rand(Kout, C, 1, 1, device="cuda") * 2 - 1) * 0.1).to(torch.bfloat16) \
    .contiguous(memory_format=torch.channels_last)
dy = (torch.rand(N, Kout, HO, HO, device="cuda") * 2 - 1).to(torch.bfloat16) \
    .contiguous(memory_format=torch.channels_last)

for _ in range(5):
    dx = ext.conv2d_dgrad(dy, w, H, W, st, 0)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(50):
    dx = ext.conv2d_dgrad(dy, w, H, W, st, 0)
torch.cuda.synchronize()
dt = (time.perf_counter() - t0) / 50
gf = 2.0 * N * HO * HO * Kout * C / 1e12
print(f"s2 1x1 dgrad: {dt*1e6:.1f} us  {gf/dt:.1f} TF")

# correctness
ref = torch.nn.grad.conv2d_input((N, C, H, W), w.float(), dy.float(), stride=st)
err = (dx.float() - ref).norm().item() / ref.norm().item()
print(f"relerr {err:.4f}")
