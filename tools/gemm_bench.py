#!/usr/bin/env python3
"""GEMM TFLOP/s microbench for the hand-written MFMA kernels — isolates the
tile structure's efficiency from the conv gather loaders. Run on a GPU box:
    python tools/gemm_bench.py
Reference points (guide, 4096^3 bf16): 128^2-tile register-staged ~500 TF,
glds 2-buffer ~874 TF, 8-phase 256^2 ~1320 TF; peak 2.5 PF."""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from mpi_operator_amd.ops import hip_ext  # noqa: E402


def bench(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    ext = hip_ext()
    shapes = [
        (4096, 4096, 4096, "square 4k"),
        (8192, 8192, 8192, "square 8k"),
        (50176, 256, 512, "conv3 1x1-ish (M=N*28^2)"),
        (12544, 512, 1024, "conv4 1x1-ish"),
        (802816, 64, 64, "stem-scale skinny"),
    ]
    for M, N, K, name in shapes:
        a = (torch.rand(M, K, device="cuda") * 2 - 1).to(torch.bfloat16)
        b = (torch.rand(N, K, device="cuda") * 2 - 1).to(torch.bfloat16)
        dt = bench(lambda: ext.gemm_nt(a, b, False))
        tf = 2.0 * M * N * K / dt / 1e12
        print(f"nt_gemm  {name:28s} M={M:<7d} N={N:<5d} K={K:<5d} {dt*1e3:8.3f} ms  {tf:7.1f} TF")
        del a, b
    torch.cuda.empty_cache()

    # real ResNet conv shapes through the full fwd/dgrad/wgrad paths
    convs = [
        (64, 256, 56, 56, 64, 1, 1, 0, "b1 1x1 down (256->64 @56)"),
        (64, 64, 56, 56, 64, 3, 1, 1, "b1 3x3 (64 @56)"),
        (64, 512, 28, 28, 128, 1, 1, 0, "b2 1x1 down"),
        (64, 128, 28, 28, 128, 3, 1, 1, "b2 3x3"),
        (64, 256, 28, 28, 1024, 1, 1, 0, "b3 1x1 up (256->1024 @28... )"),
        (64, 1024, 14, 14, 256, 1, 1, 0, "b3 1x1 down"),
        (64, 256, 14, 14, 256, 3, 1, 1, "b3 3x3"),
        (64, 512, 7, 7, 512, 3, 1, 1, "b4 3x3"),
        (64, 256, 56, 56, 128, 1, 2, 0, "s2 1x1 (stage transition)"),
    ]
    for N, C, H, W, Kout, R, st, pad, name in convs:
        HO = (H + 2 * pad - R) // st + 1
        x = (torch.rand(N, C, H, W, device="cuda") * 2 - 1).to(torch.bfloat16) \
            .contiguous(memory_format=torch.channels_last)
        w = ((torch.rand(Kout, C, R, R, device="cuda") * 2 - 1) * 0.1).to(torch.bfloat16) \
            .contiguous(memory_format=torch.channels_last)
        dy = (torch.rand(N, Kout, HO, HO, device="cuda") * 2 - 1).to(torch.bfloat16) \
            .contiguous(memory_format=torch.channels_last)
        gf = 2.0 * N * HO * HO * Kout * R * R * C / 1e12
        tfw = gf / bench(lambda: ext.conv2d_fwd(x, w, st, pad))
        tfd = gf / bench(lambda: ext.conv2d_dgrad(dy, w, H, W, st, pad))
        tfg = gf / bench(lambda: ext.conv2d_wgrad(x, dy, R, R, st, pad))
        print(f"conv     {name:28s} fwd {tfw:6.1f}  dgrad {tfd:6.1f}  wgrad {tfg:6.1f} TF")
        del x, w, dy
        torch.cuda.empty_cache()


if __name__ == "__main__":
    main()
