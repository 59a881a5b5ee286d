import os, sys, time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from mpi_operator_amd.ops import hip_ext
ext = hip_ext()
M, N, K = 4096, 4096, 1024
dy = torch.randn(M, K, device="cuda").to(torch.bfloat16)
w = torch.randn(K, N, device="cuda").to(torch.bfloat16) * 0.05
pre = torch.randn(M, N, device="cuda").to(torch.bfloat16)
for _ in range(10):
    dh = ext.linear_gelu_dgrad(dy, w, pre)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(200):
    dh = ext.linear_gelu_dgrad(dy, w, pre)
torch.cuda.synchronize()
dt = (time.perf_counter() - t0) / 200
print(f"linear_gelu_dgrad {M}x{N}x{K}: {dt*1e6:.1f} us/call, "
      f"{2*M*N*K/dt/1e12:.0f} TF")
