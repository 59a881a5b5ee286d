"""MFMA layout probe: prints the hardware fragment mapping evidence.

Run on a GPU box: python tools/gpu_probe.py
If the assumed mapping is right, both checks print MATCH; otherwise the
printed matrices identify the true lane->element mapping (value i*100+k at
position [i][j] reveals which A element fed row i / which B element fed col j).
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402

from mpi_operator_amd.ops import hip_ext  # noqa: E402


def main():
    ext = hip_ext()
    dev = "cuda"
    # Case 1: A unique values, B = "identity" (B[n][k] = 1 iff n==k).
    a = torch.zeros(16, 32, device=dev)
    for i in range(16):
        for k in range(32):
            a[i, k] = i + (k + 1) * 0.0625  # exactly representable in bf16
    b = torch.zeros(16, 32, device=dev)
    for n in range(16):
        b[n, n] = 1.0
    d1 = ext.mfma_probe(a.to(torch.bfloat16), b.to(torch.bfloat16))
    ref1 = a[:, :16].t().t()  # A @ B^T = A[:, :16]
    ok1 = torch.allclose(d1.cpu(), a[:, :16].cpu(), atol=1e-2)
    print("case1 A-unique B-eye MATCH:", ok1)
    if not ok1:
        print(d1.cpu().numpy())
    # Case 2: A = eye, B unique.
    a2 = torch.zeros(16, 32, device=dev)
    for i in range(16):
        a2[i, i] = 1.0
    b2 = torch.zeros(16, 32, device=dev)
    for n in range(16):
        for k in range(32):
            b2[n, k] = n + (k + 1) * 0.0625
    d2 = ext.mfma_probe(a2.to(torch.bfloat16), b2.to(torch.bfloat16))
    # A @ B^T = B^T[:16] → D[i][j] = B[j][i]
    ok2 = torch.allclose(d2.cpu(), b2[:, :16].t().cpu(), atol=1e-2)
    print("case2 A-eye B-unique MATCH:", ok2)
    if not ok2:
        print(d2.cpu().numpy())


if __name__ == "__main__":
    main()
