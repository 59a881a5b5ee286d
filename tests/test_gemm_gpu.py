"""GPU numerics: the NT-GEMM MFMA tile vs torch.matmul (fp32 golden).

Transpose-detecting by construction (§: asymmetric operands): A and B are
independent random matrices, so a swapped fragment layout or C-map cannot
pass."""
import pytest
import torch

pytestmark = pytest.mark.gpu


def _gemm_case(M, N, K, c_f32, seed=0):
    from mpi_operator_amd.ops import hip_ext
    torch.manual_seed(seed)
    a = (torch.rand(M, K, device="cuda") * 2 - 1).to(torch.bfloat16)
    b = (torch.rand(N, K, device="cuda") * 2 - 1).to(torch.bfloat16)
    c = hip_ext().gemm_nt(a, b, c_f32)
    ref = a.float() @ b.float().t()
    err = (c.float() - ref).abs().max().item()
    scale = ref.abs().max().item() + 1e-6
    tol = (0.002 if c_f32 else 0.02) * scale + 0.02
    assert err < tol, f"M{M} N{N} K{K} f32={c_f32}: err {err} scale {scale}"


@pytest.mark.parametrize("shape", [
    (128, 128, 64),     # single tile, single K-step
    (256, 256, 256),    # multi-tile multi-K
    (64, 1000, 2048),   # classifier head shape (edge N)
    (200, 72, 136),     # every dim ragged (M,N edge; K%8==0 only)
    (512, 384, 576),    # conv-like K=9*64
    (512, 256, 64),     # full tiles below the 256-route occupancy floor
    (768, 512, 96),     # odd K-tile count (128² pipeline path)
    (2048, 1024, 128),  # XCD-swizzled 128² grid
    (8192, 1024, 128),  # TRUE 256² pipe route: nwg=128 ≥ occupancy floor
    (8192, 1024, 192),  # 256² route, odd K-tile count (K%64==0)
])
@pytest.mark.parametrize("c_f32", [False, True])
def test_gemm_nt_matches_matmul(shape, c_f32):
    _gemm_case(*shape, c_f32)


def test_gemm_identity_asymmetric():
    """A=I with asymmetric B: catches row/col-swapped C writes."""
    from mpi_operator_amd.ops import hip_ext
    K = 128
    a = torch.eye(K, device="cuda").to(torch.bfloat16)  # A [M=K][K]
    b = torch.arange(K * K, device="cuda", dtype=torch.float32).reshape(K, K)
    b = ((b % 37) - 18 + (b // K) * 0.25).to(torch.bfloat16)  # asymmetric
    c = hip_ext().gemm_nt(a, b, True)
    ref = b.float().t()  # I @ B^T
    assert torch.allclose(c, ref, atol=1e-2), (c - ref).abs().max()


def test_gemm_splitk_via_wgrad_path():
    """conv_wgrad's split-K path on a plain GEMM-shaped problem."""
    import torch.nn.functional as F
    torch.manual_seed(1)
    N, C, H, W, Kout = 4, 16, 16, 16, 32
    x = (torch.rand(N, C, H, W, device="cuda") * 2 - 1).to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    dy = (torch.rand(N, Kout, H, W, device="cuda") * 2 - 1).to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last)
    from mpi_operator_amd.ops import hip_ext
    dw = hip_ext().conv2d_wgrad(x, dy, 3, 3, 1, 1)
    ref = torch.nn.grad.conv2d_weight(x.float(), (Kout, C, 3, 3), dy.float(),
                                      stride=1, padding=1)
    err = (dw.float() - ref).abs().max().item()
    scale = ref.abs().max().item()
    assert err < 0.02 * scale + 0.05, (err, scale)
