"""Fused bottleneck vs the unfused per-op autograd path: forward values,
input grad, and every parameter grad must match (same kernels, same order —
tolerances only cover bf16 rounding of the fused join)."""
import copy

import pytest
import torch

pytestmark = pytest.mark.gpu


def relerr(a, b):
    a, b = a.float(), b.float()
    return (a - b).norm().item() / (b.norm().item() + 1e-12)


def run_block(block, x, fused: bool):
    from mpi_operator_amd.ops import add_relu
    block.zero_grad(set_to_none=True)
    x = x.detach().clone().requires_grad_(True)
    if fused:
        out = block(x)  # training + cuda → fused path
    else:
        identity = x if block.downsample is None else block.downsample(x)
        h = block.bn1(block.conv1(x))
        h = block.bn2(block.conv2(h))
        h = block.bn3(block.conv3(h))
        out = add_relu(h, identity)
    out.float().pow(2).mean().backward()
    grads = {n: p.grad.detach().clone() for n, p in block.named_parameters()}
    return out.detach(), x.grad.detach().clone(), grads


@pytest.mark.parametrize("stride,ds", [(1, False), (2, True), (1, True)])
def test_fused_bottleneck_matches_unfused(stride, ds):
    from mpi_operator_amd.models.resnet import Bottleneck
    from mpi_operator_amd.ops import BatchNormReLU, Conv2d
    import torch.nn as nn

    torch.manual_seed(0)
    width = 32
    out_ch = width * 4
    # identity blocks (no downsample) require in_ch == out_ch, as in ResNet
    in_ch = out_ch if not ds else 64
    downsample = None
    if ds:
        downsample = nn.Sequential(Conv2d(in_ch, out_ch, 1, stride=stride),
                                   BatchNormReLU(out_ch, relu=False))
    blk = Bottleneck(in_ch, width, stride, downsample)
    blk = blk.to("cuda")
    for m in blk.modules():
        if isinstance(m, Conv2d):
            m.to(torch.bfloat16)
    blk = blk.to(memory_format=torch.channels_last)
    blk.train()

    x = (torch.rand(4, in_ch, 16, 16, device="cuda") * 2 - 1).to(torch.bfloat16)
    x = x.contiguous(memory_format=torch.channels_last)

    blk_ref = copy.deepcopy(blk)
    out_f, dx_f, g_f = run_block(blk, x, fused=True)
    out_u, dx_u, g_u = run_block(blk_ref, x, fused=False)

    assert relerr(out_f, out_u) < 0.02, relerr(out_f, out_u)
    assert relerr(dx_f, dx_u) < 0.03, relerr(dx_f, dx_u)
    for n in g_u:
        assert relerr(g_f[n], g_u[n]) < 0.03, (n, relerr(g_f[n], g_u[n]))
    # running stats must update identically (one train fwd each)
    for (n, b_f), (_, b_u) in zip(blk.named_buffers(), blk_ref.named_buffers()):
        assert relerr(b_f, b_u) < 1e-3, n
