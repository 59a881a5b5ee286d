"""CPU tests of the model zoo + training loop (PyTorch-reference op path)."""
import torch

from mpi_operator_amd import models
from mpi_operator_amd.optim import FusedSGD


def test_resnet50_forward_backward():
    torch.manual_seed(0)
    m = models.resnet50(num_classes=10)
    x = torch.randn(2, 3, 64, 64)
    y = torch.randint(0, 10, (2,))
    loss = m.loss(m(x), y)
    loss.backward()
    assert torch.isfinite(loss)
    for p in m.parameters():
        assert p.grad is not None and torch.isfinite(p.grad).all()


def test_resnet101_structure():
    m = models.resnet101()
    n_params = sum(p.numel() for p in m.parameters())
    # ResNet101 (1000 classes) ≈ 44.5M params — reference headline model
    # (tf_cnn_benchmarks resnet101). The stem pad 3→8 adds ~16k.
    assert 44e6 < n_params < 46e6
    n_bottlenecks = sum(1 for mod in m.modules() if type(mod).__name__ == "Bottleneck")
    assert n_bottlenecks == 33  # 3 + 4 + 23 + 3


def test_simple_cnn_loss_decreases():
    torch.manual_seed(0)
    m = models.SimpleCNN(in_ch=1, num_classes=10)
    m.train()
    x = torch.randn(16, 1, 28, 28)
    y = torch.randint(0, 10, (16,))
    opt = FusedSGD(m.parameters(), lr=0.05, momentum=0.9)
    first = None
    for i in range(12):
        opt.zero_grad()
        loss = m.loss(m(x), y)
        loss.backward()
        opt.step()
        if first is None:
            first = float(loss)
    assert float(loss) < first * 0.9, (first, float(loss))


def test_bn_running_stats_update():
    from mpi_operator_amd.ops import BatchNormReLU
    torch.manual_seed(0)
    bn = BatchNormReLU(8, relu=False, momentum=0.5)
    x = torch.randn(4, 8, 5, 5) * 3 + 1
    bn.train()
    bn(x)
    assert not torch.allclose(bn.running_mean, torch.zeros(8))
    ref = torch.nn.BatchNorm2d(8, momentum=0.5)
    ref.train()
    ref(x)
    assert torch.allclose(bn.running_mean, ref.running_mean, atol=1e-4)
    assert torch.allclose(bn.running_var, ref.running_var, atol=1e-3)


def test_ops_match_torch_bn_conv():
    """CPU reference ops vs torch built-ins on fp32 (the same golden model
    the GPU numerics tests use)."""
    import torch.nn.functional as F
    from mpi_operator_amd.ops import functional as Fx

    torch.manual_seed(1)
    x = torch.randn(2, 8, 9, 9, requires_grad=True)
    w = torch.randn(16, 8, 3, 3, requires_grad=True)
    y = Fx.conv2d(x, w, stride=2, padding=1)
    yr = F.conv2d(x, w, stride=2, padding=1)
    assert torch.allclose(y, yr, atol=1e-5)
    g = torch.randn_like(y)
    y.backward(g)
    x2 = x.detach().clone().requires_grad_()
    w2 = w.detach().clone().requires_grad_()
    F.conv2d(x2, w2, stride=2, padding=1).backward(g)
    assert torch.allclose(x.grad, x2.grad, atol=1e-5)
    assert torch.allclose(w.grad, w2.grad, atol=1e-4)
