"""FusedSGD must match torch.optim.SGD exactly on fp32 params."""
import torch

from mpi_operator_amd.optim import FusedSGD


def _models():
    torch.manual_seed(3)
    a = torch.nn.Linear(8, 8)
    b = torch.nn.Linear(8, 8)
    b.load_state_dict(a.state_dict())
    return a, b


def test_fused_sgd_matches_torch_sgd():
    a, b = _models()
    oa = FusedSGD(a.parameters(), lr=0.1, momentum=0.9, weight_decay=0.01)
    ob = torch.optim.SGD(b.parameters(), lr=0.1, momentum=0.9, weight_decay=0.01)
    x = torch.randn(4, 8)
    for _ in range(5):
        oa.zero_grad(); ob.zero_grad()
        a(x).pow(2).mean().backward()
        b(x).pow(2).mean().backward()
        oa.step(); ob.step()
    for pa, pb in zip(a.parameters(), b.parameters()):
        assert torch.allclose(pa, pb, atol=1e-6), (pa - pb).abs().max()


def test_fused_sgd_bf16_master_weights():
    torch.manual_seed(4)
    m = torch.nn.Linear(16, 4).to(torch.bfloat16)
    opt = FusedSGD(m.parameters(), lr=0.05, momentum=0.9)
    x = torch.randn(8, 16, dtype=torch.bfloat16)
    for _ in range(3):
        opt.zero_grad()
        m(x).float().pow(2).mean().backward()
        opt.step()
    st = opt.state[m.weight]
    assert st["master"].dtype == torch.float32
    # bf16 working copy tracks the fp32 master
    assert torch.allclose(st["master"].to(torch.bfloat16), m.weight.data)
