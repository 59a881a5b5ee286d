"""BERT on MI355X: HIP NT-GEMM linear path numerics vs fp32 torch reference,
odd/ragged GEMM shapes through the mixed-staging kernels, and a full
BERT-base train step (loss finite & decreasing)."""
import pytest
import torch

pytestmark = pytest.mark.gpu


def relerr(a, b):
    a, b = a.float(), b.float()
    return (a - b).norm().item() / (b.norm().item() + 1e-12)


def test_bert_linear_gpu_matches_fp32():
    from mpi_operator_amd.models.bert import BertLinear
    torch.manual_seed(0)
    lin = BertLinear(1024, 4096).to("cuda", torch.bfloat16)
    x = (torch.rand(4, 128, 1024, device="cuda") * 2 - 1).to(torch.bfloat16)
    y = lin(x)
    yr = torch.nn.functional.linear(x.float(), lin.weight.float(), lin.bias.float())
    assert relerr(y, yr) < 0.02


@pytest.mark.parametrize("M,N,K", [(512, 1000, 2048), (96, 24, 40), (256, 3072, 768),
                                   (8192, 256, 512),  # split-K dw (16 slabs)
                                   (4096, 1024, 1024)])  # attn-out shape (8 slabs)
def test_linear_bwd_ragged_shapes(M, N, K):
    """dx/dw via the TN-staged kernels on shapes incl. non-%128, non-%8."""
    from mpi_operator_amd.ops import hip_ext
    ext = hip_ext()
    torch.manual_seed(1)
    x = ((torch.rand(M, K, device="cuda") * 2 - 1)).to(torch.bfloat16)
    w = ((torch.rand(N, K, device="cuda") * 2 - 1) * 0.05).to(torch.bfloat16)
    dy = ((torch.rand(M, N, device="cuda") * 2 - 1)).to(torch.bfloat16)
    dx, dw, db = ext.linear_bwd(dy, x, w)
    assert relerr(dx, dy.float() @ w.float()) < 0.02
    assert relerr(dw, dy.float().t() @ x.float()) < 0.02
    assert relerr(db, dy.float().sum(0)) < 0.01


def test_bert_base_train_step_gpu():
    from mpi_operator_amd.models.bert import bert_base, to_mi355x_bert
    from mpi_operator_amd.optim import FusedSGD
    torch.manual_seed(2)
    m = to_mi355x_bert(bert_base(), "cuda")
    m.train()
    opt = FusedSGD(m.parameters(), lr=5e-3, momentum=0.9)
    ids = torch.randint(0, m.cfg.vocab_size, (2, 64), device="cuda")
    mlm_labels = ids.clone()
    nsp = torch.randint(0, 2, (2,), device="cuda")
    losses = []
    for _ in range(8):
        opt.zero_grad()
        mlm_logits, nsp_logits = m(ids)
        loss = m.loss(mlm_logits, nsp_logits, mlm_labels, nsp)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert all(l == l for l in losses), losses  # no NaN
    assert losses[-1] < losses[0], losses


@pytest.mark.parametrize("M,N", [(4096, 1024), (100, 768), (7, 2048)])
def test_layernorm_fwd_bwd_matches_fp32(M, N):
    from mpi_operator_amd.ops import functional as Fx
    torch.manual_seed(3)
    x = ((torch.rand(M, N, device="cuda") * 2 - 1) * 3).to(torch.bfloat16)
    g = torch.rand(N, device="cuda") + 0.5
    b = torch.randn(N, device="cuda") * 0.2
    x1 = x.clone().requires_grad_(True)
    y = Fx.layer_norm(x1, g.clone().requires_grad_(False), b, 1e-12)
    xr = x.float().clone().requires_grad_(True)
    gr = g.clone().requires_grad_(True)
    br = b.clone().requires_grad_(True)
    yr = torch.nn.functional.layer_norm(xr, (N,), gr, br, 1e-12)
    assert relerr(y, yr) < 0.02
    dy = ((torch.rand(M, N, device="cuda") * 2 - 1)).to(torch.bfloat16)
    # grads through our Fn (need g/b as leaf tensors for grad check)
    g2 = g.clone().requires_grad_(True)
    b2 = b.clone().requires_grad_(True)
    x2 = x.clone().requires_grad_(True)
    y2 = Fx.layer_norm(x2, g2, b2, 1e-12)
    y2.backward(dy)
    yr.backward(dy.float())
    assert relerr(x2.grad, xr.grad) < 0.03
    assert relerr(g2.grad, gr.grad) < 0.02
    assert relerr(b2.grad, br.grad) < 0.02
