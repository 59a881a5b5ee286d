"""BERT family CPU tests: shapes, gradient flow, loss decrease on a tiny
config, and BertLinear numerics vs torch.nn.functional.linear."""
import torch
import torch.nn.functional as F

from mpi_operator_amd.models.bert import (BertConfig, BertForPreTraining, BertLinear)


def tiny_cfg():
    return BertConfig(vocab_size=97, hidden=32, layers=2, heads=4,
                      intermediate=64, max_seq=32)


def test_bert_linear_matches_torch():
    torch.manual_seed(0)
    lin = BertLinear(16, 24)
    x = torch.randn(4, 5, 16)
    y = lin(x)
    assert y.shape == (4, 5, 24)
    ref = F.linear(x, lin.weight, lin.bias)
    torch.testing.assert_close(y, ref, rtol=1e-5, atol=1e-5)


def test_bert_forward_shapes():
    torch.manual_seed(0)
    cfg = tiny_cfg()
    m = BertForPreTraining(cfg)
    ids = torch.randint(0, cfg.vocab_size, (2, 16))
    mlm, nsp = m(ids)
    assert mlm.shape == (2, 16, cfg.vocab_size)
    assert nsp.shape == (2, 2)


def test_bert_attention_mask_blocks_padding():
    torch.manual_seed(0)
    cfg = tiny_cfg()
    m = BertForPreTraining(cfg)
    m.eval()
    ids = torch.randint(1, cfg.vocab_size, (1, 8))
    mask = torch.ones(1, 8)
    with torch.no_grad():
        base_mlm, _ = m(ids, attn_mask=mask)
        # changing a masked-out (padding) token must not change other outputs
        ids2 = ids.clone()
        ids2[0, 7] = (ids[0, 7] + 1) % cfg.vocab_size
        mask2 = mask.clone()
        mask2[0, 7] = 0
        out_a, _ = m(ids, attn_mask=mask2)
        out_b, _ = m(ids2, attn_mask=mask2)
    torch.testing.assert_close(out_a[:, :7], out_b[:, :7], rtol=1e-4, atol=1e-4)
    assert not torch.allclose(base_mlm[:, :7], out_a[:, :7])


def test_bert_loss_decreases():
    torch.manual_seed(0)
    cfg = tiny_cfg()
    m = BertForPreTraining(cfg)
    opt = torch.optim.Adam(m.parameters(), lr=3e-3)
    ids = torch.randint(0, cfg.vocab_size, (4, 16))
    mlm_labels = torch.full_like(ids, -100)
    mlm_labels[:, ::4] = ids[:, ::4]
    nsp = torch.randint(0, 2, (4,))
    losses = []
    for _ in range(15):
        opt.zero_grad()
        mlm_logits, nsp_logits = m(ids)
        loss = m.loss(mlm_logits, nsp_logits, mlm_labels, nsp)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert losses[-1] < losses[0] * 0.7, losses


def test_bert_all_params_receive_grads():
    cfg = tiny_cfg()
    m = BertForPreTraining(cfg)
    ids = torch.randint(0, cfg.vocab_size, (2, 8))
    mlm_labels = ids.clone()
    nsp = torch.randint(0, 2, (2,))
    mlm_logits, nsp_logits = m(ids, type_ids=torch.zeros_like(ids))
    m.loss(mlm_logits, nsp_logits, mlm_labels, nsp).backward()
    missing = [n for n, p in m.named_parameters()
               if p.grad is None and "pos" not in n and "typ" not in n]
    assert not missing, missing


def test_labels_in_forward_matches_loss_composition():
    """m(ids, mlm_labels=..., nsp_labels=...) (HF-style) equals the
    logits + m.loss(...) composition on the CPU path."""
    torch.manual_seed(5)
    cfg = tiny_cfg()
    m = BertForPreTraining(cfg)
    ids = torch.randint(0, cfg.vocab_size, (3, 12))
    mlm_labels = torch.full_like(ids, -100)
    mlm_labels[:, ::3] = ids[:, ::3]
    nsp = torch.randint(0, 2, (3,))
    loss_fwd = m(ids, mlm_labels=mlm_labels, nsp_labels=nsp)
    mlm_logits, nsp_logits = m(ids)
    loss_ref = m.loss(mlm_logits, nsp_logits, mlm_labels, nsp)
    assert abs(loss_fwd.item() - loss_ref.item()) < 1e-5 * abs(loss_ref.item()) + 1e-6
