"""BERT-Large for MI355X — the config-4 workload of BASELINE.json
("BERT-Large Horovod, 2 worker pods × 4 GPUs each"). The reference ships no
BERT code (its workload layer is out-of-tree Horovod images); this is the
stack's own transformer family, built the same MI355X-first way as ResNet:

- every GEMM-shaped op (QKV projection, attention output, FFN in/out, MLM
  head) goes through the hand-written MFMA NT-GEMM HIP kernel
  (ops/csrc/gemm.hip) via ops.functional.linear;
- bf16 activations/params end-to-end (fp32 master weights live in FusedSGD);
- attention scores use torch.matmul (rocBLAS batched GEMM on ROCm) — the
  batched-GEMM shapes are library-friendly; fusing them by hand is a later
  optimization, and the dispatch seam is ops.functional.
"""
from __future__ import annotations

import math
from dataclasses import dataclass

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import functional as Fx


import os as _os

# default ON since the LN-bwd split + per-channel reduce: 929 vs 893
# seq/s same-box (MPIAMD_FUSED_LN=0 reverts to torch-native LN)
_FUSED_LN = _os.environ.get("MPIAMD_FUSED_LN", "1") == "1"


class LayerNorm(nn.Module):
    """LayerNorm with fp32 statistics on the hand-written HIP kernel
    (ops/csrc/layernorm.hip). Default since round 2: the backward's dx
    pass was un-capped from the dgamma/dbeta slab grid (split kernels) and
    the slab reduce made per-channel — fused now measures 929 vs 893
    seq/s against torch-native on BERT-Large (MPIAMD_FUSED_LN=0 reverts)."""

    def __init__(self, n: int, eps: float = 1e-12):
        super().__init__()
        self.n, self.eps = n, eps
        self.weight = nn.Parameter(torch.ones(n))
        self.bias = nn.Parameter(torch.zeros(n))

    def forward(self, x):
        if _FUSED_LN and x.is_cuda and x.dtype == torch.bfloat16:
            return Fx.layer_norm(x, self.weight.float(), self.bias.float(),
                                 self.eps)
        return F.layer_norm(x.float(), (self.n,), self.weight.float(),
                            self.bias.float(), self.eps).to(x.dtype)


@dataclass
class BertConfig:
    vocab_size: int = 30522
    hidden: int = 1024
    layers: int = 24
    heads: int = 16
    intermediate: int = 4096
    max_seq: int = 512
    type_vocab: int = 2
    dropout: float = 0.0  # benchmarks run dropout-free (synthetic data)
    eps: float = 1e-12


def bert_large() -> "BertForPreTraining":
    return BertForPreTraining(BertConfig())


def bert_base() -> "BertForPreTraining":
    return BertForPreTraining(BertConfig(hidden=768, layers=12, heads=12, intermediate=3072))


class BertLinear(nn.Module):
    """Linear over the HIP NT-GEMM for (B·S, in) × (out, in)ᵀ shapes.

    Accepts (..., in_f); flattens leading dims for the 2-D kernel."""

    def __init__(self, in_f: int, out_f: int):
        super().__init__()
        w = torch.empty(out_f, in_f)
        nn.init.normal_(w, std=0.02)
        self.weight = nn.Parameter(w)
        self.bias = nn.Parameter(torch.zeros(out_f))

    def forward(self, x):
        lead = x.shape[:-1]
        y = Fx.linear(x.reshape(-1, x.shape[-1]).contiguous(), self.weight, self.bias)
        return y.reshape(*lead, -1)


class SelfAttention(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.heads = cfg.heads
        self.head_dim = cfg.hidden // cfg.heads
        # fused QKV: one big GEMM instead of three (3× fewer kernel launches,
        # one pass over the activations)
        self.qkv = BertLinear(cfg.hidden, 3 * cfg.hidden)
        self.out = BertLinear(cfg.hidden, cfg.hidden)
        self.scale = 1.0 / math.sqrt(self.head_dim)

    def forward(self, x, attn_mask=None):
        b, s, h = x.shape
        qkv = self.qkv(x).view(b, s, 3, self.heads, self.head_dim)
        if (x.is_cuda and x.dtype == torch.bfloat16 and self.head_dim == 64
                and s <= 256 and attn_mask is None):
            # fused QKᵀ→softmax→PV kernel (one launch per step instead of
            # the batched-matmul + softmax chain); emits ctx in [b, s, h]
            # directly — no transpose/reshape passes
            return self.out(Fx.attention(qkv, self.heads, self.scale))
        q, k, v = (qkv[:, :, i].transpose(1, 2) for i in range(3))  # b, nh, s, hd
        scores = torch.matmul(q, k.transpose(-1, -2)) * self.scale
        if attn_mask is not None:
            scores = scores + attn_mask
        probs = torch.softmax(scores.float(), dim=-1).to(v.dtype)
        ctx = torch.matmul(probs, v)  # b, nh, s, hd
        ctx = ctx.transpose(1, 2).reshape(b, s, h)
        return self.out(ctx)


class BertLayer(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.attn = SelfAttention(cfg)
        self.ln1 = LayerNorm(cfg.hidden, eps=cfg.eps)
        self.fc1 = BertLinear(cfg.hidden, cfg.intermediate)
        self.fc2 = BertLinear(cfg.intermediate, cfg.hidden)
        self.ln2 = LayerNorm(cfg.hidden, eps=cfg.eps)

    def _join_ln(self, ln, a, b):
        """residual + LN; fused single-pass kernel when enabled."""
        if _FUSED_LN and a.is_cuda and a.dtype == torch.bfloat16:
            return Fx.layer_norm_add(a, b, ln.weight.float(), ln.bias.float(),
                                     ln.eps)
        return ln(a + b)

    def forward(self, x, attn_mask=None):
        if (x.is_cuda and x.dtype == torch.bfloat16 and attn_mask is None
                and self.attn.head_dim == 64 and x.shape[1] == 128
                and _os.environ.get("MPIAMD_LAYER_FUSED", "0") == "1"):
            # whole-layer composite Function (Fx.BertLayerFn): residual-join
            # backward adds fold into accumulate-dgrad GEMM epilogues.
            # Measured same-box 1400 vs 1461 seq/s AGAINST the per-op path
            # at BERT-Large bs32 — the unsplit acc-dgrad GEMM (no split-K,
            # extra C read in the epilogue) costs more than the two
            # CUDAFunctor_add joins it removes. Default OFF.
            b, s, h = x.shape
            y = Fx.bert_layer(
                x.reshape(-1, h).contiguous(), b, s, self.attn.heads,
                self.attn.scale, self.ln1.eps,
                self.attn.qkv.weight, self.attn.qkv.bias,
                self.attn.out.weight, self.attn.out.bias,
                self.ln1.weight.float(), self.ln1.bias.float(),
                self.fc1.weight, self.fc1.bias,
                self.fc2.weight, self.fc2.bias,
                self.ln2.weight.float(), self.ln2.bias.float())
            return y.view(b, s, h)
        x = self._join_ln(self.ln1, x, self.attn(x, attn_mask))
        if x.is_cuda and x.dtype == torch.bfloat16:
            # fused FFN: GELU lives in the GEMM epilogues (fwd emits
            # h_pre + gelu(h); bwd's fc2-dx multiplies by gelu'(h_pre))
            b, s, h = x.shape
            y = Fx.ffn(x.reshape(-1, h).contiguous(),
                       self.fc1.weight, self.fc1.bias,
                       self.fc2.weight, self.fc2.bias).view(b, s, h)
        else:
            y = self.fc2(F.gelu(self.fc1(x), approximate="tanh"))
        return self._join_ln(self.ln2, x, y)


class BertEmbeddings(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.tok = nn.Embedding(cfg.vocab_size, cfg.hidden)
        self.pos = nn.Embedding(cfg.max_seq, cfg.hidden)
        self.typ = nn.Embedding(cfg.type_vocab, cfg.hidden)
        self.ln = LayerNorm(cfg.hidden, eps=cfg.eps)
        for e in (self.tok, self.pos, self.typ):
            nn.init.normal_(e.weight, std=0.02)

    def forward(self, ids, type_ids=None):
        s = ids.shape[1]
        pos = torch.arange(s, device=ids.device)
        x = self.tok(ids) + self.pos(pos)[None]
        if type_ids is not None:
            x = x + self.typ(type_ids)
        return self.ln(x)


class BertModel(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.cfg = cfg
        self.embeddings = BertEmbeddings(cfg)
        self.layers = nn.ModuleList(BertLayer(cfg) for _ in range(cfg.layers))

    def forward(self, ids, type_ids=None, attn_mask=None):
        if attn_mask is not None and attn_mask.dim() == 2:
            # (b, s) 1/0 mask → additive (b, 1, 1, s)
            attn_mask = (1.0 - attn_mask[:, None, None, :].float()) * torch.finfo(torch.float32).min
        x = self.embeddings(ids, type_ids)
        for layer in self.layers:
            x = layer(x, attn_mask)
        return x


class BertForPreTraining(nn.Module):
    """MLM + NSP heads, the standard pretraining objective."""

    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.cfg = cfg
        self.bert = BertModel(cfg)
        self.mlm_transform = BertLinear(cfg.hidden, cfg.hidden)
        self.mlm_ln = LayerNorm(cfg.hidden, eps=cfg.eps)
        # decoder tied to token embeddings (standard BERT weight tying)
        self.mlm_bias = nn.Parameter(torch.zeros(cfg.vocab_size))
        self.nsp = BertLinear(cfg.hidden, 2)

    def forward(self, ids, type_ids=None, attn_mask=None, mlm_labels=None,
                nsp_labels=None):
        x = self.bert(ids, type_ids, attn_mask)
        h = self.mlm_ln(F.gelu(self.mlm_transform(x), approximate="tanh"))
        tok_w = self.bert.embeddings.tok.weight
        if mlm_labels is not None:
            # labels-in-forward (HF-style) training path: the fused MLM head
            # never materializes the unpadded [b·s, V] logits — decoder GEMM,
            # masked CE and its backward all work on one padded-vocab buffer
            if h.is_cuda and h.dtype == torch.bfloat16:
                b, s, hd = h.shape
                l_mlm = Fx.mlm_head_loss(h.reshape(-1, hd).contiguous(),
                                         tok_w, self.mlm_bias.float(),
                                         mlm_labels.view(-1))
            else:
                logits = torch.matmul(h, tok_w.t().to(h.dtype)) \
                    + self.mlm_bias.to(h.dtype)
                l_mlm = F.cross_entropy(
                    logits.float().view(-1, self.cfg.vocab_size),
                    mlm_labels.view(-1), ignore_index=-100)
            loss = l_mlm
            if nsp_labels is not None:
                nsp_logits = self.nsp(x[:, 0])
                loss = loss + F.cross_entropy(nsp_logits.float(), nsp_labels)
            return loss
        if h.is_cuda and h.dtype == torch.bfloat16:
            # decoder GEMM on the in-tree NT kernel with the bias folded
            # into the epilogue (weight tying: grads flow to tok.weight)
            b, s, hd = h.shape
            mlm_logits = Fx.linear(h.reshape(-1, hd).contiguous(), tok_w,
                                   self.mlm_bias).view(b, s, -1)
        else:
            mlm_logits = torch.matmul(h, tok_w.t().to(h.dtype)) + self.mlm_bias.to(h.dtype)
        nsp_logits = self.nsp(x[:, 0])
        return mlm_logits, nsp_logits

    def loss(self, mlm_logits, nsp_logits, mlm_labels, nsp_labels):
        """mlm_labels: (b, s) with -100 at unmasked positions."""
        if mlm_logits.is_cuda and mlm_logits.dtype == torch.bfloat16:
            # fused masked CE: no fp32 logits cast, no fp32 probs — the
            # torch path moves ~1.5 GB extra HBM at bs32·seq128·vocab30k
            l_mlm = Fx.masked_softmax_cross_entropy(
                mlm_logits.view(-1, self.cfg.vocab_size), mlm_labels.view(-1))
        else:
            l_mlm = F.cross_entropy(
                mlm_logits.float().view(-1, self.cfg.vocab_size),
                mlm_labels.view(-1), ignore_index=-100)
        l_nsp = F.cross_entropy(nsp_logits.float(), nsp_labels)
        return l_mlm + l_nsp


def to_mi355x_bert(model: BertForPreTraining, device) -> BertForPreTraining:
    """bf16 working weights on the GPU, with LayerNorm params and linear
    biases kept fp32: the HIP kernels consume them as fp32 operands anyway
    (epilogue bias add, LN affine), so fp32 residency removes ~200 per-step
    dtype-cast kernels (.float() on entry + grad .to(bf16) on exit —
    ~2.6 ms/step on BERT-Large bs32, prof8 trace)."""
    model = model.to(device=device, dtype=torch.bfloat16)
    for mod in model.modules():
        if isinstance(mod, LayerNorm):
            mod.weight.data = mod.weight.data.float()
            mod.bias.data = mod.bias.data.float()
        elif isinstance(mod, BertLinear):
            mod.bias.data = mod.bias.data.float()
    model.mlm_bias.data = model.mlm_bias.data.float()
    return model
