"""Plain-PyTorch reference implementations of every hot op.

These serve two roles:
  1. CPU execution path (tests and bring-up run on CPU-only machines).
  2. fp32 golden model for the HIP-kernel numerics tests
     (tests compare the gfx950 kernels against these at fp32).

Shapes follow the framework's layout contract: activations are logical NCHW
(PyTorch convention) and, on GPU, channels-last in memory (NHWC); conv
weights are logical [K, C, R, S] and channels-last in memory ([K][R][S][C]).
"""
from __future__ import annotations

import torch
import torch.nn.functional as F


def conv2d_fwd(x, w, stride: int, padding: int):
    return F.conv2d(x, w, bias=None, stride=stride, padding=padding)


def conv2d_dgrad(dy, w, x_shape, stride: int, padding: int):
    return torch.nn.grad.conv2d_input(x_shape, w, dy, stride=stride, padding=padding)


def conv2d_wgrad(x, dy, w_shape, stride: int, padding: int):
    return torch.nn.grad.conv2d_weight(x, w_shape, dy, stride=stride, padding=padding)


def bn_relu_fwd_train(x, gamma, beta, eps: float, relu: bool):
    """Returns (y, batch_mean, batch_invstd)."""
    dims = (0, 2, 3)
    xf = x.float()
    mean = xf.mean(dim=dims)
    var = xf.var(dim=dims, unbiased=False)
    invstd = (var + eps).rsqrt()
    xhat = (xf - mean[None, :, None, None]) * invstd[None, :, None, None]
    y = xhat * gamma.float()[None, :, None, None] + beta.float()[None, :, None, None]
    if relu:
        y = F.relu(y)
    return y.to(x.dtype), mean, invstd


def bn_relu_fwd_eval(x, gamma, beta, running_mean, running_var, eps: float, relu: bool):
    invstd = (running_var.float() + eps).rsqrt()
    scale = gamma.float() * invstd
    shift = beta.float() - running_mean.float() * scale
    y = x.float() * scale[None, :, None, None] + shift[None, :, None, None]
    if relu:
        y = F.relu(y)
    return y.to(x.dtype)


def bn_relu_bwd(dy, x, y, gamma, mean, invstd, relu: bool):
    """Returns (dx, dgamma, dbeta). ``y`` is the post-activation output
    (used for the ReLU mask); ``mean``/``invstd`` are the saved batch stats."""
    dims = (0, 2, 3)
    dyf = dy.float()
    if relu:
        dyf = dyf * (y.float() > 0)
    xf = x.float()
    xhat = (xf - mean[None, :, None, None]) * invstd[None, :, None, None]
    dbeta = dyf.sum(dim=dims)
    dgamma = (dyf * xhat).sum(dim=dims)
    n = x.numel() / x.shape[1]
    dx = (
        gamma.float()[None, :, None, None]
        * invstd[None, :, None, None]
        * (dyf - dbeta[None, :, None, None] / n - xhat * dgamma[None, :, None, None] / n)
    )
    return dx.to(dy.dtype), dgamma, dbeta


def max_pool2d_fwd(x, kernel: int, stride: int, padding: int):
    y, idx = F.max_pool2d(x, kernel, stride, padding, return_indices=True)
    return y, idx


def max_pool2d_bwd(dy, idx, x_shape, kernel: int, stride: int, padding: int):
    return F.max_unpool2d(dy, idx, kernel, stride, padding, output_size=x_shape[2:])


def global_avg_pool_fwd(x):
    return x.float().mean(dim=(2, 3)).to(x.dtype)


def global_avg_pool_bwd(dy, x_shape):
    n, c, h, w = x_shape
    return (dy.float() / (h * w))[:, :, None, None].expand(n, c, h, w).to(dy.dtype)


def linear_fwd(x, w, b):
    return F.linear(x, w, b)


def softmax_cross_entropy_fwd(logits, target):
    """Returns (loss_mean, softmax_probs) — probs saved for backward."""
    lf = logits.float()
    logp = F.log_softmax(lf, dim=-1)
    loss = F.nll_loss(logp, target)
    return loss, logp.exp()


def softmax_cross_entropy_bwd(probs, target, grad_scale: float):
    d = probs.clone()
    d[torch.arange(d.shape[0], device=d.device), target] -= 1.0
    return d * (grad_scale / d.shape[0])


def sgd_momentum_step(params32, grads32, momenta, bf16_outs, lr, momentum, weight_decay, nesterov=False):
    """fp32 master-weight SGD with momentum; writes updated fp32 masters and
    refreshed bf16 working copies in one pass (the HIP kernel fuses this)."""
    for p, g, m, out in zip(params32, grads32, momenta, bf16_outs):
        gf = g.float()
        if weight_decay:
            gf = gf.add(p, alpha=weight_decay)
        m.mul_(momentum).add_(gf)
        step = gf.add(m, alpha=momentum) if nesterov else m
        p.add_(step, alpha=-lr)
        if out is not None:
            out.copy_(p.to(out.dtype))


def masked_softmax_cross_entropy_fwd(logits, target, ignore_index=-100):
    """fp32 golden oracle for the MLM masked CE (mean over valid rows)."""
    x = logits.float()
    mx = x.max(dim=1, keepdim=True).values
    e = (x - mx).exp()
    probs = e / e.sum(dim=1, keepdim=True)
    valid = target != ignore_index
    n = int(valid.sum().item())
    if n == 0:
        return x.new_zeros(()), probs
    lse = e.sum(dim=1).log() + mx.squeeze(1)
    picked = x[valid].gather(1, target[valid].unsqueeze(1)).squeeze(1)
    loss = (lse[valid] - picked).sum() / n
    return loss, probs


def masked_softmax_cross_entropy_bwd(probs, target, grad_scale: float,
                                     ignore_index=-100):
    valid = target != ignore_index
    n = max(int(valid.sum().item()), 1)
    d = probs.clone()
    rows = torch.arange(d.shape[0], device=d.device)
    d[rows[valid], target[valid]] -= 1.0
    d[~valid] = 0.0
    return d * (grad_scale / n)


def dgelu(x):
    """tanh-approx GELU derivative (fp32) — oracle for the fused FFN bwd."""
    xf = x.float()
    c = 0.7978845608028654
    u = c * (xf + 0.044715 * xf ** 3)
    t = torch.tanh(u)
    return 0.5 * (1 + t) + 0.5 * xf * (1 - t * t) * c * (1 + 3 * 0.044715 * xf * xf)
