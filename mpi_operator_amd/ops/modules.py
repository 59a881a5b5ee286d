"""nn.Module wrappers over the MI355X-native functional ops."""
from __future__ import annotations

import math

import torch
import torch.nn as nn

from . import functional as Fx


class Conv2d(nn.Module):
    """Bias-free conv (ResNet-style). Weight logical [K,C,R,S]; on GPU the
    module keeps it channels-last ([K][R][S][C] memory) to feed the
    implicit-GEMM MFMA kernels directly."""

    def __init__(self, in_ch: int, out_ch: int, kernel: int, stride: int = 1, padding: int = 0):
        super().__init__()
        self.stride, self.padding = stride, padding
        w = torch.empty(out_ch, in_ch, kernel, kernel)
        # He initialization (fan_out, as in ResNet reference training)
        nn.init.kaiming_normal_(w, mode="fan_out", nonlinearity="relu")
        self.weight = nn.Parameter(w)

    def forward(self, x):
        return Fx.conv2d(x, self.weight, self.stride, self.padding)

    def _apply(self, fn, recurse=True):
        mod = super()._apply(fn, recurse)
        if self.weight.is_cuda:
            with torch.no_grad():
                self.weight.data = self.weight.data.contiguous(memory_format=torch.channels_last)
        return mod


class BatchNormReLU(nn.Module):
    """Fused BN + optional ReLU. gamma/beta/stats are fp32 regardless of the
    activation dtype (bf16 on MI355X)."""

    def __init__(self, num_ch: int, eps: float = 1e-5, momentum: float = 0.1, relu: bool = True):
        super().__init__()
        self.eps, self.momentum, self.relu = eps, momentum, relu
        self.weight = nn.Parameter(torch.ones(num_ch))
        self.bias = nn.Parameter(torch.zeros(num_ch))
        self.register_buffer("running_mean", torch.zeros(num_ch))
        self.register_buffer("running_var", torch.ones(num_ch))

    def forward(self, x):
        if self.training:
            y, _, _ = Fx.bn_relu_train(x, self.weight, self.bias, self.eps,
                                       self.relu, self.running_mean,
                                       self.running_var, self.momentum)
            return y
        return Fx.bn_relu_eval(x, self.weight, self.bias, self.running_mean,
                               self.running_var, self.eps, self.relu)

    def _load_from_state_dict(self, *args, **kw):  # keep fp32 stats on cast models
        super()._load_from_state_dict(*args, **kw)

    def half_compatible_params(self):
        return [self.weight, self.bias]


class MaxPool2d(nn.Module):
    def __init__(self, kernel: int = 3, stride: int = 2, padding: int = 1):
        super().__init__()
        self.kernel, self.stride, self.padding = kernel, stride, padding

    def forward(self, x):
        return Fx.max_pool2d(x, self.kernel, self.stride, self.padding)


class GlobalAvgPool(nn.Module):
    def forward(self, x):
        return Fx.global_avg_pool(x)


class Linear(nn.Module):
    """Linear with fp32 bias (classifier head)."""

    def __init__(self, in_f: int, out_f: int):
        super().__init__()
        w = torch.empty(out_f, in_f)
        nn.init.uniform_(w, -1.0 / math.sqrt(in_f), 1.0 / math.sqrt(in_f))
        self.weight = nn.Parameter(w)
        self.bias = nn.Parameter(torch.zeros(out_f))

    def forward(self, x):
        return Fx.linear(x, self.weight, self.bias)
