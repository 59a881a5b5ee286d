"""ResNet-50/101 built on the MI355X-native ops.

This is the workload the reference only documents (tf_cnn_benchmarks
ResNet101 via an external Horovod/CUDA image — reference
examples/v2beta1/tensorflow-benchmarks/Dockerfile:1, README.md:96-143);
here it is first-class, with every hot op (conv / BN+ReLU / pool / FC /
softmax-CE / SGD) backed by hand-written CDNA4 HIP kernels on GPU.

Architecture is the standard v1.5 bottleneck ResNet (stride-2 in the 3x3 of
downsampling bottlenecks), matching what tf_cnn_benchmarks trains.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import (BatchNormReLU, Conv2d, GlobalAvgPool, Linear, MaxPool2d,
                   add_relu, softmax_cross_entropy)


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, in_ch: int, width: int, stride: int = 1, downsample: nn.Module | None = None):
        super().__init__()
        out_ch = width * self.expansion
        self.conv1 = Conv2d(in_ch, width, 1)
        self.bn1 = BatchNormReLU(width, relu=True)
        self.conv2 = Conv2d(width, width, 3, stride=stride, padding=1)
        self.bn2 = BatchNormReLU(width, relu=True)
        self.conv3 = Conv2d(width, out_ch, 1)
        self.bn3 = BatchNormReLU(out_ch, relu=False)
        self.downsample = downsample

    def forward(self, x):
        if x.is_cuda and self.training and torch.is_grad_enabled():
            # fused block: residual add folded into bn3's apply pass, join
            # gradient summed inside conv1's dgrad epilogue (ops/fused_block)
            from ..ops.fused_block import fused_bottleneck
            return fused_bottleneck(x, self)
        identity = x if self.downsample is None else self.downsample(x)
        out = self.bn1(self.conv1(x))
        out = self.bn2(self.conv2(out))
        out = self.bn3(self.conv3(out))
        return add_relu(out, identity)


class ResNet(nn.Module):
    def __init__(self, layers: list[int], num_classes: int = 1000):
        super().__init__()
        self.in_ch = 64
        # Stem takes 8 input channels: RGB is zero-padded 3→8 so every conv
        # satisfies the NHWC C%8==0 16 B-per-lane layout contract (the extra
        # channels are zeros — identical math, coalesced loads).
        self.conv1 = Conv2d(8, 64, 7, stride=2, padding=3)
        self.bn1 = BatchNormReLU(64, relu=True)
        self.maxpool = MaxPool2d(3, 2, 1)
        self.layer1 = self._make_layer(64, layers[0])
        self.layer2 = self._make_layer(128, layers[1], stride=2)
        self.layer3 = self._make_layer(256, layers[2], stride=2)
        self.layer4 = self._make_layer(512, layers[3], stride=2)
        self.gap = GlobalAvgPool()
        self.fc = Linear(512 * Bottleneck.expansion, num_classes)

    def _make_layer(self, width: int, blocks: int, stride: int = 1):
        downsample = None
        out_ch = width * Bottleneck.expansion
        if stride != 1 or self.in_ch != out_ch:
            downsample = nn.Sequential(
                Conv2d(self.in_ch, out_ch, 1, stride=stride),
                BatchNormReLU(out_ch, relu=False),
            )
        layers = [Bottleneck(self.in_ch, width, stride, downsample)]
        self.in_ch = out_ch
        for _ in range(1, blocks):
            layers.append(Bottleneck(self.in_ch, width))
        return nn.Sequential(*layers)

    def forward(self, x):
        if x.shape[1] != 8:
            x = F.pad(x, (0, 0, 0, 0, 0, 8 - x.shape[1]))
            if x.is_cuda:
                x = x.contiguous(memory_format=torch.channels_last)
        x = self.maxpool(self.bn1(self.conv1(x)))
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        return self.fc(self.gap(x))

    def loss(self, logits, target):
        return softmax_cross_entropy(logits, target)


def resnet50(num_classes: int = 1000) -> ResNet:
    return ResNet([3, 4, 6, 3], num_classes)


def resnet101(num_classes: int = 1000) -> ResNet:
    return ResNet([3, 4, 23, 3], num_classes)


def to_mi355x(model: nn.Module, device="cuda") -> nn.Module:
    """Move a model to the MI355X layout/dtype contract: channels-last, with
    conv/linear weights bf16 and BN parameters fp32."""
    model = model.to(device=device, memory_format=torch.channels_last)
    for m in model.modules():
        if isinstance(m, (Conv2d, Linear)):
            m.to(torch.bfloat16)
        if isinstance(m, Linear):
            m.bias.data = m.bias.data.float()  # classifier bias stays fp32
    for m in model.modules():
        if isinstance(m, BatchNormReLU):
            m.float()
    return model
