"""Stock-PyTorch ResNet (nn.Conv2d/nn.BatchNorm2d/MIOpen) — the A/B
comparison baseline for the hand-written CDNA4 kernel path (`bench.py
--impl torch`). Not the judged path."""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F


class TorchBottleneck(nn.Module):
    expansion = 4

    def __init__(self, in_ch, width, stride=1, downsample=None):
        super().__init__()
        out_ch = width * self.expansion
        self.conv1 = nn.Conv2d(in_ch, width, 1, bias=False)
        self.bn1 = nn.BatchNorm2d(width)
        self.conv2 = nn.Conv2d(width, width, 3, stride=stride, padding=1, bias=False)
        self.bn2 = nn.BatchNorm2d(width)
        self.conv3 = nn.Conv2d(width, out_ch, 1, bias=False)
        self.bn3 = nn.BatchNorm2d(out_ch)
        self.downsample = downsample

    def forward(self, x):
        identity = x if self.downsample is None else self.downsample(x)
        out = F.relu(self.bn1(self.conv1(x)))
        out = F.relu(self.bn2(self.conv2(out)))
        out = self.bn3(self.conv3(out))
        return F.relu(out + identity)


class TorchResNet(nn.Module):
    def __init__(self, layers, num_classes=1000):
        super().__init__()
        self.in_ch = 64
        self.conv1 = nn.Conv2d(3, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = nn.BatchNorm2d(64)
        self.maxpool = nn.MaxPool2d(3, 2, 1)
        self.layer1 = self._make_layer(64, layers[0])
        self.layer2 = self._make_layer(128, layers[1], 2)
        self.layer3 = self._make_layer(256, layers[2], 2)
        self.layer4 = self._make_layer(512, layers[3], 2)
        self.fc = nn.Linear(512 * 4, num_classes)

    def _make_layer(self, width, blocks, stride=1):
        downsample = None
        out_ch = width * 4
        if stride != 1 or self.in_ch != out_ch:
            downsample = nn.Sequential(
                nn.Conv2d(self.in_ch, out_ch, 1, stride=stride, bias=False),
                nn.BatchNorm2d(out_ch))
        layers = [TorchBottleneck(self.in_ch, width, stride, downsample)]
        self.in_ch = out_ch
        layers += [TorchBottleneck(self.in_ch, width) for _ in range(1, blocks)]
        return nn.Sequential(*layers)

    def forward(self, x):
        x = self.maxpool(F.relu(self.bn1(self.conv1(x))))
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        x = torch.flatten(F.adaptive_avg_pool2d(x, 1), 1)
        return self.fc(x)

    def loss(self, logits, target):
        return F.cross_entropy(logits.float(), target)


def resnet50(num_classes=1000):
    return TorchResNet([3, 4, 6, 3], num_classes)


def resnet101(num_classes=1000):
    return TorchResNet([3, 4, 23, 3], num_classes)
