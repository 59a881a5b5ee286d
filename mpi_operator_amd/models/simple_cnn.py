"""Small MNIST-class CNN — parity with the reference Horovod example model
(conv→pool→conv→pool→fc→fc, reference examples/v2beta1/horovod/
tensorflow_mnist.py:38-73), built on the MI355X-native ops. Used by the
elastic-training example and as a fast end-to-end test model."""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import (Conv2d, BatchNormReLU, GlobalAvgPool, Linear, MaxPool2d,
                   softmax_cross_entropy)


class SimpleCNN(nn.Module):
    def __init__(self, in_ch: int = 1, num_classes: int = 10, width: int = 32):
        super().__init__()
        self.in_ch_pad = max(8, (in_ch + 7) // 8 * 8)
        self.conv1 = Conv2d(self.in_ch_pad, width, 3, stride=1, padding=1)
        self.bn1 = BatchNormReLU(width, relu=True)
        self.pool1 = MaxPool2d(3, 2, 1)
        self.conv2 = Conv2d(width, width * 2, 3, stride=1, padding=1)
        self.bn2 = BatchNormReLU(width * 2, relu=True)
        self.pool2 = MaxPool2d(3, 2, 1)
        self.gap = GlobalAvgPool()
        self.fc1 = Linear(width * 2, 128)
        self.relu = nn.ReLU()
        self.fc2 = Linear(128, num_classes)

    def forward(self, x):
        if x.shape[1] != self.in_ch_pad:
            x = F.pad(x, (0, 0, 0, 0, 0, self.in_ch_pad - x.shape[1]))
            if x.is_cuda:
                x = x.contiguous(memory_format=torch.channels_last)
        x = self.pool1(self.bn1(self.conv1(x)))
        x = self.pool2(self.bn2(self.conv2(x)))
        x = self.gap(x)
        return self.fc2(self.relu(self.fc1(x)))

    def loss(self, logits, target):
        return softmax_cross_entropy(logits, target)
