"""MI355X-native training ops: hand-written gfx950 HIP kernels with PyTorch
reference fallbacks on CPU (see functional.py for the dispatch contract)."""
from ._backend import hip_available, hip_ext
from .functional import (
    add_relu,
    bn_relu_eval,
    bn_relu_train,
    conv2d,
    global_avg_pool,
    linear,
    max_pool2d,
    sgd_momentum_step,
    softmax_cross_entropy,
)
from .modules import BatchNormReLU, Conv2d, GlobalAvgPool, Linear, MaxPool2d

__all__ = [
    "hip_available", "hip_ext",
    "conv2d", "add_relu", "bn_relu_train", "bn_relu_eval", "max_pool2d", "global_avg_pool",
    "linear", "softmax_cross_entropy", "sgd_momentum_step",
    "Conv2d", "BatchNormReLU", "MaxPool2d", "GlobalAvgPool", "Linear",
]
