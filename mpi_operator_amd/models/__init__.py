from .resnet import ResNet, resnet50, resnet101, to_mi355x
from .simple_cnn import SimpleCNN

__all__ = ["ResNet", "resnet50", "resnet101", "to_mi355x", "SimpleCNN"]
