from .resnet import ResNet, resnet50, resnet101, to_mi355x
from .simple_cnn import SimpleCNN
from .bert import BertConfig, BertForPreTraining, bert_base, bert_large, to_mi355x_bert

__all__ = ["ResNet", "resnet50", "resnet101", "to_mi355x", "SimpleCNN",
           "BertConfig", "BertForPreTraining", "bert_base", "bert_large", "to_mi355x_bert"]
