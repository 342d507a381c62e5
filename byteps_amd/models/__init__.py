from .resnet import resnet50, resnet101, ResNet
from .vgg import vgg16, VGG
from .bert import bert_large, bert_base, BertConfig, BertForPreTraining
from .mlp import mnist_mlp

__all__ = ["resnet50", "resnet101", "ResNet", "vgg16", "VGG", "bert_large",
           "bert_base", "BertConfig", "BertForPreTraining", "mnist_mlp"]
