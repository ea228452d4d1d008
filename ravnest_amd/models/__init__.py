from .cnn import CNN
from .bert import BertConfig, BertForMLM
from .gpt import GPTConfig, GPT
from .resnet import ResNet, resnet50, resnet18_ish
from .inception import Inception3
