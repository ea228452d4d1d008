from .cnn import CNN
from .bert import BertConfig, BertForMLM
from .gpt import GPTConfig, GPT
