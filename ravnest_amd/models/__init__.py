from .cnn import CNN
