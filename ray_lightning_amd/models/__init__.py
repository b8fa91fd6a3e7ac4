from .benchmark import (GPT2LM, ResNet50Classifier, SyntheticImageNet,
                        SyntheticTokens)
from .gpt2 import GPT2, GPT2Config, gpt2_xl
from .resnet import ResNet, resnet18_like, resnet50

__all__ = [
    "GPT2", "GPT2Config", "gpt2_xl", "ResNet", "resnet18_like",
    "resnet50", "GPT2LM", "ResNet50Classifier", "SyntheticImageNet",
    "SyntheticTokens",
]
