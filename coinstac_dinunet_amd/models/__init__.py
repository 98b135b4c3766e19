from .mlp import FreeSurferMLP
from .resnet import ResNet18
from .vbm import VBMNet

__all__ = ['FreeSurferMLP', 'VBMNet', 'ResNet18']
