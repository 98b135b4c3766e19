from .mlp import FreeSurferMLP
from .resnet import ResNet18
from .unet import UNet3D
from .vbm import VBMNet

__all__ = ['FreeSurferMLP', 'VBMNet', 'ResNet18', 'UNet3D']
