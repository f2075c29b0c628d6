from .lenet import LeNet
from .resnet_dwt import Bottleneck, ResNetDWT, resnet50
from . import checkpoint

__all__ = ["LeNet", "Bottleneck", "ResNetDWT", "resnet50", "checkpoint"]
