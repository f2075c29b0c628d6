from .transforms import (Compose, Lambda, Normalize, RandomCrop,
                         RandomHorizontalFlip, Resize, ToTensor)
from .usps import USPS
from .mnist import MNIST
from .folder import ImageFolder, DatasetFolder
from .synthetic import SyntheticDigits, SyntheticOfficeHome

__all__ = [
    "Compose", "Lambda", "Normalize", "RandomCrop", "RandomHorizontalFlip",
    "Resize", "ToTensor", "USPS", "MNIST", "ImageFolder", "DatasetFolder",
    "SyntheticDigits", "SyntheticOfficeHome",
]
