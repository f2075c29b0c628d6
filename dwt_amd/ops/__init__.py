from .whitening import WTransform2d, WhiteningScaleShift
from .batch_norm import DomainBatchNorm1d, DomainBatchNorm2d, DomainBatchNorm3d
from .losses import MinEntropyConsensusLoss, EntropyLoss
from .optim import FusedAdam, FusedSGD
from .pooling import MaxPool2dDWT, global_avg_pool, max_pool2d

__all__ = [
    "WTransform2d",
    "WhiteningScaleShift",
    "DomainBatchNorm1d",
    "DomainBatchNorm2d",
    "DomainBatchNorm3d",
    "MinEntropyConsensusLoss",
    "EntropyLoss",
    "FusedAdam",
    "FusedSGD",
    "MaxPool2dDWT",
    "global_avg_pool",
    "max_pool2d",
]
