from .whitening import WTransform2d, WhiteningScaleShift
from .batch_norm import DomainBatchNorm1d, DomainBatchNorm2d, DomainBatchNorm3d
from .losses import MinEntropyConsensusLoss, EntropyLoss

__all__ = [
    "WTransform2d",
    "WhiteningScaleShift",
    "DomainBatchNorm1d",
    "DomainBatchNorm2d",
    "DomainBatchNorm3d",
    "MinEntropyConsensusLoss",
    "EntropyLoss",
]
