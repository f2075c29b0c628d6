"""dwt_amd — MI355X-native Domain-Whitening-Transform (DWT) + MEC framework.

A from-scratch, gfx950-first implementation of the capabilities of
roysubhankar/dwt-domain-adaptation (CVPR 2019 "Unsupervised Domain Adaptation
using Feature-Whitening and Consensus Loss"): grouped feature whitening
(DWT), domain-specific BatchNorm, Min-Entropy-Consensus loss, the LeNet /
DWT-ResNet50 model families, and the two-stream (digits) / three-stream
(Office-Home) training loops — with the novel compute implemented as
hand-written CDNA4 HIP kernels and data parallelism over RCCL/xGMI.

Layering:
    ops/      WTransform2d, DomainBatchNorm, losses (+ oracle + HIP dispatch)
    kernels/  HIP/C++ extension for gfx950 (built in-tree)
    models/   LeNet-DWT, DWT-ResNet50, checkpoint layout compat
    data/     USPS/MNIST loaders, dual-transform ImageFolder, augmentations,
              synthetic datasets (offline benchmarking)
    engine/   train/eval loops, stats re-estimation, metrics
    parallel/ bucketed-gradient data-parallel engine over torch.distributed
    utils/    config, seeding, profiling helpers
"""

__version__ = "0.1.0"

from . import ops  # noqa: F401
