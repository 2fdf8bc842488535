"""npairloss_amd — MI355X-native deep-metric-learning trainer.

A from-scratch framework with the capabilities of the reference Caffe
``NPairMultiClassLoss`` layer (quziyan/NPairLoss): multi-class N-pair loss
with configurable GLOBAL/LOCAL x HARD/EASY/RAND/RELATIVE_{HARD,EASY}
positive/negative pair mining, cross-GPU embedding all-gather so mining and
the loss see the whole-node batch, and online Recall@k retrieval metrics —
plus the surrounding trainer the reference's usage prototxts imply
(P x K batch sampler, augmentation, GoogLeNet/ResNet/ViT embedding
backbones, L2-normalize, SGD solver, checkpointing).

Compute path: PyTorch-ROCm autograd for backbones + hand-written HIP/CDNA4
kernels (gfx950) for every loss-path op + RCCL (torch.distributed "nccl"
backend) collectives over xGMI for multi-GPU.
"""

from .version import __version__

from .config.params import (
    MiningRegion,
    MiningMethod,
    NPairLossConfig,
    SolverConfig,
)
from .ops.npair_loss import NPairMultiClassLoss, NPairLossOutput
from .ops.l2norm import L2Normalize

__all__ = [
    "__version__",
    "MiningRegion",
    "MiningMethod",
    "NPairLossConfig",
    "SolverConfig",
    "NPairMultiClassLoss",
    "NPairLossOutput",
    "L2Normalize",
]
