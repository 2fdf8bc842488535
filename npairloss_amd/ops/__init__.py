from .npair_loss import NPairMultiClassLoss, NPairLossOutput
from .l2norm import L2Normalize, l2_normalize

__all__ = ["NPairMultiClassLoss", "NPairLossOutput", "L2Normalize", "l2_normalize"]
