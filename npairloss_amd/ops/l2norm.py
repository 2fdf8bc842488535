"""Row-wise L2 normalization — the reference fork's `L2Normalize` layer
(usage/def.prototxt:115-120) that feeds the loss its unit-norm embeddings.

GPU tensors run the hand-written HIP kernel (one fused pass per direction);
CPU tensors use the equivalent torch math.  y = x / max(||x||_2, eps) per
row; backward dx = (dy - y * <y, dy>) / max(||x||, eps).
"""

from __future__ import annotations

import torch
from torch import nn

from . import _backend

_EPS = 1e-12


class _L2NormalizeFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor) -> torch.Tensor:
        if x.is_cuda:
            y, inv_norm = _backend.ext().l2norm_fwd(x.contiguous())
        else:
            norm = x.norm(dim=1, keepdim=True).clamp_min(_EPS)
            inv_norm = norm.reciprocal().squeeze(1)
            y = x * inv_norm.unsqueeze(1)
        ctx.save_for_backward(y, inv_norm)
        return y

    @staticmethod
    def backward(ctx, dy: torch.Tensor):
        y, inv_norm = ctx.saved_tensors
        if dy.is_cuda:
            dx = _backend.ext().l2norm_bwd(y, inv_norm, dy.contiguous())
        else:
            dot = (y * dy).sum(dim=1, keepdim=True)
            dx = (dy - y * dot) * inv_norm.unsqueeze(1)
        return dx


def l2_normalize(x: torch.Tensor) -> torch.Tensor:
    """Differentiable row-wise L2 normalize (2D input: N x D)."""
    if x.dim() != 2:
        x = x.flatten(1)
    return _L2NormalizeFn.apply(x)


class L2Normalize(nn.Module):
    def forward(self, x: torch.Tensor) -> torch.Tensor:  # noqa: D102
        return l2_normalize(x)
