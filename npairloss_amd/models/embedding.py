"""EmbeddingNet: backbone -> L2-normalized embedding, the trainable net the
reference's def.prototxt implies (conv stack + L2Normalize feeding the
loss, def.prototxt:85-120)."""

from __future__ import annotations

import torch
from torch import nn

from ..ops.l2norm import l2_normalize
from .googlenet import GoogLeNet
from .resnet import ResNet50
from .vit import ViTB16


class EmbeddingNet(nn.Module):
    def __init__(self, backbone: nn.Module, normalize: bool = True):
        super().__init__()
        self.backbone = backbone
        self.normalize = normalize
        self.embed_dim = getattr(backbone, "embed_dim", None)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        f = self.backbone(x)
        if f.dim() != 2:
            f = f.flatten(1)
        if self.normalize:
            # L2Normalize runs in fp32 (autocast-safe): unit-norm embeddings
            # feed the fp32 loss path.
            f = l2_normalize(f.float())
        return f


def build_embedding_model(name: str, embed_dim: int = None, normalize: bool = True,
                          init: str = "caffe") -> EmbeddingNet:
    name = name.lower()
    if name in ("googlenet", "inception_v1"):
        import os
        fused = os.environ.get("NPAIR_FUSED_CONV", "1") != "0"
        bb = GoogLeNet(fused_bias_relu=fused, init=init)
    elif name in ("resnet50", "resnet-50"):
        bb = ResNet50(embed_dim=embed_dim or 128)
    elif name in ("vit", "vit-b/16", "vitb16"):
        bb = ViTB16(embed_dim=embed_dim or 768)
    else:
        raise ValueError(f"unknown backbone {name!r}")
    return EmbeddingNet(bb, normalize=normalize)
