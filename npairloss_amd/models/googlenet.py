"""GoogLeNet (Inception v1) embedding backbone, Caffe-layout.

The reference's usage/def.prototxt trains a GoogLeNet whose conv stack is
elided in the file; the named endpoints (conv1/7x7_s2 at def.prototxt:86,
pool5/7x7_s1 at :115-117) identify it as the standard BVLC GoogLeNet v1.
This is a from-scratch PyTorch implementation in the Caffe layout
(LRN after pool1/conv2, ceil-mode pooling, ReLU everywhere, 7x7 average
pool -> 1024-d embedding) with the original Caffe layer names preserved in
`caffe_names()` so `.caffemodel` checkpoints load by name
(utils/checkpoint.py).
"""

from __future__ import annotations

from typing import Dict

import torch
from torch import nn

from ..ops.vision import (Conv1x1BiasReLU, ConvBiasReLU, CrossChannelLRN,
                          MaxPool3x3)


class ConvReLU(nn.Module):
    def __init__(self, cin: int, cout: int, k: int, stride: int = 1, pad: int = 0):
        super().__init__()
        self.conv = nn.Conv2d(cin, cout, k, stride=stride, padding=pad, bias=True)
        self.relu = nn.ReLU(inplace=True)

    def forward(self, x):
        return self.relu(self.conv(x))


def _conv_block(cin, cout, k, stride=1, pad=0, fused=True):
    """conv+bias+relu unit.  fused=True (default) keeps bias OUT of the
    conv and runs the fused BiasReLU kernel, whose backward collapses
    torch's threshold_backward + the generic bias-grad reduce into one
    pass (csrc/biasrelu.hip).  1x1 convs go through Conv1x1BiasReLU, whose
    strategy (NPAIR_CONV1X1: off / hybrid / custom) additionally lets the
    forward or data-gradient run our MFMA GEMM (csrc/conv1x1.hip) — the
    measured default this round is "off" (MIOpen GEMMs,
    profiles/kb_conv1x1_v2.log).  fused=False is plain conv(bias)+ReLU."""
    if fused and k == 1 and stride == 1 and pad == 0:
        return Conv1x1BiasReLU(cin, cout)
    if fused:
        return ConvBiasReLU(cin, cout, k, stride=stride, pad=pad)
    return ConvReLU(cin, cout, k, stride=stride, pad=pad)


class Inception(nn.Module):
    """The 4-branch inception block: 1x1 | 1x1->3x3 | 1x1->5x5 | pool->1x1."""

    def __init__(self, cin: int, c1: int, c3r: int, c3: int, c5r: int, c5: int, cp: int,
                 fused: bool = True):
        super().__init__()
        self.b1 = _conv_block(cin, c1, 1, fused=fused)
        self.b3_reduce = _conv_block(cin, c3r, 1, fused=fused)
        self.b3 = _conv_block(c3r, c3, 3, pad=1, fused=fused)
        self.b5_reduce = _conv_block(cin, c5r, 1, fused=fused)
        self.b5 = _conv_block(c5r, c5, 5, pad=2, fused=fused)
        self.pool = MaxPool3x3(stride=1)  # fused gfx950 kernel on GPU
        self.pool_proj = _conv_block(cin, cp, 1, fused=fused)

    def forward(self, x):
        return torch.cat(
            [self.b1(x), self.b3(self.b3_reduce(x)), self.b5(self.b5_reduce(x)),
             self.pool_proj(self.pool(x))], dim=1)


# (c1, c3r, c3, c5r, c5, cp) per block — standard BVLC GoogLeNet v1
_INCEPTION_CFG = {
    "3a": (64, 96, 128, 16, 32, 32),
    "3b": (128, 128, 192, 32, 96, 64),
    "4a": (192, 96, 208, 16, 48, 64),
    "4b": (160, 112, 224, 24, 64, 64),
    "4c": (128, 128, 256, 24, 64, 64),
    "4d": (112, 144, 288, 32, 64, 64),
    "4e": (256, 160, 320, 32, 128, 128),
    "5a": (256, 160, 320, 32, 128, 128),
    "5b": (384, 192, 384, 48, 128, 128),
}
_INCEPTION_IN = {"3a": 192, "3b": 256, "4a": 480, "4b": 512, "4c": 512,
                 "4d": 512, "4e": 528, "5a": 832, "5b": 832}


class GoogLeNet(nn.Module):
    """Input B x 3 x 224 x 224 -> 1024-d embedding (pool5/7x7_s1 output,
    the reference's embedding endpoint, def.prototxt:115-120)."""

    embed_dim = 1024

    def __init__(self, dropout: float = 0.4, fused_bias_relu: bool = True,
                 init: str = "caffe"):
        """init: "caffe" reproduces the reference's filler exactly (xavier
        weights + constant 0.2 bias, def.prototxt:109-112) — which starts
        the embedding COLLAPSED (all-positive activations -> all pairwise
        sims ~1; the reference's 2M-iteration schedule absorbs this).
        "modern" = kaiming fan-out + zero bias: same architecture, trains
        from iteration 1 (the documented init deviation for demos)."""
        super().__init__()
        assert init in ("caffe", "modern")
        self._init_mode = init
        fused = fused_bias_relu
        self.conv1 = _conv_block(3, 64, 7, stride=2, pad=3, fused=fused)
        self.pool1 = MaxPool3x3(stride=2)
        self.norm1 = CrossChannelLRN(5, alpha=1e-4, beta=0.75)
        self.conv2_reduce = _conv_block(64, 64, 1, fused=fused)
        self.conv2 = _conv_block(64, 192, 3, pad=1, fused=fused)
        self.norm2 = CrossChannelLRN(5, alpha=1e-4, beta=0.75)
        self.pool2 = MaxPool3x3(stride=2)
        self.inception = nn.ModuleDict({
            name: Inception(_INCEPTION_IN[name], *cfg, fused=fused)
            for name, cfg in _INCEPTION_CFG.items()
        })
        self.pool3 = MaxPool3x3(stride=2)
        self.pool4 = MaxPool3x3(stride=2)
        self.pool5 = nn.AdaptiveAvgPool2d(1)  # 7x7 avg at 224 input
        self.dropout = nn.Dropout(dropout)
        self._init_weights()

    def _init_weights(self):
        caffe = self._init_mode == "caffe"
        bias_v = 0.2 if caffe else 0.0
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                if caffe:
                    nn.init.xavier_uniform_(m.weight)
                else:
                    nn.init.kaiming_normal_(m.weight, mode="fan_out",
                                            nonlinearity="relu")
                if m.bias is not None:
                    nn.init.constant_(m.bias, bias_v)  # def.prototxt:109-112 filler
            elif isinstance(m, (ConvBiasReLU, Conv1x1BiasReLU)):
                nn.init.constant_(m.bias, bias_v)  # same filler, bias outside conv

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = self.norm1(self.pool1(self.conv1(x)))
        x = self.pool2(self.norm2(self.conv2(self.conv2_reduce(x))))
        x = self.inception["3a"](x)
        x = self.inception["3b"](x)
        x = self.pool3(x)
        for name in ("4a", "4b", "4c", "4d", "4e"):
            x = self.inception[name](x)
        x = self.pool4(x)
        x = self.inception["5a"](x)
        x = self.inception["5b"](x)
        x = self.pool5(x)
        x = torch.flatten(x, 1)
        return self.dropout(x)

    # -- Caffe name mapping (for .caffemodel loading) ------------------------

    def caffe_names(self) -> Dict[str, nn.Module]:
        """Caffe layer name -> module holding its (weight, bias).  For
        ConvBiasReLU the module itself carries .weight (property into the
        bias-free conv) + .bias, so checkpoint IO is layout-agnostic."""
        def unit(mod):
            return mod.conv if isinstance(mod, ConvReLU) else mod

        m: Dict[str, nn.Module] = {
            "conv1/7x7_s2": unit(self.conv1),
            "conv2/3x3_reduce": unit(self.conv2_reduce),
            "conv2/3x3": unit(self.conv2),
        }
        branch_names = {
            "b1": "1x1",
            "b3_reduce": "3x3_reduce",
            "b3": "3x3",
            "b5_reduce": "5x5_reduce",
            "b5": "5x5",
            "pool_proj": "pool_proj",
        }
        for blk, mod in self.inception.items():
            for attr, suffix in branch_names.items():
                m[f"inception_{blk}/{suffix}"] = unit(getattr(mod, attr))
        return m
