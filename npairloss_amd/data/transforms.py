"""DataTransformer-equivalent augmentation pipeline.

The reference fork's `DataTransformer` layer (usage/def.prototxt:61-84)
applies per-sample random rotation (rotate_angle_scope, radians), x/y
translation (translation_w/h_scope, pixels), anisotropic scaling
(scale_w/h_scope, max factor), and horizontal flip (h_flip) on the GPU
between the data layer and the backbone; plus the standard Caffe
`transform_param` mean subtraction / crop / mirror (def.prototxt:10-16).

Implemented with one batched affine grid_sample (all four geometric
transforms compose into a single 2x3 affine per sample) so it runs as two
kernels on device, not per-image host code.
"""

from __future__ import annotations

from dataclasses import dataclass

import torch
import torch.nn.functional as F


@dataclass
class TransformConfig:
    # data_transformer_l_param (def.prototxt:68-83)
    rotate_angle_scope: float = 0.0   # radians, uniform [-a, a]
    translation_w_scope: float = 0.0  # pixels
    translation_h_scope: float = 0.0
    scale_w_scope: float = 1.0        # factor in [1/s, s]
    scale_h_scope: float = 1.0
    h_flip: bool = False
    # transform_param (def.prototxt:10-16)
    crop_size: int = 0
    mean_values: tuple = (104.0, 117.0, 123.0)
    mirror: bool = False

    @classmethod
    def from_message(cls, msg) -> "TransformConfig":
        kw = {}
        for f in ("rotate_angle_scope", "translation_w_scope", "translation_h_scope",
                  "scale_w_scope", "scale_h_scope", "h_flip"):
            if msg.has(f):
                kw[f] = msg.get(f)
        return cls(**kw)


class DataTransformer:
    """Batched geometric augmentation: x (B,3,H,W) -> augmented (B,3,H,W)."""

    def __init__(self, cfg: TransformConfig, generator: torch.Generator = None):
        self.cfg = cfg
        self.generator = generator

    def _rand(self, n, device, lo, hi):
        if self.generator is not None:
            u = torch.rand(n, device="cpu", generator=self.generator).to(device)
        else:
            # device-side RNG: no H2D sync, and hipGraph-capture-safe (the
            # philox offset advances per replay so augmentations stay random)
            u = torch.rand(n, device=device)
        return lo + (hi - lo) * u

    def __call__(self, x: torch.Tensor) -> torch.Tensor:
        cfg = self.cfg
        B, C, H, W = x.shape
        dev = x.device
        ang = self._rand(B, dev, -cfg.rotate_angle_scope, cfg.rotate_angle_scope)
        # normalized translation: pixels -> [-1,1] grid units
        tx = self._rand(B, dev, -cfg.translation_w_scope, cfg.translation_w_scope) * (2.0 / max(W, 1))
        ty = self._rand(B, dev, -cfg.translation_h_scope, cfg.translation_h_scope) * (2.0 / max(H, 1))
        sw = self._sample_scale(B, dev, cfg.scale_w_scope)
        sh = self._sample_scale(B, dev, cfg.scale_h_scope)
        if cfg.h_flip:
            # arithmetic form (no host-constant tensors: H2D copies of
            # pageable memory are forbidden during hipGraph capture)
            flip = 1.0 - 2.0 * (self._rand(B, dev, 0.0, 1.0) < 0.5).float()
        else:
            flip = torch.ones(B, device=dev)
        cos, sin = torch.cos(ang), torch.sin(ang)
        # affine: rotation * scale (+flip in x) then translation
        theta = torch.zeros(B, 2, 3, device=dev, dtype=x.dtype)
        theta[:, 0, 0] = cos / sw * flip
        theta[:, 0, 1] = -sin / sh
        theta[:, 1, 0] = sin / sw * flip
        theta[:, 1, 1] = cos / sh
        theta[:, 0, 2] = tx
        theta[:, 1, 2] = ty
        grid = F.affine_grid(theta, x.shape, align_corners=False)
        return F.grid_sample(x, grid, mode="bilinear", padding_mode="zeros", align_corners=False)

    def _sample_scale(self, n, device, scope):
        if scope <= 1.0:
            return torch.ones(n, device=device)
        lo, hi = 1.0 / scope, scope
        return self._rand(n, device, lo, hi)


def preprocess(x: torch.Tensor, cfg: TransformConfig) -> torch.Tensor:
    """Caffe transform_param: mean subtraction (+ optional center crop)."""
    mean = torch.tensor(cfg.mean_values, device=x.device, dtype=x.dtype).view(1, -1, 1, 1)
    x = x - mean
    if cfg.crop_size and (x.shape[-1] != cfg.crop_size or x.shape[-2] != cfg.crop_size):
        h0 = (x.shape[-2] - cfg.crop_size) // 2
        w0 = (x.shape[-1] - cfg.crop_size) // 2
        x = x[..., h0 : h0 + cfg.crop_size, w0 : w0 + cfg.crop_size]
    return x
