"""ViT-B/16 embedding backbone (from scratch).  Used by BASELINE.json
config 5: 768-d embeddings, fp8-capable MFMA similarity path, global batch
8192 sized for the 288 GB HBM3E per MI355X."""

from __future__ import annotations

import torch
from torch import nn


class Block(nn.Module):
    def __init__(self, dim: int, heads: int, mlp_ratio: float = 4.0):
        super().__init__()
        self.norm1 = nn.LayerNorm(dim)
        self.attn = nn.MultiheadAttention(dim, heads, batch_first=True)
        self.norm2 = nn.LayerNorm(dim)
        hidden = int(dim * mlp_ratio)
        self.mlp = nn.Sequential(nn.Linear(dim, hidden), nn.GELU(), nn.Linear(hidden, dim))

    def forward(self, x):
        a, _ = self.attn(self.norm1(x), self.norm1(x), self.norm1(x), need_weights=False)
        x = x + a
        return x + self.mlp(self.norm2(x))


class ViTB16(nn.Module):
    def __init__(self, image_size: int = 224, patch: int = 16, dim: int = 768,
                 depth: int = 12, heads: int = 12, embed_dim: int = 768):
        super().__init__()
        self.embed_dim = embed_dim
        self.patch_embed = nn.Conv2d(3, dim, patch, stride=patch)
        n_patches = (image_size // patch) ** 2
        self.cls = nn.Parameter(torch.zeros(1, 1, dim))
        self.pos = nn.Parameter(torch.zeros(1, n_patches + 1, dim))
        self.blocks = nn.Sequential(*[Block(dim, heads) for _ in range(depth)])
        self.norm = nn.LayerNorm(dim)
        self.proj = nn.Identity() if embed_dim == dim else nn.Linear(dim, embed_dim)
        nn.init.trunc_normal_(self.pos, std=0.02)
        nn.init.trunc_normal_(self.cls, std=0.02)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = self.patch_embed(x).flatten(2).transpose(1, 2)  # B x N x D
        x = torch.cat([self.cls.expand(x.shape[0], -1, -1), x], dim=1) + self.pos
        x = self.norm(self.blocks(x))
        return self.proj(x[:, 0])
