"""Synthetic datasets (there is no network access for real datasets; the
bench contract requires synthetic data of the production shape).

SyntheticImageDataset: class-conditional random images — each class has a
fixed random texture pattern plus per-sample noise, so metric learning has
actual signal and the online Recall@k metrics move during smoke training.
"""

from __future__ import annotations

import numpy as np
import torch
from torch.utils.data import Dataset


class SyntheticImageDataset(Dataset):
    def __init__(self, num_classes: int = 256, per_class: int = 16,
                 image_size: int = 224, seed: int = 0, noise: float = 0.3,
                 cache: bool = None):
        self.num_classes = num_classes
        self.per_class = per_class
        self.image_size = image_size
        self.noise = noise
        self.labels = np.repeat(np.arange(num_classes), per_class).tolist()
        self.seed = seed
        # low-res class patterns, upsampled on access (keeps memory small)
        g = torch.Generator().manual_seed(seed)
        self._patterns = torch.randn(num_classes, 3, 8, 8, generator=g)
        # generating a 224^2 image (interpolate + randn) costs ~ms on CPU,
        # which starves the GPU; cache generated images for small datasets
        # (deterministic per index, so caching changes nothing else)
        if cache is None:
            cache = len(self.labels) * image_size * image_size * 12 <= 2 << 30
        self._cache = {} if cache else None

    def __len__(self):
        return len(self.labels)

    def __getitem__(self, idx):
        if self._cache is not None:
            hit = self._cache.get(idx)
            if hit is not None:
                return hit
        lab = self.labels[idx]
        g = torch.Generator().manual_seed(self.seed * 1000003 + idx)
        base = torch.nn.functional.interpolate(
            self._patterns[lab : lab + 1], size=(self.image_size, self.image_size),
            mode="bilinear", align_corners=False)[0]
        img = base + self.noise * torch.randn(3, self.image_size, self.image_size, generator=g)
        if self._cache is not None:
            self._cache[idx] = (img, lab)
        return img, lab


class DeviceSyntheticBatches:
    """Device-resident P x K batch stream for throughput-parity training
    (the CPU DataLoader path pays ~60 ms/iter in generation + collate +
    worker IPC at batch 120, starving a ~11 ms GPU step).  Pre-generates
    `n_distinct` class-structured batches ON DEVICE and cycles them with
    fresh label permutations — same batch statistics as the sampler path,
    zero per-step host work.  Iterable like a DataLoader (train.py
    --synthetic-device)."""

    def __init__(self, identities_per_batch: int, imgs_per_identity: int,
                 image_size: int = 224, num_classes: int = 256,
                 device=None, n_distinct: int = 8, seed: int = 0,
                 noise: float = 0.3):
        self.P, self.K = identities_per_batch, imgs_per_identity
        B = self.P * self.K
        device = device or torch.device("cuda" if torch.cuda.is_available() else "cpu")
        g = torch.Generator().manual_seed(seed)
        patterns = torch.randn(num_classes, 3, 8, 8, generator=g)
        self.batches = []
        for i in range(n_distinct):
            cls = torch.randperm(num_classes, generator=g)[: self.P]
            base = torch.nn.functional.interpolate(
                patterns[cls], size=(image_size, image_size), mode="bilinear",
                align_corners=False).repeat_interleave(self.K, dim=0)
            x = base + noise * torch.randn(B, 3, image_size, image_size, generator=g)
            lab = cls.repeat_interleave(self.K)
            self.batches.append((x.to(device).to(memory_format=torch.channels_last),
                                 lab.to(device)))
        self._step = 0

    def __len__(self):
        return len(self.batches)

    def __iter__(self):
        for x, lab in self.batches:
            perm = torch.randperm(x.shape[0], device=x.device)
            yield x[perm], lab[perm]


class SyntheticEmbeddingDataset(Dataset):
    """Clustered unit-norm embeddings (backbone-free loss testing)."""

    def __init__(self, num_classes: int = 64, per_class: int = 8, dim: int = 64,
                 seed: int = 0, noise: float = 0.3):
        rng = np.random.default_rng(seed)
        centers = rng.standard_normal((num_classes, dim))
        centers /= np.linalg.norm(centers, axis=1, keepdims=True)
        feats, labels = [], []
        for c in range(num_classes):
            x = centers[c] + noise * rng.standard_normal((per_class, dim))
            x /= np.linalg.norm(x, axis=1, keepdims=True)
            feats.append(x)
            labels.extend([c] * per_class)
        self.features = torch.from_numpy(np.concatenate(feats)).float()
        self.labels = labels

    def __len__(self):
        return len(self.labels)

    def __getitem__(self, idx):
        return self.features[idx], self.labels[idx]
