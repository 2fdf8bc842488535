"""P x K identity batch sampler — the `MultibatchData` layer's sampling
policy (reference usage/def.prototxt:17-31): each batch holds
`identity_num_per_batch` (P) identities x `img_num_per_identity` (K)
images, identities drawn randomly (`rand_identity: true`), shuffled.
The production config is P=60, K=2 -> batch 120 (:25-26); test P=15, K=2.

N-pair mining needs >= 2 samples per class in a batch — this sampler
guarantees exactly K per class.
"""

from __future__ import annotations

import random
from collections import defaultdict
from typing import Dict, Iterator, List, Sequence


class PKBatchSampler:
    """Yields lists of dataset indices, P identities x K samples each.

    labels: per-index class id for the whole dataset.
    Classes with fewer than K samples are sampled with replacement (the
    reference's data layer reads K images per identity regardless).
    """

    def __init__(self, labels: Sequence[int], identities_per_batch: int,
                 imgs_per_identity: int, shuffle: bool = True,
                 rand_identity: bool = True, seed: int = 0,
                 batches_per_epoch: int = None):
        self.labels = list(labels)
        self.P = identities_per_batch
        self.K = imgs_per_identity
        self.shuffle = shuffle
        self.rand_identity = rand_identity
        self.base_seed = seed
        self.rng = random.Random(seed)
        self.by_class: Dict[int, List[int]] = defaultdict(list)
        for idx, lab in enumerate(self.labels):
            self.by_class[int(lab)].append(idx)
        self.classes = sorted(self.by_class.keys())
        if len(self.classes) < self.P:
            raise ValueError(
                f"need >= {self.P} identities, dataset has {len(self.classes)}")
        self.batches_per_epoch = (
            batches_per_epoch
            if batches_per_epoch is not None
            else max(1, len(self.labels) // (self.P * self.K))
        )

    def set_epoch(self, epoch: int) -> None:
        """Reseed deterministically from (base_seed, epoch) so runs are
        reproducible; base_seed should already be rank-folded (the builder
        passes seed = random_seed + rank so ranks draw disjoint batches)."""
        self.rng = random.Random(self.base_seed * 1_000_003 + epoch)

    def __len__(self) -> int:
        return self.batches_per_epoch

    def __iter__(self) -> Iterator[List[int]]:
        for _ in range(self.batches_per_epoch):
            if self.rand_identity:
                ids = self.rng.sample(self.classes, self.P)
            else:
                ids = self.classes[: self.P]
            batch: List[int] = []
            for c in ids:
                pool = self.by_class[c]
                if len(pool) >= self.K:
                    batch.extend(self.rng.sample(pool, self.K))
                else:
                    batch.extend(self.rng.choices(pool, k=self.K))
            if self.shuffle:
                self.rng.shuffle(batch)
            yield batch
