"""Shared test helpers: synthetic embedding batches with controlled class
structure (P identities x K images, like the reference's MultibatchData
sampler, def.prototxt:25-26)."""

import numpy as np


def make_batch(num_classes=8, per_class=4, dim=64, seed=0, normalize=True, dtype=np.float32):
    """Clustered random embeddings so mining has structure: class centers on
    the sphere + per-sample noise."""
    rng = np.random.default_rng(seed)
    centers = rng.standard_normal((num_classes, dim))
    centers /= np.linalg.norm(centers, axis=1, keepdims=True)
    feats = []
    labels = []
    for c in range(num_classes):
        x = centers[c] + 0.3 * rng.standard_normal((per_class, dim))
        feats.append(x)
        labels.extend([c] * per_class)
    f = np.concatenate(feats).astype(dtype)
    if normalize:
        f /= np.linalg.norm(f, axis=1, keepdims=True)
    lab = np.array(labels, dtype=np.int64)
    perm = rng.permutation(len(lab))
    return f[perm], lab[perm]


ALL_CONFIGS = []


def _build_config_grid():
    from npairloss_amd.config.params import MiningMethod, MiningRegion, NPairLossConfig

    grid = []
    for region in (MiningRegion.GLOBAL, MiningRegion.LOCAL):
        for method in MiningMethod:
            grid.append(
                NPairLossConfig(
                    margin_ident=0.05,
                    margin_diff=-0.05,
                    identsn=-0.3,
                    diffsn=2.0,
                    ap_mining_region=region,
                    ap_mining_method=method,
                    an_mining_region=MiningRegion.LOCAL if region == MiningRegion.GLOBAL else MiningRegion.GLOBAL,
                    an_mining_method=method,
                )
            )
    # the production config (usage/def.prototxt:137-146)
    from npairloss_amd.config.params import NPairLossConfig as C

    grid.append(
        C(
            margin_ident=0.0,
            margin_diff=-0.05,
            identsn=-0.0,
            diffsn=-0.3,
            ap_mining_region="GLOBAL",
            ap_mining_method="RELATIVE_HARD",
            an_mining_region="LOCAL",
            an_mining_method="HARD",
        )
    )
    return grid


def config_grid():
    global ALL_CONFIGS
    if not ALL_CONFIGS:
        ALL_CONFIGS = _build_config_grid()
    return ALL_CONFIGS
