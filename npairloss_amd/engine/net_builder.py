"""Build a complete training setup from the reference's prototxt files.

Reads a Caffe net definition (def.prototxt) + solver (solver.prototxt) and
assembles the MI355X-native equivalents: MultibatchData -> PKBatchSampler
(+ synthetic dataset in this offline environment, or an image-folder
dataset when root_folder/source exist), DataTransformer -> batched affine
augmentation, the GoogLeNet conv stack, L2Normalize -> ops.L2Normalize,
NPairMultiClassLoss -> the HIP-kernel loss, SGD solver -> CaffeSGD +
Trainer.  `.caffemodel` weights load by layer name.
"""

from __future__ import annotations

import os
from dataclasses import dataclass
from typing import Optional

import torch

from ..config.params import (NPairLossConfig, SolverConfig,
                             parse_net_prototxt)
from ..data import PKBatchSampler, SyntheticImageDataset
from ..data.transforms import DataTransformer, TransformConfig
from ..models import build_embedding_model
from ..ops.npair_loss import NPairMultiClassLoss
from ..utils.caffemodel import load_caffemodel_into
from .trainer import Trainer


def _clean_prototxt(text: str) -> str:
    # the reference's def.prototxt elides the conv stack with bare "." lines
    return "\n".join(l for l in text.splitlines() if l.strip() != ".")


@dataclass
class DataSpec:
    batch_size: int = 120
    identities_per_batch: int = 60
    imgs_per_identity: int = 2
    shuffle: bool = True
    rand_identity: bool = True
    new_height: int = 224
    new_width: int = 224
    root_folder: str = ""
    source: str = ""

    @classmethod
    def from_layer(cls, layer) -> "DataSpec":
        p = layer.raw.get("multi_batch_data_param")
        kw = {}
        if p is not None:
            for f in ("batch_size", "shuffle", "new_height", "new_width",
                      "root_folder", "source", "rand_identity"):
                if p.has(f):
                    kw[f] = p.get(f)
            if p.has("identity_num_per_batch"):
                kw["identities_per_batch"] = p.get("identity_num_per_batch")
            if p.has("img_num_per_identity"):
                kw["imgs_per_identity"] = p.get("img_num_per_identity")
        return cls(**kw)


def build_trainer_from_prototxt(
    net_text: str,
    solver: SolverConfig,
    device: Optional[torch.device] = None,
    synthetic_classes: int = 256,
    synthetic_per_class: int = 4,
    image_size: Optional[int] = None,
    amp_dtype: Optional[torch.dtype] = None,
    caffemodel: Optional[str] = None,
    num_workers: int = 0,
    channels_last: Optional[bool] = None,
    pure_bf16: bool = False,
    backbone: Optional[str] = None,
    hip_graph: Optional[bool] = None,
    timers: bool = False,
    init: str = "caffe",
) -> Trainer:
    net = parse_net_prototxt(_clean_prototxt(net_text))

    # --- data layer (TRAIN phase)
    data_layers = [l for l in net.find("MultibatchData") if l.phase in (None, "TRAIN")]
    spec = DataSpec.from_layer(data_layers[0]) if data_layers else DataSpec()
    test_layers = [l for l in net.find("MultibatchData") if l.phase == "TEST"]
    test_spec = DataSpec.from_layer(test_layers[0]) if test_layers else None

    img = image_size or spec.new_height
    # transform_param on the data layer (crop/mean/mirror)
    tcfg = TransformConfig()
    if data_layers is not None and data_layers and data_layers[0].raw.has("transform_param"):
        tp = data_layers[0].raw.get("transform_param")
        if tp.has("crop_size"):
            tcfg.crop_size = tp.get("crop_size")
            img = image_size or tcfg.crop_size
        mv = tp.get_all("mean_value")
        if mv:
            tcfg.mean_values = tuple(float(v) for v in mv)
        if tp.has("mirror"):
            tcfg.mirror = bool(tp.get("mirror"))

    # --- DataTransformer layer
    aug = None
    dt_layers = [l for l in net.find("DataTransformer") if l.phase in (None, "TRAIN")]
    if dt_layers and dt_layers[0].raw.has("data_transformer_l_param"):
        aug_cfg = TransformConfig.from_message(dt_layers[0].raw.get("data_transformer_l_param"))
        aug = DataTransformer(aug_cfg)

    # --- backbone: the reference's conv stack is GoogLeNet v1 (overridable);
    # L2-normalize the embedding iff the net has an L2Normalize layer
    has_l2 = bool(net.find("L2Normalize"))
    model = build_embedding_model(backbone or "googlenet", normalize=has_l2, init=init)
    if caffemodel:
        load_caffemodel_into(model, caffemodel)

    # --- loss layer
    loss_layers = net.find("NPairMultiClassLoss")
    if loss_layers and loss_layers[0].raw.has("npair_loss_param"):
        lcfg = NPairLossConfig.from_message(loss_layers[0].raw.get("npair_loss_param"))
    else:
        lcfg = NPairLossConfig()
    loss_mod = NPairMultiClassLoss(lcfg)

    # --- dataset: real list-file folder when it exists, else synthetic.
    # Seeds are rank-folded (solver.random_seed + rank): with the loss
    # all-gathering embeddings across ranks, identical per-rank batches would
    # inject world_size duplicates of every sample into the global batch —
    # perfect sim~1.0 "positives" that pair_is_self does not exclude,
    # corrupting mining and Recall@k.  Distinct seeds keep ranks drawing
    # independent batches like the reference's per-rank MPI data layers.
    from ..parallel import collectives as comm

    base_seed = solver.random_seed if solver.random_seed is not None else 0
    rank_seed = base_seed + comm.rank()
    use_real = spec.root_folder and os.path.isdir(spec.root_folder) and os.path.isfile(spec.source)
    if use_real:
        from ..data.folder import FolderListDataset

        ds = FolderListDataset(spec.root_folder, spec.source,
                               new_height=spec.new_height, new_width=spec.new_width)
    else:
        # rank-folded dataset seed: the synthetic pool is small, so index
        # collisions across ranks would be frequent — rank-unique images
        # (with the shared label space) avoid exact-duplicate embeddings
        n_cls = max(synthetic_classes, spec.identities_per_batch)
        ds = SyntheticImageDataset(num_classes=n_cls,
                                   per_class=max(synthetic_per_class, spec.imgs_per_identity),
                                   image_size=img, seed=base_seed * 131 + comm.rank())
    sampler = PKBatchSampler(ds.labels, spec.identities_per_batch, spec.imgs_per_identity,
                             shuffle=spec.shuffle, rand_identity=spec.rand_identity,
                             seed=rank_seed)
    loader = torch.utils.data.DataLoader(ds, batch_sampler=sampler, num_workers=num_workers)

    test_loader = None
    if test_spec is not None:
        tds = SyntheticImageDataset(num_classes=max(synthetic_classes, test_spec.identities_per_batch),
                                    per_class=max(synthetic_per_class, test_spec.imgs_per_identity),
                                    image_size=img, seed=(base_seed + 1) * 131 + comm.rank())
        tsampler = PKBatchSampler(tds.labels, test_spec.identities_per_batch,
                                  test_spec.imgs_per_identity, seed=rank_seed + 7919)
        test_loader = torch.utils.data.DataLoader(tds, batch_sampler=tsampler, num_workers=num_workers)

    if channels_last is None:
        channels_last = torch.cuda.is_available()
    trainer = Trainer(model, loss_mod, solver, loader, test_loader=test_loader,
                      device=device, amp_dtype=amp_dtype,
                      channels_last=channels_last, pure_bf16=pure_bf16,
                      hip_graph=hip_graph, timers=timers)
    trainer.augment = aug  # applied in train_step before the model (TRAIN phase)
    # Caffe transform_param (mean subtraction / crop) applies to real 0-255
    # image sources; synthetic data is already zero-mean at crop size.
    has_tp = bool(data_layers) and data_layers[0].raw.has("transform_param")
    trainer.preprocess = tcfg if (use_real and has_tp) else None
    return trainer
