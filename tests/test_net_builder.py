"""End-to-end: build a trainer from the reference's actual prototxt files
and run a few iterations on CPU (tiny images for speed)."""

import os

import pytest
import torch

from npairloss_amd.config.params import SolverConfig
from npairloss_amd.engine.net_builder import build_trainer_from_prototxt

REF_DEF = "/root/reference/usage/def.prototxt"
REF_SOLVER = "/root/reference/usage/solver.prototxt"

MINI_NET = """
name: "MiniNet"
layer {
    name: "data_mb"
    type: "MultibatchData"
    top: "data_mb"
    top: "label_mb"
    include { phase: TRAIN }
    transform_param { crop_size: 64 mean_value: 104 mean_value: 117 mean_value: 123 }
    multi_batch_data_param {
        batch_size: 16
        identity_num_per_batch: 8
        img_num_per_identity: 2
        shuffle: true
        new_height: 64
        new_width: 64
        rand_identity: true
    }
}
layer {
    name: "norm"
    type: "L2Normalize"
    bottom: "pool5/7x7_s1"
    top: "feat"
}
layer {
    name: "loss"
    type: "NPairMultiClassLoss"
    bottom: "feat"
    bottom: "label_mb"
    top: "loss"
    npair_loss_param {
        margin_diff: -0.05
        ap_mining_region: GLOBAL
        ap_mining_method: RELATIVE_HARD
        an_mining_region: LOCAL
        an_mining_method: HARD
        identsn: -0.0
    }
}
"""


def test_mini_net_builds_and_trains():
    solver = SolverConfig(base_lr=0.001, momentum=0.9, max_iter=2, display=0)
    tr = build_trainer_from_prototxt(MINI_NET, solver, device=torch.device("cpu"),
                                     synthetic_classes=16, image_size=64)
    tr.fit(max_iter=2)
    assert tr.iter == 2
    assert tr.loss.cfg.ap_mining_method.name == "RELATIVE_HARD"
    assert tr.loss.cfg.margin_diff == pytest.approx(-0.05)


@pytest.mark.skipif(not os.path.exists(REF_DEF), reason="reference not mounted")
def test_reference_prototxts_build():
    solver = SolverConfig.from_prototxt(open(REF_SOLVER).read())
    tr = build_trainer_from_prototxt(open(REF_DEF).read(), solver,
                                     device=torch.device("cpu"),
                                     synthetic_classes=64, image_size=64)
    # production config flowed through
    assert tr.loss.cfg.ap_mining_region.name == "GLOBAL"
    assert tr.loss.cfg.an_mining_method.name == "HARD"
    # sampler follows the data layer: P=60, K=2
    assert tr.train_loader.batch_sampler.P == 60
    assert tr.train_loader.batch_sampler.K == 2
    # test-phase loader exists (test data layer present)
    assert tr.test_loader is not None
    # augmentation layer found
    assert tr.augment is not None
    # one iteration end to end
    tr.fit(max_iter=1)
    assert tr.iter == 1
