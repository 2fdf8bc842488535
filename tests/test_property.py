"""Property-based tests (hypothesis) for oracle invariants that hold for
ANY input/config, complementing the example-based suites."""

import numpy as np
import pytest

hyp = pytest.importorskip("hypothesis")
from hypothesis import given, settings, strategies as st

from npairloss_amd.config.params import MiningMethod, MiningRegion, NPairLossConfig
from npairloss_amd.ops import oracle


@given(sn=st.floats(min_value=-1.5, max_value=20, allow_nan=False),
       size=st.integers(min_value=0, max_value=100000))
@settings(max_examples=200, deadline=None)
def test_relative_index_in_range(sn, size):
    pos = oracle.relative_index(sn, size)
    if size <= 0:
        assert pos == -1
    else:
        assert 0 <= pos <= size - 1


@st.composite
def batch_and_cfg(draw):
    ncls = draw(st.integers(2, 5))
    per = draw(st.integers(2, 4))
    dim = draw(st.integers(2, 8))
    seed = draw(st.integers(0, 1000))
    cfg = NPairLossConfig(
        margin_ident=draw(st.floats(-0.5, 0.5)),
        margin_diff=draw(st.floats(-0.5, 0.5)),
        identsn=draw(st.floats(-0.99, 3.0)),
        diffsn=draw(st.floats(-0.99, 3.0)),
        ap_mining_region=draw(st.sampled_from(list(MiningRegion))),
        ap_mining_method=draw(st.sampled_from(list(MiningMethod))),
        an_mining_region=draw(st.sampled_from(list(MiningRegion))),
        an_mining_method=draw(st.sampled_from(list(MiningMethod))),
    )
    rng = np.random.default_rng(seed)
    f = rng.standard_normal((ncls * per, dim))
    f /= np.linalg.norm(f, axis=1, keepdims=True)
    lab = np.repeat(np.arange(ncls), per)
    rng.shuffle(lab)
    return f, lab, cfg


@given(bc=batch_and_cfg())
@settings(max_examples=60, deadline=None)
def test_forward_invariants(bc):
    f, lab, cfg = bc
    fwd = oracle.npair_forward(f, lab, f, lab, cfg, rank=0)
    B = f.shape[0]
    # loss is a mean of -log(p) with p in (0, 1]: finite and >= 0 (up to fp eps)
    assert np.isfinite(fwd.loss)
    assert fwd.loss >= -1e-9
    # selected counts never exceed the raw mask counts
    assert (fwd.ident_num <= fwd.same.sum(1)).all()
    assert (fwd.diff_num <= fwd.diff.sum(1)).all()
    # RAND selects every pair in its class
    if cfg.ap_mining_method == MiningMethod.RAND:
        assert (fwd.ident_num == fwd.same.sum(1)).all()
    if cfg.an_mining_method == MiningMethod.RAND:
        assert (fwd.diff_num == fwd.diff.sum(1)).all()
    # loss_sum = loss_ident + loss over negatives >= loss_ident
    assert (fwd.loss_sum >= fwd.loss_ident - 1e-12).all()
    # recall is a fraction and non-decreasing in k
    ks = sorted(fwd.recall)
    vals = [fwd.recall[k] for k in ks]
    assert all(0.0 <= v <= 1.0 for v in vals)
    assert all(a <= b + 1e-12 for a, b in zip(vals, vals[1:]))


@given(bc=batch_and_cfg(), lw=st.floats(0.1, 3.0))
@settings(max_examples=30, deadline=None)
def test_backward_scales_linearly(bc, lw):
    f, lab, cfg = bc
    fwd = oracle.npair_forward(f, lab, f, lab, cfg, rank=0)
    b1 = oracle.npair_backward_local(fwd, f, f, loss_weight=1.0)
    b2 = oracle.npair_backward_local(fwd, f, f, loss_weight=lw)
    np.testing.assert_allclose(b2.dF_local, lw * b1.dF_local, rtol=1e-10, atol=1e-12)
    np.testing.assert_allclose(b2.dF_total, lw * b1.dF_total, rtol=1e-10, atol=1e-12)
    # W rows are bounded: each part is a sub-probability distribution
    assert np.abs(b1.W).max() <= 2.0 + 1e-9


@given(world=st.integers(1, 4), seed=st.integers(0, 100))
@settings(max_examples=20, deadline=None)
def test_multirank_similarity_slicing(world, seed):
    rng = np.random.default_rng(seed)
    G = world * 6
    f = rng.standard_normal((G, 5))
    lab = rng.integers(0, 3, G)
    fwds, _ = oracle.npair_loss_multirank(f, lab, NPairLossConfig(), world)
    S_full = f @ f.T
    B = G // world
    for r, fw in enumerate(fwds):
        np.testing.assert_allclose(fw.S, S_full[r * B:(r + 1) * B], rtol=1e-12)


@given(bc=batch_and_cfg())
@settings(max_examples=40, deadline=None)
def test_torch_fp64_path_matches_oracle_any_config(bc):
    """Differential harness: the torch (CPU, fp64) module path must
    reproduce the NumPy oracle to ~machine precision for ANY mining
    region x method x margins x order-statistic combination — loss,
    recalls, and gradient."""
    import torch

    from npairloss_amd.ops.npair_loss import NPairMultiClassLoss

    f, lab, cfg = bc
    F = torch.from_numpy(f).requires_grad_(True)
    out = NPairMultiClassLoss(cfg, sim_dtype="fp64")(F, torch.from_numpy(lab))
    fwds, grads = oracle.npair_loss_multirank(f, lab, cfg, num_gpu=1)
    np.testing.assert_allclose(float(out.loss.detach()), fwds[0].loss,
                               rtol=1e-10, atol=1e-12)
    np.testing.assert_allclose(float(out.retrieve_top1), fwds[0].recall[1], atol=0)
    out.loss.backward()
    np.testing.assert_allclose(F.grad.numpy(), grads[0], rtol=1e-8, atol=1e-12)


@given(bc=batch_and_cfg(), world=st.sampled_from([2, 4]))
@settings(max_examples=25, deadline=None)
def test_torch_rank_slices_match_multirank_oracle(bc, world):
    """Rank-local torch computation on gathered features == the oracle's
    per-rank result, for any config (the SURVEY §4 multi-rank invariant,
    exercised WITHOUT processes by direct rank slicing)."""
    import torch

    from npairloss_amd.ops import npair_loss as NL

    f, lab, cfg = bc
    G = f.shape[0]
    if G % world != 0:
        return
    B = G // world
    fwds, _ = oracle.npair_loss_multirank(f, lab, cfg, num_gpu=world)
    F_g = torch.from_numpy(f)
    lab_g = torch.from_numpy(lab)
    for r in range(world):
        F_l = F_g[r * B:(r + 1) * B]
        lab_l = lab_g[r * B:(r + 1) * B]
        loss, recalls, _ = NL._forward_torch(F_l, lab_l, F_g, lab_g, r, cfg,
                                             (1, 5, 10))
        np.testing.assert_allclose(float(loss), fwds[r].loss, rtol=1e-10, atol=1e-12)
        np.testing.assert_allclose(float(recalls[0]), fwds[r].recall[1], atol=1e-9)
