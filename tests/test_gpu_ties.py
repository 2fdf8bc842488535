"""Deliberate-tie tests (GPU, atol=0): quantized similarity values force
exact duplicates at every decision boundary — the (k+1)-th retrieval
threshold with strict-> hits (.cu:190-203), the relative order-statistic
picks over lists with repeated values (.cu:282-336), and the mining select
comparisons at s == thr (.cu:79-120).  All must match the torch/NumPy
reference semantics EXACTLY (no tolerance)."""

import pytest
import torch

pytestmark = pytest.mark.gpu

from npairloss_amd.ops import _backend
from npairloss_amd.ops import npair_loss as NL

from util import config_grid


def _C():
    return _backend.ext()


def quantized_sg(B=48, G=192, ncls=6, levels=9, seed=0, device="cuda"):
    """S values drawn from a tiny grid (multiples of 0.25 in [-1, 1]):
    every row contains many exact duplicates."""
    g = torch.Generator().manual_seed(seed)
    q = torch.randint(0, levels, (B, G), generator=g).float()
    S = (q - (levels - 1) / 2) / ((levels - 1) / 2)
    lab_g = torch.randint(0, ncls, (G,), generator=g, dtype=torch.int64)
    lab_l = lab_g[:B].clone()
    return S.to(device).contiguous(), lab_l.to(device), lab_g.to(device)


@pytest.mark.parametrize("bg", [(16, 32), (48, 192), (64, 512)])
def test_recall_with_ties_exact(bg):
    B, G = bg
    S, lab_l, lab_g = quantized_sg(B, G, seed=B)
    ks = [1, 5, 10]
    ks_t = torch.tensor(ks, dtype=torch.int32, device="cuda")
    hits = _C().recall_hits(S, lab_l.int(), lab_g.int(), 0, ks_t, max(ks))
    ref = torch.round(NL._recall_torch(S, lab_l, lab_g, 0, ks) * B)
    torch.testing.assert_close(hits.float(), ref, atol=0, rtol=0)


def test_recall_all_equal_sims():
    """Every similarity identical: threshold == every value, strict->
    means NO query retrieves anything, at any k."""
    B, G = 8, 64
    S = torch.full((B, G), 0.5, device="cuda")
    lab = torch.arange(G, device="cuda") % 4
    ks_t = torch.tensor([1, 5, 10], dtype=torch.int32, device="cuda")
    hits = _C().recall_hits(S.contiguous(), lab[:B].int().contiguous(),
                            lab.int().contiguous(), 0, ks_t, 10)
    assert hits.sum().item() == 0


@pytest.mark.parametrize("use_same", [True, False])
@pytest.mark.parametrize("sn", [-0.0, -0.3, -0.5, -0.99, 0.0, 1.0, 3.0])
def test_local_relative_thr_with_ties(use_same, sn):
    B, G = 48, 192
    S, lab_l, lab_g = quantized_sg(B, G, seed=17)
    thr = _C().local_relative_thr(S, lab_l.int(), lab_g.int(), 0, use_same, sn)
    same, diff = NL._masks(lab_l, lab_g, 0)
    ref = NL._local_relative_thr(S, same if use_same else diff, sn)
    fmax = torch.finfo(torch.float32).max
    torch.testing.assert_close(thr, ref.clamp(-fmax, fmax), atol=0, rtol=0)


@pytest.mark.parametrize("use_same", [True, False])
@pytest.mark.parametrize("sn", [-0.0, -0.3, -0.7, 0.0, 2.0])
def test_global_relative_thr_with_ties(use_same, sn):
    B, G = 48, 192
    S, lab_l, lab_g = quantized_sg(B, G, seed=23)
    thr = _C().global_relative_thr(S, lab_l.int(), lab_g.int(), 0, use_same, sn)
    same, diff = NL._masks(lab_l, lab_g, 0)
    ref = NL._global_relative_thr(S, same if use_same else diff, sn)
    fmax = torch.finfo(torch.float32).max
    assert thr.item() == ref.clamp(-fmax, fmax).item()


@pytest.mark.parametrize("cfg_idx", range(len(config_grid())))
def test_select_counts_at_boundary_exact(cfg_idx):
    """Pair counts when many s == thr exactly: the <= vs < vs >= branches
    (.cu:79-120) must agree bit-for-bit with the reference rule."""
    cfg = config_grid()[cfg_idx]
    B, G = 48, 192
    S, lab_l, lab_g = quantized_sg(B, G, seed=cfg_idx)
    same, diff = NL._masks(lab_l, lab_g, 0)
    mnw, mxb, mxa = NL._row_stats(S, same, diff)
    thr_p, thr_n = NL._thresholds_torch(S, same, diff, mnw, mxb, cfg)
    fmax_t = torch.nan_to_num(mxa, neginf=0.0, posinf=0.0)
    fin = torch.finfo(torch.float32)
    inum, dnum, li, ls, lg = _C().fused_fwd(
        S, lab_l.int(), lab_g.int(), 0,
        thr_p.clamp(fin.min, fin.max).contiguous(),
        thr_n.clamp(fin.min, fin.max).contiguous(), fmax_t,
        cfg.margin_ident, cfg.margin_diff,
        int(cfg.ap_mining_method), int(cfg.an_mining_method))
    sel_p, sel_n = NL._select(S, same, diff, thr_p, thr_n, cfg)
    torch.testing.assert_close(inum, sel_p.sum(1).float(), atol=0, rtol=0)
    torch.testing.assert_close(dnum, sel_n.sum(1).float(), atol=0, rtol=0)
