"""Numerics tests for each gfx950 HIP kernel against the plain PyTorch fp32
reference of the same op (run on MI355X via gpurun; skipped without GPU)."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

from npairloss_amd.config.params import MiningMethod, MiningRegion, NPairLossConfig
from npairloss_amd.ops import _backend
from npairloss_amd.ops import npair_loss as NL

from util import make_batch, config_grid


def _C():
    return _backend.ext()


def rand_sg(B=128, G=256, ncls=16, seed=0, device="cuda"):
    g = torch.Generator(device="cpu").manual_seed(seed)
    S = torch.randn(B, G, generator=g).float().to(device)
    lab_g = torch.randint(0, ncls, (G,), generator=g, dtype=torch.int64).to(device)
    lab_l = lab_g[:B].clone()
    return S.contiguous(), lab_l, lab_g


def test_extension_loaded():
    assert _backend.has_extension()
    assert torch.cuda.is_available()


@pytest.mark.parametrize("shape", [(4, 8), (128, 1024), (256, 1024), (37, 129)])
def test_l2norm_fwd_bwd(shape):
    N, D = shape
    x = torch.randn(N, D, device="cuda") * 3
    y, inv = _C().l2norm_fwd(x)
    ref = torch.nn.functional.normalize(x, dim=1, eps=1e-12)
    torch.testing.assert_close(y, ref, rtol=1e-5, atol=1e-6)
    dy = torch.randn_like(y)
    dx = _C().l2norm_bwd(y, inv, dy)
    xr = x.clone().requires_grad_(True)
    torch.nn.functional.normalize(xr, dim=1, eps=1e-12).backward(dy)
    torch.testing.assert_close(dx, xr.grad, rtol=1e-4, atol=1e-5)


@pytest.mark.parametrize("mnk", [(32, 32, 32), (256, 2048, 1024), (120, 960, 1024),
                                 (33, 65, 17), (64, 64, 4096)])
def test_sim_gemm_nt(mnk):
    M, N, K = mnk
    A = torch.randn(M, K, device="cuda")
    B = torch.randn(N, K, device="cuda")
    C = _C().sim_gemm_nt(A, B)
    # compare against an fp64 reference: both the MFMA kernel and rocBLAS
    # fp32 carry O(sqrt(K)*eps) accumulation noise of their own
    ref = (A.double() @ B.double().t()).float()
    torch.testing.assert_close(C, ref, rtol=1e-4, atol=1e-3)


@pytest.mark.parametrize("mnk", [(256, 1024, 2048), (33, 65, 127), (120, 1024, 960)])
def test_gemm_nn_tn(mnk):
    M, N, K = mnk
    A = torch.randn(M, K, device="cuda")
    B = torch.randn(K, N, device="cuda")
    torch.testing.assert_close(_C().gemm_nn(A, B), A @ B, rtol=2e-5, atol=2e-4)
    At = torch.randn(K, M, device="cuda")  # gemm_tn computes At^T @ B
    torch.testing.assert_close(_C().gemm_tn(At, B), At.t() @ B, rtol=2e-5, atol=2e-4)


@pytest.mark.parametrize("bg", [(8, 16), (128, 256), (256, 2048), (120, 960)])
def test_rowstats(bg):
    B, G = bg
    S, lab_l, lab_g = rand_sg(B, G, seed=B)
    mnw, mxb, mxa = _C().rowstats(S, lab_l.int(), lab_g.int(), 0)
    same, diff = NL._masks(lab_l, lab_g, 0)
    r_mnw, r_mxb, r_mxa = NL._row_stats(S, same, diff)
    # empty-set sentinels: kernel uses +-FLT_MAX, torch uses +-inf
    fmax = torch.finfo(torch.float32).max
    torch.testing.assert_close(mnw, r_mnw.clamp(-fmax, fmax), atol=0, rtol=0)
    torch.testing.assert_close(mxb, r_mxb.clamp(-fmax, fmax), atol=0, rtol=0)
    torch.testing.assert_close(mxa, r_mxa.clamp(-fmax, fmax), atol=0, rtol=0)


@pytest.mark.parametrize("use_same", [True, False])
@pytest.mark.parametrize("sn", [-0.0, -0.3, -0.99, 0.0, 2.0, 5.0])
def test_local_relative_thr(use_same, sn):
    B, G = 64, 512
    S, lab_l, lab_g = rand_sg(B, G, ncls=8, seed=7)
    thr = _C().local_relative_thr(S, lab_l.int(), lab_g.int(), 0, use_same, sn)
    same, diff = NL._masks(lab_l, lab_g, 0)
    ref = NL._local_relative_thr(S, same if use_same else diff, sn)
    fmax = torch.finfo(torch.float32).max
    torch.testing.assert_close(thr, ref.clamp(-fmax, fmax), atol=0, rtol=0)


@pytest.mark.parametrize("use_same", [True, False])
@pytest.mark.parametrize("sn", [-0.0, -0.3, -0.99, 0.0, 2.0])
def test_global_relative_thr(use_same, sn):
    B, G = 96, 768
    S, lab_l, lab_g = rand_sg(B, G, ncls=12, seed=11)
    thr = _C().global_relative_thr(S, lab_l.int(), lab_g.int(), 0, use_same, sn)
    same, diff = NL._masks(lab_l, lab_g, 0)
    ref = NL._global_relative_thr(S, same if use_same else diff, sn)
    fmax = torch.finfo(torch.float32).max
    assert thr.item() == ref.clamp(-fmax, fmax).item()


def test_global_relative_thr_empty():
    # all labels equal -> diff list empty
    S = torch.randn(8, 8, device="cuda")
    lab = torch.zeros(8, dtype=torch.int32, device="cuda")
    thr = _C().global_relative_thr(S, lab, lab, 0, False, -0.3)
    assert thr.item() == -torch.finfo(torch.float32).max


@pytest.mark.parametrize("cfg_idx", range(len(config_grid())))
def test_fused_fwd_vs_torch(cfg_idx):
    cfg = config_grid()[cfg_idx]
    B, G = 64, 256
    S, lab_l, lab_g = rand_sg(B, G, ncls=8, seed=cfg_idx)
    same, diff = NL._masks(lab_l, lab_g, 0)
    mnw, mxb, mxa = NL._row_stats(S, same, diff)
    thr_p, thr_n = NL._thresholds_torch(S, same, diff, mnw, mxb, cfg)
    fmax_t = torch.nan_to_num(mxa, neginf=0.0, posinf=0.0)
    fin = torch.finfo(torch.float32)
    tp = thr_p.clamp(fin.min, fin.max).contiguous()
    tn = thr_n.clamp(fin.min, fin.max).contiguous()
    inum, dnum, li, ls, lg = _C().fused_fwd(
        S, lab_l.int(), lab_g.int(), 0, tp, tn, fmax_t,
        cfg.margin_ident, cfg.margin_diff, int(cfg.ap_mining_method), int(cfg.an_mining_method))
    sel_p, sel_n = NL._select(S, same, diff, thr_p, thr_n, cfg)
    E = torch.exp(S - fmax_t.unsqueeze(1))
    r_li = (E * sel_p).sum(1)
    r_ls = r_li + (E * sel_n).sum(1)
    torch.testing.assert_close(inum, sel_p.sum(1).float(), atol=0, rtol=0)
    torch.testing.assert_close(dnum, sel_n.sum(1).float(), atol=0, rtol=0)
    torch.testing.assert_close(li, r_li, rtol=1e-5, atol=1e-6)
    torch.testing.assert_close(ls, r_ls, rtol=1e-5, atol=1e-6)


@pytest.mark.parametrize("cfg_idx", range(0, len(config_grid()), 2))
def test_bwd_weights_vs_torch(cfg_idx):
    cfg = config_grid()[cfg_idx]
    B, G = 64, 256
    S, lab_l, lab_g = rand_sg(B, G, ncls=8, seed=cfg_idx + 50)
    # build the forward state with the torch path on the same S
    same, diff = NL._masks(lab_l, lab_g, 0)
    mnw, mxb, mxa = NL._row_stats(S, same, diff)
    thr_p, thr_n = NL._thresholds_torch(S, same, diff, mnw, mxb, cfg)
    fmax_t = torch.nan_to_num(mxa, neginf=0.0, posinf=0.0)
    sel_p, sel_n = NL._select(S, same, diff, thr_p, thr_n, cfg)
    E = torch.exp(S - fmax_t.unsqueeze(1))
    li = (E * sel_p).sum(1)
    ls = li + (E * sel_n).sum(1)
    fin = torch.finfo(torch.float32)
    W = _C().bwd_weights(S, lab_l.int(), lab_g.int(), 0,
                         thr_p.clamp(fin.min, fin.max).contiguous(),
                         thr_n.clamp(fin.min, fin.max).contiguous(),
                         fmax_t, li, ls, cfg.margin_ident, cfg.margin_diff,
                         int(cfg.ap_mining_method), int(cfg.an_mining_method), 1.0 / B)
    Wref = NL._bwd_weights_torch(S, lab_l, lab_g, 0, thr_p, thr_n, fmax_t, li, ls, cfg, 1.0 / B)
    torch.testing.assert_close(W, Wref, rtol=1e-5, atol=1e-7)


@pytest.mark.parametrize("bg", [(32, 64), (128, 1024), (120, 960)])
def test_recall_vs_torch(bg):
    B, G = bg
    S, lab_l, lab_g = rand_sg(B, G, ncls=6, seed=B + 1)
    ks = [1, 5, 10]
    ks_t = torch.tensor(ks, dtype=torch.int32, device="cuda")
    hits = _C().recall_hits(S, lab_l.int(), lab_g.int(), 0, ks_t, max(ks))
    # exact integer hit counts: same threshold extraction + strict->
    # semantics (tie fixtures with deliberate duplicates: test_gpu_ties.py).
    # round() undoes the reference's fp32 /B*B wobble (e.g. 77/120*120)
    ref = torch.round(NL._recall_torch(S, lab_l, lab_g, 0, ks) * B)
    torch.testing.assert_close(hits.float(), ref, atol=0, rtol=0)


@pytest.mark.parametrize("cfg_idx", range(len(config_grid())))
def test_module_gpu_vs_oracle(cfg_idx):
    """End-to-end GPU module (HIP path) vs the NumPy oracle: loss, metrics,
    and gradient."""
    from npairloss_amd.ops import oracle

    cfg = config_grid()[cfg_idx]
    f, lab = make_batch(num_classes=8, per_class=4, dim=64, seed=cfg_idx + 70)
    fwds, grads = oracle.npair_loss_multirank(f.astype(np.float64), lab, cfg, num_gpu=1)

    ft = torch.from_numpy(f).float().cuda().requires_grad_(True)
    lt = torch.from_numpy(lab).cuda()
    mod = NL.NPairMultiClassLoss(cfg)
    out = mod(ft, lt)
    out.loss.backward()
    assert out.loss.item() == pytest.approx(fwds[0].loss, rel=2e-4, abs=1e-6)
    assert out.retrieve_top1.item() == pytest.approx(fwds[0].recall[1], abs=1e-6)
    assert out.retrieve_top5.item() == pytest.approx(fwds[0].recall[5], abs=1e-6)
    assert out.retrieve_top10.item() == pytest.approx(fwds[0].recall[10], abs=1e-6)
    assert out.feature_asum.item() == pytest.approx(fwds[0].feature_asum, rel=1e-5)
    np.testing.assert_allclose(ft.grad.cpu().numpy(), grads[0], rtol=5e-4, atol=5e-6)


def test_module_gpu_production_config_batch120():
    """The reference's production shape: batch 120 = 60 ids x 2 imgs,
    1024-d embeddings, GLOBAL RELATIVE_HARD ap + LOCAL HARD an."""
    from npairloss_amd.ops import oracle

    cfg = NPairLossConfig(
        margin_ident=0.0, margin_diff=-0.05, identsn=-0.0, diffsn=-0.3,
        ap_mining_region=MiningRegion.GLOBAL, ap_mining_method=MiningMethod.RELATIVE_HARD,
        an_mining_region=MiningRegion.LOCAL, an_mining_method=MiningMethod.HARD)
    f, lab = make_batch(num_classes=60, per_class=2, dim=1024, seed=99)
    fwds, grads = oracle.npair_loss_multirank(f.astype(np.float64), lab, cfg, num_gpu=1)
    ft = torch.from_numpy(f).float().cuda().requires_grad_(True)
    lt = torch.from_numpy(lab).cuda()
    out = NL.NPairMultiClassLoss(cfg)(ft, lt)
    out.loss.backward()
    assert out.loss.item() == pytest.approx(fwds[0].loss, rel=2e-4, abs=1e-6)
    np.testing.assert_allclose(ft.grad.cpu().numpy(), grads[0], rtol=1e-3, atol=1e-6)


def test_offline_recall_gpu_matches_cpu():
    from npairloss_amd.eval import recall_at_k

    f, lab = make_batch(num_classes=12, per_class=6, dim=64, seed=17)
    ft = torch.from_numpy(f).float()
    lt = torch.from_numpy(lab)
    cpu = recall_at_k(ft, lt, ks=(1, 5, 10), chunk=24)
    gpu = recall_at_k(ft.cuda(), lt.cuda(), ks=(1, 5, 10), chunk=24)
    for k in (1, 5, 10):
        assert gpu[k] == pytest.approx(cpu[k], abs=1e-9)


def test_loss_path_deterministic():
    """Two identical forward+backward passes produce bit-identical loss and
    gradients (fixed reduction orders; recall uses count-only atomics)."""
    from npairloss_amd.config.params import NPairLossConfig
    from npairloss_amd.ops.npair_loss import NPairMultiClassLoss

    cfg = NPairLossConfig(margin_diff=-0.05, ap_mining_region="GLOBAL",
                          ap_mining_method="RELATIVE_HARD", identsn=-0.0,
                          an_mining_region="LOCAL", an_mining_method="HARD")
    f, lab = make_batch(num_classes=30, per_class=4, dim=512, seed=77)
    outs = []
    for _ in range(2):
        ft = torch.from_numpy(f).float().cuda().requires_grad_(True)
        lt = torch.from_numpy(lab).cuda()
        out = NPairMultiClassLoss(cfg)(ft, lt)
        out.loss.backward()
        outs.append((out.loss.item(), out.retrieve_top1.item(), ft.grad.clone()))
    assert outs[0][0] == outs[1][0]
    assert outs[0][1] == outs[1][1]
    torch.testing.assert_close(outs[0][2], outs[1][2], rtol=0, atol=0)
