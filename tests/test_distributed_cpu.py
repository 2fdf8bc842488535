"""Multi-process distributed correctness on CPU (gloo, world_size 2/4):
each rank's loss/metrics/gradient must equal the single-process multi-rank
oracle simulation on the same monolithic batch (SURVEY.md section 4 — the
reference's math guarantees this identity)."""

import os
import pickle
import tempfile

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp

from util import make_batch, config_grid


def _worker(rank, world, tmpdir, cfg_idx, seed):
    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ.setdefault("MASTER_PORT", "29531")
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from npairloss_amd.ops.npair_loss import NPairMultiClassLoss

        cfg = config_grid()[cfg_idx]
        f, lab = make_batch(num_classes=8, per_class=4, dim=32, seed=seed)
        G = f.shape[0]
        B = G // world
        fl = torch.from_numpy(f[rank * B:(rank + 1) * B]).float().requires_grad_(True)
        ll = torch.from_numpy(lab[rank * B:(rank + 1) * B])

        mod = NPairMultiClassLoss(cfg)
        out = mod(fl, ll)
        out.loss.backward()

        result = dict(
            rank=rank,
            loss=out.loss.item(),
            top1=out.retrieve_top1.item(),
            top5=out.retrieve_top5.item(),
            top10=out.retrieve_top10.item(),
            asum=out.feature_asum.item(),
            grad=fl.grad.numpy(),
        )
        with open(os.path.join(tmpdir, f"rank{rank}.pkl"), "wb") as fh:
            pickle.dump(result, fh)
    finally:
        dist.destroy_process_group()


def _world_grid():
    # full config grid at world 2/4 + one 8-rank case mirroring the
    # 8-GPU node topology (G = 8B; GLOBAL mining spans all ranks)
    return ([(w, c) for w in (2, 4) for c in (0, 3, len(config_grid()) - 1)]
            + [(8, 0)])


@pytest.mark.parametrize("world,cfg_idx", _world_grid())
def test_gloo_matches_multirank_oracle(world, cfg_idx):
    from npairloss_amd.ops import oracle

    seed = 100 + cfg_idx
    cfg = config_grid()[cfg_idx]
    with tempfile.TemporaryDirectory() as tmpdir:
        os.environ["MASTER_PORT"] = str(29600 + world * 10 + cfg_idx)
        ctx = mp.get_context("spawn")
        procs = [ctx.Process(target=_worker, args=(r, world, tmpdir, cfg_idx, seed)) for r in range(world)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=180)
            assert p.exitcode == 0, f"worker failed with exit code {p.exitcode}"
        results = {}
        for r in range(world):
            with open(os.path.join(tmpdir, f"rank{r}.pkl"), "rb") as fh:
                d = pickle.load(fh)
                results[d["rank"]] = d

    f, lab = make_batch(num_classes=8, per_class=4, dim=32, seed=seed)
    fwds, grads = oracle.npair_loss_multirank(f.astype(np.float64), lab, cfg, num_gpu=world)
    for r in range(world):
        res = results[r]
        assert res["loss"] == pytest.approx(fwds[r].loss, rel=1e-4, abs=1e-6), f"rank {r} loss"
        assert res["top1"] == pytest.approx(fwds[r].recall[1], abs=1e-6)
        assert res["top5"] == pytest.approx(fwds[r].recall[5], abs=1e-6)
        assert res["top10"] == pytest.approx(fwds[r].recall[10], abs=1e-6)
        assert res["asum"] == pytest.approx(fwds[r].feature_asum, rel=1e-5)
        np.testing.assert_allclose(res["grad"], grads[r], rtol=5e-4, atol=1e-6)
