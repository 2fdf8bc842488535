"""Multi-process gloo test of the bucketed gradient reducer: after
finalize(), every rank holds the average of all ranks' gradients, equal to
the single-process gradient on the concatenated batch."""

import os
import pickle
import tempfile

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp


def _make_model(seed=0):
    torch.manual_seed(seed)
    return torch.nn.Sequential(
        torch.nn.Linear(16, 32), torch.nn.ReLU(), torch.nn.Linear(32, 4))


def _worker(rank, world, tmpdir):
    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from npairloss_amd.parallel.ddp import BucketedGradReducer

        model = _make_model(seed=rank)  # deliberately different init per rank
        red = BucketedGradReducer(model, bucket_mb=0.0001)  # force many buckets
        red.broadcast_params()  # now identical to rank 0's init

        torch.manual_seed(100 + rank)
        x = torch.randn(8, 16)
        y = model(x).pow(2).mean()
        y.backward()
        red.finalize()
        grads = {n: p.grad.clone().numpy() for n, p in model.named_parameters()}
        with open(os.path.join(tmpdir, f"r{rank}.pkl"), "wb") as fh:
            pickle.dump({"x": x.numpy(), "grads": grads}, fh)
    finally:
        dist.destroy_process_group()


@pytest.mark.parametrize("world", [2])
def test_reducer_averages_grads(world):
    with tempfile.TemporaryDirectory() as tmpdir:
        os.environ["MASTER_PORT"] = "29711"
        ctx = mp.get_context("spawn")
        procs = [ctx.Process(target=_worker, args=(r, world, tmpdir)) for r in range(world)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=120)
            assert p.exitcode == 0
        res = []
        for r in range(world):
            with open(os.path.join(tmpdir, f"r{r}.pkl"), "rb") as fh:
                res.append(pickle.load(fh))

    # all ranks hold identical (averaged) grads
    for n in res[0]["grads"]:
        np.testing.assert_allclose(res[0]["grads"][n], res[1]["grads"][n], rtol=1e-6)

    # equals the average of per-rank local grads computed single-process
    model = _make_model(seed=0)  # broadcast made everyone rank-0's params
    per_rank = []
    for r in range(world):
        m = _make_model(seed=0)
        m.load_state_dict(model.state_dict())
        x = torch.from_numpy(res[r]["x"])
        m(x).pow(2).mean().backward()
        per_rank.append({n: p.grad.clone() for n, p in m.named_parameters()})
    for n in res[0]["grads"]:
        avg = sum(pr[n] for pr in per_rank) / world
        np.testing.assert_allclose(res[0]["grads"][n], avg.numpy(), rtol=1e-5, atol=1e-7)


def test_reducer_flat_views_single_process():
    """World=1: grads live as views in persistent flat buckets; zero_grad
    reinstalls views; accumulation across micro-steps sums in place."""
    from npairloss_amd.parallel.ddp import BucketedGradReducer

    model = _make_model(seed=3)
    red = BucketedGradReducer(model, bucket_mb=0.0001)
    red.zero_grad()
    params = list(model.parameters())
    # grads are views into the flat buffers (shared storage)
    flats = {g.untyped_storage().data_ptr() for g in red._flat}
    for p in params:
        assert p.grad is not None
        assert p.grad.untyped_storage().data_ptr() in flats

    x = torch.randn(4, 16)
    model(x).pow(2).mean().backward()
    g1 = {n: p.grad.clone() for n, p in model.named_parameters()}
    # second backward WITHOUT zero_grad accumulates into the same views
    model(x).pow(2).mean().backward()
    for n, p in model.named_parameters():
        torch.testing.assert_close(p.grad, 2 * g1[n])
        assert p.grad.untyped_storage().data_ptr() in flats
    # zero_grad zeroes in place and keeps the views installed
    red.zero_grad()
    for p in params:
        assert p.grad.abs().sum() == 0
        assert p.grad.untyped_storage().data_ptr() in flats


def _accum_worker(rank, world, tmpdir):
    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from npairloss_amd.parallel.ddp import BucketedGradReducer

        model = _make_model(seed=0)
        red = BucketedGradReducer(model, bucket_mb=0.0001)
        red.broadcast_params()
        torch.manual_seed(10 + rank)
        xs = [torch.randn(4, 16) for _ in range(3)]
        red.zero_grad()
        red.set_accumulate(True)
        for x in xs[:-1]:  # micro-steps: no comm
            model(x).pow(2).mean().backward()
        red.set_accumulate(False)
        model(xs[-1]).pow(2).mean().backward()  # boundary: comm fires
        red.finalize()
        grads = {n: p.grad.clone().numpy() for n, p in model.named_parameters()}
        with open(os.path.join(tmpdir, f"a{rank}.pkl"), "wb") as fh:
            pickle.dump({"xs": [x.numpy() for x in xs], "grads": grads}, fh)
    finally:
        dist.destroy_process_group()


def test_reducer_grad_accumulation_2rank():
    """3 micro-steps x 2 ranks: final grads == average over ranks of the
    summed micro-step grads (communication only on the boundary step)."""
    world = 2
    with tempfile.TemporaryDirectory() as tmpdir:
        os.environ["MASTER_PORT"] = "29713"
        ctx = mp.get_context("spawn")
        procs = [ctx.Process(target=_accum_worker, args=(r, world, tmpdir)) for r in range(world)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(timeout=120)
            assert p.exitcode == 0
        res = []
        for r in range(world):
            with open(os.path.join(tmpdir, f"a{r}.pkl"), "rb") as fh:
                res.append(pickle.load(fh))

    for n in res[0]["grads"]:
        np.testing.assert_allclose(res[0]["grads"][n], res[1]["grads"][n], rtol=1e-6)

    model = _make_model(seed=0)
    per_rank = []
    for r in range(world):
        m = _make_model(seed=0)
        m.load_state_dict(model.state_dict())
        for xn in res[r]["xs"]:
            m(torch.from_numpy(xn)).pow(2).mean().backward()
        per_rank.append({n: p.grad.clone() for n, p in m.named_parameters()})
    for n in res[0]["grads"]:
        avg = sum(pr[n] for pr in per_rank) / world
        np.testing.assert_allclose(res[0]["grads"][n], avg.numpy(), rtol=1e-5, atol=1e-7)
