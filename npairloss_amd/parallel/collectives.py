"""Collectives for the loss-level global-batch parallelism.

The reference staged everything through HOST buffers with MPI
(npair_multi_class_loss.cu:17-43 Allgather of cpu_data, .cu:462-489
Allreduce of cpu_diff) — a D2H + H2D round trip per iteration.  Here the
collectives run on DEVICE buffers over RCCL/xGMI (torch.distributed "nccl"
backend == RCCL on ROCm); on CPU (tests) the gloo backend serves the same
calls.

Key structural change vs the reference: the backward's
allreduce-then-take-my-slice (.cu:467,494) is replaced by a
reduce-scatter, which is mathematically identical (each rank only ever
reads its own B-row slice of the summed G x D gradient) and moves
1/world_size of the bytes per link.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist


def is_dist() -> bool:
    return dist.is_available() and dist.is_initialized()


def world_size(group: Optional[dist.ProcessGroup] = None) -> int:
    return dist.get_world_size(group) if is_dist() else 1


def rank(group: Optional[dist.ProcessGroup] = None) -> int:
    return dist.get_rank(group) if is_dist() else 0


def _backend_is_gloo(group) -> bool:
    try:
        return dist.get_backend(group) == "gloo"
    except Exception:  # noqa: BLE001
        return False


def all_gather_rows(t: torch.Tensor, group: Optional[dist.ProcessGroup] = None) -> torch.Tensor:
    """All-gather along dim 0: (B, ...) -> (world*B, ...). Identity when not
    distributed.  Device buffers — no host staging."""
    if not is_dist() or world_size(group) == 1:
        return t
    ws = world_size(group)
    t = t.contiguous()
    out = t.new_empty((ws * t.shape[0],) + tuple(t.shape[1:]))
    if _backend_is_gloo(group):
        chunks = list(out.chunk(ws, dim=0))
        dist.all_gather(chunks, t, group=group)
    else:
        dist.all_gather_into_tensor(out, t, group=group)
    return out


def reduce_scatter_rows(t: torch.Tensor, group: Optional[dist.ProcessGroup] = None) -> torch.Tensor:
    """Sum-reduce-scatter along dim 0: (world*B, ...) -> this rank's (B, ...)
    slice of the sum.  Equivalent to allreduce+slice (the reference's
    .cu:467+494 pattern) at 1/world the traffic.  Identity when not
    distributed."""
    if not is_dist() or world_size(group) == 1:
        return t
    ws = world_size(group)
    t = t.contiguous()
    assert t.shape[0] % ws == 0
    out = t.new_empty((t.shape[0] // ws,) + tuple(t.shape[1:]))
    if _backend_is_gloo(group):
        # gloo has no reduce_scatter: allreduce then slice (CPU test path).
        dist.all_reduce(t, group=group)
        r = rank(group)
        out.copy_(t[r * out.shape[0] : (r + 1) * out.shape[0]])
    else:
        dist.reduce_scatter_tensor(out, t, group=group)
    return out


def all_reduce_sum(t: torch.Tensor, group: Optional[dist.ProcessGroup] = None) -> torch.Tensor:
    if is_dist() and world_size(group) > 1:
        dist.all_reduce(t, group=group)
    return t
