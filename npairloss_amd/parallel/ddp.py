"""Bucketed gradient all-reduce overlapped with backward.

The reference relies on its Caffe fork's solver to all-reduce backbone
gradients over MPI after the whole backward (host-staged, serial).  Here
the backbone's data-parallel gradient exchange is a hand-rolled bucketed
all-reduce: parameters are grouped into flat buckets in reverse
registration order (the approximate order grads become ready), each bucket
launches an async RCCL all-reduce the moment its last gradient lands
(post-accumulate hooks), overlapping communication with the rest of the
backward on RCCL's side stream over xGMI.  `finalize()` waits and writes
back the averaged gradients before the optimizer step.

Works over both the RCCL ("nccl") device backend and gloo (CPU tests).
"""

from __future__ import annotations

from typing import Dict, List, Optional

import torch
import torch.distributed as dist

from . import collectives as comm


class BucketedGradReducer:
    def __init__(self, module: torch.nn.Module, bucket_mb: float = 25.0,
                 group: Optional[object] = None):
        self.module = module
        self.group = group
        self.world = comm.world_size(group)
        self.enabled = comm.is_dist() and self.world > 1
        self.params: List[torch.nn.Parameter] = [
            p for p in module.parameters() if p.requires_grad]
        self._hooks = []
        self.buckets: List[List[torch.nn.Parameter]] = []
        self._param_bucket: Dict[int, int] = {}
        self._pending: List[int] = []
        self._works: List[Optional[object]] = []
        self._flat: List[Optional[torch.Tensor]] = []
        if not self.enabled:
            return
        # reverse order: later layers' grads arrive first during backward
        cap = int(bucket_mb * 1024 * 1024)
        cur: List[torch.nn.Parameter] = []
        size = 0
        for p in reversed(self.params):
            cur.append(p)
            size += p.numel() * p.element_size()
            if size >= cap:
                self.buckets.append(cur)
                cur, size = [], 0
        if cur:
            self.buckets.append(cur)
        for bi, bucket in enumerate(self.buckets):
            for p in bucket:
                self._param_bucket[id(p)] = bi
                self._hooks.append(p.register_post_accumulate_grad_hook(self._on_grad))
        self._reset()

    def _reset(self):
        n = len(self.buckets)
        self._pending = [len(b) for b in self.buckets]
        self._works = [None] * n
        self._flat = [None] * n

    def broadcast_params(self):
        """Rank-0 parameters (and buffers) to all ranks at startup."""
        if not self.enabled:
            return
        for t in list(self.module.parameters()) + list(self.module.buffers()):
            dist.broadcast(t.data, src=0, group=self.group)

    def _on_grad(self, param: torch.nn.Parameter):
        bi = self._param_bucket[id(param)]
        self._pending[bi] -= 1
        if self._pending[bi] == 0:
            bucket = self.buckets[bi]
            flat = torch._utils._flatten_dense_tensors([p.grad for p in bucket])
            self._works[bi] = dist.all_reduce(flat, group=self.group, async_op=True)
            self._flat[bi] = flat

    def finalize(self):
        """Wait for all bucket reductions, write averaged grads back."""
        if not self.enabled:
            return
        for bi, bucket in enumerate(self.buckets):
            if self._works[bi] is None:
                # grads never produced this step (e.g. frozen path): skip
                if self._pending[bi] != len(bucket):
                    raise RuntimeError("bucket %d incomplete: %d grads missing"
                                       % (bi, self._pending[bi]))
                continue
            self._works[bi].wait()
            flat = self._flat[bi]
            flat.div_(self.world)
            for p, g in zip(bucket, torch._utils._unflatten_dense_tensors(flat, [p.grad for p in bucket])):
                p.grad.copy_(g)
        self._reset()

    def remove(self):
        for h in self._hooks:
            h.remove()
        self._hooks = []
