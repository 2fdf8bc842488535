"""SGD solver with Caffe semantics.

The reference trains with the fork's Caffe solver (usage/solver.prototxt:
SGD, base_lr 0.001, step x0.5/10k, momentum 0.9, weight_decay 2e-5).
Caffe's update rule differs from torch.optim.SGD in where lr enters:
    Caffe:  v = momentum * v + lr * (grad + wd * w);  w -= v
    torch:  v = momentum * v + (grad + wd * w);       w -= lr * v
(the two diverge whenever lr changes mid-run, which the step policy does
every 10k iterations), so CaffeSGD implements the Caffe rule exactly.
"""

from __future__ import annotations

from typing import Iterable

import torch

from ..config.params import SolverConfig


class CaffeSGD(torch.optim.Optimizer):
    """Exact Caffe SGD.  With `master_weights=True` (for pure-bf16 models)
    the optimizer keeps an fp32 master copy per parameter: gradients are
    upcast, the velocity/update run in fp32, and the bf16 parameter is
    refreshed from the master each step — removing autocast's ~2 weight
    casts per conv per forward."""

    def __init__(self, params: Iterable, lr: float = 0.01, momentum: float = 0.0,
                 weight_decay: float = 0.0, master_weights: bool = False):
        defaults = dict(lr=lr, momentum=momentum, weight_decay=weight_decay,
                        master_weights=master_weights)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            lr = group["lr"]
            mom = group["momentum"]
            wd = group["weight_decay"]
            params = [p for p in group["params"] if p.grad is not None]
            if not params:
                continue
            master = group["master_weights"]
            grads = [p.grad for p in params]
            vs, ws = [], []
            for p in params:
                st = self.state[p]
                if "v" not in st:
                    if master and p.dtype != torch.float32:
                        st["master"] = p.detach().float().clone()
                    st["v"] = torch.zeros_like(st.get("master", p))
                vs.append(st["v"])
                ws.append(st.get("master", p))
            if master:
                grads = [g.float() if g.dtype != torch.float32 else g for g in grads]
            # fused multi-tensor update: v = mom*v + lr*g + (lr*wd)*w; w -= v
            # — 4 foreach kernels total, no temporaries (the reference-rule
            # lr*(g + wd*w) distributes so wd folds into a second add)
            torch._foreach_mul_(vs, mom)
            torch._foreach_add_(vs, grads, alpha=lr)
            if wd != 0:
                torch._foreach_add_(vs, ws, alpha=lr * wd)
            torch._foreach_sub_(ws, vs)
            if master:
                for p, w in zip(params, ws):
                    if w is not p:
                        p.copy_(w)  # bf16 <- fp32 master
        return loss

    def set_lr(self, lr: float):
        for group in self.param_groups:
            group["lr"] = lr


def build_optimizer(model: torch.nn.Module, solver: SolverConfig,
                    master_weights: bool = False) -> CaffeSGD:
    return CaffeSGD(model.parameters(), lr=solver.base_lr,
                    momentum=solver.momentum, weight_decay=solver.weight_decay,
                    master_weights=master_weights)
