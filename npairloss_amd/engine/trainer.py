"""Training loop with Caffe-solver semantics.

Drives the implied trainer of the reference's prototxts: per-iteration
forward (backbone -> L2Normalize -> NPairMultiClassLoss) and backward,
Caffe-rule SGD with the step LR policy, `display`-interval logging with
`average_loss`-window smoothing (solver.prototxt:5-6), periodic TEST-phase
evaluation (test_interval x test_iter, :2-3), and snapshot/resume
(:15-16).  Multi-GPU: one process per GPU; the loss module all-gathers
embeddings internally; backbone grads go through the bucketed RCCL
all-reduce overlapped with backward (parallel/ddp.py).
"""

from __future__ import annotations

import collections
import json
import os
import time
from typing import Callable, Iterable, Optional

import numpy as np
import torch

from ..config.params import SolverConfig
from ..ops.npair_loss import NPairMultiClassLoss
from ..parallel import collectives as comm
from ..parallel.ddp import BucketedGradReducer
from .solver import build_optimizer


class AverageWindow:
    """Caffe's average_loss smoothing (running mean over the last N)."""

    def __init__(self, n: int):
        self.buf = collections.deque(maxlen=max(1, n))

    def add(self, v: float) -> float:
        self.buf.append(v)
        return sum(self.buf) / len(self.buf)


class Trainer:
    def __init__(self, model: torch.nn.Module, loss: NPairMultiClassLoss,
                 solver: SolverConfig, train_loader: Iterable,
                 test_loader: Optional[Iterable] = None,
                 device: Optional[torch.device] = None,
                 amp_dtype: Optional[torch.dtype] = None,
                 log_fn: Callable[[str], None] = print,
                 channels_last: bool = False,
                 pure_bf16: bool = False,
                 hip_graph: Optional[bool] = None,
                 timers: bool = False):
        self.device = device or (torch.device("cuda") if torch.cuda.is_available()
                                 else torch.device("cpu"))
        # hipGraph whole-step replay (the bench's headline mode): capture
        # after `graph_warmup` eager steps, replay thereafter; off on CPU.
        self.use_graph = hip_graph if hip_graph is not None else False
        self.graph_warmup = 12
        self._graph_state = None
        self._graph_warmup_left = None  # latched from graph_warmup on first step
        from ..utils.profiling import PhaseTimers
        self.timers = PhaseTimers(enabled=timers, use_cuda=self.device.type == "cuda")
        self.model = model.to(self.device)
        if channels_last:
            self.model = self.model.to(memory_format=torch.channels_last)
        self.channels_last = channels_last
        self.pure_bf16 = pure_bf16
        if pure_bf16:
            self.model = self.model.to(torch.bfloat16)
        self.loss = loss
        self.solver = solver
        self.train_loader = train_loader
        self.test_loader = test_loader
        self.amp_dtype = amp_dtype
        self.log = log_fn
        self.optimizer = build_optimizer(self.model, solver, master_weights=pure_bf16)
        self.reducer = BucketedGradReducer(self.model)
        self.reducer.broadcast_params()
        self.iter = 0
        self.avg = AverageWindow(solver.average_loss)
        self.history = []
        self.augment = None  # optional TRAIN-phase batch augmentation (DataTransformer)
        # optional Caffe transform_param (TransformConfig): mean subtraction +
        # center crop applied to every batch BEFORE augmentation, matching the
        # reference data layer (usage/def.prototxt:10-16).  Set by the net
        # builder when training from a real 0-255 image source.
        self.preprocess = None
        # failure detection: raise on NaN/Inf loss every N iters (0 = off).
        # (the reference had none — an NaN would silently poison the run)
        self.divergence_check = 25

    # -- checkpointing ------------------------------------------------------

    def snapshot(self, prefix: Optional[str] = None):
        if comm.rank() != 0:
            return None
        prefix = prefix or self.solver.snapshot_prefix or "./snap/model_"
        os.makedirs(os.path.dirname(prefix) or ".", exist_ok=True)
        path = f"{prefix}iter_{self.iter}.pt"
        torch.save({
            "iter": self.iter,
            "model": self.model.state_dict(),
            "optimizer": self.optimizer.state_dict(),
            "solver": self.solver.__dict__,
        }, path)
        return path

    def restore(self, path: str):
        ck = torch.load(path, map_location=self.device, weights_only=False)
        self.model.load_state_dict(ck["model"])
        self.optimizer.load_state_dict(ck["optimizer"])
        self.iter = ck["iter"]
        # optimizer load_state_dict REPLACES state tensors; a captured graph
        # would keep updating the orphaned ones — force recapture
        self._graph_state = None

    # -- Caffe-format snapshots (.caffemodel + .solverstate) ----------------

    def snapshot_caffe(self, prefix: Optional[str] = None) -> Optional[tuple]:
        """Write `<prefix>iter_N.caffemodel` (weights by layer name) plus
        `<prefix>iter_N.solverstate` (iter + SGD momentum history in
        canonical learnable-parameter order), like the reference solver
        (usage/solver.prototxt:15-16)."""
        if comm.rank() != 0:
            return None
        from ..utils.caffemodel import (CaffeSolverState, caffe_param_order,
                                        save_caffemodel, write_solverstate)

        prefix = prefix or self.solver.snapshot_prefix or "./snap/model_"
        os.makedirs(os.path.dirname(prefix) or ".", exist_ok=True)
        mpath = f"{prefix}iter_{self.iter}.caffemodel"
        spath = f"{prefix}iter_{self.iter}.solverstate"
        save_caffemodel(self.model, mpath)
        history = []
        for _, p in caffe_param_order(self.model):
            v = self.optimizer.state.get(p, {}).get("v")
            if v is None:
                v = torch.zeros_like(p, dtype=torch.float32)
            # .contiguous(): channels_last tensors serialize in logical
            # (NCHW) element order, matching Caffe blob storage
            history.append(v.detach().float().cpu().contiguous().numpy())
        write_solverstate(spath, CaffeSolverState(
            iter=self.iter, learned_net=os.path.basename(mpath),
            history=history))
        return mpath, spath

    def restore_caffe(self, solverstate_path: str,
                      caffemodel_path: Optional[str] = None):
        """Resume mid-training from Caffe artifacts: weights from the
        .caffemodel (name-matched), iteration + SGD momentum from the
        .solverstate (canonical parameter order)."""
        from ..utils.caffemodel import (caffe_param_order, load_caffemodel_into,
                                        read_solverstate)

        st = read_solverstate(solverstate_path)
        if caffemodel_path is None and st.learned_net:
            cand = os.path.join(os.path.dirname(solverstate_path), st.learned_net)
            caffemodel_path = cand if os.path.exists(cand) else None
        if caffemodel_path:
            load_caffemodel_into(self.model, caffemodel_path)
        params = caffe_param_order(self.model)
        if len(st.history) != len(params):
            raise ValueError(
                f"solverstate history has {len(st.history)} blobs, model has "
                f"{len(params)} learnable parameters")
        with torch.no_grad():
            for (name, p), h in zip(params, st.history):
                if h.size != p.numel():
                    raise ValueError(f"history blob for {name}: {h.size} elements "
                                     f"vs parameter {p.numel()}")
                stp = self.optimizer.state.setdefault(p, {})
                master = stp.get("master")
                ref = master if master is not None else p
                t = torch.from_numpy(np.ascontiguousarray(h)).reshape(
                    p.shape).to(device=ref.device, dtype=ref.dtype)
                if (t.dim() == 4 and not ref.is_contiguous()
                        and ref.is_contiguous(memory_format=torch.channels_last)):
                    t = t.contiguous(memory_format=torch.channels_last)
                stp["v"] = t
        self.iter = st.iter
        self._graph_state = None  # state tensors replaced: recapture

    # -- one training iteration --------------------------------------------

    def _prepare(self, images: torch.Tensor) -> torch.Tensor:
        """Input preprocessing run EAGERLY (outside any captured graph):
        Caffe transform_param, DataTransformer augmentation, layout/dtype."""
        if self.preprocess is not None and images.dim() == 4:
            from ..data.transforms import preprocess as _pp
            images = _pp(images, self.preprocess)
        if self.augment is not None and images.dim() == 4:
            images = self.augment(images)
        if self.channels_last and images.dim() == 4:
            images = images.to(memory_format=torch.channels_last)
        if self.pure_bf16:
            images = images.to(torch.bfloat16)
        return images

    def _micro_step(self, images: torch.Tensor, labels: torch.Tensor,
                    loss_scale: float = 1.0):
        """Forward + loss + (scaled) backward only — gradients accumulate
        into the flat buckets; no zeroing, comm wait, or optimizer here."""
        if self.amp_dtype is not None and self.device.type == "cuda":
            with torch.autocast("cuda", dtype=self.amp_dtype):
                feats = self.model(images)
            out = self.loss(feats.float(), labels)
        else:
            feats = self.model(images)
            out = self.loss(feats, labels)
        loss = out.loss if loss_scale == 1.0 else out.loss * loss_scale
        loss.backward()
        return out

    def _compute_step(self, images: torch.Tensor, labels: torch.Tensor):
        """Device-only step body (no host reads): forward + loss + backward
        + grad comm + optimizer.  hipGraph-capturable — inputs must already
        be prepared (_prepare) and sit at fixed addresses during capture."""
        self.reducer.zero_grad()  # zeroes the persistent flat grad buckets
        out = self._micro_step(images, labels)
        self.reducer.finalize()
        self.optimizer.step()
        return out

    def train_step(self, images: torch.Tensor, labels: torch.Tensor,
                   extra_batches: Optional[list] = None) -> dict:
        """One solver iteration.  With solver.iter_size > 1 (Caffe's
        gradient accumulation), `extra_batches` carries the remaining
        iter_size-1 micro-batches: losses are scaled by 1/iter_size and
        gradients accumulate in the flat buckets (no communication) until
        the final micro-batch, exactly like Caffe's normalized update."""
        self.model.train()
        images = images.to(self.device, non_blocking=True)
        labels = labels.to(self.device, non_blocking=True)
        lr = self.solver.lr_at(self.iter)
        self.optimizer.set_lr(lr)
        micro = [(images, labels)] + list(extra_batches or [])
        n_micro = max(len(micro), 1)
        with self.timers.phase("data_prep"):
            images = self._prepare(images)
        with self.timers.phase("step"):
            if n_micro == 1:
                out = self._compute_step(images, labels)
            else:
                self.reducer.zero_grad()
                self.reducer.set_accumulate(True)
                try:
                    for i, (mx, mlab) in enumerate(micro):
                        if i > 0:
                            mx = self._prepare(mx.to(self.device, non_blocking=True))
                            mlab = mlab.to(self.device, non_blocking=True)
                        else:
                            mx = images
                            mlab = labels
                        if i == n_micro - 1:
                            self.reducer.set_accumulate(False)
                        out = self._micro_step(mx, mlab, 1.0 / n_micro)
                finally:
                    self.reducer.set_accumulate(False)
                self.reducer.finalize()
                self.optimizer.step()
        self.iter += 1
        if self.divergence_check and self.iter % self.divergence_check == 0:
            lv = float(out.loss.detach())
            if not (lv == lv and abs(lv) != float("inf")):
                raise FloatingPointError(
                    f"training diverged: loss={lv} at iter {self.iter}")
        # detach EVERYTHING: a non-detached Function output held across a
        # later hipGraph capture keeps the autograd node (and its stream
        # state) alive and hipGraphInstantiate segfaults (bisect: tools/
        # debug_capture.py — hold-the-stats-dict was the exact trigger)
        return {
            "loss": out.loss.detach(), "top1": out.retrieve_top1.detach(),
            "top5": out.retrieve_top5.detach(), "top10": out.retrieve_top10.detach(),
            "asum": out.feature_asum.detach(), "lr": lr,
        }

    # -- hipGraph-captured iteration ----------------------------------------

    def _graph_step(self, images: torch.Tensor, labels: torch.Tensor) -> dict:
        """Like train_step, but the whole compute step replays as ONE
        hipGraph: batches are copied into static device tensors, the
        captured work (fwd + loss + bwd + RCCL comm + optimizer) re-executes
        with zero per-kernel launch gaps.  Capture happens lazily after
        `graph_warmup` eager steps (MIOpen find must have run) and is
        re-taken whenever the LR-policy value changes (lr is baked into the
        captured optimizer kernels)."""
        self.model.train()
        lr = self.solver.lr_at(self.iter)
        g = self._graph_state
        if g is not None and g["lr"] != lr:
            g = self._graph_state = None  # lr changed: recapture
        if self._graph_warmup_left is None:
            self._graph_warmup_left = self.graph_warmup
        if g is None:
            if self._graph_warmup_left > 0:
                self._graph_warmup_left -= 1
                return self.train_step(images, labels)
            self.optimizer.set_lr(lr)
            static_img = self._prepare(images.to(self.device, non_blocking=True)).clone()
            static_lab = labels.to(self.device, non_blocking=True).clone()
            torch.cuda.synchronize()
            if comm.is_dist() and comm.world_size() > 1:
                import torch.distributed as dist
                dist.barrier()  # align ranks: every rank captures the same comm
            side = torch.cuda.Stream()
            side.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(side):
                self._compute_step(static_img, static_lab)  # allocator warmup
            torch.cuda.current_stream().wait_stream(side)
            torch.cuda.synchronize()
            try:
                graph = torch.cuda.CUDAGraph()
                # thread_local: only THIS thread's HIP calls are captured;
                # background threads (DataLoader machinery, autograd engine
                # workers from earlier eager iterations) otherwise poison
                # the capture and hipGraphInstantiate crashes
                with torch.cuda.graph(graph, capture_error_mode="thread_local"):
                    out = self._compute_step(static_img, static_lab)
            except Exception as e:  # noqa: BLE001
                self.log(f"hipGraph capture failed ({e!r}); continuing eager")
                graph = None
            if comm.is_dist() and comm.world_size() > 1:
                # consensus: if any rank failed capture, all ranks go eager
                # (a mid-capture failure may have desynced RCCL state)
                import torch.distributed as dist
                ok = torch.tensor([0 if graph is None else 1], device=self.device)
                dist.all_reduce(ok, op=dist.ReduceOp.MIN)
                if int(ok.item()) == 0:
                    graph = None
            if graph is None:
                self.use_graph = False
                return self.train_step(images, labels)
            self._graph_state = g = dict(graph=graph, img=static_img,
                                         lab=static_lab, out=out, lr=lr)
            self.iter += 1
            return self._stats_from(g, lr)
        if images.shape != g["img"].shape or labels.shape != g["lab"].shape:
            return self.train_step(images, labels)  # odd-sized batch: eager
        g["img"].copy_(self._prepare(images.to(self.device, non_blocking=True)))
        g["lab"].copy_(labels.to(self.device, non_blocking=True))
        g["graph"].replay()
        self.iter += 1
        if self.divergence_check and self.iter % self.divergence_check == 0:
            lv = float(g["out"].loss.detach())
            if not (lv == lv and abs(lv) != float("inf")):
                raise FloatingPointError(
                    f"training diverged: loss={lv} at iter {self.iter}")
        return self._stats_from(g, lr)

    def _stats_from(self, g, lr) -> dict:
        out = g["out"]
        # detach EVERYTHING: a non-detached Function output held across a
        # later hipGraph capture keeps the autograd node (and its stream
        # state) alive and hipGraphInstantiate segfaults (bisect: tools/
        # debug_capture.py — hold-the-stats-dict was the exact trigger)
        return {
            "loss": out.loss.detach(), "top1": out.retrieve_top1.detach(),
            "top5": out.retrieve_top5.detach(), "top10": out.retrieve_top10.detach(),
            "asum": out.feature_asum.detach(), "lr": lr,
        }

    # -- evaluation ---------------------------------------------------------

    @torch.no_grad()
    def evaluate(self, max_batches: Optional[int] = None) -> dict:
        if self.test_loader is None:
            return {}
        self.model.eval()
        n = 0
        acc = collections.defaultdict(float)
        for images, labels in self.test_loader:
            images = images.to(self.device, non_blocking=True)
            if self.preprocess is not None and images.dim() == 4:
                from ..data.transforms import preprocess as _pp
                images = _pp(images, self.preprocess)
            labels = labels.to(self.device, non_blocking=True)
            feats = self.model(images)
            out = self.loss(feats, labels)
            for k, v in (("loss", out.loss), ("top1", out.retrieve_top1),
                         ("top5", out.retrieve_top5), ("top10", out.retrieve_top10)):
                acc[k] += float(v)
            n += 1
            if max_batches is not None and n >= max_batches:
                break
        return {k: v / max(n, 1) for k, v in acc.items()}

    # -- main loop ----------------------------------------------------------

    def fit(self, max_iter: Optional[int] = None):
        max_iter = max_iter or self.solver.max_iter
        t0 = time.time()
        it_timer = time.time()
        epoch = 0
        data_iter = iter(self.train_loader)
        while self.iter < max_iter:
            def next_batch():
                nonlocal data_iter, epoch
                try:
                    return next(data_iter)
                except StopIteration:
                    epoch += 1
                    bs = getattr(self.train_loader, "batch_sampler", None)
                    if bs is not None and hasattr(bs, "set_epoch"):
                        bs.set_epoch(epoch)  # deterministic (base_seed, epoch)
                    data_iter = iter(self.train_loader)
                    return next(data_iter)

            images, labels = next_batch()
            iter_size = max(1, self.solver.iter_size)
            if iter_size > 1:
                # Caffe gradient accumulation: iter_size micro-batches per
                # solver iteration (always eager — the captured graph holds
                # exactly one accumulate+step sequence)
                extra = [next_batch() for _ in range(iter_size - 1)]
                stats = self.train_step(images, labels, extra_batches=extra)
            elif self.use_graph and self.device.type == "cuda":
                stats = self._graph_step(images, labels)
            else:
                stats = self.train_step(images, labels)
            if self.solver.display and self.iter % self.solver.display == 0:
                loss_v = float(stats["loss"])
                sm = self.avg.add(loss_v)
                dt = time.time() - it_timer
                it_timer = time.time()
                if comm.rank() == 0:
                    self.log(
                        f"iter {self.iter} lr {stats['lr']:.3g} loss {loss_v:.4f} "
                        f"(avg {sm:.4f}) top1 {float(stats['top1']):.3f} "
                        f"top5 {float(stats['top5']):.3f} asum {float(stats['asum']):.3f} "
                        f"[{dt:.1f}s]")
                self.history.append({"iter": self.iter, "loss": loss_v, "lr": stats["lr"]})
            if (self.solver.test_interval and self.test_loader is not None
                    and self.iter % self.solver.test_interval == 0):
                ev = self.evaluate(max_batches=self.solver.test_iter or None)
                if comm.rank() == 0 and ev:
                    self.log(f"TEST iter {self.iter}: " + json.dumps(ev))
            if self.solver.snapshot and self.iter % self.solver.snapshot == 0:
                self.snapshot()
        return time.time() - t0
