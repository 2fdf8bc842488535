#!/usr/bin/env python3
"""Binary-search the trainer hipGraph-capture segfault: each variant runs in
a SUBPROCESS (a segfault kills only that variant).  Usage on the GPU box:
    python tools/debug_capture.py          # run all variants, print verdicts
    python tools/debug_capture.py --variant N   # internal (single run)
"""

import argparse
import os
import subprocess
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

VARIANTS = {
    1: "trainer capture, loader batch, benchmark=False (repro)",
    2: "trainer capture, loader batch, benchmark=True",
    3: "trainer capture, synthetic batch (no DataLoader alive), benchmark=True",
    4: "trainer capture, synthetic batch, no test_loader, benchmark=True",
    5: "bare capture of trainer._compute_step outside fit, benchmark=True",
    6: "trainer capture, loader batch, workers=0, benchmark=True",
    7: "manual loop: live loader iterator, SYNTHETIC tensors to _graph_step",
    8: "manual loop: pre-materialized loader batches (iterator dead)",
    9: "manual loop: synthetic tensors, stats dict held across calls",
    10: "manual loop: live iterator batches passed to _graph_step",
}


def run_variant(v: int):
    import torch

    torch.backends.cudnn.benchmark = v != 1
    from npairloss_amd.config.params import SolverConfig
    from npairloss_amd.engine.net_builder import build_trainer_from_prototxt

    root = os.path.join(os.path.dirname(os.path.abspath(__file__)), "..",
                        "examples", "googlenet_npair")
    solver = SolverConfig.from_prototxt(open(os.path.join(root, "sgd_solver.prototxt")).read())
    solver.display = 0
    solver.test_interval = 0
    net_text = open(os.path.join(root, "train_net.prototxt")).read()
    workers = 0 if v == 6 else 4
    tr = build_trainer_from_prototxt(net_text, solver, num_workers=workers,
                                     amp_dtype=torch.bfloat16, hip_graph=True)
    tr.graph_warmup = 6
    if v == 4:
        tr.test_loader = None

    if v in (3, 4):
        import itertools

        x = torch.randn(120, 3, 224, 224)
        lab = torch.arange(60).repeat_interleave(2)
        for i in range(10):
            if tr.use_graph:
                tr._graph_step(x, lab)
            else:
                tr.train_step(x, lab)
        print(f"variant {v}: captured={tr._graph_state is not None} OK")
        return
    if v == 5:
        x = torch.randn(120, 3, 224, 224, device="cuda")
        lab = torch.arange(60).repeat_interleave(2).cuda()
        xs = tr._prepare(x).clone()
        for _ in range(6):
            tr._compute_step(xs, lab)
        torch.cuda.synchronize()
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            tr._compute_step(xs, lab)
        torch.cuda.current_stream().wait_stream(side)
        torch.cuda.synchronize()
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            tr._compute_step(xs, lab)
        g.replay()
        torch.cuda.synchronize()
        print("variant 5: bare capture OK")
        return
    if v in (7, 9, 10):
        import itertools

        x = torch.randn(120, 3, 224, 224)
        lab = torch.arange(60).repeat_interleave(2)
        it = iter(tr.train_loader) if v in (7, 10) else None
        stats = None
        for i in range(10):
            if v == 10:
                try:
                    bx, blab = next(it)
                except StopIteration:
                    it = iter(tr.train_loader)
                    bx, blab = next(it)
            else:
                if it is not None:
                    next(it)  # keep the iterator hot but ignore its batch
                bx, blab = x, lab
            s = tr._graph_step(bx, blab)
            if v == 9:
                stats = s  # held across the capture call
        print(f"variant {v}: captured={tr._graph_state is not None} OK")
        return
    if v == 8:
        import itertools

        batches = list(itertools.islice(iter(tr.train_loader), 10))
        for bx, blab in batches:
            tr._graph_step(bx, blab)
        print(f"variant 8: captured={tr._graph_state is not None} OK")
        return
    # 1, 2, 6: run fit for a few iters (captures at graph_warmup)
    tr.fit(max_iter=10)
    print(f"variant {v}: captured={tr._graph_state is not None} OK")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--variant", type=int, default=None)
    args = ap.parse_args()
    if args.variant is not None:
        run_variant(args.variant)
        return
    for v, desc in VARIANTS.items():
        r = subprocess.run([sys.executable, __file__, "--variant", str(v)],
                           capture_output=True, text=True, timeout=240)
        status = "OK" if r.returncode == 0 else f"FAIL rc={r.returncode}"
        print(f"[{status}] variant {v}: {desc}")
        tail = (r.stdout + r.stderr).strip().splitlines()[-3:]
        for line in tail:
            print("    " + line)


if __name__ == "__main__":
    main()
