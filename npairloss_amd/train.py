"""Training CLI: `python -m npairloss_amd.train --solver solver.prototxt`.

Drives the whole reference workflow from the reference's own config files:
the solver prototxt names the net prototxt (`net:` field), which defines
the data layer, augmentation, backbone, L2Normalize and the
NPairMultiClassLoss layer with its mining config.  Multi-GPU: launch with
torch.distributed.run, one rank per GPU (RCCL).
"""

from __future__ import annotations

import argparse
import os

import torch


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--solver", required=True, help="solver.prototxt path")
    p.add_argument("--net", default=None, help="override net prototxt path")
    p.add_argument("--weights", default=None, help=".caffemodel or .pt to load")
    p.add_argument("--max-iter", type=int, default=None)
    p.add_argument("--synthetic-classes", type=int, default=256)
    p.add_argument("--amp", choices=["off", "bf16", "fp16"], default="bf16")
    p.add_argument("--pure-bf16", action="store_true",
                   help="bf16 model + fp32 master weights (replaces --amp)")
    p.add_argument("--num-workers", type=int, default=2)
    p.add_argument("--backbone", default=None,
                   help="override the backbone (googlenet/resnet50/vit)")
    p.add_argument("--graph", dest="graph", action="store_true", default=None,
                   help="hipGraph whole-step capture+replay (default ON on "
                        "GPU; the bench's headline mode — lazy capture after "
                        "warmup, recapture on LR-policy steps)")
    p.add_argument("--no-graph", dest="graph", action="store_false")
    p.add_argument("--timers", action="store_true",
                   help="per-phase HIP-event timing (eager steps only)")
    p.add_argument("--base-lr", type=float, default=None,
                   help="override solver base_lr (e.g. warmup demos)")
    p.add_argument("--init", choices=["caffe", "modern"], default="caffe",
                   help="weight init: caffe = the reference's exact filler "
                        "(xavier + 0.2 bias; starts collapsed — the reference "
                        "schedules 2M iters), modern = kaiming/zero-bias "
                        "(documented deviation; converges from iter 1)")
    p.add_argument("--display", type=int, default=None,
                   help="override solver display interval")
    p.add_argument("--synthetic-device", action="store_true",
                   help="device-resident synthetic batches (bench-parity "
                        "throughput: skips the CPU DataLoader, which costs "
                        "~60 ms/iter in generation + collate at batch 120)")
    args = p.parse_args(argv)

    from .config.params import SolverConfig
    from .engine.net_builder import build_trainer_from_prototxt

    solver = SolverConfig.from_prototxt(open(args.solver).read())
    if args.base_lr is not None:
        solver.base_lr = args.base_lr
    if args.display is not None:
        solver.display = args.display
    net_path = args.net or solver.net
    if net_path and not os.path.isabs(net_path):
        cand = os.path.join(os.path.dirname(os.path.abspath(args.solver)), net_path)
        net_path = cand if os.path.exists(cand) else net_path
    net_text = open(net_path).read()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world > 1:
        import datetime

        import torch.distributed as dist

        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        tmo = int(os.environ.get("NPAIR_COMM_TIMEOUT_S", "300"))
        dist.init_process_group("nccl" if torch.cuda.is_available() else "gloo",
                                timeout=datetime.timedelta(seconds=tmo))
        if torch.cuda.is_available():
            torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", "0")))

    amp_dtype = None if args.pure_bf16 else {"off": None, "bf16": torch.bfloat16,
                                              "fp16": torch.float16}[args.amp]
    caffemodel = args.weights if (args.weights and args.weights.endswith(".caffemodel")) else None
    use_graph = args.graph if args.graph is not None else torch.cuda.is_available()
    trainer = build_trainer_from_prototxt(
        net_text, solver, synthetic_classes=args.synthetic_classes,
        amp_dtype=amp_dtype, caffemodel=caffemodel,
        num_workers=args.num_workers, pure_bf16=args.pure_bf16,
        backbone=args.backbone, hip_graph=use_graph, timers=args.timers,
        init=args.init)
    if args.synthetic_device:
        from .data.synthetic import DeviceSyntheticBatches

        bs = getattr(trainer.train_loader, "batch_sampler", None)
        P = getattr(bs, "P", 60)
        K = getattr(bs, "K", 2)
        rank = int(os.environ.get("RANK", "0"))
        trainer.train_loader = DeviceSyntheticBatches(
            P, K, image_size=224,
            num_classes=max(args.synthetic_classes, P),
            device=trainer.device, seed=(solver.random_seed or 0) * 131 + rank)
    if args.weights and args.weights.endswith(".pt"):
        trainer.restore(args.weights)
    elif args.weights and args.weights.endswith(".solverstate"):
        # mid-training resume from Caffe artifacts (weights via the
        # learned_net reference next to the .solverstate, momentum + iter
        # from the state blobs)
        trainer.restore_caffe(args.weights)
    trainer.fit(max_iter=args.max_iter)


if __name__ == "__main__":
    main()
