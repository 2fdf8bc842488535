#!/usr/bin/env python3
"""Flagship benchmark: GoogLeNet N-pair metric-learning training step.

Measures the BASELINE.json headline metric — images/sec (whole node) for
GoogLeNet 1024-d N-pair training at batch 256/GPU with the production
mining config (GLOBAL RELATIVE_HARD ap + LOCAL HARD an), synthetic data,
random-init weights, bf16 autocast backbone + fp32 loss path.

Single GPU:   python bench.py --gpus 1 --steps 30 --warmup 5
Multi GPU:    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
                  --master-addr 127.0.0.1 bench.py --gpus N ...
(one rank per GPU over RCCL; reads RANK/LOCAL_RANK/WORLD_SIZE/MASTER_*)

Rank 0 prints ONE JSON line with the whole-job aggregate.
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import torch


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--batch", type=int, default=256, help="per-GPU batch")
    p.add_argument("--model", type=str, default="googlenet")
    p.add_argument("--image", type=int, default=224)
    p.add_argument("--ids-per-batch", type=int, default=None,
                   help="P identities per batch (default batch//2)")
    p.add_argument("--no-amp", action="store_true")
    p.add_argument("--pure-bf16", action="store_true",
                   help="cast the model to bf16 with fp32 master weights in the "
                        "optimizer (no per-step autocast weight casts)")
    p.add_argument("--mining", choices=["production", "hard", "all"], default="production",
                   help="production = GLOBAL RELATIVE_HARD ap + LOCAL HARD an (def.prototxt); "
                        "hard = semi-hard negatives only; all = no mining (RAND)")
    p.add_argument("--sim-dtype", choices=["fp32", "bf16", "fp8", "fp64"], default="fp32",
                   help="similarity-GEMM precision on GPU (MFMA for "
                        "fp32/bf16/fp8; rocBLAS DGEMM + templated fp64 kernels)")
    p.add_argument("--timers", action="store_true",
                   help="per-phase HIP-event timing report on stderr (rank 0)")
    p.add_argument("--graph", dest="graph", action="store_true", default=None,
                   help="capture the whole train step in a hipGraph and replay "
                        "(default ON on GPU, any world size — RCCL collectives "
                        "are captured too; falls back to eager if capture fails)")
    p.add_argument("--no-graph", dest="graph", action="store_false")
    p.add_argument("--data-pipeline", action="store_true",
                   help="include the input pipeline in the timed region: rotate "
                        "8 distinct pinned-host uint8 batches through H2D copy + "
                        "device-side DataTransformer augmentation every step "
                        "(graph mode: copy_ into captured static tensors)")
    return p.parse_args()


def main():
    args = parse_args()
    env_ws = int(os.environ.get("WORLD_SIZE", "1"))
    world = max(env_ws, 1)
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    use_cuda = torch.cuda.is_available()
    # modulo lets a multi-rank smoke run on fewer GPUs than ranks (e.g. the
    # 2-rank RCCL path check on a 1-GPU box); production launches have
    # local_rank < device_count so this is the identity there
    dev_idx = local_rank % max(1, torch.cuda.device_count()) if use_cuda else 0
    device = torch.device(f"cuda:{dev_idx}" if use_cuda else "cpu")
    if use_cuda:
        torch.cuda.set_device(device)
        torch.backends.cudnn.benchmark = True  # MIOpen find-best for fixed shapes

    if world > 1:
        import datetime

        import torch.distributed as dist

        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29517")
        # failure detection: a hung peer fails the collective after this
        # timeout instead of deadlocking the job (reference: MPI = job hang)
        tmo = int(os.environ.get("NPAIR_COMM_TIMEOUT_S", "300"))
        dist.init_process_group("nccl" if use_cuda else "gloo",
                                rank=rank, world_size=world,
                                timeout=datetime.timedelta(seconds=tmo))

    from npairloss_amd.config.params import NPairLossConfig
    from npairloss_amd.models import build_embedding_model
    from npairloss_amd.ops.npair_loss import NPairMultiClassLoss
    from npairloss_amd.engine.solver import CaffeSGD
    from npairloss_amd.parallel.ddp import BucketedGradReducer

    torch.manual_seed(1234 + rank)
    B = args.batch
    P = args.ids_per_batch or max(2, B // 2)
    K = B // P

    model = build_embedding_model(args.model).to(device)
    model = model.to(memory_format=torch.channels_last)
    if args.pure_bf16:
        model = model.to(torch.bfloat16)
    model.train()
    if args.mining == "production":
        cfg = NPairLossConfig(
            margin_ident=0.0, margin_diff=-0.05, identsn=-0.0, diffsn=-0.3,
            ap_mining_region="GLOBAL", ap_mining_method="RELATIVE_HARD",
            an_mining_region="LOCAL", an_mining_method="HARD")
    elif args.mining == "hard":
        cfg = NPairLossConfig(
            margin_diff=-0.05, ap_mining_method="RAND",
            an_mining_region="LOCAL", an_mining_method="HARD")
    else:
        cfg = NPairLossConfig()  # RAND/RAND: every pair
    loss_mod = NPairMultiClassLoss(cfg, sim_dtype=args.sim_dtype)
    opt = CaffeSGD(model.parameters(), lr=0.001, momentum=0.9, weight_decay=2e-5,
                   master_weights=args.pure_bf16)
    reducer = BucketedGradReducer(model)
    reducer.broadcast_params()

    amp = use_cuda and not args.no_amp and not args.pure_bf16

    # synthetic data: two alternating P x K-labelled image batches on device
    def make_batch(seed):
        g = torch.Generator(device="cpu").manual_seed(seed)
        x = torch.randn(B, 3, args.image, args.image, generator=g)
        lab = torch.arange(P).repeat_interleave(K)[:B]
        perm = torch.randperm(B, generator=g)
        return (x.to(device).to(memory_format=torch.channels_last),
                lab[perm].to(device))
    batches = [make_batch(1000 + rank * 10 + i) for i in range(2)]

    # --data-pipeline: N distinct uint8 host batches (pinned), H2D-copied into
    # static device tensors + augmented on device INSIDE the timed region.
    pipe = None
    if args.data_pipeline:
        from npairloss_amd.data.transforms import DataTransformer, TransformConfig

        NPIPE = 8
        g = torch.Generator(device="cpu").manual_seed(4321 + rank)
        host_x, host_lab = [], []
        for i in range(NPIPE):
            x = torch.randint(0, 256, (B, 3, args.image, args.image),
                              dtype=torch.uint8, generator=g)
            x = x.to(memory_format=torch.channels_last)
            lab = torch.arange(P).repeat_interleave(K)[:B]
            lab = lab[torch.randperm(B, generator=g)]
            if use_cuda:
                x, lab = x.pin_memory(), lab.pin_memory()
            host_x.append(x)
            host_lab.append(lab)
        static_u8 = host_x[0].to(device, non_blocking=True)
        static_lab = host_lab[0].to(device, non_blocking=True)
        static_x = torch.empty(B, 3, args.image, args.image, device=device)
        static_x = static_x.to(memory_format=torch.channels_last)
        aug = DataTransformer(TransformConfig(
            rotate_angle_scope=0.18, translation_w_scope=8.0,
            translation_h_scope=8.0, scale_w_scope=1.1, scale_h_scope=1.1,
            h_flip=True))
        pipe = dict(host_x=host_x, host_lab=host_lab, static_u8=static_u8,
                    static_lab=static_lab, static_x=static_x, aug=aug, n=NPIPE)

    last_out = {}
    from npairloss_amd.utils.profiling import PhaseTimers
    timers = PhaseTimers(enabled=args.timers, use_cuda=use_cuda)

    def feed(i):
        """Input pipeline work that stays OUTSIDE the captured graph (still
        inside the timed region): H2D copy of batch i + decode + device-side
        random affine augmentation into the static input tensor."""
        if pipe is not None:
            j = i % pipe["n"]
            pipe["static_u8"].copy_(pipe["host_x"][j], non_blocking=True)
            pipe["static_lab"].copy_(pipe["host_lab"][j], non_blocking=True)
            x = pipe["static_u8"].float().sub_(127.5).mul_(1.0 / 64.0)
            x = pipe["aug"](x).to(memory_format=torch.channels_last)
            pipe["static_x"].copy_(x)

    def step(i):
        nonlocal last_out
        if pipe is not None:
            x, lab = pipe["static_x"], pipe["static_lab"]
        else:
            x, lab = batches[i % 2]
        if args.pure_bf16:
            x = x.to(torch.bfloat16)
        reducer.zero_grad()
        with timers.phase("forward"):
            if amp:
                with torch.autocast("cuda", dtype=torch.bfloat16):
                    feats = model(x)
            else:
                feats = model(x)
        with timers.phase("loss"):
            out = loss_mod(feats.float(), lab)
        with timers.phase("backward"):
            out.loss.backward()
        with timers.phase("comm"):
            reducer.finalize()
        with timers.phase("optimizer"):
            opt.step()
        last_out = {"loss": out.loss.detach(), "top1": out.retrieve_top1.detach()}

    def barrier_sync():
        if world > 1:
            import torch.distributed as dist
            dist.barrier()
        if use_cuda:
            torch.cuda.synchronize()

    graph = None
    use_graph = args.graph if args.graph is not None else use_cuda
    if use_graph and not use_cuda:
        if rank == 0 and args.graph:
            print("--graph requires CUDA; ignoring", file=sys.stderr)
        use_graph = False
    if use_graph:
        # hipGraph capture of the full step (fwd + loss + bwd + comm +
        # optimizer): the launch-bound inner loop becomes one graph replay.
        # Inputs stay at fixed addresses (batches[0] / the static pipeline
        # tensors); gradients live in the reducer's persistent flat buckets,
        # so the RCCL all-reduces capture with fixed buffers at world>1.
        # MIOpen find must have run during warmup (cudnn.benchmark per shape).
        for i in range(max(args.warmup, 3)):
            feed(i)
            step(0)
        torch.cuda.synchronize()
        if world > 1:
            import torch.distributed as dist
            dist.barrier()  # align ranks so every rank captures the same comm
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            step(0)  # warm the allocator on the capture stream
        torch.cuda.current_stream().wait_stream(side)
        torch.cuda.synchronize()
        timers.enabled = False  # timing events cannot record during capture
        try:
            graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(graph):
                step(0)
        except Exception as e:  # noqa: BLE001
            graph = None
            print(f"hipGraph capture failed ({e!r}); running eager", file=sys.stderr)
        if world > 1:
            # consensus: replaying on some ranks while others run eager
            # keeps the collective SEQUENCE identical, but a rank whose
            # capture failed mid-way may have desynced RCCL state — if any
            # rank failed, every rank falls back to eager
            import torch.distributed as dist
            ok = torch.tensor([0 if graph is None else 1],
                              device=device if use_cuda else "cpu")
            dist.all_reduce(ok, op=dist.ReduceOp.MIN)
            if int(ok.item()) == 0 and graph is not None:
                graph = None
                print("peer rank failed hipGraph capture; running eager",
                      file=sys.stderr)
            dist.barrier()

    def run_step(i):
        feed(i)
        if graph is not None:
            graph.replay()
        else:
            step(i)

    for i in range(args.warmup):
        run_step(i)
    timers.report(reset=True)  # drop warmup phases
    barrier_sync()
    t0 = time.time()
    for i in range(args.steps):
        run_step(args.warmup + i)
    barrier_sync()
    elapsed = time.time() - t0

    # MAX elapsed over ranks
    if world > 1:
        import torch.distributed as dist
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if use_cuda else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    images_per_sec = world * B * args.steps / elapsed
    if args.timers and rank == 0:
        ph = timers.report()
        per_step = {k: v / args.steps for k, v in sorted(ph.items(), key=lambda kv: -kv[1])}
        print("phases ms/step: " + json.dumps({k: round(v, 2) for k, v in per_step.items()}),
              file=sys.stderr)
    if rank == 0:
        result = {
            "metric": "images/sec (whole node), GoogLeNet N-pair batch=256/GPU",
            "value": images_per_sec,
            "unit": "images/sec",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000.0,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if (amp or args.pure_bf16) else "fp32",
            "data": "synthetic+pipeline" if pipe is not None else "synthetic",
            "config": {
                "model": args.model,
                "global_batch": world * B,
                "batch_per_gpu": B,
                "image": args.image,
                "embed_dim": 1024 if args.model == "googlenet" else None,
                "mining": args.mining,
                "sim_dtype": args.sim_dtype,
                "parallelism": f"dp{world}",
                "hip_graph": graph is not None,
                "data_pipeline": pipe is not None,
                # 288 GB HBM3E sizing evidence (BASELINE config 5)
                "peak_mem_gb": round(torch.cuda.max_memory_allocated() / 2**30, 2)
                               if use_cuda else None,
                "recall_top1_last_step": float(last_out["top1"]) if last_out else None,
            },
        }
        print(json.dumps(result))

    if world > 1:
        import torch.distributed as dist
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
