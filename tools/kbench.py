#!/usr/bin/env python3
"""Microbenchmark for every npairloss_amd HIP kernel at the flagship bench
shapes.  HIP-event timing; prints ms + achieved GB/s (memory-bound ops) or
GFLOP/s (GEMMs).  Also the target for rocprofv3 --pmc counter capture
(profiles/): run only these kernels so counters attribute cleanly.

Usage (GPU box): python tools/kbench.py [--iters 50] [--only PATTERN]
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch


def timeit(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    e0 = torch.cuda.Event(enable_timing=True)
    e1 = torch.cuda.Event(enable_timing=True)
    e0.record()
    for _ in range(iters):
        fn()
    e1.record()
    torch.cuda.synchronize()
    return e0.elapsed_time(e1) / iters


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=50)
    ap.add_argument("--only", type=str, default="")
    args = ap.parse_args()

    from npairloss_amd.ops import _backend

    C = _backend.ext()
    dev = "cuda"
    results = []

    def bench(name, fn, bytes_moved=None, flops=None):
        if args.only and args.only not in name:
            return
        ms = timeit(fn, args.iters)
        extra = ""
        if bytes_moved:
            extra = f"{bytes_moved / ms / 1e6:8.0f} GB/s"
        if flops:
            extra = f"{flops / ms / 1e9:8.1f} TFLOP/s"
        results.append((name, ms, extra))
        print(f"{name:42s} {ms:9.3f} ms {extra}")

    B, G, D = 256, 2048, 1024
    F_l = torch.nn.functional.normalize(torch.randn(B, D, device=dev), dim=1)
    F_g = torch.nn.functional.normalize(torch.randn(G, D, device=dev), dim=1)
    lab_l = torch.randint(0, 128, (B,), device=dev, dtype=torch.int32)
    lab_g = torch.randint(0, 128, (G,), device=dev, dtype=torch.int32)
    S = C.sim_gemm_nt(F_l, F_g)
    W = torch.randn(B, G, device=dev)

    gemm_flops = 2.0 * B * G * D
    bench("sim_gemm_nt fp32  256x2048x1024", lambda: C.sim_gemm_nt(F_l, F_g), flops=gemm_flops)
    Fb_l, Fb_g = F_l.bfloat16(), F_g.bfloat16()
    bench("sim_gemm_nt bf16", lambda: C.sim_gemm_nt_bf16(Fb_l, Fb_g), flops=gemm_flops)
    f8l, f8g = C.cast_fp8(F_l), C.cast_fp8(F_g)
    bench("sim_gemm_nt fp8", lambda: C.sim_gemm_nt_fp8(f8l, f8g), flops=gemm_flops)
    bench("gemm_nn (dF_l) 256x1024x2048", lambda: C.gemm_nn(W, F_g), flops=gemm_flops)
    bench("gemm_tn (dF_g) 2048x1024x256", lambda: C.gemm_tn(W, F_l), flops=gemm_flops)

    sbytes = S.numel() * 4
    bench("rowstats 256x2048", lambda: C.rowstats(S, lab_l, lab_g, 0), bytes_moved=sbytes)
    mnw, mxb, mxa = C.rowstats(S, lab_l, lab_g, 0)
    bench("fused_fwd", lambda: C.fused_fwd(S, lab_l, lab_g, 0, mnw, mxb, mxa,
                                           0.0, -0.05, 3, 0), bytes_moved=sbytes)
    li = torch.rand(B, device=dev) + 1
    ls = li + 1
    bench("bwd_weights", lambda: C.bwd_weights(S, lab_l, lab_g, 0, mnw, mxb, mxa,
                                               li, ls, 0.0, -0.05, 3, 0, 1.0 / B),
          bytes_moved=2 * sbytes)
    ks = torch.tensor([1, 5, 10], dtype=torch.int32, device=dev)
    bench("recall_hits k=1,5,10", lambda: C.recall_hits(S, lab_l, lab_g, 0, ks, 10),
          bytes_moved=2 * sbytes)
    bench("local_relative_thr (LDS bitonic)",
          lambda: C.local_relative_thr(S, lab_l, lab_g, 0, True, -0.3), bytes_moved=sbytes)
    bench("global_relative_thr (radix select)",
          lambda: C.global_relative_thr(S, lab_l, lab_g, 0, True, -0.3), bytes_moved=6 * sbytes)
    x = torch.randn(B, D, device=dev)
    bench("l2norm_fwd 256x1024", lambda: C.l2norm_fwd(x), bytes_moved=2 * x.numel() * 4)

    # vision kernels at bench shapes (bf16 NHWC)
    def cl(t):
        return t.to(memory_format=torch.channels_last)

    for name, shape, stride in (("pool1 s2 112^2x64", (256, 64, 112, 112), 2),
                                ("inception pool s1 28^2x512", (256, 512, 28, 28), 1),
                                ("inception pool s1 14^2x528", (256, 528, 14, 14), 1)):
        xx = cl(torch.randn(*shape, device=dev, dtype=torch.bfloat16))
        nb = xx.numel() * 2
        y, idx = C.maxpool3_fwd(xx, stride, True)
        dy = torch.randn_like(y)
        bench(f"maxpool3_fwd {name}", lambda xx=xx, s=stride: C.maxpool3_fwd(xx, s, True),
              bytes_moved=nb + y.numel() * 3)
        bench(f"maxpool3_bwd {name}",
              lambda dy=dy, idx=idx, s=stride, h=shape[2], w=shape[3]: C.maxpool3_bwd(dy, idx, s, h, w),
              bytes_moved=nb + y.numel() * 3)

    # fused conv-unit kernels (round 2)
    for name, (Cc, H) in (("56^2x64", (64, 56)), ("28^2x256", (256, 28)),
                          ("7^2x832", (832, 7))):
        xx = cl(torch.randn(256, Cc, H, H, device=dev, dtype=torch.bfloat16))
        bb = torch.randn(Cc, device=dev)
        nb = xx.numel() * 2
        yy = C.biasrelu_fwd(xx, bb)
        dyy = torch.randn_like(yy)
        bench(f"biasrelu_fwd {name}", lambda xx=xx, bb=bb: C.biasrelu_fwd(xx, bb),
              bytes_moved=2 * nb)
        bench(f"biasrelu_bwd {name}", lambda yy=yy, dyy=dyy: C.biasrelu_bwd(yy, dyy),
              bytes_moved=3 * nb)

    for name, (K, H, N) in (("conv1x1 56^2 K64 N64", (64, 56, 64)),
                            ("conv1x1 28^2 K192 N128", (192, 28, 128)),
                            ("conv1x1 7^2 K832 N384", (832, 7, 384))):
        M = 256 * H * H
        xm = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
        ww = torch.randn(N, K, device=dev, dtype=torch.bfloat16) * 0.05
        bb = torch.randn(N, device=dev)
        fl = 2.0 * M * N * K
        bench(f"{name} fused v2 fwd", lambda xm=xm, ww=ww, bb=bb:
              C.conv1x1_bias_relu_fwd(xm, ww, bb), flops=fl)
        gg = torch.randn(M, N, device=dev, dtype=torch.bfloat16)
        wt = ww.t().contiguous()
        bench(f"{name} dgrad v2", lambda gg=gg, wt=wt: C.conv1x1_dgrad(gg, wt), flops=fl)

    for name, shape in (("norm1 112^2x64", (256, 64, 112, 112)),
                        ("norm2 56^2x192", (256, 192, 56, 56))):
        xx = cl(torch.randn(*shape, device=dev, dtype=torch.bfloat16))
        dy = torch.randn_like(xx)
        nb = xx.numel() * 2
        bench(f"lrn_fwd {name}", lambda xx=xx: C.lrn_fwd(xx, 5, 1e-4, 0.75, 1.0),
              bytes_moved=2 * nb)
        bench(f"lrn_bwd {name}", lambda xx=xx, dy=dy: C.lrn_bwd(xx, dy, 5, 1e-4, 0.75, 1.0),
              bytes_moved=3 * nb)

    print(f"\n{len(results)} kernels benchmarked")


if __name__ == "__main__":
    main()
