#!/usr/bin/env python3
"""Microbenchmark: fused 1x1-conv MFMA GEMM vs MIOpen conv2d and hipBLASLt
matmul at the GoogLeNet bench shapes (B=256, bf16, channels_last).

Usage (GPU box): python tools/bench_conv1x1.py
"""

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch
import torch.nn.functional as F


def timeit(fn, iters=30, warmup=10):
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
    from npairloss_amd.ops import _backend

    C = _backend.ext()
    torch.backends.cudnn.benchmark = True
    B = 256
    # (K=Cin, H, N=Cout): the dominant GoogLeNet 1x1 shapes
    shapes = [(64, 56, 64), (192, 28, 128), (256, 28, 128), (480, 14, 192),
              (512, 14, 160), (528, 14, 256), (832, 7, 384), (1024, 7, 256)]
    hdr = (f"{'K,H,N':>14} {'fused ms':>9} {'TF':>6} {'conv+br ms':>11} "
           f"{'matmul+br ms':>13} {'dgrad ms':>9} {'blaslt ms':>10} "
           f"{'mio_bwd':>10} {'blt_bwd':>10}")
    print(hdr)
    tot = [0.0] * 7
    for K, H, N in shapes:
        M = B * H * H
        x = torch.randn(B, K, H, H, device="cuda", dtype=torch.bfloat16)
        x = x.to(memory_format=torch.channels_last)
        xm = x.permute(0, 2, 3, 1).reshape(M, K)
        w4 = torch.randn(N, K, 1, 1, device="cuda", dtype=torch.bfloat16) * 0.05
        w = w4.reshape(N, K).contiguous()
        b = torch.randn(N, device="cuda")
        bb = b.to(torch.bfloat16)

        t_fused = timeit(lambda: C.conv1x1_bias_relu_fwd(xm, w, b))
        t_conv = timeit(lambda: C.biasrelu_fwd(
            F.conv2d(x, w4), b))
        t_mm = timeit(lambda: C.biasrelu_fwd(
            (xm @ w.t()).view(B, H, H, N).permute(0, 3, 1, 2), b))
        g = torch.randn(M, N, device="cuda", dtype=torch.bfloat16)
        g4 = g.view(B, H, H, N).permute(0, 3, 1, 2)
        wt = w.t().contiguous()
        t_dg = timeit(lambda: C.conv1x1_dgrad(g, wt))
        t_dgb = timeit(lambda: g @ wt.t())
        # full backward: MIOpen convolution_backward vs two hipBLASLt GEMMs
        t_mio_bwd = timeit(lambda: torch.ops.aten.convolution_backward(
            g4, x, w4, None, (1, 1), (0, 0), (1, 1), False, (0, 0), 1,
            [True, True, False]))
        t_blt_bwd = timeit(lambda: (g @ w, g.t() @ xm))
        tf = 2.0 * M * N * K / t_fused / 1e9
        print(f"{K:>4},{H:>3},{N:>4} {t_fused:9.3f} {tf:6.1f} {t_conv:11.3f} "
              f"{t_mm:13.3f} {t_dg:9.3f} {t_dgb:10.3f} {t_mio_bwd:10.3f} {t_blt_bwd:10.3f}")
        for i, v in enumerate((t_fused, t_conv, t_mm, t_dg, t_dgb, t_mio_bwd, t_blt_bwd)):
            tot[i] += v
    print(f"{'TOTAL':>14} {tot[0]:9.3f} {'':6} {tot[1]:11.3f} {tot[2]:13.3f} "
          f"{tot[3]:9.3f} {tot[4]:10.3f} {tot[5]:10.3f} {tot[6]:10.3f}")


if __name__ == "__main__":
    main()
