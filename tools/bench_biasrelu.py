#!/usr/bin/env python3
"""Microbenchmark the fused BiasReLU kernels at GoogLeNet bench shapes
(B=256, bf16, channels_last) vs the torch ops they replace
(threshold_backward + bias-grad reduce).

Usage (GPU box): python tools/bench_biasrelu.py
"""

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch


def timeit(fn, iters=30, warmup=8):
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

    C_ext = _backend.ext()
    B = 256
    # (C, H) for the heavy GoogLeNet activation shapes
    shapes = [(64, 112), (192, 56), (256, 28), (480, 28), (512, 14),
              (528, 14), (832, 7), (1024, 7)]
    print(f"{'shape':>18} {'fwd ms':>8} {'fwd GB/s':>9} {'bwd ms':>8} {'bwd GB/s':>9} "
          f"{'torch relu_bwd+bias ms':>22}")
    tot_f = tot_b = tot_t = 0.0
    for C, H in shapes:
        x = torch.randn(B, C, H, H, device="cuda", dtype=torch.bfloat16)
        x = x.to(memory_format=torch.channels_last)
        b = torch.randn(C, device="cuda", dtype=torch.float32)
        y = C_ext.biasrelu_fwd(x, b)
        dy = torch.randn_like(y)
        t_f = timeit(lambda: C_ext.biasrelu_fwd(x, b))
        t_b = timeit(lambda: C_ext.biasrelu_bwd(y, dy))
        # torch equivalent backward: dReLU then bias reduce over dx
        def torch_bwd():
            dx = torch.ops.aten.threshold_backward(dy, y, 0)
            dx.sum(dim=(0, 2, 3), dtype=torch.float32)
        t_t = timeit(torch_bwd)
        nbytes = x.numel() * x.element_size()
        gbs_f = 2 * nbytes / t_f / 1e6
        gbs_b = 3 * nbytes / t_b / 1e6
        print(f"{B}x{C}x{H}x{H:>4} {t_f:8.3f} {gbs_f:9.0f} {t_b:8.3f} {gbs_b:9.0f} {t_t:22.3f}")
        tot_f += t_f
        tot_b += t_b
        tot_t += t_t
    print(f"{'TOTAL':>18} {tot_f:8.3f} {'':>9} {tot_b:8.3f} {'':>9} {tot_t:22.3f}")


if __name__ == "__main__":
    main()
