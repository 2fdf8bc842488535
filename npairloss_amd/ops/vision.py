"""Vision ops backed by the fused gfx950 kernels (GPU) or torch (CPU).

CrossChannelLRN and MaxPool3x3 replace torch's eager LRN chain and
atomic-based max-pool backward in the GoogLeNet path — rocprof showed
those two dominating the flagship bench step (profiles/).
"""

from __future__ import annotations

import torch
from torch import nn
import torch.nn.functional as F

from . import _backend


def _gpu_safe_dtype(x):
    """The HIP kernels store bf16/fp32; route half through fp32."""
    if x.is_cuda and x.dtype == torch.float16:
        return x.float(), torch.float16
    return x, None


class _LRNFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, size, alpha, beta, k):
        ctx.params = (size, alpha, beta, k)
        if x.is_cuda:
            xc, half = _gpu_safe_dtype(x)
            y = _backend.ext().lrn_fwd(xc, size, alpha, beta, k)
            ctx.save_for_backward(xc)
            return y.half() if half else y
        ctx.save_for_backward(x)
        # CPU avg_pool3d lacks a bf16 kernel: compute in fp32, cast back
        return F.local_response_norm(x.float(), size, alpha=alpha, beta=beta, k=k).to(x.dtype)

    @staticmethod
    def backward(ctx, dy):
        (x,) = ctx.saved_tensors
        size, alpha, beta, k = ctx.params
        if x.is_cuda:
            dyc, half = _gpu_safe_dtype(dy)
            dx = _backend.ext().lrn_bwd(x, dyc, size, alpha, beta, k)
            if half:
                dx = dx.half()
        else:
            with torch.enable_grad():
                xr = x.detach().float().requires_grad_(True)
                y = F.local_response_norm(xr, size, alpha=alpha, beta=beta, k=k)
                (dx,) = torch.autograd.grad(y, xr, dy.float())
                dx = dx.to(x.dtype)
        return dx, None, None, None, None


class CrossChannelLRN(nn.Module):
    """Caffe across-channel LRN: y = x * (k + alpha/n * sum_win x^2)^-beta."""

    def __init__(self, size: int = 5, alpha: float = 1e-4, beta: float = 0.75,
                 k: float = 1.0):
        super().__init__()
        self.size, self.alpha, self.beta, self.k = size, alpha, beta, k

    def forward(self, x):
        return _LRNFn.apply(x, self.size, self.alpha, self.beta, self.k)

    def extra_repr(self):
        return f"size={self.size}, alpha={self.alpha}, beta={self.beta}, k={self.k}"


class _MaxPool3Fn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, stride, ceil_mode):
        if x.is_cuda:
            xc, half = _gpu_safe_dtype(x)
            y, idx = _backend.ext().maxpool3_fwd(xc, stride, ceil_mode)
            ctx.save_for_backward(idx)
            ctx.meta = (stride, x.shape[2], x.shape[3], True)
            return y.half() if half else y
        ctx.save_for_backward(x)
        ctx.meta = (stride, x.shape[2], x.shape[3], False)
        ctx.ceil_mode = ceil_mode
        return F.max_pool2d(x, 3, stride=stride, padding=1, ceil_mode=ceil_mode)

    @staticmethod
    def backward(ctx, dy):
        (saved,) = ctx.saved_tensors
        stride, H, W, gpu = ctx.meta
        if gpu:
            dyc, half = _gpu_safe_dtype(dy)
            dx = _backend.ext().maxpool3_bwd(dyc, saved, stride, H, W)
            if half:
                dx = dx.half()
        else:
            with torch.enable_grad():
                xr = saved.detach().requires_grad_(True)
                y = F.max_pool2d(xr, 3, stride=stride, padding=1, ceil_mode=ctx.ceil_mode)
                (dx,) = torch.autograd.grad(y, xr, dy)
        return dx, None, None


class MaxPool3x3(nn.Module):
    """3x3 max pool, padding 1, stride 1 or 2, Caffe ceil_mode."""

    def __init__(self, stride: int = 1, ceil_mode: bool = True):
        super().__init__()
        assert stride in (1, 2)
        self.stride = stride
        self.ceil_mode = ceil_mode

    def forward(self, x):
        return _MaxPool3Fn.apply(x, self.stride, self.ceil_mode)

    def extra_repr(self):
        return f"stride={self.stride}, ceil_mode={self.ceil_mode}"


class _BiasReLUFn(torch.autograd.Function):
    """Fused y = relu(x + b[c]) with a single-pass backward producing both
    dx and the bias gradient (csrc/biasrelu.hip — vectorized deterministic
    NHWC kernels).  CPU path is the exact same math in torch."""

    @staticmethod
    def forward(ctx, x, bias):
        ctx.bias_dtype = bias.dtype
        if x.is_cuda:
            xc, half = _gpu_safe_dtype(x)
            y = _backend.ext().biasrelu_fwd(xc, bias.float().contiguous())
            ctx.save_for_backward(y)
            ctx.gpu = True
            ctx.half = half
            return y.half() if half else y
        y = torch.relu(x + bias.view(1, -1, 1, 1).to(x.dtype))
        ctx.save_for_backward(y)
        ctx.gpu = False
        return y

    @staticmethod
    def backward(ctx, dy):
        (y,) = ctx.saved_tensors
        if ctx.gpu:
            dyc, half = _gpu_safe_dtype(dy)
            dx, db = _backend.ext().biasrelu_bwd(y, dyc)
            if half:
                dx = dx.half()
        else:
            mask = y > 0
            dx = dy * mask
            db = dx.float().sum(dim=(0, 2, 3))
        return dx, db.to(ctx.bias_dtype)


class ConvBiasReLU(nn.Module):
    """conv2d(bias=False) + fused BiasReLU.  Exposes `.weight`/`.bias` so
    caffe_names() checkpoint mapping keeps working."""

    def __init__(self, cin, cout, k, stride=1, pad=0):
        super().__init__()
        self.conv = nn.Conv2d(cin, cout, k, stride=stride, padding=pad, bias=False)
        self.bias = nn.Parameter(torch.zeros(cout))

    @property
    def weight(self):
        return self.conv.weight

    def forward(self, x):
        return _BiasReLUFn.apply(self.conv(x), self.bias)


class _Conv1x1BiasReLUFn(torch.autograd.Function):
    """1x1 conv + bias + ReLU as ONE fused MFMA GEMM pass (csrc/conv1x1.hip).

    In NHWC the conv is y[M,N] = x[M,K] @ w[N,K]^T (M = B*H*W): the kernel
    adds bias and applies ReLU in the epilogue, so the layer forward makes a
    single pass over x with no intermediate conv output in HBM.  Backward:
    one fused dReLU+bias-grad pass (biasrelu_bwd), the data gradient on the
    same MFMA GEMM, and the weight gradient as a plain hipBLASLt TN GEMM
    (torch.matmul — a library GEMM, not a fusion target)."""

    @staticmethod
    @torch.amp.custom_fwd(device_type="cuda", cast_inputs=torch.bfloat16)
    def forward(ctx, x, w, bias):
        ctx.dtypes = (x.dtype, w.dtype, bias.dtype)
        use_hip = (x.is_cuda and x.dtype == torch.bfloat16
                   and x.is_contiguous(memory_format=torch.channels_last))
        if use_hip:
            C = _backend.ext()
            B, K, H, W = x.shape
            N = w.shape[0]
            xm = x.permute(0, 2, 3, 1).reshape(B * H * W, K)  # free view (NHWC)
            w2 = w.reshape(N, K)
            if w2.dtype != torch.bfloat16:
                w2 = w2.to(torch.bfloat16)
            w2 = w2.contiguous()
            y2 = C.conv1x1_bias_relu_fwd(xm, w2, bias)
            y = y2.view(B, H, W, N).permute(0, 3, 1, 2)  # channels_last view
            ctx.save_for_backward(x, w2, y)
            ctx.hip = True
            return y
        y = torch.relu(F.conv2d(x, w, bias.to(x.dtype)))
        ctx.save_for_backward(x, w, y)
        ctx.hip = False
        return y

    @staticmethod
    @torch.amp.custom_bwd(device_type="cuda")
    def backward(ctx, dy):
        x, w, y = ctx.saved_tensors
        x_dt, w_dt, b_dt = ctx.dtypes
        if ctx.hip:
            C = _backend.ext()
            B, K, H, W = x.shape
            N = w.shape[0]
            g4, db = C.biasrelu_bwd(y, dy)  # one pass: dReLU mask + bias grad
            gm = g4.permute(0, 2, 3, 1).reshape(B * H * W, N)
            wt = w.t().contiguous()  # tiny K x N copy
            dx = C.conv1x1_dgrad(gm, wt).view(B, H, W, K).permute(0, 3, 1, 2)
            # wgrad: MIOpen wgrad-only (split-K tuned; an unsplit library TN
            # GEMM loses 10x on the M-site reduction — profiles/kb_conv1x1)
            w4 = w.view(N, K, 1, 1)
            dw = torch.ops.aten.convolution_backward(
                g4, x, w4, None, (1, 1), (0, 0), (1, 1), False, (0, 0), 1,
                [False, True, False])[1].view(N, K, 1, 1)
        else:
            mask = y > 0
            g = dy * mask
            db = g.float().sum(dim=(0, 2, 3))
            dx = F.conv_transpose2d(g, w)
            dw = torch.nn.grad.conv2d_weight(x, w.shape, g)
        return dx.to(x_dt), dw.to(w_dt), db.to(b_dt)


class _Conv1x1MM(torch.autograd.Function):
    """Bias-free 1x1 conv with a mixed backward.  Forward stays on MIOpen
    (measured fastest overall at the bench shapes — profiles/kb_conv1x1*);
    backward: the data gradient runs our v2 MFMA GEMM
    dx[M,K] = g[M,N] @ (w^T)[K,N]^T (beats hipBLASLt at EVERY bench shape,
    profiles/kb_conv1x1_v2.log), the weight gradient stays on MIOpen's
    wgrad-only convolution_backward (an unsplit library TN GEMM loses 10x
    on the M=800k reduction)."""

    @staticmethod
    @torch.amp.custom_fwd(device_type="cuda", cast_inputs=torch.bfloat16)
    def forward(ctx, x, w):
        ctx.dtypes = (x.dtype, w.dtype)
        y = F.conv2d(x, w)
        ctx.save_for_backward(x, w)
        return y

    @staticmethod
    @torch.amp.custom_bwd(device_type="cuda")
    def backward(ctx, g):
        x, w = ctx.saved_tensors
        x_dt, w_dt = ctx.dtypes
        if (x.is_cuda and x.dtype == torch.bfloat16
                and x.is_contiguous(memory_format=torch.channels_last)
                and g.is_contiguous(memory_format=torch.channels_last)):
            B, K, H, W = x.shape
            N = w.shape[0]
            gm = g.permute(0, 2, 3, 1).reshape(-1, N)
            w2 = w.reshape(N, K)
            wt = w2.t().contiguous()  # tiny K x N copy
            dx = (_backend.ext().conv1x1_dgrad(gm, wt)
                  .view(B, H, W, K).permute(0, 3, 1, 2))
            dw = torch.ops.aten.convolution_backward(
                g, x, w, None, (1, 1), (0, 0), (1, 1), False, (0, 0), 1,
                [False, True, False])[1]
        else:
            dx = F.conv_transpose2d(g, w)
            dw = torch.nn.grad.conv2d_weight(x, w.shape, g)
        return dx.to(x_dt), dw.to(w_dt)


class Conv1x1BiasReLU(nn.Module):
    """1x1 conv + bias + ReLU.  GPU strategies (NPAIR_CONV1X1 env):
      off (default) — plain conv2d + fused BiasReLU (MIOpen both ways;
                      measured fastest overall this round)
      hybrid        — MIOpen fwd + fused BiasReLU; backward = fused
                      dReLU+bias-grad + our v2 MFMA dgrad + MIOpen wgrad
      custom        — our MFMA GEMM with bias+relu epilogue forward, same
                      backward chain (csrc/conv1x1.hip v2, 115-245 TF)
      auto          — per-shape: custom where the fused v2 forward measured
                      faster than MIOpen+BiasReLU (large M), hybrid below
                      (profiles/kb_conv1x1_v2.log; round-3 default candidate)
    Exposes `.weight`/`.bias` for caffe_names() checkpoint IO."""

    # fused v2 fwd wins at the 56^2 shape (M=802816) and ties ~28^2;
    # MIOpen keeps the 14^2/7^2 tiles (kb_conv1x1_v2.log)
    AUTO_MIN_SITES = 28 * 28 * 256

    def __init__(self, cin, cout):
        super().__init__()
        self.conv = nn.Conv2d(cin, cout, 1, bias=False)
        self.bias = nn.Parameter(torch.zeros(cout))
        import os
        self.mode = os.environ.get("NPAIR_CONV1X1", "off")

    @property
    def weight(self):
        return self.conv.weight

    def forward(self, x):
        mode = self.mode
        if mode == "auto":
            sites = x.shape[0] * x.shape[2] * x.shape[3] if x.dim() == 4 else 0
            mode = "custom" if sites > self.AUTO_MIN_SITES else "hybrid"
        if mode == "custom":
            return _Conv1x1BiasReLUFn.apply(x, self.conv.weight, self.bias)
        if mode == "hybrid":
            return _BiasReLUFn.apply(_Conv1x1MM.apply(x, self.conv.weight), self.bias)
        return _BiasReLUFn.apply(self.conv(x), self.bias)
