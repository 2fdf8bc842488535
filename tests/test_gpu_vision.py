"""GPU numerics tests for the fused vision kernels (LRN, 3x3 max pool)
against the plain PyTorch fp32 references."""

import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

from npairloss_amd.ops.vision import CrossChannelLRN, MaxPool3x3


def _maybe_cl(x, cl):
    return x.to(memory_format=torch.channels_last) if cl else x


@pytest.mark.parametrize("cl", [False, True])
@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("shape", [(2, 64, 16, 16), (3, 192, 28, 28), (2, 5, 7, 9)])
def test_lrn_fwd_bwd(cl, dtype, shape):
    x = torch.randn(*shape, device="cuda", dtype=dtype) * 2
    x = _maybe_cl(x, cl).requires_grad_(True)
    mod = CrossChannelLRN(5, alpha=1e-4, beta=0.75)
    y = mod(x)
    dy = torch.randn_like(y)
    y.backward(dy)

    xr = x.detach().float().clone().requires_grad_(True)
    yr = F.local_response_norm(xr, 5, alpha=1e-4, beta=0.75, k=1.0)
    yr.backward(dy.float())

    tol = dict(rtol=1e-4, atol=1e-5) if dtype == torch.float32 else dict(rtol=2e-2, atol=2e-2)
    torch.testing.assert_close(y.float(), yr.detach(), **tol)
    torch.testing.assert_close(x.grad.float(), xr.grad, **tol)


@pytest.mark.parametrize("cl", [False, True])
@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("stride", [1, 2])
@pytest.mark.parametrize("shape", [(2, 64, 56, 56), (2, 32, 15, 17)])
def test_maxpool3_fwd_bwd(cl, dtype, stride, shape):
    x = torch.randn(*shape, device="cuda", dtype=dtype)
    x = _maybe_cl(x, cl).requires_grad_(True)
    mod = MaxPool3x3(stride=stride)
    y = mod(x)
    dy = torch.randn_like(y)
    y.backward(dy)

    xr = x.detach().float().clone().requires_grad_(True)
    yr = F.max_pool2d(xr, 3, stride=stride, padding=1, ceil_mode=True)
    yr.backward(dy.float())

    assert y.shape == yr.shape
    tol = dict(rtol=1e-6, atol=1e-6) if dtype == torch.float32 else dict(rtol=1e-2, atol=1e-2)
    torch.testing.assert_close(y.float(), yr.detach(), **tol)
    # backward: ties can route gradient to a different (equally max) element;
    # random floats make ties measure-zero
    torch.testing.assert_close(x.grad.float(), xr.grad, **tol)


def test_googlenet_gpu_matches_cpu():
    """Whole backbone on GPU (fused kernels) vs CPU (torch ops), fp32."""
    from npairloss_amd.models import GoogLeNet

    torch.manual_seed(0)
    m = GoogLeNet(dropout=0.0)
    x = torch.randn(2, 3, 224, 224)
    m.eval()
    y_cpu = m(x)
    mg = m.cuda()
    y_gpu = mg(x.cuda())
    torch.testing.assert_close(y_gpu.cpu(), y_cpu, rtol=1e-3, atol=1e-4)


def test_fp16_routes_through_fp32():
    """half inputs upcast transparently (kernels store bf16/fp32)."""
    x = torch.randn(2, 64, 16, 16, device="cuda", dtype=torch.float16,
                    requires_grad=True)
    y = CrossChannelLRN(5)(x)
    assert y.dtype == torch.float16
    y.sum().backward()
    assert x.grad.dtype == torch.float16 and torch.isfinite(x.grad).all()
    x2 = torch.randn(2, 32, 14, 14, device="cuda", dtype=torch.float16,
                     requires_grad=True)
    y2 = MaxPool3x3(stride=2)(x2)
    assert y2.dtype == torch.float16
    y2.sum().backward()
    assert torch.isfinite(x2.grad).all()
