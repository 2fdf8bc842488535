"""GPU tests for the fused BiasReLU kernels (csrc/biasrelu.hip) — the
DEFAULT conv+bias+relu path in GoogLeNet since round 2 (bias kept out of
the conv; backward fuses dReLU + bias-grad into one pass).

Covers both code paths: the vectorized fixed-channel NHWC kernel
(C % (16/sizeof(T)) == 0) and the scalar fallback (odd C or NCHW), plus
bitwise determinism of the bias gradient (ordered reduces, no atomics in
the vector path; cross-block finalize ordered in both)."""

import pytest
import torch

pytestmark = [pytest.mark.gpu]


def _ref(x, b, dy):
    xr = x.detach().float().clone().requires_grad_(True)
    br = b.detach().float().clone().requires_grad_(True)
    yr = torch.relu(xr + br.view(1, -1, 1, 1))
    yr.backward(dy.float())
    return yr.detach(), xr.grad, br.grad


@pytest.mark.parametrize("cl", [False, True])
@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("C", [48, 64, 33])  # 48/64 hit the vector path (NHWC), 33 scalar
def test_biasrelu_fwd_bwd(cl, dtype, C):
    from npairloss_amd.ops.vision import _BiasReLUFn

    x = torch.randn(3, C, 14, 14, device="cuda", dtype=dtype)
    if cl:
        x = x.to(memory_format=torch.channels_last)
    x.requires_grad_(True)
    b = torch.randn(C, device="cuda", requires_grad=True)
    y = _BiasReLUFn.apply(x, b)
    dy = torch.randn_like(y)
    y.backward(dy)

    yr, dxr, dbr = _ref(x, b, dy)
    tol = dict(rtol=1e-5, atol=1e-5) if dtype == torch.float32 else dict(rtol=2e-2, atol=2e-2)
    torch.testing.assert_close(y.float(), yr, **tol)
    torch.testing.assert_close(x.grad.float(), dxr, **tol)
    torch.testing.assert_close(b.grad, dbr, rtol=1e-3, atol=1e-3)


@pytest.mark.parametrize("C", [64, 192, 288])  # GoogLeNet channel counts
def test_biasrelu_bias_grad_deterministic(C):
    """Two identical backward passes must produce BITWISE-identical db —
    the vector path's fixed-channel register accumulate + ordered reduces."""
    from npairloss_amd.ops import _backend

    torch.manual_seed(0)
    y = torch.randn(8, C, 28, 28, device="cuda", dtype=torch.bfloat16).relu()
    y = y.to(memory_format=torch.channels_last)
    dy = torch.randn_like(y)
    _, db1 = _backend.ext().biasrelu_bwd(y, dy)
    _, db2 = _backend.ext().biasrelu_bwd(y, dy)
    assert torch.equal(db1, db2)
    # and against the fp32 reference within bf16 accumulation tolerance
    ref = (dy.float() * (y.float() > 0)).sum(dim=(0, 2, 3))
    torch.testing.assert_close(db1, ref, rtol=1e-3, atol=1e-2)


def test_biasrelu_conv_module_gpu():
    """ConvBiasReLU end-to-end vs conv(bias)+relu on GPU, channels_last bf16
    (the production layout)."""
    from npairloss_amd.ops.vision import ConvBiasReLU

    torch.manual_seed(1)
    m = ConvBiasReLU(16, 32, 3, pad=1).cuda().to(memory_format=torch.channels_last)
    with torch.no_grad():
        m.bias.uniform_(-0.5, 0.5)
    ref = torch.nn.Sequential(
        torch.nn.Conv2d(16, 32, 3, padding=1, bias=True), torch.nn.ReLU()).cuda()
    with torch.no_grad():
        ref[0].weight.copy_(m.conv.weight)
        ref[0].bias.copy_(m.bias)
    x = torch.randn(4, 16, 20, 20, device="cuda").to(memory_format=torch.channels_last)
    x1 = x.clone().requires_grad_(True)
    x2 = x.clone().requires_grad_(True)
    y1 = m(x1)
    y2 = ref(x2)
    torch.testing.assert_close(y1, y2, rtol=1e-4, atol=1e-4)
    g = torch.randn_like(y1)
    y1.backward(g)
    y2.backward(g)
    torch.testing.assert_close(x1.grad, x2.grad, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(m.bias.grad, ref[0].bias.grad, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(m.conv.weight.grad, ref[0].weight.grad, rtol=1e-4, atol=1e-4)
