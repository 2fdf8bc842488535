"""GPU numerics for the fused 1x1-conv MFMA GEMM (csrc/conv1x1.hip):
forward (GEMM+bias+relu epilogue) and full backward (fused dReLU+bias-grad
pass + dgrad GEMM + hipBLASLt wgrad) against fp32 torch conv references."""

import pytest
import torch

pytestmark = [pytest.mark.gpu]


@pytest.mark.parametrize("shape", [(4, 64, 28, 28, 96), (2, 192, 14, 14, 48),
                                   (3, 33, 10, 10, 17)])  # odd K/N tile edges
def test_conv1x1_kernel_fwd(shape):
    from npairloss_amd.ops import _backend

    B, K, H, W, N = shape
    torch.manual_seed(0)
    x = torch.randn(B, K, H, W, device="cuda", dtype=torch.bfloat16)
    x = x.to(memory_format=torch.channels_last)
    w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16) * 0.1
    b = torch.randn(N, device="cuda")
    xm = x.permute(0, 2, 3, 1).reshape(-1, K)
    y = _backend.ext().conv1x1_bias_relu_fwd(xm, w, b)
    ref = torch.relu(xm.float() @ w.float().t() + b)
    torch.testing.assert_close(y.float(), ref, rtol=2e-2, atol=2e-2)


def test_conv1x1_dgrad_kernel():
    from npairloss_amd.ops import _backend

    torch.manual_seed(1)
    M, N, K = 1000, 96, 192
    g = torch.randn(M, N, device="cuda", dtype=torch.bfloat16)
    wt = torch.randn(K, N, device="cuda", dtype=torch.bfloat16) * 0.1
    dx = _backend.ext().conv1x1_dgrad(g, wt)
    ref = g.float() @ wt.float().t()
    torch.testing.assert_close(dx.float(), ref, rtol=2e-2, atol=2e-2)


@pytest.mark.parametrize("amp", [True, False])
def test_conv1x1_module_end_to_end(amp):
    """Conv1x1BiasReLU vs conv2d(bias)+relu: fwd + dx + dw + db."""
    from npairloss_amd.ops.vision import Conv1x1BiasReLU

    torch.manual_seed(2)
    B, K, H, W, N = 4, 64, 14, 14, 96
    m = Conv1x1BiasReLU(K, N).cuda().to(memory_format=torch.channels_last)
    with torch.no_grad():
        m.bias.uniform_(-0.2, 0.2)
    ref_conv = torch.nn.Conv2d(K, N, 1, bias=True).cuda()
    with torch.no_grad():
        ref_conv.weight.copy_(m.conv.weight)
        ref_conv.bias.copy_(m.bias)

    dt = torch.bfloat16 if not amp else torch.float32
    x = torch.randn(B, K, H, W, device="cuda", dtype=torch.float32)
    x = x.to(memory_format=torch.channels_last)
    x1 = x.clone().to(dt if not amp else torch.float32).requires_grad_(True)
    x2 = x.clone().requires_grad_(True)

    if amp:
        with torch.autocast("cuda", dtype=torch.bfloat16):
            y1 = m(x1)
    else:
        mb = m.to(torch.bfloat16)
        y1 = mb(x1.to(torch.bfloat16))
    y2 = torch.relu(ref_conv(x2))
    torch.testing.assert_close(y1.float(), y2, rtol=3e-2, atol=3e-2)

    g = torch.randn_like(y2)
    y1.backward(g.to(y1.dtype))
    # mask-consistent reference backward: near-zero pre-activations can flip
    # the ReLU mask between bf16 and fp32, giving whole-gradient diffs at
    # ~3% of sites — use the MODULE's mask so the GEMM chain is what's tested
    mask = (y1 > 0).float()
    # reference from the SAME bf16-rounded operands the module consumed
    # (fp32 operands would differ by input rounding, amplified by the
    # M-site reduction in dw)
    gm = (g * mask).to(torch.bfloat16).float()
    xb = x1.detach().to(torch.bfloat16).float()
    wb = ref_conv.weight.detach().to(torch.bfloat16).float()
    ref_dx = torch.nn.functional.conv_transpose2d(gm, wb)
    ref_dw = torch.nn.grad.conv2d_weight(xb, ref_conv.weight.shape, gm)
    ref_db = gm.sum(dim=(0, 2, 3))
    torch.testing.assert_close(x1.grad.float(), ref_dx, rtol=3e-2, atol=3e-2)
    torch.testing.assert_close(m.weight.grad.float().flatten(),
                               ref_dw.flatten(), rtol=3e-2, atol=3e-2)
    torch.testing.assert_close(m.bias.grad.float(), ref_db, rtol=2e-2, atol=2e-1)


def test_googlenet_fused_close_to_reference_gpu():
    """Whole GoogLeNet with fused 1x1 GEMM convs vs the unfused build with
    identical weights: embeddings must agree to bf16 tolerance."""
    from npairloss_amd.models.googlenet import GoogLeNet

    torch.manual_seed(3)
    fused = GoogLeNet(fused_bias_relu=True).cuda().to(memory_format=torch.channels_last).eval()
    plain = GoogLeNet(fused_bias_relu=False).cuda().to(memory_format=torch.channels_last).eval()
    # copy weights fused -> plain by caffe names (bias lives differently)
    fmap, pmap = fused.caffe_names(), plain.caffe_names()
    with torch.no_grad():
        for name, fm in fmap.items():
            pm = pmap[name]
            pm.weight.copy_(fm.weight)
            pm.bias.copy_(fm.bias)
    x = torch.randn(2, 3, 224, 224, device="cuda").to(memory_format=torch.channels_last)
    with torch.autocast("cuda", dtype=torch.bfloat16):
        yf = fused(x)
        yp = plain(x)
    torch.testing.assert_close(yf.float(), yp.float(), rtol=5e-2, atol=5e-2)
