"""GPU tests for the bf16 / fp8-e4m3 MFMA similarity GEMMs and the
sim_dtype loss option.  Asymmetric operands (per the CDNA guide: symmetric
inputs hide transposed C-writes)."""

import pytest
import torch

pytestmark = pytest.mark.gpu

from npairloss_amd.ops import _backend


def _C():
    return _backend.ext()


@pytest.mark.parametrize("mnk", [(64, 64, 32), (256, 2048, 1024), (120, 960, 1024),
                                 (33, 65, 40), (16, 16, 1024)])
def test_sim_gemm_bf16(mnk):
    M, N, K = mnk
    A = torch.randn(M, K, device="cuda").bfloat16()
    B = torch.randn(N, K, device="cuda").bfloat16()
    C = _C().sim_gemm_nt_bf16(A, B)
    ref = (A.float() @ B.float().t())
    torch.testing.assert_close(C, ref, rtol=1e-2, atol=1e-2)


def test_sim_gemm_bf16_asymmetric_exact():
    # integer-valued bf16 operands: MFMA must be exact, catching any
    # fragment-map/transpose error bit-for-bit
    M, N, K = 32, 48, 64
    A = torch.randint(-4, 5, (M, K), device="cuda").bfloat16()
    B = torch.arange(N * K, device="cuda").reshape(N, K).remainder(7).sub(3).bfloat16()
    C = _C().sim_gemm_nt_bf16(A, B)
    ref = A.float() @ B.float().t()
    torch.testing.assert_close(C, ref, rtol=0, atol=0)


@pytest.mark.parametrize("mnk", [(64, 64, 32), (128, 256, 1024), (33, 65, 40)])
def test_sim_gemm_fp8(mnk):
    M, N, K = mnk
    # unit-norm-ish rows: e4m3 holds [-1,1] with ~2 decimal digits
    A = torch.nn.functional.normalize(torch.randn(M, K, device="cuda"), dim=1)
    B = torch.nn.functional.normalize(torch.randn(N, K, device="cuda"), dim=1)
    a8 = _C().cast_fp8(A)
    b8 = _C().cast_fp8(B)
    C = _C().sim_gemm_nt_fp8(a8, b8)
    ref = A @ B.t()
    # e4m3 has ~3 bits mantissa; row dot of unit vectors stays within ~0.05
    torch.testing.assert_close(C, ref, rtol=0.1, atol=0.06)


def test_cast_fp8_roundtrip_values():
    x = torch.tensor([0.0, 0.5, -0.5, 1.0, -1.0, 0.25, 0.875], device="cuda")
    b = _C().cast_fp8(x)
    # these values are exactly representable in e4m3
    back = b.cpu().numpy()
    import numpy as np
    # decode manually: sign(1) exp(4, bias 7) mant(3)
    def dec(u):
        s = -1.0 if (u >> 7) else 1.0
        e = (u >> 3) & 0xF
        m = u & 7
        if e == 0:
            return s * (m / 8.0) * 2.0 ** -6
        return s * (1 + m / 8.0) * 2.0 ** (e - 7)
    vals = [dec(int(u)) for u in back]
    np.testing.assert_allclose(vals, x.cpu().numpy(), rtol=0, atol=0)


@pytest.mark.parametrize("sim_dtype", ["bf16", "fp8"])
def test_loss_module_sim_dtype(sim_dtype):
    """The loss with low-precision similarity stays close to the fp32 loss
    and trains (finite grads)."""
    import numpy as np

    from npairloss_amd.config.params import NPairLossConfig
    from npairloss_amd.ops.npair_loss import NPairMultiClassLoss

    from util import make_batch

    cfg = NPairLossConfig(margin_diff=-0.05, ap_mining_region="GLOBAL",
                          ap_mining_method="RELATIVE_HARD", identsn=-0.0,
                          an_mining_region="LOCAL", an_mining_method="HARD")
    f, lab = make_batch(num_classes=16, per_class=4, dim=256, seed=3)
    ft = torch.from_numpy(f).float().cuda().requires_grad_(True)
    lt = torch.from_numpy(lab).cuda()

    out32 = NPairMultiClassLoss(cfg, sim_dtype="fp32")(ft, lt)
    outlp = NPairMultiClassLoss(cfg, sim_dtype=sim_dtype)(ft, lt)
    tol = 0.02 if sim_dtype == "bf16" else 0.15
    assert abs(float(outlp.loss) - float(out32.loss)) < tol * max(1.0, abs(float(out32.loss)))
    outlp.loss.backward()
    assert torch.isfinite(ft.grad).all()
