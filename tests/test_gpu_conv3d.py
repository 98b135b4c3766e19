"""Conv3d implicit-GEMM MFMA kernels vs torch fp32 references (MI355X)."""
import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from coinstac_dinunet_amd import ops
    C = ops.require_native()


@pytest.fixture(scope='module')
def dev():
    assert torch.cuda.is_available()
    return torch.device('cuda:0')


def test_mfma_probe_layout(dev):
    """Pin the bf16 16x16x32 fragment layout with an ASYMMETRIC B
    (catches transposes — guide G9)."""
    torch.manual_seed(0)
    A = (torch.randn(16, 32, device=dev) * 0.5).round_()
    B = (torch.randn(32, 16, device=dev) * 0.5).round_()
    B[3, 7] += 5.0  # asymmetry spike
    out = C.mfma_probe_gemm(A, B)
    ref = A @ B
    torch.testing.assert_close(out, ref, rtol=1e-2, atol=1e-2)


CASES = [
    # (N, Cin, Cout, D, H, W, stride)
    (2, 1, 32, 16, 16, 16, 1),
    (2, 32, 32, 16, 16, 16, 1),
    (2, 32, 64, 16, 16, 16, 2),
    (1, 64, 128, 8, 8, 8, 2),
    (3, 4, 8, 9, 11, 13, 1),   # odd sizes
    (1, 16, 16, 7, 7, 7, 2),
    (1, 32, 64, 32, 32, 32, 2),  # routes the parity-spatial s2 dgrad
]


def _ref_conv(x, w, stride):
    return torch.nn.functional.conv3d(x.float(), w.float(), stride=stride,
                                      padding=1)


@pytest.mark.parametrize('case', CASES)
def test_conv3d_fwd(dev, case):
    N, Cin, Cout, D, H, W, s = case
    torch.manual_seed(1)
    x = torch.randn(N, Cin, D, H, W, device=dev, dtype=torch.bfloat16)
    w = torch.randn(Cout, Cin, 3, 3, 3, device=dev, dtype=torch.bfloat16) * 0.2
    out = C.conv3d_fwd(x, w, s)
    ref = _ref_conv(x, w, s)
    torch.testing.assert_close(out.float(), ref, rtol=5e-2,
                               atol=5e-2 * (Cin * 27) ** 0.5 * 0.2)
    if W % 8 == 0 and ((W + 2 - 3) // s + 1) % 8 == 0:
        out2 = C.conv3d_fwd_spatial(x, w, s)
        torch.testing.assert_close(out2.float(), ref, rtol=5e-2,
                                   atol=5e-2 * (Cin * 27) ** 0.5 * 0.2)


@pytest.mark.parametrize('case', CASES)
def test_conv3d_dgrad(dev, case):
    N, Cin, Cout, D, H, W, s = case
    torch.manual_seed(2)
    x = torch.randn(N, Cin, D, H, W, device=dev, requires_grad=True)
    w = torch.randn(Cout, Cin, 3, 3, 3, device=dev) * 0.2
    ref_out = _ref_conv(x, w, s)
    go = torch.randn_like(ref_out)
    ref_out.backward(go)
    dx = C.conv3d_dgrad(go.to(torch.bfloat16), w.to(torch.bfloat16),
                        list(x.shape), s)
    torch.testing.assert_close(dx.float(), x.grad, rtol=5e-2,
                               atol=5e-2 * (Cout * 27) ** 0.5 * 0.2)
    if s == 1 and W % 8 == 0:
        dx2 = C.conv3d_dgrad_spatial(go.to(torch.bfloat16),
                                     w.to(torch.bfloat16), list(x.shape))
        torch.testing.assert_close(dx2.float(), x.grad, rtol=5e-2,
                                   atol=5e-2 * (Cout * 27) ** 0.5 * 0.2)
    if s == 2 and ((W + 1) // 2) % 8 == 0 and Cout >= 32:
        dx3 = C.conv3d_dgrad_s2_spatial(go.to(torch.bfloat16),
                                        w.to(torch.bfloat16), list(x.shape))
        torch.testing.assert_close(dx3.float(), x.grad, rtol=5e-2,
                                   atol=5e-2 * (Cout * 27) ** 0.5 * 0.2)


@pytest.mark.parametrize('case', CASES)
def test_conv3d_wgrad(dev, case):
    N, Cin, Cout, D, H, W, s = case
    torch.manual_seed(3)
    x = torch.randn(N, Cin, D, H, W, device=dev) * 0.3
    w = torch.randn(Cout, Cin, 3, 3, 3, device=dev, requires_grad=True) * 0.2
    w = w.detach().requires_grad_(True)
    ref_out = _ref_conv(x, w, s)
    go = torch.randn_like(ref_out) * 0.1
    ref_out.backward(go)
    dw = C.conv3d_wgrad(x.to(torch.bfloat16), go.to(torch.bfloat16), s)
    # wgrad sums over many positions: scale tolerance with sqrt(M)
    M = ref_out.numel() // Cout
    torch.testing.assert_close(dw.float(), w.grad, rtol=5e-2,
                               atol=3e-2 * M ** 0.5 * 0.03)


def test_ops_conv3d_module_autograd(dev):
    """Full module fwd+bwd vs torch conv3d (bf16-tolerance)."""
    from coinstac_dinunet_amd.ops.conv import OpsConv3d
    torch.manual_seed(4)
    m = OpsConv3d(8, 16, 3, stride=2, padding=1, bias=True).to(dev)
    x = torch.randn(2, 8, 12, 12, 12, device=dev, requires_grad=True)
    out = m(x)
    ref = torch.nn.functional.conv3d(x.float(), m.weight.float(),
                                     m.bias.float(), stride=2, padding=1)
    torch.testing.assert_close(out.float(), ref, rtol=5e-2, atol=0.5)
    out.sum().backward()
    assert x.grad is not None and m.weight.grad is not None \
        and m.bias.grad is not None
    assert torch.isfinite(m.weight.grad).all()


@pytest.mark.parametrize('case', [(2, 1, 32, 16, 16, 16),
                                  (1, 1, 8, 16, 32, 32),
                                  (2, 4, 16, 8, 16, 16)])
def test_conv3d_fwd_spatial_ctile1(dev, case):
    """CTILE=1 single-channel spatial instances (default routing for the
    Cin<16 first layer since r2 — measured A/B in profiles/r2_scaffold_ab.md)."""
    N, Cin, Cout, D, H, W = case
    torch.manual_seed(4)
    x = torch.randn(N, Cin, D, H, W, device=dev, dtype=torch.bfloat16)
    w = torch.randn(Cout, Cin, 3, 3, 3, device=dev, dtype=torch.bfloat16) * 0.2
    out = C.conv3d_fwd_spatial(x, w, 1, 1)
    ref = _ref_conv(x, w, 1)
    torch.testing.assert_close(out.float(), ref, rtol=5e-2,
                               atol=5e-2 * (Cin * 27) ** 0.5 * 0.2)


@pytest.mark.parametrize('case', [(2, 32, 32, 12, 16, 16),
                                  (1, 48, 64, 8, 16, 32),
                                  (1, 16, 32, 6, 8, 8)])
def test_conv3d_fwd_spatial_double_buffered(dev, case):
    """Double-buffered CTILE=16 instances (compiled, routing off by default).
    Must agree with the validated single-buffered CTILE=32 kernel."""
    N, Cin, Cout, D, H, W = case
    torch.manual_seed(6)
    x = torch.randn(N, Cin, D, H, W, device=dev, dtype=torch.bfloat16)
    w = torch.randn(Cout, Cin, 3, 3, 3, device=dev, dtype=torch.bfloat16) * 0.2
    out_db = C.conv3d_fwd_spatial(x, w, 1, 16)
    out_sb = C.conv3d_fwd_spatial(x, w, 1, 0)
    ref = _ref_conv(x, w, 1)
    tol = 5e-2 * (Cin * 27) ** 0.5 * 0.2
    torch.testing.assert_close(out_db.float(), ref, rtol=5e-2, atol=tol)
    # and bitwise-compatible accumulation order vs the single-buffered
    # CTILE=16 math is not guaranteed; only the reference bound is.
    torch.testing.assert_close(out_db.float(), out_sb.float(),
                               rtol=5e-2, atol=tol)


@pytest.mark.parametrize('case', [(2, 32, 32, 12, 16, 16),
                                  (1, 48, 64, 8, 16, 32),
                                  (2, 16, 32, 6, 8, 8)])
def test_conv3d_wgrad_double_buffered(dev, case):
    """Double-buffered stride-1 wgrad (compiled, routing off by default) vs the
    validated single-buffered kernel and the torch fp32 reference."""
    N, Cin, Cout, D, H, W = case
    torch.manual_seed(8)
    x = torch.randn(N, Cin, D, H, W, device=dev, dtype=torch.bfloat16)
    go_shape = (N, Cout, D, H, W)
    go = torch.randn(*go_shape, device=dev, dtype=torch.bfloat16) * 0.1
    dw_db = C.conv3d_wgrad(x, go, 1, 1)
    dw_sb = C.conv3d_wgrad(x, go, 1, 0)
    # torch reference
    xf = x.float().requires_grad_(True)
    w0 = torch.zeros(Cout, Cin, 3, 3, 3, device=dev, requires_grad=True)
    out = torch.nn.functional.conv3d(xf, w0, padding=1)
    out.backward(go.float())
    ref = w0.grad
    m = (N * D * H * W) ** 0.5
    torch.testing.assert_close(dw_db.float(), ref, rtol=5e-2,
                               atol=5e-2 * m * 0.1)
    torch.testing.assert_close(dw_db.float(), dw_sb.float(), rtol=2e-2,
                               atol=2e-2 * m * 0.1)


@pytest.mark.parametrize('case', [(2, 32, 8, 16, 32),
                                  (1, 16, 6, 10, 64),
                                  (1, 8, 9, 7, 32)])   # OH % OHT != 0
def test_conv3d_wgrad_ci1_specialized(dev, case):
    """Cin=1 slab tap-reuse wgrad (first-layer specialization) vs torch
    fp32; exercises the dedicated conv3d_wgrad_ci1_kernel routing."""
    N, Co, D, H, W = case
    torch.manual_seed(12)
    x = torch.randn(N, 1, D, H, W, device=dev, dtype=torch.bfloat16)
    go = torch.randn(N, Co, D, H, W, device=dev, dtype=torch.bfloat16) * 0.1
    go[0, 0, 0, 0, 3] += 1.5
    dw = C.conv3d_wgrad(x, go, 1, 0)
    xf = x.float().requires_grad_(True)
    w0 = torch.zeros(Co, 1, 3, 3, 3, device=dev, requires_grad=True)
    torch.nn.functional.conv3d(xf, w0, padding=1).backward(go.float())
    m = (N * D * H * W) ** 0.5
    torch.testing.assert_close(dw.float(), w0.grad, rtol=5e-2,
                               atol=5e-2 * m * 0.1)
