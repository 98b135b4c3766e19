"""2D conv implicit-GEMM MFMA kernels vs torch fp32 (MI355X) — the
ResNet-18 hot path."""
import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from coinstac_dinunet_amd import ops
    C = ops.require_native()


@pytest.fixture(scope='module')
def dev():
    assert torch.cuda.is_available()
    return torch.device('cuda:0')


CASES = [
    # (N, Cin, Cout, H, W, stride) — ResNet-ish shapes + odd sizes
    (2, 64, 64, 56, 56, 1),
    (2, 64, 128, 56, 56, 2),
    (2, 256, 512, 14, 14, 2),
    (1, 512, 512, 7, 7, 1),
    (3, 5, 8, 9, 13, 1),
    (1, 16, 16, 10, 10, 2),
]


@pytest.mark.parametrize('case', CASES)
def test_conv2d_fwd(dev, case):
    N, Ci, Co, H, W, s = case
    torch.manual_seed(61)
    x = torch.randn(N, Ci, H, W, device=dev, dtype=torch.bfloat16)
    w = torch.randn(Co, Ci, 3, 3, device=dev, dtype=torch.bfloat16) * 0.1
    w[min(1, Co - 1), min(2, Ci - 1), 0, 2] += 2.0  # asymmetry spike
    out = C.conv2d_fwd(x, w, s)
    ref = F.conv2d(x.float(), w.float(), stride=s, padding=1)
    tol = 5e-2 * (Ci * 9) ** 0.5 * 0.1
    torch.testing.assert_close(out.float(), ref, rtol=5e-2, atol=tol + 0.3)


@pytest.mark.parametrize('case', CASES)
def test_conv2d_backward(dev, case):
    N, Ci, Co, H, W, s = case
    torch.manual_seed(62)
    x = torch.randn(N, Ci, H, W, device=dev, dtype=torch.bfloat16)
    w = torch.randn(Co, Ci, 3, 3, device=dev, dtype=torch.bfloat16) * 0.1
    OH, OW = (H + 2 - 3) // s + 1, (W + 2 - 3) // s + 1
    go = torch.randn(N, Co, OH, OW, device=dev, dtype=torch.bfloat16) * 0.1
    go[0, 0, 0, min(3, OW - 1)] += 1.5

    gx = C.conv2d_dgrad(go, w, list(x.shape), s)
    gw = C.conv2d_wgrad(x, go, s)

    xr = x.float().requires_grad_(True)
    wr = w.float().requires_grad_(True)
    F.conv2d(xr, wr, stride=s, padding=1).backward(go.float())
    m = (N * OH * OW) ** 0.5
    torch.testing.assert_close(gx.float(), xr.grad, rtol=5e-2,
                               atol=5e-2 * (Co * 9) ** 0.5 * 0.1 + 0.2)
    torch.testing.assert_close(gw.view_as(wr), wr.grad, rtol=5e-2,
                               atol=5e-2 * m * 0.1 + 0.2)


def test_opsconv2d_module_routing(dev):
    from coinstac_dinunet_amd.ops.conv import OpsConv2d
    torch.manual_seed(63)
    m3 = OpsConv2d(16, 32, 3, padding=1, bias=True).to(dev)
    x = torch.randn(2, 16, 12, 12, device=dev, dtype=torch.bfloat16,
                    requires_grad=True)
    out = m3(x)
    out.sum().backward()
    assert out.dtype == torch.bfloat16
    assert m3.weight.grad is not None and m3.bias.grad is not None
    ref = F.conv2d(x.detach().float(), m3.weight.detach().float(),
                   m3.bias.detach().float(), padding=1)
    torch.testing.assert_close(out.float(), ref, rtol=5e-2, atol=0.5)


def test_opsconv2d_1x1_downsample(dev):
    """Stride-2 1x1 (ResNet downsample) -> strided view + pointwise GEMM."""
    from coinstac_dinunet_amd.ops.conv import OpsConv2d
    torch.manual_seed(64)
    m = OpsConv2d(64, 128, 1, stride=2, bias=False).to(dev)
    x = torch.randn(2, 64, 14, 14, device=dev, dtype=torch.bfloat16,
                    requires_grad=True)
    out = m(x)
    assert out.shape == (2, 128, 7, 7)
    out.sum().backward()
    assert m.weight.grad is not None and x.grad is not None
    ref = F.conv2d(x.detach().float(), m.weight.detach().float(), stride=2)
    torch.testing.assert_close(out.float(), ref, rtol=5e-2, atol=0.3)


def test_conv2d_7x7_stem(dev):
    """7x7/pad-3/stride-2 stem fwd + wgrad (KS=7 igemm instances)."""
    torch.manual_seed(66)
    x = torch.randn(2, 3, 32, 32, device=dev, dtype=torch.bfloat16)
    w = torch.randn(16, 3, 7, 7, device=dev, dtype=torch.bfloat16) * 0.05
    w[1, 2, 0, 6] += 1.0
    out = C.conv2d_fwd(x, w, 2)
    ref = F.conv2d(x.float(), w.float(), stride=2, padding=3)
    torch.testing.assert_close(out.float(), ref, rtol=5e-2, atol=0.3)
    go = torch.randn_like(ref).bfloat16() * 0.1
    gw = C.conv2d_wgrad(x, go, 2, ks=7)
    wr = w.float().requires_grad_(True)
    F.conv2d(x.float(), wr, stride=2, padding=3).backward(go.float())
    torch.testing.assert_close(gw.view_as(wr), wr.grad, rtol=5e-2,
                               atol=5e-2 * (2 * 16 * 16) ** 0.5 * 0.1 + 0.2)


def test_resnet18_trains_on_hip_kernels(dev):
    """Whole ResNet-18 fwd/bwd on GPU: loss decreases, all grads flow."""
    from coinstac_dinunet_amd.models import ResNet18
    torch.manual_seed(65)
    net = ResNet18(in_channels=3, num_class=4,
                   widths=(16, 32, 64, 128)).to(dev)
    x = torch.randn(8, 3, 64, 64, device=dev)
    y = (torch.arange(8) % 4).to(dev)
    for k in range(4):
        x[y == k] += 0.25 * k
    opt = torch.optim.Adam(net.parameters(), lr=2e-3)
    losses = []
    for _ in range(25):
        opt.zero_grad(set_to_none=True)
        with torch.autocast('cuda', dtype=torch.bfloat16):
            out = net(x)
        loss = ops.cross_entropy(out.float(), y)
        loss.backward()
        opt.step()
        losses.append(float(loss.detach()))
    assert losses[-1] < losses[0] * 0.8, losses[::5]
    for n, p in net.named_parameters():
        assert p.grad is not None, n


def test_conv_bn2d_fused_matches_composed(dev):
    """2D BN(+ReLU)->conv normalize-on-load vs the composed ops."""
    from coinstac_dinunet_amd.ops.bnorm import OpsBatchNorm2d
    from coinstac_dinunet_amd.ops.conv import OpsConv2d, conv_bn2d
    torch.manual_seed(93)
    for Ci, Co, H, W, s in [(32, 32, 28, 28, 1), (64, 128, 28, 28, 2),
                            (64, 64, 14, 14, 1)]:
        bn = OpsBatchNorm2d(Ci, relu=True).to(dev)
        with torch.no_grad():
            bn.weight.mul_(0).add_(torch.rand(Ci, device=dev) + 0.5)
            bn.bias.add_(torch.randn(Ci, device=dev) * 0.1)
        conv = OpsConv2d(Ci, Co, 3, stride=s, padding=1, bias=False).to(dev)
        bn.train(), conv.train()
        x0 = torch.randn(2, Ci, H, W, device=dev, dtype=torch.bfloat16)
        go = None

        def run(fused):
            nonlocal go
            bn.running_mean.zero_(), bn.running_var.fill_(1.0)
            x = x0.clone().requires_grad_(True)
            for p in list(bn.parameters()) + list(conv.parameters()):
                p.grad = None
            out = conv_bn2d(x, bn, conv) if fused else conv(bn(x))
            if go is None:
                go = torch.randn_like(out) * 0.1
            out.backward(go)
            return (out.float(), x.grad.float(), conv.weight.grad.clone())

        yf, gxf, gwf = run(True)
        yu, gxu, gwu = run(False)
        tol = 5e-2 * (Ci * 9) ** 0.5 * 0.3
        torch.testing.assert_close(yf, yu, rtol=3e-2, atol=tol)
        torch.testing.assert_close(gxf, gxu, rtol=5e-2, atol=5e-2)
        m = (2 * H * W) ** 0.5
        torch.testing.assert_close(gwf, gwu, rtol=5e-2, atol=5e-2 * m * 0.1)
