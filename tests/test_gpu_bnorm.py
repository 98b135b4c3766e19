"""Fused BN3d(+ReLU) kernels vs torch BatchNorm3d fp32 reference."""
import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from coinstac_dinunet_amd import ops
    C = ops.require_native()
    from coinstac_dinunet_amd.ops.bnorm import OpsBatchNorm3d


@pytest.fixture(scope='module')
def dev():
    assert torch.cuda.is_available()
    return torch.device('cuda:0')


@pytest.mark.parametrize('relu', [False, True])
def test_bn3d_forward_backward(dev, relu):
    torch.manual_seed(0)
    N, Cc, D = 4, 8, 10
    x = torch.randn(N, Cc, D, D, D, device=dev, requires_grad=True)
    gamma = torch.randn(Cc, device=dev).abs() + 0.5
    beta = torch.randn(Cc, device=dev)
    gamma.requires_grad_(True)
    beta.requires_grad_(True)

    from coinstac_dinunet_amd.ops.bnorm import _BN3dFn
    y = _BN3dFn.apply(x, gamma, beta, None, None, 0.1, 1e-5, relu)

    # reference on the SAME bf16-quantized input so the ReLU mask agrees
    xr = x.detach().to(torch.bfloat16).float().requires_grad_(True)
    gr = gamma.detach().clone().requires_grad_(True)
    br = beta.detach().clone().requires_grad_(True)
    z = torch.nn.functional.batch_norm(
        xr, None, None, gr, br, training=True, eps=1e-5)
    ref = torch.relu(z) if relu else z
    # elements with z ~ 0 can flip the mask under bf16 rounding: exclude
    keep = (z.abs() > 1e-2).detach()
    torch.testing.assert_close(y.float()[keep], ref[keep],
                               rtol=5e-2, atol=5e-2)

    g = torch.randn_like(ref)
    y.backward(g.to(y.dtype))
    ref.backward(g)
    torch.testing.assert_close(x.grad[keep], xr.grad[keep],
                               rtol=5e-2, atol=5e-2)
    torch.testing.assert_close(gamma.grad, gr.grad, rtol=5e-2, atol=0.5)
    torch.testing.assert_close(beta.grad, br.grad, rtol=5e-2, atol=0.5)


def test_bn3d_module_running_stats(dev):
    torch.manual_seed(1)
    m = OpsBatchNorm3d(6, relu=False).to(dev)
    ref = torch.nn.BatchNorm3d(6).to(dev)
    x = torch.randn(3, 6, 8, 8, 8, device=dev) * 2 + 1
    m.train()
    ref.train()
    m(x)
    ref(x.float())
    torch.testing.assert_close(m.running_mean, ref.running_mean,
                               rtol=2e-2, atol=2e-2)
    torch.testing.assert_close(m.running_var, ref.running_var,
                               rtol=2e-2, atol=5e-2)
    # eval path uses running stats
    m.eval()
    ref.eval()
    y = m(x)
    yr = ref(x.float())
    torch.testing.assert_close(y.float(), yr, rtol=5e-2, atol=5e-2)
