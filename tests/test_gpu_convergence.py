"""End-to-end GPU sanity: the full native stack trains (loss decreases)."""
import pytest
import torch

pytestmark = pytest.mark.gpu


def test_vbm_stack_learns():
    """VBMNet + fused BN/conv kernels + fused Adam on a learnable synthetic
    task: mean-intensity-shifted classes. Loss must drop substantially."""
    assert torch.cuda.is_available()
    from coinstac_dinunet_amd import ops
    from coinstac_dinunet_amd.models import VBMNet
    from coinstac_dinunet_amd.parallel.engine import FlatGradBuffer

    torch.manual_seed(0)
    dev = torch.device('cuda:0')
    net = VBMNet(in_channels=1, num_class=2, widths=(16, 32, 64, 64)).to(dev)
    buf = FlatGradBuffer(net.parameters(), world_size=1)
    opt = ops.FusedAdam(net.parameters(), lr=3e-3)

    B = 16
    y = torch.arange(B, device=dev) % 2
    x = torch.randn(B, 1, 32, 32, 32, device=dev) * 0.5
    x += y.view(B, 1, 1, 1, 1).float() * 1.0  # class-1 brighter

    losses = []
    for step in range(30):
        buf.zero_()
        buf.begin_round(sync=False)
        with torch.autocast('cuda', dtype=torch.bfloat16):
            out = net(x)
        loss = ops.cross_entropy(out.float(), y)
        loss.backward()
        opt.step()
        losses.append(float(loss))
    assert losses[-1] < 0.25 * losses[0] or losses[-1] < 0.1, losses[::5]
    # training accuracy should be high
    with torch.no_grad(), torch.autocast('cuda', dtype=torch.bfloat16):
        pred = ops.argmax_rows(net(x).float())
    assert (pred == y).float().mean().item() >= 0.9
