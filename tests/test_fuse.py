"""Inference BN folding: fused model == original at eval, BN gone."""
import torch
import torch.nn as nn

from coinstac_dinunet_amd.ops.fuse import fuse_conv_bn_eval
from coinstac_dinunet_amd.models import UNet3D, VBMNet


def _train_a_little(net, shape):
    opt = torch.optim.SGD(net.parameters(), lr=1e-2)
    for _ in range(3):  # give BN running stats something real
        x = torch.randn(*shape)
        out = net(x)
        out.float().mean().backward()
        opt.step(), opt.zero_grad()


def _count(net, cls):
    return sum(isinstance(m, cls) for m in net.modules())


def test_fuse_vbm_matches_and_removes_bn():
    torch.manual_seed(0)
    net = VBMNet(in_channels=1, num_class=2)
    _train_a_little(net, (2, 1, 16, 16, 16))
    net.eval()
    fused = fuse_conv_bn_eval(net)
    assert _count(fused, nn.BatchNorm3d) == 0
    assert _count(net, nn.BatchNorm3d) > 0  # original untouched
    x = torch.randn(2, 1, 16, 16, 16)
    with torch.no_grad():
        torch.testing.assert_close(fused(x), net(x), rtol=1e-4, atol=1e-5)


def test_fuse_unet_matches():
    torch.manual_seed(1)
    net = UNet3D(in_channels=1, num_class=2, widths=(4, 8))
    _train_a_little(net, (2, 1, 12, 12, 12))
    net.eval()
    fused = fuse_conv_bn_eval(net)
    assert _count(fused, nn.BatchNorm3d) == 0
    x = torch.randn(1, 1, 12, 12, 12)
    with torch.no_grad():
        torch.testing.assert_close(fused(x), net(x), rtol=1e-4, atol=1e-5)


def test_fuse_generic_sequential():
    torch.manual_seed(2)
    net = nn.Sequential(nn.Conv2d(3, 8, 3, padding=1), nn.BatchNorm2d(8),
                        nn.ReLU(), nn.Conv2d(8, 4, 3, padding=1),
                        nn.BatchNorm2d(4))
    _train_a_little(net, (2, 3, 8, 8))
    net.eval()
    fused = fuse_conv_bn_eval(net)
    assert _count(fused, nn.BatchNorm2d) == 0
    x = torch.randn(2, 3, 8, 8)
    with torch.no_grad():
        torch.testing.assert_close(fused(x), net(x), rtol=1e-4, atol=1e-5)


def test_fuse_bn_eval_flag_in_protocol(tmp_path):
    """cache['fuse_bn_eval']=True: validation/test run on fused copies and
    the run still completes with the originals restored for training."""
    import os
    import sys
    sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
    from computations import TabularDataset, TabularTrainer, make_site_data
    from coinstac_dinunet_amd import COINNLocal, COINNRemote
    from coinstac_dinunet_amd.config.keys import Key, Mode
    from coinstac_dinunet_amd.simulator import LoopbackCluster

    cluster = LoopbackCluster(
        str(tmp_path), n_sites=2,
        site_data=lambda s: make_site_data(s.as_dict(), n_samples=16,
                                           seed=int(s.clientId[-1])))
    kw = dict(task_id='tab', mode=Mode.TRAIN, batch_size=4, epochs=1,
              validation_epochs=1, local_iterations=1,
              split_ratio=(0.6, 0.2, 0.2), data_dir='data', num_class=2,
              seed_all=True, patience=1, verbose=False, fuse_bn_eval=True)
    success, _ = cluster.run(
        lambda cache, input, state: COINNLocal(cache=cache, input=input,
                                               state=state, **kw),
        lambda cache, input, state: COINNRemote(cache=cache, input=input,
                                                state=state),
        TabularTrainer, dataset_cls=TabularDataset, max_rounds=300)
    assert success
    assert cluster.remote_cache[Key.GLOBAL_TEST_SERIALIZABLE]
    # originals (with BN) are still what lives in the cache
    import torch.nn as nn
    net = cluster.site_caches[0]['nn']['net']
    assert any(isinstance(m, nn.BatchNorm1d) for m in net.modules())
