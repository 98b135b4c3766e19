"""RCCL engine tests on CPU: flat grad arena + 2-process gloo cluster."""
import os
import socket
import sys

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp

TESTS_DIR = os.path.dirname(os.path.abspath(__file__))


def test_flat_grad_buffer_views_alias_grads():
    from coinstac_dinunet_amd.parallel.engine import FlatGradBuffer
    net = torch.nn.Sequential(torch.nn.Linear(8, 4), torch.nn.Linear(4, 2))
    buf = FlatGradBuffer(net.parameters(), bucket_bytes=40, world_size=1)
    assert buf.flat.numel() == sum(p.numel() for p in net.parameters())
    assert len(buf.buckets) >= 2
    buf.zero_()
    buf.begin_round(sync=False)
    x = torch.randn(3, 8)
    net(x).sum().backward()
    # autograd accumulated straight into the arena
    total = buf.flat.abs().sum().item()
    manual = sum(p.grad.abs().sum().item() for p in net.parameters())
    assert total == pytest.approx(manual, rel=1e-6)
    for p in net.parameters():
        assert p.grad.data_ptr() >= buf.flat.data_ptr()
        assert p.grad.data_ptr() < buf.flat.data_ptr() + buf.flat.numel() * 4
    # second backward accumulates (sum over micro-batches)
    net(x).sum().backward()
    assert buf.flat.abs().sum().item() == pytest.approx(2 * manual, rel=1e-6)
    buf.zero_()
    assert buf.flat.abs().sum().item() == 0.0


def _rank_main(rank, world_size, port, root, result_dir):
    os.environ['MASTER_ADDR'] = '127.0.0.1'
    os.environ['MASTER_PORT'] = str(port)
    os.environ['RANK'] = str(rank)
    os.environ['WORLD_SIZE'] = str(world_size)
    os.environ['LOCAL_RANK'] = str(rank)
    sys.path.insert(0, TESTS_DIR)
    from computations import TabularDataset, TabularTrainer, make_site_data
    from coinstac_dinunet_amd.config.keys import Mode
    from coinstac_dinunet_amd.parallel.cluster import RcclCluster

    local_kw = dict(task_id='tab', mode=Mode.TRAIN, batch_size=4, epochs=1,
                    validation_epochs=1, local_iterations=1,
                    split_ratio=(0.6, 0.2, 0.2), data_dir='data', num_class=2,
                    seed_all=True, patience=1, verbose=False)
    cluster = RcclCluster(root, local_kw=local_kw)
    make_site_data(cluster.site.as_dict(), n_samples=20, seed=rank)
    success, out = cluster.run(TabularTrainer, dataset_cls=TabularDataset,
                               max_rounds=200)
    assert success, f'rank {rank} did not reach SUCCESS'
    # dump final weights for cross-rank comparison
    net = cluster.site_cache['nn']['net']
    flat = torch.cat([p.detach().reshape(-1) for p in net.parameters()])
    np.save(os.path.join(result_dir, f'weights_rank{rank}.npy'), flat.numpy())
    import torch.distributed as dist
    dist.destroy_process_group()


def _free_port():
    s = socket.socket()
    s.bind(('127.0.0.1', 0))
    port = s.getsockname()[1]
    s.close()
    return port


def test_rccl_cluster_two_ranks_gloo(tmp_path):
    """Full phase machine on a 2-process gloo group (the GPU code path,
    CPU backend). Lock-step dSGD => identical final weights on all ranks."""
    result_dir = str(tmp_path / 'results')
    os.makedirs(result_dir)
    port = _free_port()
    mp.spawn(_rank_main, args=(2, port, str(tmp_path / 'cluster'), result_dir),
             nprocs=2, join=True)
    w0 = np.load(os.path.join(result_dir, 'weights_rank0.npy'))
    w1 = np.load(os.path.join(result_dir, 'weights_rank1.npy'))
    np.testing.assert_allclose(w0, w1, rtol=1e-5, atol=1e-6)


def _rank_kfold(rank, world, port, root, result_dir):
    os.environ.update(MASTER_ADDR='127.0.0.1', MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank))
    sys.path.insert(0, TESTS_DIR)
    from computations import TabularDataset, TabularTrainer, make_site_data
    from coinstac_dinunet_amd.config.keys import Key, Mode
    from coinstac_dinunet_amd.parallel.cluster import RcclCluster

    local_kw = dict(task_id='tab', mode=Mode.TRAIN, batch_size=4, epochs=1,
                    validation_epochs=1, local_iterations=1, num_folds=3,
                    data_dir='data', num_class=2, seed_all=True, patience=1,
                    verbose=False)
    cluster = RcclCluster(root, local_kw=local_kw)
    make_site_data(cluster.site.as_dict(), n_samples=18, seed=rank)
    success, out = cluster.run(TabularTrainer, dataset_cls=TabularDataset,
                               max_rounds=600)
    assert success, f'rank {rank}: k-fold run did not finish'
    if rank == 0:
        assert len(cluster.remote_cache[Key.GLOBAL_TEST_SERIALIZABLE]) == 3
    # per-fold checkpoints exist on this rank
    for fold in range(3):
        d = os.path.join(cluster.site.outputDirectory, 'tab', f'fold_{fold}')
        assert f'latest.tab-{fold}.pt' in os.listdir(d)
    import torch.distributed as dist
    dist.destroy_process_group()


def test_rccl_cluster_kfold_gloo(tmp_path):
    """3-fold cross-validation on the persistent process group: model
    re-init per fold, grad-arena rebuild, per-fold checkpoints."""
    result_dir = str(tmp_path / 'results2')
    os.makedirs(result_dir)
    port = _free_port()
    mp.spawn(_rank_kfold, args=(2, port, str(tmp_path / 'kf'), result_dir),
             nprocs=2, join=True)


def test_rccl_cluster_three_ranks_gloo(tmp_path):
    """Odd world size (3 sites, unequal data): quorum and the all-reduce
    mean must not assume a power-of-2 group."""
    result_dir = str(tmp_path / 'results3')
    os.makedirs(result_dir)
    port = _free_port()
    mp.spawn(_rank_main, args=(3, port, str(tmp_path / 'cluster3'),
                               result_dir), nprocs=3, join=True)
    w0 = np.load(os.path.join(result_dir, 'weights_rank0.npy'))
    for r in (1, 2):
        wr = np.load(os.path.join(result_dir, f'weights_rank{r}.npy'))
        np.testing.assert_allclose(w0, wr, rtol=1e-5, atol=1e-6)


def test_adaptive_bucket_sizing():
    """Default bucket size = arena/4 clamped to [2, 50] MB (r2: a fixed
    50 MB put every BASELINE model in ONE bucket — no overlap)."""
    import torch
    from coinstac_dinunet_amd.parallel.engine import FlatGradBuffer

    # 14 MB arena (VBM-like) -> ~4 buckets
    params = [torch.nn.Parameter(torch.zeros(875_000)) for _ in range(4)]
    buf = FlatGradBuffer(params)
    assert 3 <= len(buf.buckets) <= 5, len(buf.buckets)
    # tiny arena -> single bucket at the 2 MB floor
    small = [torch.nn.Parameter(torch.zeros(1000))]
    buf2 = FlatGradBuffer(small)
    assert len(buf2.buckets) == 1
    # explicit size still honored
    buf3 = FlatGradBuffer(params, bucket_bytes=64 * 1024 * 1024)
    assert len(buf3.buckets) == 1
