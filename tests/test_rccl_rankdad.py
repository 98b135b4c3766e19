"""RCCL-native rankDAD over a 2-process gloo group: single-round factor
exchange must keep site weights bit-aligned (seeded recompression)."""
import os
import socket
import sys

import numpy as np
import torch
import torch.multiprocessing as mp

TESTS_DIR = os.path.dirname(os.path.abspath(__file__))


def _free_port():
    s = socket.socket()
    s.bind(('127.0.0.1', 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _rank_main(rank, world, port, root, result_dir):
    os.environ.update(MASTER_ADDR='127.0.0.1', MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank))
    sys.path.insert(0, TESTS_DIR)
    from computations import TabularDataset, TabularTrainer, make_site_data
    from coinstac_dinunet_amd.config.keys import Mode
    from coinstac_dinunet_amd.parallel.cluster import RcclCluster
    from coinstac_dinunet_amd.parallel.rankdad import (RcclDADLearner,
                                                       RcclDADReducer)

    local_kw = dict(task_id='tab', mode=Mode.TRAIN, batch_size=4, epochs=1,
                    validation_epochs=1, local_iterations=1,
                    split_ratio=(0.6, 0.2, 0.2), data_dir='data', num_class=2,
                    seed_all=True, patience=1, verbose=False,
                    agg_engine='rcclDAD',
                    dad_reduction_rank=3,  # cat width 2*3 > 3 -> recompress
                    dad_num_pow_iters=5, seed=3)
    cluster = RcclCluster(root, local_kw=local_kw)
    make_site_data(cluster.site.as_dict(), n_samples=16, seed=rank)

    import torch.distributed as dist
    from coinstac_dinunet_amd.distrib.nodes.local import COINNLocal

    success, out = False, {}
    for _ in range(200):
        local = COINNLocal(cache=cluster.site_cache,
                           input=dict(cluster.input),
                           state=cluster.site.as_dict(), **cluster.local_kw)
        result = local(None, TabularTrainer, dataset_cls=TabularDataset,
                       learner_cls=RcclDADLearner)
        my_out = result['output']
        gathered = [None] * world
        dist.all_gather_object(gathered, (cluster.site.clientId, my_out))
        site_outs = dict(gathered)
        bcast = [None]
        if rank == 0:
            from coinstac_dinunet_amd.distrib.nodes.remote import COINNRemote
            cluster._site_transfer_to_remote(site_outs)
            remote = COINNRemote(cache=cluster.remote_cache, input=site_outs,
                                 state=cluster.remote_state)
            rres = remote(None, TabularTrainer, reducer_cls=RcclDADReducer)
            cluster._remote_transfer_to_sites(site_outs)
            bcast = [(rres['output'], rres.get('success', False))]
        dist.broadcast_object_list(bcast, src=0)
        remote_out, success = bcast[0]
        cluster.input = dict(remote_out)
        if success:
            break
    assert success, f'rank {rank}: no SUCCESS'
    assert cluster.site_cache['dad_iter'] > 0, 'DAD rounds never stepped'
    net = cluster.site_cache['nn']['net']
    flat = torch.cat([p.detach().reshape(-1) for p in net.parameters()])
    np.save(os.path.join(result_dir, f'w{rank}.npy'), flat.numpy())
    dist.destroy_process_group()


def test_rccl_rankdad_two_ranks(tmp_path):
    result_dir = str(tmp_path / 'res')
    os.makedirs(result_dir)
    mp.spawn(_rank_main, args=(2, _free_port(), str(tmp_path / 'c'),
                               result_dir), nprocs=2, join=True)
    w0 = np.load(os.path.join(result_dir, 'w0.npy'))
    w1 = np.load(os.path.join(result_dir, 'w1.npy'))
    np.testing.assert_allclose(w0, w1, rtol=1e-5, atol=1e-6)


def test_seeded_recompression_is_deterministic():
    """Two independent processes recompressing identical gathered factors
    with the same seed must produce bit-identical results — the property
    RcclDADLearner relies on to skip a second collective."""
    from coinstac_dinunet_amd.distrib.rankdad import power_iteration_BC
    torch.manual_seed(11)
    true_rank = 6
    B = torch.randn(24, true_rank) @ torch.randn(true_rank, 40)  # [n, k]
    C = torch.randn(18, true_rank) @ torch.randn(true_rank, 40)  # [m, k]
    g1 = torch.Generator().manual_seed(123)
    g2 = torch.Generator().manual_seed(123)
    bf1, cf1 = power_iteration_BC(B, C, rank=3, generator=g1)
    bf2, cf2 = power_iteration_BC(B, C, rank=3, generator=g2)
    assert torch.equal(bf1, bf2) and torch.equal(cf1, cf2)
    # and the factors still approximate the dominant structure
    G = B @ C.t()
    approx = bf1 @ cf1.t()
    u, s, v = torch.linalg.svd(G)
    best3 = (u[:, :3] * s[:3]) @ v[:3, :]
    assert torch.norm(approx - G) <= 1.5 * torch.norm(best3 - G) + 1e-4
