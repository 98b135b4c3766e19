"""RCCL-native PowerSGD over a 2-process gloo group."""
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
    from coinstac_dinunet_amd.parallel.powersgd import (RcclPowerSGDLearner,
                                                        RcclPowerSGDReducer)

    local_kw = dict(task_id='tab', mode=Mode.TRAIN, batch_size=4, epochs=1,
                    validation_epochs=1, local_iterations=1,
                    split_ratio=(0.6, 0.2, 0.2), data_dir='data', num_class=2,
                    seed_all=True, patience=1, verbose=False,
                    agg_engine='rcclPowerSGD',  # falls through to learner_cls
                    matrix_approximation_rank=2, start_powerSGD_iter=2,
                    seed=3)
    cluster = RcclCluster(root, local_kw=local_kw)
    make_site_data(cluster.site.as_dict(), n_samples=16, seed=rank)

    # drive rounds manually so both learner_cls and reducer_cls inject
    import torch.distributed as dist
    from coinstac_dinunet_amd.distrib.nodes.local import COINNLocal

    success, out = False, {}
    for _ in range(200):
        local = COINNLocal(cache=cluster.site_cache,
                           input=dict(cluster.input),
                           state=cluster.site.as_dict(), **cluster.local_kw)
        result = local(None, TabularTrainer, dataset_cls=TabularDataset,
                       learner_cls=RcclPowerSGDLearner)
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
            rres = remote(None, TabularTrainer,
                          reducer_cls=RcclPowerSGDReducer)
            cluster._remote_transfer_to_sites(site_outs)
            bcast = [(rres['output'], rres.get('success', False))]
        dist.broadcast_object_list(bcast, src=0)
        remote_out, success = bcast[0]
        cluster.input = dict(remote_out)
        if success:
            break
    assert success, f'rank {rank}: no SUCCESS'
    assert cluster.site_cache['powerSGD_iter'] > 2, 'compression never engaged'
    net = cluster.site_cache['nn']['net']
    flat = torch.cat([p.detach().reshape(-1) for p in net.parameters()])
    np.save(os.path.join(result_dir, f'w{rank}.npy'), flat.numpy())
    dist.destroy_process_group()


def test_rccl_powersgd_two_ranks(tmp_path):
    result_dir = str(tmp_path / 'res')
    os.makedirs(result_dir)
    mp.spawn(_rank_main, args=(2, _free_port(), str(tmp_path / 'c'),
                               result_dir), nprocs=2, join=True)
    w0 = np.load(os.path.join(result_dir, 'w0.npy'))
    w1 = np.load(os.path.join(result_dir, 'w1.npy'))
    np.testing.assert_allclose(w0, w1, rtol=1e-4, atol=1e-5)
