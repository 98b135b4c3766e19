"""Mid-run checkpoint/resume of the persistent cluster (gloo, 2 ranks)."""
import os
import socket
import sys

import torch
import torch.multiprocessing as mp

TESTS_DIR = os.path.dirname(os.path.abspath(__file__))


def _free_port():
    s = socket.socket()
    s.bind(('127.0.0.1', 0))
    p = s.getsockname()[1]
    s.close()
    return p


def _rank_main(rank, world, port, root, ckpt_dir):
    os.environ.update(MASTER_ADDR='127.0.0.1', MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank))
    sys.path.insert(0, TESTS_DIR)
    import torch.distributed as dist
    from computations import TabularDataset, TabularTrainer, make_site_data
    from coinstac_dinunet_amd.config.keys import Mode
    from coinstac_dinunet_amd.parallel.cluster import (RcclCluster,
                                                       load_cluster_state,
                                                       save_cluster_state)

    local_kw = dict(task_id='tab', mode=Mode.TRAIN, batch_size=4, epochs=2,
                    validation_epochs=1, local_iterations=1,
                    split_ratio=(0.6, 0.2, 0.2), data_dir='data', num_class=2,
                    seed_all=True, patience=2, verbose=False)
    cluster = RcclCluster(root, local_kw=local_kw)
    make_site_data(cluster.site.as_dict(), n_samples=16, seed=rank)

    # phase 1: run a handful of rounds, snapshot, remember weights
    for _ in range(6):
        cluster.run_round(TabularTrainer, dataset_cls=TabularDataset)
    ckpt = os.path.join(ckpt_dir, f'state_rank{rank}.pt')
    save_cluster_state(cluster, ckpt)
    w_before = torch.cat([p.detach().reshape(-1)
                          for p in cluster.site_cache['nn']['net'].parameters()])

    # phase 2: fresh cluster object, resume, verify weights survive and
    # the protocol still reaches SUCCESS
    cluster2 = RcclCluster(root, local_kw=local_kw)
    load_cluster_state(cluster2, ckpt, TabularTrainer,
                       dataset_cls=TabularDataset)
    w_after = torch.cat([p.detach().reshape(-1)
                         for p in cluster2.site_cache['nn']['net'].parameters()])
    assert torch.equal(w_before, w_after), 'weights lost across resume'

    success, _ = cluster2.run(TabularTrainer, dataset_cls=TabularDataset,
                              max_rounds=400)
    assert success, f'rank {rank}: resumed run never finished'
    dist.destroy_process_group()


def test_cluster_checkpoint_resume(tmp_path):
    ckpt_dir = str(tmp_path / 'ckpt')
    os.makedirs(ckpt_dir)
    mp.spawn(_rank_main, args=(2, _free_port(), str(tmp_path / 'c'),
                               ckpt_dir), nprocs=2, join=True)
