"""BASELINE config 5: ResNet-18 with the custom FedAvg reducer, driven
through the persistent cluster (gloo 2-rank here; RCCL on the node)."""
import os
import socket
import sys

import numpy as np
import torch

TESTS_DIR = os.path.dirname(os.path.abspath(__file__))


def _free_port():
    s = socket.socket()
    s.bind(('127.0.0.1', 0))
    p = s.getsockname()[1]
    s.close()
    return p


def _rank_main(rank, world, port, root):
    os.environ.update(MASTER_ADDR='127.0.0.1', MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank))
    sys.path.insert(0, TESTS_DIR)
    import torch.distributed as dist
    from coinstac_dinunet_amd import COINNDataset, COINNTrainer, ops
    from coinstac_dinunet_amd.config.keys import Key, Mode
    from coinstac_dinunet_amd.distrib.fedavg import (FedAvgLearner,
                                                     FedAvgReducer)
    from coinstac_dinunet_amd.models import ResNet18
    from coinstac_dinunet_amd.parallel.cluster import RcclCluster

    class TinyImgDataset(COINNDataset):
        def load_index(self, file):
            self.indices.append(file)

        def __getitem__(self, ix):
            rec = np.load(os.path.join(self.state['baseDirectory'],
                                       self.cache['data_dir'],
                                       self.indices[ix]),
                          allow_pickle=True).item()
            return {'inputs': torch.from_numpy(rec['x']),
                    'labels': torch.tensor(rec['y'], dtype=torch.long)}

    class ResNetTrainer(COINNTrainer):
        def _init_nn_model(self):
            self.nn['net'] = ResNet18(in_channels=3, num_class=2,
                                      widths=(8, 8, 16, 16))

        def iteration(self, batch):
            dev = self.device['gpu']
            x = batch['inputs'].to(dev).float()
            y = batch['labels'].to(dev).long()
            out = self.nn['net'](x)
            loss = ops.cross_entropy(out, y)
            avg = self.new_averages()
            avg.add(loss.item(), len(x))
            metrics = self.new_metrics()
            metrics.add(torch.argmax(out, 1), y)
            return {'loss': loss, 'averages': avg, 'metrics': metrics,
                    'output': out}

    cluster = RcclCluster(root, local_kw=dict(
        task_id='rn', mode=Mode.TRAIN, batch_size=4, epochs=1,
        validation_epochs=1, local_iterations=2,
        split_ratio=(0.6, 0.2, 0.2), data_dir='data', num_class=2,
        patience=1, verbose=False, agg_engine='fedAvg'))

    rng = np.random.RandomState(rank)
    data_dir = os.path.join(cluster.site.baseDirectory, 'data')
    os.makedirs(data_dir, exist_ok=True)
    for i in range(12):
        y = i % 2
        x = (rng.randn(3, 32, 32) * 0.5 + y).astype(np.float32)
        np.save(os.path.join(data_dir, f's_{i:03d}.npy'),
                {'x': x, 'y': y}, allow_pickle=True)

    success, _ = cluster.run(ResNetTrainer, dataset_cls=TinyImgDataset,
                             max_rounds=300, learner_cls=FedAvgLearner,
                             reducer_cls=FedAvgReducer)
    assert success, f'rank {rank}: ResNet/FedAvg run did not finish'
    if rank == 0:
        assert cluster.remote_cache[Key.GLOBAL_TEST_SERIALIZABLE]
    dist.destroy_process_group()


def test_resnet18_fedavg_cluster(tmp_path):
    import torch.multiprocessing as mp
    try:
        mp.spawn(_rank_main, args=(2, _free_port(), str(tmp_path / 'rn')),
                 nprocs=2, join=True)
    except Exception:
        # one retry on a fresh port/dir: the heaviest spawn test in the
        # suite occasionally hits transient rendezvous failures under full
        # -suite load; a deterministic failure fails both attempts
        mp.spawn(_rank_main, args=(2, _free_port(), str(tmp_path / 'rn2')),
                 nprocs=2, join=True)
