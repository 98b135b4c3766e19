"""End-to-end example: FreeSurfer-style tabular MLP classification.

The analog of the reference's dinunet_implementations FreeSurfer demo: a
per-subject feature vector (e.g. 66 FreeSurfer volumetric features) and a
binary label, trained across sites with any aggregation engine.

Three ways to run the same computation:

  1. CPU platform emulation (the reference's file/JSON relay), any machine:
       python examples/freesurfer_mlp_computation.py --mode loopback

  2. N persistent GPU-sites over RCCL (one rank per MI355X):
       python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
           --master-addr 127.0.0.1 examples/freesurfer_mlp_computation.py \
           --mode rccl

  3. Single-site standalone debugging (no aggregator at all):
       python examples/freesurfer_mlp_computation.py --mode site

Pass --engine powerSGD to train with low-rank gradient compression
(two-phase over loopback; single-round collectives over RCCL).
"""
import argparse
import os
import sys

import numpy as np
import torch

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                '..'))

from coinstac_dinunet_amd import (COINNDataset, COINNLocal, COINNRemote,
                                  COINNTrainer, ops)  # noqa: E402
from coinstac_dinunet_amd.config.keys import Mode  # noqa: E402
from coinstac_dinunet_amd.models import FreeSurferMLP  # noqa: E402

N_FEATURES = 66


class FreeSurferDataset(COINNDataset):
    """One .npy record per subject: {'x': float32 [66], 'y': 0/1}."""

    def load_index(self, file):
        self.indices.append(file)

    def __getitem__(self, ix):
        rec = np.load(os.path.join(self.path(), self.cache['data_dir'],
                                   self.indices[ix]), allow_pickle=True).item()
        return {'inputs': torch.from_numpy(rec['x']),
                'labels': torch.tensor(rec['y'], dtype=torch.long)}


class FreeSurferTrainer(COINNTrainer):
    def _init_nn_model(self):
        self.nn['fs'] = FreeSurferMLP(in_features=N_FEATURES,
                                      hidden_sizes=(256, 128, 64),
                                      num_class=self.cache.get('num_class', 2),
                                      dropout=0.3)

    def iteration(self, batch):
        dev = self.device['gpu']
        x = batch['inputs'].to(dev).float()
        y = batch['labels'].to(dev).long()
        out = self.nn['fs'](x)
        loss = ops.cross_entropy(out, y)
        pred = torch.argmax(out, 1)
        avg, metrics = self.new_averages(), self.new_metrics()
        avg.add(loss.item(), len(x))
        metrics.add(pred, y)
        return {'loss': loss, 'averages': avg, 'metrics': metrics,
                'output': pred}


def synthesize(base_dir, n=64, seed=0):
    rng = np.random.RandomState(seed)
    data_dir = os.path.join(base_dir, 'data')
    os.makedirs(data_dir, exist_ok=True)
    w = rng.randn(N_FEATURES)
    for i in range(n):
        x = rng.randn(N_FEATURES).astype(np.float32)
        np.save(os.path.join(data_dir, f'subject_{i:04d}.npy'),
                {'x': x, 'y': int(x @ w > 0)}, allow_pickle=True)


def local_kw(args):
    return dict(task_id='fs', mode=Mode.TRAIN, batch_size=16,
                epochs=args.epochs, validation_epochs=1, local_iterations=1,
                split_ratio=(0.7, 0.15, 0.15), data_dir='data', num_class=2,
                seed_all=True, patience=5, agg_engine=args.engine,
                verbose=True)


def run_loopback(args):
    from coinstac_dinunet_amd.simulator import LoopbackCluster
    cluster = LoopbackCluster(
        args.root, n_sites=args.sites,
        site_data=lambda s: synthesize(s.baseDirectory,
                                       seed=int(s.clientId[-1])))
    kw = local_kw(args)
    success, out = cluster.run(
        lambda cache, input, state: COINNLocal(cache=cache, input=input,
                                               state=state, **kw),
        lambda cache, input, state: COINNRemote(cache=cache, input=input,
                                                state=state),
        FreeSurferTrainer, dataset_cls=FreeSurferDataset, max_rounds=2000)
    print('success:', success)


def run_rccl(args):
    from coinstac_dinunet_amd.parallel.cluster import RcclCluster
    cluster = RcclCluster(args.root, local_kw=local_kw(args))
    synthesize(cluster.site.baseDirectory,
               seed=int(os.environ.get('RANK', 0)))
    success, _ = cluster.run(FreeSurferTrainer,
                             dataset_cls=FreeSurferDataset, max_rounds=2000)
    print('success:', success)


def run_site(args):
    from coinstac_dinunet_amd.site_runner import SiteRunner
    runner = SiteRunner(task_id='fs', data_path=args.root,
                        **{k: v for k, v in local_kw(args).items()
                           if k not in ('task_id',)})
    synthesize(os.path.join(args.root, 'input', 'local0', 'simulatorRun'))
    runner.run(FreeSurferTrainer, dataset_cls=FreeSurferDataset)


if __name__ == '__main__':
    ap = argparse.ArgumentParser()
    ap.add_argument('--mode', choices=['loopback', 'rccl', 'site'],
                    default='loopback')
    ap.add_argument('--root', default='/tmp/fs_run')
    ap.add_argument('--sites', type=int, default=2)
    ap.add_argument('--epochs', type=int, default=5)
    ap.add_argument('--engine', default='dSGD',
                    choices=['dSGD', 'powerSGD', 'rankDAD', 'fedAvg'])
    args = ap.parse_args()
    {'loopback': run_loopback, 'rccl': run_rccl, 'site': run_site}[args.mode](args)
