"""End-to-end example: VBM 3D-CNN k-fold training on N GPU-sites.

This is what a user computation looks like (the analog of the reference's
external dinunet_implementations repos): a COINNDataset that reads one
volume per file, a COINNTrainer with the model + iteration hook, and the
cluster launcher.

Run (1 node, N GPUs = N sites):
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 examples/vbm_computation.py --root /tmp/vbmrun

With no real data present, --synthesize writes random 48^3 volumes so the
whole pipeline (splits, folds, lock-step dSGD over RCCL, checkpointing,
global metrics, results zip) can be exercised anywhere.
"""
import argparse
import os
import sys

import numpy as np
import torch

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                '..'))

from coinstac_dinunet_amd import COINNDataset, COINNTrainer  # noqa: E402
from coinstac_dinunet_amd import ops  # noqa: E402
from coinstac_dinunet_amd.config.keys import Mode  # noqa: E402
from coinstac_dinunet_amd.models import VBMNet  # noqa: E402
from coinstac_dinunet_amd.parallel.cluster import RcclCluster  # noqa: E402


class VBMDataset(COINNDataset):
    def load_index(self, file):
        self.indices.append(file)

    def __getitem__(self, ix):
        rec = np.load(os.path.join(self.path(), self.cache['data_dir'],
                                   self.indices[ix]), allow_pickle=True).item()
        return {'inputs': torch.from_numpy(rec['x']).unsqueeze(0),
                'labels': torch.tensor(rec['y'], dtype=torch.long)}


class VBMTrainer(COINNTrainer):
    def _init_nn_model(self):
        self.nn['vbm'] = VBMNet(in_channels=1,
                                num_class=self.cache.get('num_class', 2))

    def iteration(self, batch):
        dev = self.device['gpu']
        x = batch['inputs'].to(dev).float()
        y = batch['labels'].to(dev).long()
        if dev.type == 'cuda':
            with torch.autocast('cuda', dtype=torch.bfloat16):
                out = self.nn['vbm'](x)
            loss = ops.cross_entropy(out.float(), y)
            pred = ops.argmax_rows(out.float())
        else:
            out = self.nn['vbm'](x)
            loss = ops.cross_entropy(out, y)
            pred = torch.argmax(out, 1)
        avg = self.new_averages()
        avg.add(loss.item(), len(x))
        metrics = self.new_metrics()
        metrics.add(pred, y)
        return {'loss': loss, 'averages': avg, 'metrics': metrics,
                'output': pred}


def synthesize(data_dir, n=40, side=48, seed=0):
    rng = np.random.RandomState(seed)
    os.makedirs(data_dir, exist_ok=True)
    for i in range(n):
        y = i % 2
        x = (rng.randn(side, side, side) * 0.5 + y).astype(np.float32)
        np.save(os.path.join(data_dir, f'vol_{i:04d}.npy'),
                {'x': x, 'y': y}, allow_pickle=True)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--root', default='/tmp/vbmrun')
    ap.add_argument('--folds', type=int, default=3)
    ap.add_argument('--epochs', type=int, default=2)
    ap.add_argument('--batch', type=int, default=8)
    ap.add_argument('--synthesize', action='store_true', default=True)
    args = ap.parse_args()

    cluster = RcclCluster(args.root, local_kw=dict(
        task_id='vbm', mode=Mode.TRAIN, batch_size=args.batch,
        epochs=args.epochs, validation_epochs=1, num_folds=args.folds,
        data_dir='data', num_class=2, patience=args.epochs, verbose=False))
    if args.synthesize:
        synthesize(os.path.join(cluster.site.baseDirectory, 'data'),
                   seed=cluster.rank)
    success, out = cluster.run(VBMTrainer, dataset_cls=VBMDataset)
    if cluster.rank == 0:
        zips = [f for f in os.listdir(cluster.site.outputDirectory)
                if f.endswith('.zip')]
        print('SUCCESS' if success else 'DID NOT FINISH',
              '| results zip:', zips[0] if zips else None)


if __name__ == '__main__':
    main()
