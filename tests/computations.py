"""Shared synthetic computations for tests: tabular MLP classification.

Mimics a user computation repo: a COINNDataset reading per-sample .npy
files from baseDirectory/data_dir, and a COINNTrainer with an MLP.
"""
import os

import numpy as np
import torch

from coinstac_dinunet_amd import COINNDataset, COINNTrainer
from coinstac_dinunet_amd.models import FreeSurferMLP

N_FEATURES = 16


def make_site_data(state, n_samples=24, n_features=N_FEATURES, seed=0,
                   n_classes=2):
    """Write per-sample .npy files (features + label) into baseDirectory."""
    rng = np.random.RandomState(seed)
    data_dir = os.path.join(state['baseDirectory']
                            if isinstance(state, dict) else state.baseDirectory,
                            'data')
    os.makedirs(data_dir, exist_ok=True)
    w = rng.randn(n_features, n_classes - 1)
    for i in range(n_samples):
        x = rng.randn(n_features).astype(np.float32)
        if n_classes == 2:
            y = int(x @ w[:, 0] > 0)
        else:
            scores = np.concatenate([[0.0], x @ w])
            y = int(np.argmax(scores))
        np.save(os.path.join(data_dir, f'sample_{i:03d}.npy'),
                {'x': x, 'y': y}, allow_pickle=True)


class TabularDataset(COINNDataset):
    def load_index(self, file):
        self.indices.append(file)

    def __getitem__(self, ix):
        file = self.indices[ix]
        rec = np.load(os.path.join(self.state['baseDirectory'],
                                   self.cache.get('data_dir', 'data'), file),
                      allow_pickle=True).item()
        return {'inputs': torch.from_numpy(rec['x']),
                'labels': torch.tensor(rec['y'], dtype=torch.long)}


class TabularTrainer(COINNTrainer):
    def _init_nn_model(self):
        self.nn['net'] = FreeSurferMLP(
            in_features=N_FEATURES,
            hidden_sizes=tuple(self.cache.get('hidden_sizes', (32, 16))),
            num_class=self.cache.get('num_class', 2),
            dropout=0.0)

    def iteration(self, batch):
        from coinstac_dinunet_amd import ops
        dev = self.device.get('gpu', torch.device('cpu'))
        inputs = batch['inputs'].to(dev).float()
        labels = batch['labels'].to(dev).long()
        out = self.nn['net'](inputs)
        loss = ops.cross_entropy(out, labels)
        pred = torch.argmax(out, 1)
        avg = self.new_averages()
        avg.add(loss.item(), len(inputs))
        metrics = self.new_metrics()
        if self.cache.get('monitor_metric') == 'auc':
            metrics.add(torch.softmax(out.detach(), 1)[:, 1], labels)
        else:
            metrics.add(pred, labels)
        return {'loss': loss, 'averages': avg, 'metrics': metrics,
                'output': pred}
