"""FedAvg engine: weight averaging instead of gradient averaging.

This is the custom-engine extension point exercised (the reference lets
users inject learner_cls/reducer_cls through compute() and fall through
the AGG_Engine map — local.py:276-287, README.md:102). FedAvg ships the
model WEIGHTS after local_iterations micro-batches of local training and
the remote publishes the parameter-wise mean, which every site loads.

Used by BASELINE.json config 5 (ResNet-18 + custom FedAvg reducer).
"""
from os import sep as _sep

import numpy as _np
import torch as _torch

from ..utils import tensorutils as _tu
from .learner import COINNLearner
from .reducer import COINNReducer

WEIGHTS_FILE = 'fedavg_weights.npy'
AVG_WEIGHTS_FILE = 'fedavg_avg_weights.npy'


class FedAvgLearner(COINNLearner):
    def step(self):
        """Load the remote's averaged weights into the model."""
        out = {}
        avg = _tu.load_arrays(self.state['baseDirectory'] + _sep +
                              self.input['avg_weights_file'])
        model = self.trainer.nn[self.first_model]
        with _torch.no_grad():
            for p, w in zip(model.parameters(), avg):
                p.copy_(_torch.tensor(_np.asarray(w), dtype=p.dtype,
                                      device=p.device))
        return out

    def to_reduce(self):
        """Local steps (fwd/bwd/optim per micro-batch), then ship weights."""
        out = {}
        model = self.trainer.nn[self.first_model]
        optim = self.trainer.optimizer[self.first_optim]
        model.train()
        its = []
        for _ in range(self.cache.get('local_iterations', 1)):
            optim.zero_grad()
            batch, nxt_iter_out = self.trainer.data_handle.next_iter()
            it = self.trainer.iteration(batch)
            it['loss'].backward()
            optim.step()
            its.append(it)
            out.update(**nxt_iter_out)
            if nxt_iter_out.get('mode'):
                break
        weights = [p.detach().cpu().numpy().astype(self.dtype)
                   for p in model.parameters()]
        _tu.save_arrays(self.state['transferDirectory'] + _sep + WEIGHTS_FILE,
                        weights)
        out['weights_file_fedavg'] = WEIGHTS_FILE
        out['reduce'] = True
        return self.trainer.reduce_iteration(its), out


class FedAvgReducer(COINNReducer):
    def reduce(self):
        out = {'avg_weights_file': AVG_WEIGHTS_FILE}
        _tu.save_arrays(self.state['transferDirectory'] + _sep +
                        AVG_WEIGHTS_FILE,
                        self._average('weights_file_fedavg'))
        out['update'] = True
        return out
