"""dSGD site-side learner.

API-parity: /root/reference/coinstac_dinunet/distrib/learner.py:9-59
(COINNLearner.step()->dict, backward()->(it,out), to_reduce()->(it,out)).
This class implements the loopback (file) hand-off used by the CPU plumbing
tests and by custom user engines; the MI355X hot path is
parallel.engine.RcclLearner, which overrides the grad exchange with a
fused-bucket RCCL all-reduce(avg) over xGMI (no host round trip, overlapped
with backward on a side HIP stream).
"""
from os import sep as _sep

import numpy as _np
import torch as _torch

from .. import config as _conf
from ..utils import tensorutils as _tu


class COINNLearner:
    def __init__(self, trainer=None, mp_pool=None, **kw):
        self.cache = trainer.cache
        self.input = trainer.input
        self.state = trainer.state
        self.trainer = trainer
        self.global_modes = self.input.get('global_modes', {})
        self.pool = mp_pool
        self.dtype = f"float{self.cache.get('precision_bits', 32)}"
        self.device = trainer.device.get('gpu', _torch.device('cpu'))

    @property
    def first_model(self):
        return list(self.trainer.nn.keys())[0]

    @property
    def first_optim(self):
        return list(self.trainer.optimizer.keys())[0]

    def step(self):
        """Apply the remote's averaged gradients, then optimizer.step()."""
        out = {}
        grads = _tu.load_arrays(self.state['baseDirectory'] + _sep +
                                self.input['avg_grads_file'])
        model = self.trainer.nn[self.first_model]
        for i, param in enumerate(model.parameters()):
            param.grad = _torch.tensor(_np.asarray(grads[i]),
                                       dtype=_torch.float32).to(self.device)
        self.trainer.optimizer[self.first_optim].step()
        return out

    def backward(self):
        """local_iterations micro-batches of fwd/bwd; grads accumulate."""
        out = {}
        self.trainer.nn[self.first_model].train()
        self.trainer.optimizer[self.first_optim].zero_grad()
        its = []
        for _ in range(self.cache.get('local_iterations', 1)):
            batch, nxt_iter_out = self.trainer.data_handle.next_iter()
            it = self.trainer.iteration(batch)
            it['loss'].backward()
            its.append(it)
            out.update(**nxt_iter_out)
        return self.trainer.reduce_iteration(its), out

    def to_reduce(self):
        """backward + ship gradients (grads.npy into transferDirectory)."""
        it, out = self.backward()
        out['grads_file'] = _conf.grads_file
        grads = _tu.extract_grads(self.trainer.nn[self.first_model],
                                  dtype=self.dtype)
        _tu.save_arrays(self.state['transferDirectory'] + _sep +
                        out['grads_file'], grads)
        out['reduce'] = True
        return it, out
