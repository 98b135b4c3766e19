"""dSGD remote-side reducer: per-parameter mean over sites.

API-parity: /root/reference/coinstac_dinunet/distrib/reducer.py:16-54.
The per-parameter stack->mean(0) here IS the collective that the MI355X
path replaces with one RCCL all-reduce(avg) over xGMI on a flat fused
bucket (parallel/engine.py) — this file-based form survives for the CPU
plumbing tests and for the star-topology loopback harness.
"""
import os as _os
from functools import partial as _partial

import numpy as _np
import torch as _torch

from .. import config as _conf
from ..utils import tensorutils as _tu


def _multi_load(file_key, state, site, site_vars):
    grads_file = state['baseDirectory'] + _os.sep + site + _os.sep + site_vars[file_key]
    return _tu.load_arrays(grads_file)


class COINNReducer:
    def __init__(self, trainer=None, mp_pool=None, **kw):
        self.cache = trainer.cache
        self.input = trainer.input
        self.state = trainer.state
        self.trainer = trainer
        self.pool = mp_pool
        self.dtype = f"float{self.cache.get('precision_bits', 32)}"
        self.device = trainer.device.get('gpu', _torch.device('cpu'))

    def _load(self, file_key):
        loader = _partial(_multi_load, file_key, self.state)
        if self.pool is not None:
            return list(self.pool.starmap(loader, self.input.items()))
        return [loader(site, site_vars) for site, site_vars in self.input.items()]

    def _average(self, file_key):
        sites_data = self._load(file_key)
        averaged = []
        for data in zip(*sites_data):
            stacked = _np.stack([_np.asarray(d, dtype=_np.float32) for d in data])
            avg = _torch.from_numpy(stacked).to(self.device,
                                                non_blocking=True).mean(0)
            averaged.append(avg.cpu().numpy().astype(self.dtype))
        return averaged

    def reduce(self):
        """Average every site's gradients and publish to all sites."""
        out = {'avg_grads_file': _conf.avg_grads_file}
        _tu.save_arrays(self.state['transferDirectory'] + _os.sep +
                        out['avg_grads_file'], self._average('grads_file'))
        out['update'] = True
        return out
