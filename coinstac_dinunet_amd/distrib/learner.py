"""dSGD site-side learner (loopback transport).

Behavior-parity target: /root/reference/coinstac_dinunet/distrib/learner.py:9-59
(COINNLearner: step() -> dict, backward() -> (it, out), to_reduce() -> (it, out)).
The body is deliberately structured differently from the reference: the wire
I/O lives in two private helpers (`_ship_grads` / `_adopt_grads`) shared with
subclasses, the micro-batch loop is an iterator expression over the cursor
handle, and gradient assignment aligns by the *same* grad-present filter used
on the extract side — the reference assigns positionally over all parameters
(learner.py:25-26) which silently shifts every index after a frozen/grad-less
parameter (documented deviation; fixes ADVICE r1 finding on tensorutils).

This class implements the file hand-off used by the CPU plumbing tests and by
custom user engines; the MI355X hot path is parallel.engine.RcclLearner, which
replaces the grad exchange with a fused-bucket RCCL all-reduce(avg) over xGMI
(no host round trip, overlapped with backward on a side HIP stream).
"""
import os as _os

import numpy as _np
import torch as _torch

from .. import config as _conf
from ..utils import tensorutils as _tu


class COINNLearner:
    """One aggregation engine's site half: run local micro-batches, ship
    gradients out, adopt the remote's average back in."""

    def __init__(self, trainer=None, mp_pool=None, **kw):
        self.trainer = trainer
        self.cache = trainer.cache
        self.input = trainer.input
        self.state = trainer.state
        self.global_modes = self.input.get('global_modes', {})
        self.pool = mp_pool
        self.dtype = f"float{self.cache.get('precision_bits', 32)}"
        self.device = trainer.device.get('gpu', _torch.device('cpu'))

    # ---- model/optimizer handles ---------------------------------------
    @property
    def first_model(self):
        return next(iter(self.trainer.nn))

    @property
    def first_optim(self):
        return next(iter(self.trainer.optimizer))

    def _model(self):
        return self.trainer.nn[self.first_model]

    # ---- wire helpers (loopback .npy format) ---------------------------
    def _inbox(self, file_key):
        """Path of a file the remote placed in our baseDirectory."""
        return _os.path.join(self.state['baseDirectory'], self.input[file_key])

    def _outbox(self, name):
        """Path for a file we emit into transferDirectory."""
        return _os.path.join(self.state['transferDirectory'], name)

    def _ship_grads(self):
        """Serialize local gradients for the remote (grad-present filter)."""
        grads = _tu.extract_grads(self._model(), dtype=self.dtype)
        _tu.save_arrays(self._outbox(_conf.grads_file), grads)

    def _adopt_grads(self, grads):
        """Write averaged gradient arrays into param.grad, aligned by the
        grad-present filter (see module docstring)."""
        with_grad = [p for p in self._model().parameters()
                     if p.grad is not None]
        targets = with_grad if len(with_grad) == len(grads) \
            else list(self._model().parameters())
        for param, g in zip(targets, grads):
            param.grad = _torch.tensor(
                _np.asarray(g), dtype=_torch.float32).to(self.device)

    # ---- engine protocol -----------------------------------------------
    def step(self):
        """Apply the remote's averaged gradients, then optimizer.step()."""
        grads = _tu.load_arrays(self._inbox('avg_grads_file'))
        self._adopt_grads(grads)
        self.trainer.optimizer[self.first_optim].step()
        return {}

    def backward(self):
        """local_iterations micro-batches of fwd/bwd; grads accumulate."""
        out = {}
        self._model().train()
        self.trainer.optimizer[self.first_optim].zero_grad()
        its = []
        for _ in range(self.cache.get('local_iterations', 1)):
            batch, cursor_out = self.trainer.data_handle.next_iter()
            it = self.trainer.iteration(batch)
            it['loss'].backward()
            its.append(it)
            out.update(**cursor_out)
        return self.trainer.reduce_iteration(its), out

    def to_reduce(self):
        """backward + ship gradients (grads.npy into transferDirectory)."""
        it, out = self.backward()
        self._ship_grads()
        out['grads_file'] = _conf.grads_file
        out['reduce'] = True
        return it, out
