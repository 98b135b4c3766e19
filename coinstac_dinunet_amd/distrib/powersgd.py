"""PowerSGD low-rank gradient-compression engine.

Protocol parity: /root/reference/coinstac_dinunet/distrib/powersgd/__init__.py
(PowerSGDLearner:51-181, PowerSGDReducer:184-219): plain dSGD during the
warmup (start_powerSGD_iter) then a two-phase round trip per step —
  phase_P_sync: backward; M = grad + error feedback; Q ~ seeded randn
                (identical across sites), orthogonalized; ship P = M @ Q;
  phase_Q_sync: receive averaged P, orthogonalize; ship Q = M^T @ P and
                the rank-1 (ndim<=1) grads;
  step:         receive averaged Q + rank-1 grads; grad ~= P @ Q^T;
                error = M - reconstruction; optimizer.step().
Reducer side: average P -> flip to Q phase; average Q + rank-1 -> update.

The skinny GEMMs (P = M@Q etc.) run through rocBLAS (plain library GEMMs);
the Gram-Schmidt orthogonalization is the fused HIP kernel K8 when the
extension is loaded (single launch per matrix vs 3*rank elementwise ops).
"""
import os as _os

import numpy as _np
import torch as _torch

from ..utils import tensorutils as _tu
from .. import ops as _ops
from .learner import COINNLearner
from .reducer import COINNReducer

P_FILE = 'powerSGD_P.npy'
Q_FILE = 'powerSGD_Q.npy'
R1_FILE = 'powerSGD_rank1.npy'
P_AGG = 'powerSGD_P_AGG.npy'
Q_AGG = 'powerSGD_Q_AGG.npy'
R1_AGG = 'powerSGD_rank1_AGG.npy'

PHASE_P = 'phase_P_sync'
PHASE_Q = 'phase_Q_sync'


def orthogonalize(mat, eps=1e-8):
    """Column-wise Gram-Schmidt in place (parity: powersgd:15-38)."""
    if mat.is_cuda:
        from .. import ops
        if ops.native_available():
            C = ops.require_native()
            if hasattr(C, 'gram_schmidt'):
                C.gram_schmidt(mat, eps)
                return mat
    n_cols = mat.shape[1]
    for i in range(n_cols):
        col = mat[:, i: i + 1]
        col.div_(_torch.norm(col) + eps)
        if i + 1 < n_cols:
            rest = mat[:, i + 1:]
            rest.sub_(col @ (col.t() @ rest))
    return mat


class PowerSGDLearner(COINNLearner):
    def __init__(self, trainer=None, mp_pool=None, **kw):
        super().__init__(trainer=trainer, mp_pool=mp_pool, **kw)
        self.rank = self.cache.get('matrix_approximation_rank', 1)
        self.start_iter = self.cache.get('start_powerSGD_iter', 10)
        self.use_error_feedback = self.cache.get('use_error_feedback', True)
        self.warm_start = self.cache.setdefault('warm_start', True)
        self.seed = self.cache.get('seed', 0)
        self.cache.setdefault('powerSGD_iter', 0)

    # ---- helpers -------------------------------------------------------
    def _param_tensors(self):
        model = self.trainer.nn[self.first_model]
        return list(model.parameters())

    def _split(self, tensors):
        """(index, tensor) lists: rank-1 (ndim<=1) vs matrices (2D views)."""
        rank1, mats = [], []
        for i, p in enumerate(tensors):
            if p.ndim <= 1:
                rank1.append((i, p))
            else:
                mats.append((i, p))
        return rank1, mats

    def _matrix_view(self, g):
        return g.reshape(g.shape[0], -1)

    @property
    def _warmup(self):
        return self.cache['powerSGD_iter'] < self.start_iter

    # ---- protocol ------------------------------------------------------
    def step(self):
        if self._warmup or not self.input.get('powerSGD_applied'):
            out = super().step()
            self.cache['powerSGD_iter'] += 1
            return out
        out = {}
        aggs = _tu.load_arrays(self.state['baseDirectory'] + _os.sep +
                               self.input['powerSGD_Q_AGG_file'])
        r1 = _tu.load_arrays(self.state['baseDirectory'] + _os.sep +
                             self.input['rank1_AGG_file'])
        params = self._param_tensors()
        rank1, mats = self._split(params)
        dev = self.device
        # rank-1 grads verbatim
        for (i, p), g in zip(rank1, r1):
            p.grad = _torch.tensor(_np.asarray(g), dtype=_torch.float32,
                                   device=dev).reshape(p.shape)
        # matrices: grad ~= P @ Q^T, error feedback update
        Ps = self.cache['powerSGD_Ps']
        error = self.cache.setdefault('powerSGD_error', {})
        Ms = self.cache['powerSGD_Ms']
        warm_qs = self.cache.setdefault('powerSGD_Qs', {})
        for (i, p), q in zip(mats, aggs):
            Q = _torch.tensor(_np.asarray(q), dtype=_torch.float32,
                              device=dev)
            P = Ps[i]
            recon = _ops.matmul_abT(P, Q)
            if self.use_error_feedback:
                error[i] = Ms[i] - recon
            if self.warm_start:  # reuse next round (reference powersgd:110)
                warm_qs[i] = Q
            p.grad = recon.reshape(p.shape).to(p.dtype)
        self.trainer.optimizer[self.first_optim].step()
        self.cache['powerSGD_iter'] += 1
        return out

    def to_reduce(self):
        if self._warmup:
            it, out = super().to_reduce()
            out['powerSGD_phase'] = PHASE_P
            return it, out
        phase = self.input.get('powerSGD_phase', PHASE_P)
        if phase == PHASE_P:
            return self._phase_P()
        return self._phase_Q()

    # NOTE: at an epoch boundary the remote's mode transition can drop a
    # half-finished P/Q round trip; training resumes at PHASE_P (the
    # reference engine behaves the same way — that batch's gradient is
    # carried by the error-feedback buffer, not lost).

    def _phase_P(self):
        it, out = self.backward()
        params = self._param_tensors()
        rank1, mats = self._split(params)
        dev = self.device
        error = self.cache.setdefault('powerSGD_error', {})
        Ms, Ps, r1 = {}, {}, []
        ship = []
        for i, p in mats:
            g = self._matrix_view(p.grad.detach().float())
            M = g + error[i] if (self.use_error_feedback and i in error) \
                else g
            warm_qs = self.cache.get('powerSGD_Qs', {})
            if self.warm_start and i in warm_qs:
                # warm start: re-orthogonalize the last averaged Q
                # (reference powersgd/__init__.py:110, 120-121)
                Q = warm_qs[i].clone()
            else:
                # generator re-seeded PER PARAM — bitwise parity with the
                # reference's Q init (powersgd/__init__.py:113-114 re-seeds
                # the global rng inside the param loop)
                gen = _torch.Generator(device='cpu').manual_seed(
                    int(self.seed) + int(self.cache['powerSGD_iter']))
                Q = _torch.randn(M.shape[1], self.rank, generator=gen).to(dev)
            orthogonalize(Q)
            P = _ops.matmul_ab(M, Q)
            Ms[i], Ps[i] = M, P
            ship.append(P.cpu().numpy().astype(self.dtype))
        for i, p in rank1:
            r1.append(p.grad.detach().float().cpu().numpy().astype(self.dtype))
        self.cache['powerSGD_Ms'] = Ms
        self.cache['powerSGD_Ps'] = Ps
        self.cache['powerSGD_rank1'] = r1
        _tu.save_arrays(self.state['transferDirectory'] + _os.sep + P_FILE,
                        ship)
        out['powerSGD_P_file'] = P_FILE
        out['powerSGD_phase'] = PHASE_P
        out['reduce'] = True
        return it, out

    def _phase_Q(self):
        """Averaged Ps arrived: orthogonalize, ship Q = M^T @ P + rank-1."""
        out = {}
        aggs = _tu.load_arrays(self.state['baseDirectory'] + _os.sep +
                               self.input['powerSGD_P_AGG_file'])
        dev = self.device
        Ms = self.cache['powerSGD_Ms']
        Ps = self.cache['powerSGD_Ps']
        ship = []
        for (i, M), p_avg in zip(sorted(Ms.items()), aggs):
            P = _torch.tensor(_np.asarray(p_avg), dtype=_torch.float32,
                              device=dev)
            orthogonalize(P)
            Ps[i] = P
            Q = _ops.matmul_aTb(M, P)
            ship.append(Q.cpu().numpy().astype(self.dtype))
        _tu.save_arrays(self.state['transferDirectory'] + _os.sep + Q_FILE,
                        ship)
        _tu.save_arrays(self.state['transferDirectory'] + _os.sep + R1_FILE,
                        self.cache.get('powerSGD_rank1', []))
        out['powerSGD_Q_file'] = Q_FILE
        out['rank1_file'] = R1_FILE
        out['powerSGD_phase'] = PHASE_Q
        out['reduce'] = True
        return {}, out


class PowerSGDReducer(COINNReducer):
    def reduce(self):
        out = {}
        some_site = list(self.input.values())[0]
        if some_site.get('powerSGD_phase') == PHASE_P and \
                some_site.get('powerSGD_P_file'):
            avg = self._average('powerSGD_P_file')
            _tu.save_arrays(self.state['transferDirectory'] + _os.sep + P_AGG,
                            avg)
            out['powerSGD_P_AGG_file'] = P_AGG
            out['powerSGD_phase'] = PHASE_Q
            return out
        if some_site.get('powerSGD_Q_file'):
            avg_q = self._average('powerSGD_Q_file')
            _tu.save_arrays(self.state['transferDirectory'] + _os.sep + Q_AGG,
                            avg_q)
            out['powerSGD_Q_AGG_file'] = Q_AGG
            avg_r1 = self._average('rank1_file')
            _tu.save_arrays(self.state['transferDirectory'] + _os.sep +
                            R1_AGG, avg_r1)
            out['rank1_AGG_file'] = R1_AGG
            out['powerSGD_phase'] = PHASE_P
            out['powerSGD_applied'] = True
            out['update'] = True
            return out
        # warmup rounds: plain dSGD averaging
        return super().reduce()
