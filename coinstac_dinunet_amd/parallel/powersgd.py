"""RCCL-native PowerSGD: the two-phase P/Q file round trip collapsed into
in-round collectives.

The loopback engine (distrib/powersgd.py) needs two COMPUTATION rounds per
step because the file relay is half-duplex. On the persistent process
group the whole compression step runs inside to_reduce():

    backward -> M = grad + error -> P = M @ Q0 -> all_reduce(P)
    -> orthogonalize(P^) -> Q = M^T @ P^ -> all_reduce(Q, rank-1 grads)
    -> grad ~= P^ @ Q^T, error = M - grad -> (step() applies optimizer)

Identical math to the reference engine (and torch's DDP PowerSGD hook);
communication drops from full-gradient to (n+m)*rank per matrix.
"""
import torch
import torch.distributed as dist

from .. import ops as _ops
from ..distrib.learner import COINNLearner
from ..distrib.powersgd import orthogonalize
from ..distrib.reducer import COINNReducer


class RcclPowerSGDLearner(COINNLearner):
    def __init__(self, trainer=None, mp_pool=None, **kw):
        super().__init__(trainer=trainer, mp_pool=mp_pool, **kw)
        self.rank_approx = self.cache.get('matrix_approximation_rank', 1)
        self.start_iter = self.cache.get('start_powerSGD_iter', 10)
        self.use_error_feedback = self.cache.get('use_error_feedback', True)
        self.seed = self.cache.get('seed', 0)
        self.cache.setdefault('powerSGD_iter', 0)
        self.world = dist.get_world_size() if dist.is_initialized() else 1

    def _params(self):
        return list(self.trainer.nn[self.first_model].parameters())

    def to_reduce(self):
        it, out = self.backward()
        if self.cache['powerSGD_iter'] >= self.start_iter:
            self._compress_round()
            out['powerSGD_applied'] = True
        else:
            self._plain_allreduce()
        out['reduce'] = True
        return it, out

    def step(self):
        self.trainer.optimizer[self.first_optim].step()
        self.cache['powerSGD_iter'] += 1
        return {}

    def _plain_allreduce(self):
        grads = [p.grad for p in self._params() if p.grad is not None]
        if self.world > 1:
            flat = torch.cat([g.reshape(-1).float() for g in grads])
            dist.all_reduce(flat)
            flat.div_(self.world)
            off = 0
            for g in grads:
                n = g.numel()
                g.copy_(flat[off:off + n].view_as(g))
                off += n

    def _compress_round(self):
        error = self.cache.setdefault('powerSGD_error', {})
        mats, rank1 = [], []
        for i, p in enumerate(self._params()):
            if p.grad is None:
                continue
            (rank1 if p.ndim <= 1 else mats).append((i, p))

        Ms, Ps = {}, {}
        for i, p in mats:
            g = p.grad.detach().float().reshape(p.shape[0], -1)
            M = g + error[i] if (self.use_error_feedback and i in error) else g
            # per-param re-seed: numerically identical Q to the loopback
            # engine and the reference (powersgd/__init__.py:113-114)
            gen = torch.Generator(device='cpu').manual_seed(
                int(self.seed) + int(self.cache['powerSGD_iter']))
            Q0 = torch.randn(M.shape[1], self.rank_approx,
                             generator=gen).to(M.device)
            orthogonalize(Q0)
            P = _ops.matmul_ab(M, Q0)  # K9: skinny GEMM on MFMA kernel
            Ms[i], Ps[i] = M, P
        # round 1: average Ps
        if self.world > 1 and Ps:
            flatP = torch.cat([P.reshape(-1) for _, P in sorted(Ps.items())])
            dist.all_reduce(flatP)
            flatP.div_(self.world)
            off = 0
            for i, P in sorted(Ps.items()):
                n = P.numel()
                Ps[i] = flatP[off:off + n].view_as(P)
                off += n
        # local: orthogonalize P^, Q = M^T P^
        Qs = {}
        for i in Ps:
            orthogonalize(Ps[i])
            Qs[i] = _ops.matmul_aTb(Ms[i], Ps[i])  # K9
        # round 2: average Qs + rank-1 grads
        r1 = [p.grad.detach().float().reshape(-1) for _, p in rank1]
        pieces = [Q.reshape(-1) for _, Q in sorted(Qs.items())] + r1
        if self.world > 1 and pieces:
            flat = torch.cat(pieces)
            dist.all_reduce(flat)
            flat.div_(self.world)
            off = 0
            for i, Q in sorted(Qs.items()):
                n = Q.numel()
                Qs[i] = flat[off:off + n].view_as(Q)
                off += n
            for (j, p), orig in zip(rank1, r1):
                n = p.numel()
                p.grad.copy_(flat[off:off + n].view_as(p))
                off += n
        # reconstruct + error feedback
        for i, p in mats:
            recon = _ops.matmul_abT(Ps[i], Qs[i])  # K9
            if self.use_error_feedback:
                error[i] = Ms[i] - recon
            p.grad.copy_(recon.view_as(p))


class RcclPowerSGDReducer(COINNReducer):
    """Compression + averaging happen on the ranks; remote only flags."""

    def reduce(self):
        return {'update': True}
