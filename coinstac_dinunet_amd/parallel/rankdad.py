"""RCCL-native rankDAD: the file round trip collapsed into in-round
collectives.

The loopback engine (distrib/rankdad.py) ships per-layer factor pairs to
the aggregator, which concatenates them along the rank axis (the sum of
site outer products) and optionally re-compresses before relaying back —
one full extra round per step. On the persistent process group the whole
exchange runs inside to_reduce():

    backward -> per leaf: (gf, af) = power_iteration_BC(grad^T, act^T)
    -> all_gather both factors (variable widths: gather sizes, pad, slice)
    -> concat along rank == sum of site outer products
    -> if total width > rank: recompress with a SEEDED generator so every
       rank extracts bit-identical factors (no second collective needed)
    -> weight.grad = gf_cat @ af_cat^T, bias.grad = gf_cat rows summed

step() then just applies the optimizer. Identical math to the reference
engine (rankdad/spi.py: dad_backward + concat-reduce + synced_param_update);
communication drops from O(out*in) to O((out+in)*rank) per layer and the
aggregator round trip disappears.
"""
import torch
import torch.distributed as dist

from .. import ops as _ops
from ..distrib.rankdad import (DADLearner, _mm_flatten, power_iteration_BC)
from ..distrib.reducer import COINNReducer


class RcclDADLearner(DADLearner):
    """DADLearner whose data plane is all_gather instead of the file relay."""

    def __init__(self, trainer=None, mp_pool=None, **kw):
        super().__init__(trainer=trainer, mp_pool=mp_pool, **kw)
        self.rank_k = self.cache.setdefault('dad_reduction_rank', 10)
        self.num_pow_iters = self.cache.setdefault('dad_num_pow_iters', 5)
        self.dad_tol = self.cache.setdefault('dad_tol', 1e-3)
        self.seed = int(self.cache.get('seed', 0) or 0)
        self.cache.setdefault('dad_iter', 0)
        self.world = dist.get_world_size() if dist.is_initialized() else 1

    def step(self):
        out = {}
        self.trainer.optimizer[self.first_optim].step()
        self.cache['dad_iter'] += 1
        return out

    def to_reduce(self):
        it, out = self.forward()  # fw+bw with DAD hooks capturing (act, grad)
        model = self.trainer.nn[self.first_model]
        covered = set()
        for lix, (name, mod) in enumerate(model._leaves()[::-1]):
            if name not in model._grads or name not in model._activations:
                continue
            grad, act = _mm_flatten(model._grads[name].detach().float(),
                                    model._activations[name].detach().float())
            gf, af = power_iteration_BC(grad.t(), act.t(), self.rank_k,
                                        self.num_pow_iters, self.dad_tol)
            gf, af = self._gather_cat(gf), self._gather_cat(af)
            if gf.shape[1] > self.rank_k:
                # fresh per-(iter, layer) generator: the draw count of the
                # LOCAL compression is data-dependent, so a shared running
                # generator would desynchronize ranks.
                gen = torch.Generator().manual_seed(
                    self.seed + 7919 * int(self.cache['dad_iter']) + lix)
                gf, af = power_iteration_BC(gf, af, self.rank_k,
                                            self.num_pow_iters, self.dad_tol,
                                            generator=gen)
            params = dict(mod.named_parameters(recurse=False))
            w = params.get('weight')
            if w is not None:
                # K11: factor reconstruction gf[out,r] @ af[in,r]^T on the
                # in-tree MFMA kernel
                g = _ops.matmul_abT(gf, af).view_as(w)
                if w.grad is None:
                    w.grad = g
                else:
                    w.grad.copy_(g)
            b = params.get('bias')
            if b is not None:
                gb = _ops.row_sum(gf)
                if b.grad is None:
                    b.grad = gb
                else:
                    b.grad.copy_(gb)
            covered.update(id(p) for p in params.values() if p is not None)
        self._sync_uncovered(model, covered)
        out['reduce'] = True
        return it, out

    def _sync_uncovered(self, model, covered):
        """Norm layers (and any leaf the hooks missed) are outside the DAD
        factorization — the reference leaves their LOCAL gradients in place,
        silently de-synchronizing site weights (rankdad/spi.py skips
        BatchNorm*/LayerNorm). Here they are tiny vectors on a live process
        group, so all-reduce them (SUM — matching the scale of the factor
        concatenation, which sums site outer products) and keep sites
        bit-aligned."""
        if self.world <= 1:
            return
        rest = [p.grad for p in model.parameters()
                if id(p) not in covered and p.grad is not None]
        if not rest:
            return
        flat = torch.cat([g.reshape(-1).float() for g in rest])
        dist.all_reduce(flat)
        off = 0
        for g in rest:
            n = g.numel()
            g.copy_(flat[off:off + n].view_as(g))
            off += n

    def _gather_cat(self, t):
        """All-gather a [n, r_i] factor with per-rank widths; concat by rank
        order (identical on every rank) along the rank axis."""
        if self.world <= 1:
            return t
        widths = [None] * self.world
        dist.all_gather_object(widths, int(t.shape[1]))
        rmax = max(widths)
        padded = torch.zeros(t.shape[0], rmax, device=t.device, dtype=t.dtype)
        padded[:, :t.shape[1]] = t
        bufs = [torch.empty_like(padded) for _ in range(self.world)]
        dist.all_gather(bufs, padded)
        return torch.cat([bufs[i][:, :widths[i]]
                          for i in range(self.world)], 1)


class RcclDADReducer(COINNReducer):
    """Factor exchange + reconstruction happen on the ranks; remote flags."""

    def reduce(self):
        return {'update': True}
