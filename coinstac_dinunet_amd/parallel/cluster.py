"""RcclCluster: the COINSTAC phase machine on a persistent process group.

One process per GPU "site" (torchrun); rank 0 doubles as the remote
aggregator. Per round:
  1. every rank runs COINNLocal.compute with learner_cls=RcclLearner
     (gradient rounds are RCCL all-reduce over xGMI — no files);
  2. the tiny control dicts are gathered to rank 0 (gloo object gather);
  3. rank 0 runs COINNRemote with reducer_cls=RcclReducer (pure control:
     epoch/mode/fold transitions, score reduction, best-model signal);
  4. rank 0 relays any cold-path artifacts (pretrained weights.tar,
     results zip) over the node-local filesystem and broadcasts the
     remote out-dict, which becomes every rank's next input.

The hot path therefore exchanges ONLY the fused gradient bucket per round;
control costs one object gather+broadcast of a few KB.

Scope: one node (the BASELINE configs — 8 GPU-sites over xGMI). The
collectives themselves are multi-node-clean (torchrun + MASTER_ADDR),
but the COLD path (pretrained weights.tar, results zip) relays over
`root`, which must then be a shared filesystem; with a node-local root,
keep the group on one node.
"""
import os
import shutil

import torch.distributed as dist

from ..distrib.nodes.local import COINNLocal
from ..distrib.nodes.remote import COINNRemote
from .engine import RcclLearner, RcclReducer, init_distributed


class RankSiteState:
    """Per-rank site directories under one shared (node-local) root."""

    def __init__(self, root, rank):
        self.clientId = f'rank{rank}'
        base = os.path.join(root, self.clientId)
        self.baseDirectory = os.path.join(base, 'input')
        self.transferDirectory = os.path.join(base, 'transfer')
        self.outputDirectory = os.path.join(base, 'output')
        for d in (self.baseDirectory, self.transferDirectory,
                  self.outputDirectory):
            os.makedirs(d, exist_ok=True)

    def as_dict(self):
        return {'clientId': self.clientId,
                'baseDirectory': self.baseDirectory,
                'transferDirectory': self.transferDirectory,
                'outputDirectory': self.outputDirectory}


class RcclCluster:
    def __init__(self, root, local_kw=None, remote_kw=None, timeout_s=300):
        # unlike the reference (a crashed site stalls the all-site quorum
        # forever — SURVEY §5.3), the process group carries a timeout: a
        # dead rank aborts the collective with an error on every survivor
        self.rank, self.world_size = init_distributed(timeout_s=timeout_s)
        self.root = root
        self.site = RankSiteState(root, self.rank)
        self.remote_state = {
            'clientId': 'remote',
            'baseDirectory': os.path.join(root, 'remote', 'input'),
            'transferDirectory': os.path.join(root, 'remote', 'transfer'),
            'outputDirectory': os.path.join(root, 'remote', 'output')}
        if self.rank == 0:
            for d in self.remote_state.values():
                if d != 'remote':
                    os.makedirs(d, exist_ok=True)
        self.local_kw = local_kw or {}
        self.remote_kw = remote_kw or {}
        self.site_cache = {}
        self.remote_cache = {}
        self.input = {}
        self.rounds = 0

    # -- filesystem relay (cold path only) --------------------------------
    def _site_transfer_to_remote(self, site_outs):
        for site_id in site_outs:
            src = os.path.join(self.root, site_id, 'transfer')
            dst = os.path.join(self.remote_state['baseDirectory'], site_id)
            os.makedirs(dst, exist_ok=True)
            for name in os.listdir(src):
                s, d = os.path.join(src, name), os.path.join(dst, name)
                if os.path.isdir(d):
                    shutil.rmtree(d)
                elif os.path.exists(d):
                    os.remove(d)
                shutil.move(s, d)

    def _remote_transfer_to_sites(self, site_outs):
        src = self.remote_state['transferDirectory']
        for name in os.listdir(src):
            for site_id in site_outs:
                dst_dir = os.path.join(self.root, site_id, 'input')
                s, d = os.path.join(src, name), os.path.join(dst_dir, name)
                if os.path.isfile(s):
                    shutil.copy(s, d)
            p = os.path.join(src, name)
            shutil.rmtree(p) if os.path.isdir(p) else os.remove(p)

    # -- one protocol round -------------------------------------------------
    def run_round(self, trainer_cls, dataset_cls=None, mp_pool=None,
                  learner_cls=RcclLearner, reducer_cls=RcclReducer, **kw):
        local = COINNLocal(cache=self.site_cache, input=dict(self.input),
                           state=self.site.as_dict(), **self.local_kw)
        result = local(mp_pool, trainer_cls, dataset_cls=dataset_cls,
                       learner_cls=learner_cls, **kw)
        my_out = result['output']

        gathered = [None] * self.world_size
        dist.all_gather_object(gathered, (self.site.clientId, my_out))
        site_outs = dict(gathered)

        bcast = [None]
        success = False
        if self.rank == 0:
            self._site_transfer_to_remote(site_outs)
            remote = COINNRemote(cache=self.remote_cache, input=site_outs,
                                 state=self.remote_state, **self.remote_kw)
            rres = remote(mp_pool, trainer_cls, reducer_cls=reducer_cls)
            self._remote_transfer_to_sites(site_outs)
            bcast = [(rres['output'], rres.get('success', False))]
        dist.broadcast_object_list(bcast, src=0)
        remote_out, success = bcast[0]
        self.input = dict(remote_out)
        self.rounds += 1
        return success, remote_out

    def run(self, trainer_cls, dataset_cls=None, mp_pool=None,
            max_rounds=10000, **kw):
        success, out = False, {}
        for _ in range(max_rounds):
            success, out = self.run_round(trainer_cls, dataset_cls=dataset_cls,
                                          mp_pool=mp_pool, **kw)
            if success:
                break
        return success, out


# --------------------------------------------------------------------------
# Mid-run checkpoint/resume (a capability the reference lacks: its "resume"
# granularity is one platform iteration and a crashed run restarts the fold
# — SURVEY §5.4). Here every rank can snapshot its whole protocol state
# (cache scalars + model/optimizer state_dicts + the pending input dict)
# and resume lock-step from the next round.
# --------------------------------------------------------------------------
_UNPICKLABLE_KEYS = ('train_loader_iter', 'nn', 'device', 'optimizer',
                     'dataset', '_rccl_grad_buffer', 'frozen_args')


def save_cluster_state(cluster, path):
    import torch as _torch
    cache = {k: v for k, v in cluster.site_cache.items()
             if k not in _UNPICKLABLE_KEYS}
    models = {k: m.state_dict()
              for k, m in cluster.site_cache.get('nn', {}).items()}
    optims = {k: o.state_dict()
              for k, o in cluster.site_cache.get('optimizer', {}).items()}
    frozen = dict(cluster.site_cache.get('frozen_args', {}))
    state = {'cache': cache, 'models': models, 'optimizers': optims,
             'frozen_args': frozen, 'input': dict(cluster.input),
             'rounds': cluster.rounds}
    if cluster.rank == 0:
        remote_cache = {k: v for k, v in cluster.remote_cache.items()
                        if k not in _UNPICKLABLE_KEYS}
        state['remote_cache'] = remote_cache
    _torch.save(state, path)


def load_cluster_state(cluster, path, trainer_cls, dataset_cls=None):
    """Restore a snapshot into a freshly constructed RcclCluster."""
    import torch as _torch
    from ..data import COINNDataHandle
    from ..utils import FrozenDict
    state = _torch.load(path, weights_only=False)
    cluster.site_cache.clear()
    cluster.site_cache.update(state['cache'])
    cluster.site_cache['cursor'] = 0  # loader iterator cannot be restored
    if state.get('frozen_args'):
        cluster.site_cache['frozen_args'] = FrozenDict(state['frozen_args'])
    cluster.input = dict(state['input'])
    cluster.rounds = state['rounds']
    if cluster.rank == 0 and 'remote_cache' in state:
        cluster.remote_cache.clear()
        cluster.remote_cache.update(state['remote_cache'])
    # rebuild model/optimizer into the cache registries
    handle = COINNDataHandle(cache=cluster.site_cache, input=cluster.input,
                             state=cluster.site.as_dict())
    trainer = trainer_cls(data_handle=handle)
    trainer.init_nn(init_model=True, init_optim=True, set_devices=True)
    for k, sd in state['models'].items():
        if k in trainer.nn:
            m = trainer.nn[k]
            (m.module if hasattr(m, 'module') else m).load_state_dict(sd)
    for k, sd in state['optimizers'].items():
        if k in trainer.optimizer:
            trainer.optimizer[k].load_state_dict(sd)
    # rebuild the dataset registry (loader iterators don't survive)
    if dataset_cls is not None and cluster.site_cache.get('split_file'):
        handle.get_train_dataset(dataset_cls)
        handle.get_validation_dataset(dataset_cls)
    return cluster
