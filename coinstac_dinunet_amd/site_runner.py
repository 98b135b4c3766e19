"""Standalone single-site harness (debug/profile without the platform).

Parity: /root/reference/coinstac_dinunet/site_runner.py:8-45 (SiteRunner):
fabricates cache/state mimicking the simulator layout
(.../local<i>/simulatorRun), runs INIT_RUNS then NEXT_RUN with
pretrain=True so the whole training happens locally via _pretrain_local.
"""
import json as _json
import os as _os

from .config.keys import Phase
from .distrib.nodes.local import COINNLocal


class SiteRunner:
    def __init__(self, task_id, data_path, site_index=0, inputspec='inputspec.json',
                 **kw):
        self.task_id = task_id
        self.data_path = data_path
        self.site_index = site_index
        self.cache = {}
        self.kw = kw

        spec_path = _os.path.join(data_path, inputspec)
        self.inputspec = {}
        if _os.path.exists(spec_path):
            with open(spec_path) as f:
                spec = _json.load(f)
            # platform layout: a LIST with one {key: {"value": ...}} dict
            # per site (reference site_runner.py:13-15); a bare dict is
            # accepted too for hand-written specs
            if isinstance(spec, list):
                spec = spec[site_index]
            for k, v in spec.items():
                self.inputspec[k] = v.get('value') if isinstance(v, dict) else v

        base = _os.path.join(data_path, 'input', f'local{site_index}', 'simulatorRun')
        out = _os.path.join(data_path, 'output', f'local{site_index}', 'simulatorRun')
        transfer = _os.path.join(data_path, 'transfer', f'local{site_index}')
        for d in (out, transfer):
            _os.makedirs(d, exist_ok=True)
        self.state = {'clientId': f'local{site_index}',
                      'baseDirectory': base,
                      'transferDirectory': transfer,
                      'outputDirectory': out,
                      'cacheDirectory': out}

    def run(self, trainer_cls, dataset_cls, datahandle_cls=None, **kw):
        from .data import COINNDataHandle
        datahandle_cls = datahandle_cls or COINNDataHandle
        pretrain_args = kw.pop('pretrain_args', None) or \
            {'epochs': self.inputspec.get('epochs', self.kw.get('epochs', 5))}

        # INIT_RUNS
        input0 = {**self.inputspec, 'phase': Phase.INIT_RUNS}
        local = COINNLocal(cache=self.cache, input=input0, state=self.state,
                           task_id=self.task_id,
                           pretrain_args=pretrain_args, **self.kw, **kw)
        local(None, trainer_cls, dataset_cls=dataset_cls,
              datahandle_cls=datahandle_cls)

        # NEXT_RUN with fake global_runs forcing local pretraining
        input1 = {'phase': Phase.NEXT_RUN,
                  'global_runs': {self.state['clientId']: {
                      'split_ix': '0', 'seed': self.cache.get('seed', 0),
                      'pretrain': True}}}
        local = COINNLocal(cache=self.cache, input=input1, state=self.state,
                           task_id=self.task_id,
                           pretrain_args=pretrain_args, **self.kw, **kw)
        local(None, trainer_cls, dataset_cls=dataset_cls,
              datahandle_cls=datahandle_cls)
        return self.cache
