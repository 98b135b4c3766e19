"""Per-site phase-machine driver.

API-parity: /root/reference/coinstac_dinunet/distrib/nodes/local.py:25-295
(COINNLocal): same constructor surface, phase dispatch, out-dict keys and
cache protocol, so reference computations drop in. Differences (deliberate):
  - engine selection additionally resolves the RCCL learner when the
    process runs inside the persistent MI355X engine (parallel/engine.py);
  - _pretrain_local actually applies pretrain_args overrides (the reference
    builds the merged cache and then discards it — local.py:155-160).
"""
import json as _json
import os as _os
import shutil as _shutil
import time as _time
import traceback as _tback
from os import sep as _sep

from ... import config as _conf
from ... import utils as _utils
from ...config.keys import AGG_Engine, Key, Mode, Phase
from ...data import COINNDataHandle as _DataHandle
from ...utils import FrozenDict as _FrozenDict
from ..learner import COINNLearner as _dSGDLearner


class COINNLocal:
    _PROMPT_TASK_ = 'Task id must be given.'
    _PROMPT_MODE_ = f'Mode must be provided and should be one of {[Mode.TRAIN, Mode.TEST]}.'

    def __init__(self, cache=None, input=None, state=None,
                 task_id='nn_task',
                 mode=None,
                 batch_size=8,
                 local_iterations=1,
                 epochs=31,
                 validation_epochs=1,
                 learning_rate=0.001,
                 gpus=None,
                 pin_memory=False,
                 num_workers=0,
                 load_limit=_conf.max_size,
                 load_sparse=False,
                 pretrained_path=None,
                 patience=None,
                 num_folds=None,
                 split_ratio=None,
                 pretrain_args=None,
                 dataloader_args=None,
                 verbose=False,
                 monitor_metric='f1',
                 metric_direction='maximize',
                 log_header='Loss|Accuracy,F1',
                 agg_engine=AGG_Engine.dSGD,
                 num_reducers=2,
                 precision_bits=32,
                 **kw):
        self.out = {}
        self.cache = cache if cache is not None else {}
        self.input = _FrozenDict(input if input is not None else {})
        self.state = _FrozenDict(state if state is not None else {})

        self._args = {
            'task_id': task_id, 'mode': mode, 'batch_size': batch_size,
            'local_iterations': local_iterations, 'epochs': epochs,
            'validation_epochs': validation_epochs,
            'learning_rate': learning_rate, 'gpus': gpus,
            'pin_memory': pin_memory, 'num_workers': num_workers,
            'load_limit': load_limit, 'load_sparse': load_sparse,
            'pretrained_path': pretrained_path,
            'patience': patience if patience else epochs,
            'split_ratio': split_ratio, 'num_folds': num_folds,
            'verbose': verbose, 'monitor_metric': monitor_metric,
            'metric_direction': metric_direction, 'log_header': log_header,
            'agg_engine': agg_engine, 'num_reducers': num_reducers,
            'precision_bits': precision_bits,
        }
        self._args.update(**kw)
        self._args = _FrozenDict(self._args)
        self._pretrain_args = pretrain_args if pretrain_args else {}
        self._dataloader_args = dataloader_args if dataloader_args else {}

        # Cache args once, with inputspec-driven overrides.
        if not self.cache.get(Key.ARGS_CACHED):
            self.cache.update(**self.input)
            task_args = self.input.get(f"{self.input.get('task_id')}_args", {})
            self.cache.update(**task_args)
            agg_engine_args = self.input.get(f"{self.input.get('agg_engine')}_args", {})
            self.cache.update(**agg_engine_args)
            data_conf = self.input.get(f"{self.input.get('task_id')}_data_conf", {})
            for k, v in data_conf.items():
                if k not in task_args and k not in agg_engine_args:
                    self.cache[k] = v
            for k in self._args:
                if self.cache.get(k) is None:
                    self.cache[k] = self._args[k]

            assert self.cache['task_id'] is not None, self._PROMPT_TASK_
            assert self.cache['mode'] in [Mode.TRAIN, Mode.TEST], self._PROMPT_MODE_
            if self.cache['mode'] == Mode.TRAIN:
                assert self.cache['split_ratio'] or self.cache['num_folds'], \
                    'Split ratio or K(num k-folds) is needed.'
            self.cache[Key.ARGS_CACHED] = True

    # ---- phase bodies ---------------------------------------------------
    def _init_runs(self, trainer):
        out = {}
        out.update(trainer.data_handle.prepare_data())
        self.cache['num_folds'] = len(self.cache['splits'])
        trainer.init_nn(set_devices=True)

        out['data_size'] = {}
        for k, sp in self.cache['splits'].items():
            with open(self.cache['split_dir'] + _sep + sp) as f:
                sp = _json.load(f)
            out['data_size'][k] = {key: len(sp.get(key, [])) for key in sp}
        return out

    def _next_run(self, trainer):
        out = {}
        self.cache.update(cursor=0)
        self.cache[Key.TRAIN_SERIALIZABLE] = []
        self.cache['split_file'] = self.cache['splits'][self.cache['split_ix']]
        self.cache['log_dir'] = _os.path.join(
            self.state['outputDirectory'], self.cache['task_id'],
            f"fold_{self.cache['split_ix']}")
        _os.makedirs(self.cache['log_dir'], exist_ok=True)

        trainer.init_nn(init_model=True, init_optim=True, set_devices=True,
                        init_weights=True)
        self.cache['best_nn_state'] = \
            f"best.{self.cache['task_id']}-{self.cache['split_ix']}.pt"
        self.cache['latest_nn_state'] = \
            f"latest.{self.cache['task_id']}-{self.cache['split_ix']}.pt"
        out['phase'] = Phase.COMPUTATION
        return out

    def _pretrain_local(self, trainer_cls, datahandle_cls, train_dataset,
                        validation_dataset):
        """Elected max-data site trains locally; its best weights ship via
        transferDirectory. Everyone then meets at PRE_COMPUTATION."""
        out = {'phase': Phase.COMPUTATION}
        pretrain_epochs = self._pretrain_args.get('epochs', 0)
        if pretrain_epochs > 0 and self.cache.get('pretrain'):
            # overlay pretrain_args onto the live cache for the duration of
            # local training, then restore (logs/best-state persist)
            overrides = dict(self.cache.get('pretrain_args',
                                            self._pretrain_args))
            saved = {k: self.cache.get(k) for k in overrides}
            self.cache.update(**overrides)
            try:
                trainer = trainer_cls(data_handle=datahandle_cls(
                    cache=self.cache, input=self.input, state=self.state,
                    dataloader_args=self._dataloader_args))
                trainer.init_nn()
                trainer.init_training_cache()
                out.update(**trainer.train_local(train_dataset,
                                                 validation_dataset))
            finally:
                self.cache.update(**saved)
            out['phase'] = Phase.PRE_COMPUTATION

        if pretrain_epochs > 0 and any(
                r.get('pretrain') for r in
                self.input.get('global_runs', {}).values()):
            out['phase'] = Phase.PRE_COMPUTATION
        return out

    # ---- main dispatch ---------------------------------------------------
    def compute(self, mp_pool, trainer_cls, dataset_cls=None,
                datahandle_cls=_DataHandle, learner_cls=_dSGDLearner, **kw):
        trainer = trainer_cls(data_handle=datahandle_cls(
            cache=self.cache, input=self.input, state=self.state,
            dataloader_args=self._dataloader_args))

        self.out['phase'] = self.input.get('phase', Phase.INIT_RUNS)
        if self.out['phase'] == Phase.INIT_RUNS:
            self.out.update(**self._init_runs(trainer))
            frozen_args = {k: self.cache[k] for k in self._args}
            self.cache['frozen_args'] = _FrozenDict(frozen_args)
            self.out['shared_args'] = dict(self.cache['frozen_args'])

        elif self.out['phase'] == Phase.NEXT_RUN:
            self.cache.update(**self.input['global_runs'][self.state['clientId']])
            self.out.update(**self._next_run(trainer))
            if self.cache['mode'] == Mode.TRAIN:
                self.out.update(**self._pretrain_local(
                    trainer_cls, datahandle_cls,
                    trainer.data_handle.get_train_dataset(dataset_cls),
                    trainer.data_handle.get_validation_dataset(dataset_cls)))

        elif self.out['phase'] == Phase.PRE_COMPUTATION:
            if self.input.get('pretrained_weights'):
                trainer.load_checkpoint(
                    file_path=self.state['baseDirectory'] + _sep +
                    self.input['pretrained_weights'])
            # Deviation from the reference (local.py:208-212): when the
            # elected site's pretraining never improved, no weights.tar
            # exists — the reference then echoes PRE_COMPUTATION forever
            # (site/remote deadlock). Proceed to COMPUTATION instead.
            self.out['phase'] = Phase.COMPUTATION

        learner = self._get_learner_cls(learner_cls)(trainer=trainer,
                                                     mp_pool=mp_pool)
        self.out['mode'] = learner.global_modes.get(
            self.state['clientId'], self.cache['mode'])

        if self.out['phase'] == Phase.COMPUTATION:
            if self.input.get('save_current_as_best'):
                learner.trainer.save_checkpoint(
                    file_path=self.cache['log_dir'] + _sep +
                    self.cache['best_nn_state'])

            if self.input.get('update'):
                self.out.update(**learner.step())

            if any(m == Mode.TRAIN for m in learner.global_modes.values()):
                # lagged sites go VALIDATION_WAITING and reshuffle; everyone
                # trains until the whole quorum is waiting.
                it, out = learner.to_reduce()
                self.out.update(**out)
                if it.get('averages') and it.get('metrics'):
                    self.cache[Key.TRAIN_SERIALIZABLE].append(
                        {'averages': it['averages'].serialize(),
                         'metrics': it['metrics'].serialize()})
                    self.out.update(**(trainer.on_iteration_end(0, 0, it) or {}))

            if all(m == Mode.VALIDATION for m in learner.global_modes.values()):
                self.out.update(**trainer.validation_distributed(dataset_cls))
                self.out[Key.TRAIN_SERIALIZABLE] = self.cache[Key.TRAIN_SERIALIZABLE]
                self.cache[Key.TRAIN_SERIALIZABLE] = []
                self.out['mode'] = Mode.TRAIN_WAITING

            if all(m == Mode.TEST for m in learner.global_modes.values()):
                self.out.update(**trainer.test_distributed(dataset_cls))
                self.out['mode'] = self.cache['frozen_args']['mode']
                self.out['phase'] = Phase.NEXT_RUN_WAITING
                trainer.save_checkpoint(
                    file_path=self.cache['log_dir'] + _sep +
                    self.cache['latest_nn_state'])
                _utils.save_cache(self.cache, self.cache['log_dir'])

        elif self.out['phase'] == Phase.SUCCESS:
            zip_path = f"{self.state['baseDirectory']}{_sep}{self.input['results_zip']}.zip"
            for i in range(3):
                _time.sleep(i)
                if _os.path.exists(zip_path):
                    _shutil.copy(zip_path,
                                 f"{self.state['outputDirectory']}{_sep}"
                                 f"{self.input['results_zip']}.zip")
                    break

    def _get_learner_cls(self, learner_cls):
        engine = self.cache.get('agg_engine')
        if engine == AGG_Engine.dSGD:
            return _dSGDLearner
        if engine == AGG_Engine.rankDAD:
            from ..rankdad import DADLearner
            return DADLearner
        if engine == AGG_Engine.powerSGD:
            from ..powersgd import PowerSGDLearner
            return PowerSGDLearner
        return learner_cls

    def __call__(self, *args, **kwargs):
        t0 = _time.time()
        try:
            self.compute(*args, **kwargs)
            timings = self.cache.setdefault('round_timings', [])
            timings.append([self.out.get('phase'), self.out.get('mode'),
                            round(_time.time() - t0, 4)])
            del timings[:-1000]  # bounded trace
            return {'output': self.out}
        except Exception:
            _tback.print_exc()
            raise Exception(self.out)
