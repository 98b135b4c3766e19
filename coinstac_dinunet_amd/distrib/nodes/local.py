"""Per-site phase-machine driver.

Behavior-parity: /root/reference/coinstac_dinunet/distrib/nodes/local.py
:25-295 — same constructor surface, phase dispatch, out-dict keys and
cache protocol, so reference computations drop in. Organized as phase
handlers instead of the reference's inline chain. Deliberate deviations:
  - PRE_COMPUTATION proceeds to COMPUTATION even when no pretrained
    weights exist (the reference deadlocks if pretraining never improved);
  - _pretrain_local really applies pretrain_args overrides (the reference
    builds the merged cache and discards it — local.py:155-160);
  - per-round wall timings are recorded into cache['round_timings'].
"""
import json as _json
import os as _os
import shutil as _shutil
import time as _time
import traceback as _tback

from ... import config as _conf
from ... import utils as _utils
from ...config.keys import AGG_Engine, Key, Mode, Phase
from ...data import COINNDataHandle as _DataHandle
from ...utils import FrozenDict as _FrozenDict
from ..learner import COINNLearner as _dSGDLearner

# constructor surface (reference local.py:29-55); values None = "take the
# inputspec's value if present, else this default"
_ARG_DEFAULTS = dict(
    task_id='nn_task', mode=None, batch_size=8, local_iterations=1,
    epochs=31, validation_epochs=1, learning_rate=0.001, gpus=None,
    pin_memory=False, num_workers=0, load_limit=_conf.max_size,
    load_sparse=False, pretrained_path=None, patience=None, num_folds=None,
    split_ratio=None, verbose=False, monitor_metric='f1',
    metric_direction='maximize', log_header='Loss|Accuracy,F1',
    agg_engine=AGG_Engine.dSGD, num_reducers=2, precision_bits=32)


class COINNLocal:
    _PROMPT_TASK_ = 'Task id must be given.'
    _PROMPT_MODE_ = (f'Mode must be provided and should be one of '
                     f'{[Mode.TRAIN, Mode.TEST]}.')

    def __init__(self, cache=None, input=None, state=None,
                 pretrain_args=None, dataloader_args=None, **kw):
        self.out = {}
        self.cache = cache if cache is not None else {}
        self.input = _FrozenDict(input if input is not None else {})
        self.state = _FrozenDict(state if state is not None else {})

        args = dict(_ARG_DEFAULTS)
        args.update(**kw)
        if not args.get('patience'):
            args['patience'] = args['epochs']
        self._args = _FrozenDict(args)
        self._pretrain_args = pretrain_args if pretrain_args else {}
        self._dataloader_args = dataloader_args if dataloader_args else {}
        self._cache_args_once()

    def _cache_args_once(self):
        """Three-source precedence resolved once per run: platform input
        (incl. nested task/engine/data-conf blocks) > ctor defaults."""
        if self.cache.get(Key.ARGS_CACHED):
            return
        self.cache.update(**self.input)
        task_args = self.input.get(f"{self.input.get('task_id')}_args", {})
        self.cache.update(**task_args)
        engine_args = self.input.get(
            f"{self.input.get('agg_engine')}_args", {})
        self.cache.update(**engine_args)
        data_conf = self.input.get(
            f"{self.input.get('task_id')}_data_conf", {})
        for k, v in data_conf.items():
            if k not in task_args and k not in engine_args:
                self.cache[k] = v
        for k in self._args:
            if self.cache.get(k) is None:
                self.cache[k] = self._args[k]

        assert self.cache['task_id'] is not None, self._PROMPT_TASK_
        assert self.cache['mode'] in (Mode.TRAIN, Mode.TEST), \
            self._PROMPT_MODE_
        if self.cache['mode'] == Mode.TRAIN:
            assert self.cache['split_ratio'] or self.cache['num_folds'], \
                'Split ratio or K(num k-folds) is needed.'
        self.cache[Key.ARGS_CACHED] = True

    # =====================================================================
    # compute
    # =====================================================================
    def compute(self, mp_pool, trainer_cls, dataset_cls=None,
                datahandle_cls=_DataHandle, learner_cls=_dSGDLearner, **kw):
        trainer = trainer_cls(data_handle=datahandle_cls(
            cache=self.cache, input=self.input, state=self.state,
            dataloader_args=self._dataloader_args))

        phase = self.out['phase'] = self.input.get('phase', Phase.INIT_RUNS)
        if phase == Phase.INIT_RUNS:
            self._handle_init_runs(trainer)
        elif phase == Phase.NEXT_RUN:
            self._handle_next_run(trainer, trainer_cls, dataset_cls,
                                  datahandle_cls)
        elif phase == Phase.PRE_COMPUTATION:
            self._handle_pre_computation(trainer)
        elif phase == Phase.SUCCESS:
            self._handle_success()

        learner = self._get_learner_cls(learner_cls)(trainer=trainer,
                                                     mp_pool=mp_pool)
        self.out['mode'] = learner.global_modes.get(
            self.state['clientId'], self.cache['mode'])

        if self.out['phase'] == Phase.COMPUTATION:
            self._handle_computation(trainer, learner, dataset_cls)

    # ---- INIT_RUNS -------------------------------------------------------
    def _handle_init_runs(self, trainer):
        self.out.update(trainer.data_handle.prepare_data() or {})
        self.cache['num_folds'] = len(self.cache['splits'])
        trainer.init_nn(set_devices=True)

        sizes = {}
        for fold_key, split_name in self.cache['splits'].items():
            with open(_os.path.join(self.cache['split_dir'],
                                    split_name)) as f:
                split = _json.load(f)
            sizes[fold_key] = {k: len(split.get(k, [])) for k in split}
        self.out['data_size'] = sizes

        # freeze + share the resolved hyperparameters with the remote
        frozen = _FrozenDict({k: self.cache[k] for k in self._args})
        self.cache['frozen_args'] = frozen
        self.out['shared_args'] = dict(frozen)

    # ---- NEXT_RUN --------------------------------------------------------
    def _handle_next_run(self, trainer, trainer_cls, dataset_cls,
                         datahandle_cls):
        my_run = self.input['global_runs'][self.state['clientId']]
        self.cache.update(**my_run)
        self.cache.update(cursor=0)
        self.cache[Key.TRAIN_SERIALIZABLE] = []
        self.cache['split_file'] = \
            self.cache['splits'][self.cache['split_ix']]
        fold_tag = f"{self.cache['task_id']}-{self.cache['split_ix']}"
        self.cache['log_dir'] = _os.path.join(
            self.state['outputDirectory'], self.cache['task_id'],
            f"fold_{self.cache['split_ix']}")
        _os.makedirs(self.cache['log_dir'], exist_ok=True)
        self.cache['best_nn_state'] = f'best.{fold_tag}.pt'
        self.cache['latest_nn_state'] = f'latest.{fold_tag}.pt'

        trainer.init_nn(init_model=True, init_optim=True, set_devices=True,
                        init_weights=True)
        self.out['phase'] = Phase.COMPUTATION

        if self.cache['mode'] == Mode.TRAIN:
            self._pretrain_local(
                trainer_cls, datahandle_cls,
                trainer.data_handle.get_train_dataset(dataset_cls),
                trainer.data_handle.get_validation_dataset(dataset_cls))

    def _pretrain_local(self, trainer_cls, datahandle_cls, train_dataset,
                        validation_dataset):
        """Elected max-data site trains locally; its best weights ship via
        transferDirectory. Everyone then meets at PRE_COMPUTATION."""
        epochs = self._pretrain_args.get('epochs', 0)
        if epochs <= 0:
            return
        if self.cache.get('pretrain'):
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
                self.out.update(**trainer.train_local(train_dataset,
                                                      validation_dataset))
            finally:
                self.cache.update(**saved)
            self.out['phase'] = Phase.PRE_COMPUTATION
        elif any(r.get('pretrain') for r in
                 self.input.get('global_runs', {}).values()):
            # someone else pretrains: meet them at PRE_COMPUTATION
            self.out['phase'] = Phase.PRE_COMPUTATION

    # ---- PRE_COMPUTATION -------------------------------------------------
    def _handle_pre_computation(self, trainer):
        if self.input.get('pretrained_weights'):
            trainer.load_checkpoint(
                file_path=_os.path.join(self.state['baseDirectory'],
                                        self.input['pretrained_weights']))
        # Deviation from the reference (local.py:208-212): when the elected
        # site's pretraining never improved, no weights.tar exists — the
        # reference then echoes PRE_COMPUTATION forever (site/remote
        # deadlock). Proceed to COMPUTATION instead.
        self.out['phase'] = Phase.COMPUTATION

    # ---- COMPUTATION -----------------------------------------------------
    def _handle_computation(self, trainer, learner, dataset_cls):
        modes = learner.global_modes.values()

        if self.input.get('save_current_as_best'):
            trainer.save_checkpoint(file_path=_os.path.join(
                self.cache['log_dir'], self.cache['best_nn_state']))

        if self.input.get('update'):
            self.out.update(**learner.step())

        if any(m == Mode.TRAIN for m in modes):
            # all sites train together; a site that exhausts its data goes
            # VALIDATION_WAITING, reshuffles, and keeps contributing until
            # the whole quorum is waiting
            it, out = learner.to_reduce()
            self.out.update(**out)
            if it.get('averages') and it.get('metrics'):
                self.cache[Key.TRAIN_SERIALIZABLE].append(
                    {'averages': it['averages'].serialize(),
                     'metrics': it['metrics'].serialize()})
                self.out.update(**(trainer.on_iteration_end(0, 0, it) or {}))

        if all(m == Mode.VALIDATION for m in modes):
            self.out.update(**trainer.validation_distributed(dataset_cls))
            self.out[Key.TRAIN_SERIALIZABLE] = \
                self.cache[Key.TRAIN_SERIALIZABLE]
            self.cache[Key.TRAIN_SERIALIZABLE] = []
            self.out['mode'] = Mode.TRAIN_WAITING

        if all(m == Mode.TEST for m in modes):
            self.out.update(**trainer.test_distributed(dataset_cls))
            self.out['mode'] = self.cache['frozen_args']['mode']
            self.out['phase'] = Phase.NEXT_RUN_WAITING
            trainer.save_checkpoint(file_path=_os.path.join(
                self.cache['log_dir'], self.cache['latest_nn_state']))
            _utils.save_cache(self.cache, self.cache['log_dir'])

    # ---- SUCCESS ---------------------------------------------------------
    def _handle_success(self):
        """Copy the remote's results zip into this site's outputs."""
        name = f"{self.input['results_zip']}.zip"
        src = _os.path.join(self.state['baseDirectory'], name)
        for attempt in range(3):
            _time.sleep(attempt)
            if _os.path.exists(src):
                _shutil.copy(src, _os.path.join(
                    self.state['outputDirectory'], name))
                break

    # ---- engine selection --------------------------------------------------
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
